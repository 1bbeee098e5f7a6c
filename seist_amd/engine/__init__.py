from .train import train_worker
from .test import test_worker
from .validate import validate
from .metrics import Metrics
from .postprocess import ResultSaver, detect_peaks, process_outputs, trigger_onset

__all__ = ["train_worker", "test_worker", "validate", "Metrics",
           "process_outputs", "ResultSaver", "detect_peaks", "trigger_onset"]
