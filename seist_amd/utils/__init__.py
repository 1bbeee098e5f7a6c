from .logger import logger
from .meters import AverageMeter, ProgressMeter
from .visualization import vis_phase_picking, vis_waves_preds_targets
from .misc import (
    cal_snr,
    count_parameters,
    get_safe_path,
    get_time_str,
    setup_seed,
    strfargs,
    strftimedelta,
)

__all__ = [
    "logger", "AverageMeter", "ProgressMeter", "cal_snr", "count_parameters",
    "get_safe_path", "get_time_str", "setup_seed", "strfargs", "strftimedelta",
    "vis_phase_picking", "vis_waves_preds_targets",
]
