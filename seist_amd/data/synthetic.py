"""Synthetic in-memory seismic dataset.

Not present in the reference (which ships no tests and requires the real
DiTing/PNW archives). Generates deterministic random waveforms with
plausible P/S arrivals and all label fields, so the full train/val/test
pipeline, the test-suite and `bench.py` run without network or datasets.
Waveform shape and sampling rate mirror DiTing (3 x L @ 50 Hz).
"""

from typing import Tuple

import numpy as np
import pandas as pd

from .base import DatasetBase
from .registry import register_dataset


class Synthetic(DatasetBase):
    _name = "synthetic"
    _part_range = None
    _channels = ["z", "n", "e"]
    _sampling_rate = 50

    def __init__(self, seed, mode, data_dir="", shuffle=True, data_split=True,
                 train_size=0.8, val_size=0.1, size: int = 256,
                 num_samples: int = 12288, **kwargs):
        self._size = int(size)
        self._num_samples = int(num_samples)
        super().__init__(seed=seed, mode=mode, data_dir=data_dir,
                         shuffle=shuffle, data_split=data_split,
                         train_size=train_size, val_size=val_size)

    def _load_meta_data(self) -> pd.DataFrame:
        n = self._size
        meta = pd.DataFrame({"idx": np.arange(n)})
        if self._shuffle:
            meta = meta.sample(frac=1, replace=False, random_state=self._seed)
        meta.reset_index(drop=True, inplace=True)
        if self._data_split:
            lo, hi = self._split_rows(n)
            meta = meta.iloc[lo:hi, :]
        return meta

    def _load_event_data(self, idx: int) -> Tuple[dict, dict]:
        row = self._meta_data.iloc[idx]
        gid = int(row["idx"])
        rng = np.random.default_rng(self._seed * 1_000_003 + gid)
        L = self._num_samples
        C = len(self._channels)

        data = rng.standard_normal((C, L)).astype(np.float32) * 0.05
        ppk = int(rng.integers(L // 8, L // 2))
        spk = int(ppk + rng.integers(L // 16, L // 4))
        # P/S wavelets: decaying sinusoids
        for pk, amp, freq in ((ppk, 1.0, 0.08), (spk, 1.6, 0.05)):
            dur = min(L - pk, int(rng.integers(200, 800)))
            t = np.arange(dur, dtype=np.float32)
            wavelet = amp * np.exp(-t / (dur / 4)) * np.sin(
                2 * np.pi * freq * t)
            data[:, pk:pk + dur] += wavelet * rng.uniform(
                0.5, 1.0, size=(C, 1)).astype(np.float32)

        event = {
            "data": data,
            "ppks": [ppk],
            "spks": [spk],
            "emg": [float(rng.uniform(1.0, 6.0))],
            "smg": [float(rng.uniform(1.0, 6.0))],
            "pmp": [int(rng.integers(0, 2))],
            "clr": [int(rng.integers(0, 2))],
            "baz": [float(rng.uniform(0.0, 360.0))],
            "dis": [float(rng.uniform(5.0, 300.0))],
            "snr": np.array([20.0, 20.0, 20.0]),
        }
        meta = {"idx": gid, "ev_id": gid}
        return event, meta


@register_dataset
def synthetic(**kwargs):
    return Synthetic(**kwargs)
