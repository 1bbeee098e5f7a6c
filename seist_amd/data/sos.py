"""SOS dataset reader — single-channel 500 Hz picks in npz files,
pre-split on disk into train/val/test directories.

Parity with /root/reference/datasets/sos.py, with its latent attribute bug
fixed (sos.py:71 reads ``self.data_dir``/``self.mode`` which do not exist;
here the private attributes are used so the reader actually works).
"""

import os
from typing import Tuple

import numpy as np
import pandas as pd

from ..utils.logger import logger
from ..utils.misc import cal_snr
from .base import DatasetBase
from .registry import register_dataset


class SOS(DatasetBase):
    _name = "sos"
    _part_range = None
    _channels = ["z"]
    _sampling_rate = 500

    def __init__(self, seed, mode, data_dir, shuffle=True, data_split=False,
                 train_size=0.8, val_size=0.1, **kwargs):
        super().__init__(seed=seed, mode=mode, data_dir=data_dir,
                         shuffle=shuffle, data_split=data_split,
                         train_size=train_size, val_size=val_size)

    def _load_meta_data(self) -> pd.DataFrame:
        if self._data_split:
            logger.warning(
                "dataset 'sos' is pre-split on disk; 'data_split' ignored.")
        csv_path = os.path.join(self._data_dir, self._mode, "_all_label.csv")
        return pd.read_csv(csv_path, dtype={"fname": str, "itp": int,
                                            "its": int})

    def _load_event_data(self, idx: int) -> Tuple[dict, dict]:
        row = self._meta_data.iloc[idx]
        fname, ppk, spk = row["fname"], row["itp"], row["its"]
        npz = np.load(os.path.join(self._data_dir, self._mode, fname))
        data = np.stack(npz["data"].astype(np.float32), axis=1)
        event = {
            "data": data,
            "ppks": [ppk] if ppk > 0 else [],
            "spks": [spk] if spk > 0 else [],
            "snr": np.array([cal_snr(data=data, pat=ppk)]) if ppk > 0
            else np.array([0.0]),
        }
        return event, row.to_dict()


@register_dataset
def sos(**kwargs):
    return SOS(**kwargs)
