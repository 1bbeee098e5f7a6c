"""PNW dataset reader (Ni et al. 2023).

Parity with /root/reference/datasets/pnw.py: seisbench-style
``trace_name = bucket$n,:c,:l`` indexing into ``comcat_waveforms.hdf5``
(pnw.py:102-110), 4-class polarity map (:131), '|'-separated snr string
parse (:136-138); channels [e, n, z] at 100 Hz.
"""

import os
from typing import Tuple

import numpy as np
import pandas as pd

from ..utils.logger import logger
from .base import DatasetBase
from .diting import _h5py
from .registry import register_dataset


class PNW(DatasetBase):
    _name = "pnw"
    _part_range = None
    _channels = ["e", "n", "z"]
    _sampling_rate = 100

    _meta_filename = "comcat_metadata.csv"

    def __init__(self, seed, mode, data_dir, shuffle=True, data_split=True,
                 train_size=0.8, val_size=0.1, **kwargs):
        super().__init__(seed=seed, mode=mode, data_dir=data_dir,
                         shuffle=shuffle, data_split=data_split,
                         train_size=train_size, val_size=val_size)

    def _load_meta_data(self) -> pd.DataFrame:
        meta_df = pd.read_csv(
            os.path.join(self._data_dir, self._meta_filename),
            low_memory=False)
        for k in meta_df.columns:
            if meta_df[k].dtype in (np.dtype("float"), np.dtype("int")):
                meta_df[k] = meta_df[k].fillna(0)
            elif meta_df[k].dtype == object:
                meta_df[k] = meta_df[k].str.replace(" ", "").fillna("")
        if self._shuffle:
            meta_df = meta_df.sample(frac=1, replace=False,
                                     random_state=self._seed)
        meta_df.reset_index(drop=True, inplace=True)
        if self._data_split:
            lo, hi = self._split_rows(meta_df.shape[0])
            meta_df = meta_df.iloc[lo:hi, :]
            logger.info(f"Data Split: {self._mode}: {lo}-{hi}")
        return meta_df

    def _load_event_data(self, idx: int) -> Tuple[dict, dict]:
        row = self._meta_data.iloc[idx]
        bucket, array = row["trace_name"].split("$")
        n, c, l = [int(i) for i in array.split(",:")]

        path = os.path.join(self._data_dir, "comcat_waveforms.hdf5")
        with _h5py().File(path, "r") as f:
            data = np.nan_to_num(
                np.array(f.get(f"data/{bucket}")[n]).astype(np.float32))

        motion = {"positive": 0, "negative": 1, "undecidable": 2, "": 3}[
            str(row["trace_P_polarity"]).lower()]
        assert str(row["preferred_source_magnitude_type"]).lower() == "ml"
        evmag = np.clip(row["preferred_source_magnitude"], 0, 8,
                        dtype=np.float32)
        snrs = [s.strip() for s in str(row["trace_snr_db"]).split("|")]
        snr = np.array([float(s) if s != "nan" else 0.0 for s in snrs])

        ppk = row["trace_P_arrival_sample"]
        spk = row["trace_S_arrival_sample"]
        event = {
            "data": data,
            "ppks": [ppk] if pd.notnull(ppk) else [],
            "spks": [spk] if pd.notnull(spk) else [],
            "emg": [evmag] if pd.notnull(evmag) else [],
            "pmp": [motion] if pd.notnull(motion) else [],
            "clr": [0],  # compatibility with other datasets
            "snr": snr,
        }
        return event, row.to_dict()


class PNW_light(PNW):
    """PNW with undecidable-polarity events removed."""

    _name = "pnw_light"
    _meta_filename = "comcat_metadata_light.csv"


@register_dataset
def pnw(**kwargs):
    return PNW(**kwargs)


@register_dataset
def pnw_light(**kwargs):
    return PNW_light(**kwargs)
