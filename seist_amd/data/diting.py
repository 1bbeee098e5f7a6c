"""DiTing dataset reader (Zhao et al. 2023).

Semantics parity with /root/reference/datasets/diting.py: 28 CSV metadata
parts + per-part HDF5 files with waveforms at ``earthquake/<key>``; the key
is zero-padded to ``6.4`` digits (diting.py:136-138); magnitude types
ms/mb are converted to ml (diting.py:183-194); polarity u/c->0, r/d->1 and
clarity i->0 else 1 (diting.py:174-178); baz wrapped to [0,360);
deterministic seeded shuffle + train/val/test split.

h5py is imported lazily so the rest of the framework works without it.
"""

import os
from typing import Tuple

import numpy as np
import pandas as pd

from ..utils.logger import logger
from .base import DatasetBase
from .registry import register_dataset

_META_DTYPES = {
    "part": np.int64, "key": str, "ev_id": np.int64, "evmag": str,
    "mag_type": str, "p_pick": np.int64, "p_clarity": str, "p_motion": str,
    "s_pick": np.int64, "net": str, "sta_id": np.int64, "dis": np.float32,
    "st_mag": str, "baz": str,
    "Z_P_amplitude_snr": np.float32, "Z_P_power_snr": np.float32,
    "Z_S_amplitude_snr": np.float32, "Z_S_power_snr": np.float32,
    "N_P_amplitude_snr": np.float32, "N_P_power_snr": np.float32,
    "N_S_amplitude_snr": np.float32, "N_S_power_snr": np.float32,
    "E_P_amplitude_snr": np.float32, "E_P_power_snr": np.float32,
    "E_S_amplitude_snr": np.float32, "E_S_power_snr": np.float32,
    "P_residual": str, "S_residual": str,
}


def _h5py():
    try:
        import h5py
        return h5py
    except ImportError as e:
        raise ImportError(
            "h5py is required to read DiTing/PNW waveforms") from e


class DiTing(DatasetBase):
    _name = "diting"
    _part_range = (0, 28)  # [start, end)
    _channels = ["z", "n", "e"]
    _sampling_rate = 50

    def __init__(self, seed, mode, data_dir, shuffle=True, data_split=True,
                 train_size=0.8, val_size=0.1, **kwargs):
        super().__init__(seed=seed, mode=mode, data_dir=data_dir,
                         shuffle=shuffle, data_split=data_split,
                         train_size=train_size, val_size=val_size)

    def _read_csvs(self):
        start, end = self._part_range
        return pd.concat([
            pd.read_csv(
                os.path.join(self._data_dir, f"DiTing330km_part_{i}.csv"),
                dtype=_META_DTYPES, low_memory=False, index_col=0)
            for i in range(start, end)
        ])

    def _load_meta_data(self) -> pd.DataFrame:
        meta_df = self._read_csvs()
        for k in meta_df.columns:
            if meta_df[k].dtype == object:
                meta_df[k] = meta_df[k].str.replace(" ", "")
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
        part = row["part"]
        # key zero-padding quirk: "123.45" -> "000123.4500"
        key_int, key_frac = str(row["key"]).split(".")
        key = key_int.rjust(6, "0") + "." + key_frac.ljust(4, "0")

        path = os.path.join(self._data_dir, f"DiTing330km_part_{part}.hdf5")
        with _h5py().File(path, "r") as f:
            data = np.array(f.get("earthquake/" + key)).astype(np.float32).T

        motion = row["p_motion"]
        if pd.notnull(motion) and str(motion).lower() not in ("", "n"):
            motion = {"u": 0, "c": 0, "r": 1, "d": 1}[str(motion).lower()]

        clarity = row["p_clarity"]
        if pd.notnull(clarity):
            clarity = 0 if str(clarity).lower() == "i" else 1

        baz = row["baz"]
        if pd.notnull(baz):
            baz = float(baz) % 360

        evmag, stmag = row["evmag"], row["st_mag"]
        if pd.notnull(evmag):
            evmag = float(evmag)
        if pd.notnull(stmag):
            stmag = float(stmag)
        mag_type = str(row["mag_type"]).lower()
        if mag_type == "ms":
            evmag = (evmag + 1.08) / 1.13
            stmag = (stmag + 1.08) / 1.13
        elif mag_type == "mb":
            evmag = (1.17 * evmag + 0.67) / 1.13
            stmag = (1.17 * stmag + 0.67) / 1.13
        elif mag_type != "ml":
            raise ValueError(f"Unknown 'mag_type' : '{row['mag_type']}'")
        evmag = np.clip(evmag, 0, 8).astype(np.float32) \
            if pd.notnull(evmag) else evmag
        stmag = np.clip(stmag, 0, 8).astype(np.float32) \
            if pd.notnull(stmag) else stmag

        snr = np.array([row["Z_P_power_snr"], row["N_S_power_snr"],
                        row["E_S_power_snr"]])
        event = {
            "data": data,
            "ppks": [row["p_pick"]] if pd.notnull(row["p_pick"]) else [],
            "spks": [row["s_pick"]] if pd.notnull(row["s_pick"]) else [],
            "emg": [evmag] if pd.notnull(evmag) else [],
            "smg": [stmag] if pd.notnull(stmag) else [],
            "pmp": [motion] if pd.notnull(motion) else [],
            "clr": [clarity] if pd.notnull(clarity) else [],
            "baz": [baz] if pd.notnull(baz) else [],
            "dis": [row["dis"]] if pd.notnull(row["dis"]) else [],
            "snr": snr,
        }
        return event, row.to_dict()


class DiTing_light(DiTing):
    """Single-CSV variant of DiTing."""

    _name = "diting_light"
    _part_range = None
    _channels = ["z", "n", "e"]
    _sampling_rate = 50

    _LIGHT_DTYPES = dict(_META_DTYPES, evmag=np.float32, st_mag=np.float32,
                         baz=np.float32, P_residual=np.float32,
                         S_residual=np.float32)

    def _read_csvs(self):
        return pd.read_csv(
            os.path.join(self._data_dir, "DiTing330km_light.csv"),
            dtype=self._LIGHT_DTYPES, low_memory=False, index_col=0)


@register_dataset
def diting(**kwargs):
    return DiTing(**kwargs)


@register_dataset
def diting_light(**kwargs):
    return DiTing_light(**kwargs)
