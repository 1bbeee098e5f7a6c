"""Dataset base contract (parity with /root/reference/datasets/base.py:5-90).

A dataset yields ``(event_dict, meta_dict)`` per index, where the event dict
carries {data (C,L) float32, ppks, spks, emg, smg, pmp, clr, baz, dis, snr}
(missing keys allowed), and exposes classmethod metadata name()/channels()/
sampling_rate().
"""

import copy
from typing import Optional, Tuple


class DatasetBase:
    _name: str
    _part_range: Optional[tuple] = None
    _channels: list
    _sampling_rate: int

    def __init__(self, seed: int, mode: str, data_dir: str,
                 shuffle: bool = True, data_split: bool = True,
                 train_size: float = 0.8, val_size: float = 0.1):
        self._seed = seed
        assert mode.lower() in ("train", "val", "test")
        self._mode = mode.lower()
        self._data_dir = data_dir
        self._shuffle = shuffle
        self._data_split = data_split
        assert train_size + val_size < 1.0, \
            f"train_size:{train_size}, val_size:{val_size}"
        self._train_size = train_size
        self._val_size = val_size
        self._meta_data = self._load_meta_data()

    # -- subclass API -------------------------------------------------
    def _load_meta_data(self):
        raise NotImplementedError

    def _load_event_data(self, idx: int) -> Tuple[dict, dict]:
        raise NotImplementedError

    def _split_rows(self, n: int) -> Tuple[int, int]:
        """Row range [lo, hi) of this mode's split over n shuffled rows."""
        t = int(self._train_size * n)
        v = t + int(self._val_size * n)
        return {"train": (0, t), "val": (t, v), "test": (v, n)}[self._mode]

    # -- public -------------------------------------------------------
    def __len__(self):
        return len(self._meta_data)

    def __getitem__(self, idx: int) -> Tuple[dict, dict]:
        return self._load_event_data(idx=idx)

    def __repr__(self):
        return (f"Dataset(name:{self._name}, part_range:{self._part_range}, "
                f"channels:{self._channels}, sampling_rate:{self._sampling_rate}, "
                f"data_dir:{self._data_dir}, shuffle:{self._shuffle}, "
                f"data_split:{self._data_split}, train_size:{self._train_size}, "
                f"val_size:{self._val_size})")

    @classmethod
    def name(cls):
        return cls._name

    @classmethod
    def sampling_rate(cls):
        return cls._sampling_rate

    @classmethod
    def channels(cls):
        return copy.deepcopy(cls._channels)
