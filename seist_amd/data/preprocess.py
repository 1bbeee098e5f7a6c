"""Preprocessing, augmentation and label generation.

Semantics parity with /root/reference/training/preprocess.py:
noise gating (:154-170), window cutting with optional fixed-P position
(:172-222), per-trace demean + max/std normalisation (:224-242), the 11
augmentations (:244-499), soft-label rasterisation in gaussian / triangle /
box / sigmoid shapes (:544-683), io-item dispatch soft/value/onehot
(:685-742) and the (inputs, loss_targets, metrics_targets, meta_json)
sample tuple (:924-953). The np.random call order inside each augmentation
matches the reference so seeded streams are comparable.
"""

import argparse
import copy
import json
import os
from operator import itemgetter
from typing import Any, List, Tuple, Union

import numpy as np
from torch.utils.data import Dataset

from ..config import Config
from ..utils.logger import logger
from .registry import build_dataset

# K20: native (C++) loader-side workers — bit-exact with the numpy path
# (numpy's pairwise summation is reproduced; label windows are computed in
# numpy and only rasterized natively). SEIST_AMD_PY_DATA=1 forces numpy.
try:
    if os.environ.get("SEIST_AMD_PY_DATA") == "1":
        _native_data = None
    else:
        from .. import _native_data  # type: ignore
except ImportError:  # pragma: no cover - extension not built
    _native_data = None

__all__ = ["DataPreprocessor", "SeismicDataset", "_pad_phases", "_pad_array"]


def _pad_phases(ppks: list, spks: list, padding_idx: int,
                num_samples: int) -> Tuple[list, list]:
    """Align P/S pick lists to equal length by padding unmatched picks with
    out-of-range sentinels (reference preprocess.py:16-35)."""
    padding_idx = abs(padding_idx)
    ppks, spks = sorted(ppks), sorted(spks)
    ppk_arr, spk_arr = np.array(ppks), np.array(spks)
    idx = 0
    while idx < min(len(ppks), len(spks)) and all(
            ppk_arr[: idx + 1] < spk_arr[-idx - 1:]):
        idx += 1
    ppks = (len(spk_arr) - idx) * [-padding_idx] + ppks
    spks = spks + len(ppk_arr[idx:]) * [num_samples + padding_idx]
    assert len(ppks) == len(spks), f"{ppks}, {spks}"
    return ppks, spks


def _pad_array(s, length: int, padding_value) -> np.ndarray:
    padding_size = int(length - len(s))
    if padding_size < 0:
        raise Exception(f"`length < len(s)` . Array:{len(s)},Target:{length}")
    return np.pad(s, (0, padding_size), mode="constant",
                  constant_values=padding_value)


class DataPreprocessor:
    """Augment raw events and rasterise labels (CPU, loader workers).

    The augmentation methods `_normalize`, `_adjust_amplitude`,
    `_scale_amplitude` and `_pre_emphasis` follow the EQTransformer
    conventions, like the reference does.
    """

    def __init__(
        self,
        data_channels,
        sampling_rate: int,
        in_samples: int,
        min_snr: float,
        p_position_ratio: float,
        coda_ratio: float,
        norm_mode: str,
        add_event_rate: float,
        add_noise_rate: float,
        add_gap_rate: float,
        drop_channel_rate: float,
        scale_amplitude_rate: float,
        pre_emphasis_rate: float,
        pre_emphasis_ratio: float,
        max_event_num: int,
        generate_noise_rate: float,
        shift_event_rate: float,
        mask_percent: float,
        noise_percent: float,
        min_event_gap_sec: float,
        soft_label_shape: str,
        soft_label_width: int,
        dtype=np.float32,
    ):
        self.sampling_rate = sampling_rate
        self.data_channels = data_channels
        self.in_samples = in_samples
        self.coda_ratio = coda_ratio
        self.norm_mode = norm_mode
        self.min_snr = min_snr
        self.p_position_ratio = p_position_ratio
        self.add_event_rate = add_event_rate
        self.add_noise_rate = add_noise_rate
        self.add_gap_rate = add_gap_rate
        self.drop_channel_rate = drop_channel_rate
        self.scale_amplitude_rate = scale_amplitude_rate
        self.pre_emphasis_rate = pre_emphasis_rate
        self.pre_emphasis_ratio = pre_emphasis_ratio
        self._max_event_num = max_event_num
        self.generate_noise_rate = generate_noise_rate
        self.shift_event_rate = shift_event_rate
        self.mask_percent = mask_percent
        self.noise_percent = noise_percent
        self.min_event_gap = int(min_event_gap_sec * sampling_rate)

        if 0 <= self.p_position_ratio <= 1:
            # fixed-P windows are incompatible with event-moving augmentations
            for attr in ("add_event_rate", "shift_event_rate",
                         "generate_noise_rate"):
                if getattr(self, attr) > 0:
                    setattr(self, attr, 0.0)
                    logger.warning(
                        f"`p_position_ratio` is {p_position_ratio}, "
                        f"`{attr}` -> 0.0")

        self.soft_label_shape = soft_label_shape
        self.soft_label_width = soft_label_width
        self.dtype = dtype

    # ------------------------------------------------------------------
    # gating / windowing / normalisation
    # ------------------------------------------------------------------

    def _clear_dict_except(self, d: dict, *keep) -> None:
        for k in set(d) - set(keep):
            v = d[k]
            if isinstance(v, (list, dict)):
                v.clear()
            elif isinstance(v, np.ndarray):
                d[k] = np.array([])
            elif isinstance(v, (int, float)):
                d[k] = 0
            elif isinstance(v, str):
                d[k] = ""
            else:
                raise TypeError(f"Got `{v}`({type(v)})")

    def _is_noise(self, data, ppks: List[int], spks: List[int], snr) -> bool:
        is_noise = (
            (len(ppks) != len(spks))
            or len(ppks) < 1
            or len(spks) < 1
            or min(ppks + spks) < 0
            or max(ppks + spks) >= data.shape[-1]
            or all(np.atleast_1d(snr) < self.min_snr)
        )
        for p, s in zip(ppks, spks):
            is_noise |= p >= s
        return bool(is_noise)

    def _cut_window(self, data, ppks: list, spks: list, window_size: int):
        input_len = data.shape[-1]
        if 0 <= self.p_position_ratio <= 1:
            new_data = np.zeros((data.shape[0], window_size), dtype=np.float32)
            tgt_l, tgt_r = 0, window_size
            p_idx = ppks[0]
            c_l = p_idx - int(window_size * self.p_position_ratio)
            c_r = c_l + window_size
            offset = -c_l
            if c_l < 0:
                tgt_l += abs(c_l)
                offset += c_l
                c_l = 0
            if c_r > input_len:
                tgt_r -= c_r - input_len
                c_r = input_len
            new_data[:, tgt_l:tgt_r] = data[:, c_l:c_r]
            offset += tgt_l
            data = new_data
            ppks = [t + offset for t in ppks if 0 <= t + offset < window_size]
            spks = [t + offset for t in spks if 0 <= t + offset < window_size]
        else:
            if input_len > window_size:
                c_l = np.random.randint(
                    0, max(min(ppks + [input_len - window_size])
                           - self.min_event_gap, 1))
                c_r = c_l + window_size
                data = data[:, c_l:c_r]
                ppks = [t - c_l for t in ppks if c_l <= t < c_r]
                spks = [t - c_l for t in spks if c_l <= t < c_r]
            elif input_len < window_size:
                data = np.concatenate(
                    [data, np.zeros((data.shape[0], window_size - input_len))],
                    axis=1)
        return data, ppks, spks

    def _normalize(self, data, mode):
        if mode not in ("", "max", "std"):
            raise ValueError(f"Supported mode: 'max','std', got '{mode}'")
        if (_native_data is not None and data.dtype == np.float32
                and data.ndim == 2 and data.flags["C_CONTIGUOUS"]):
            _native_data.normalize(data, {"": 0, "max": 1, "std": 2}[mode])
            return data
        data -= np.mean(data, axis=1, keepdims=True)
        if mode == "max":
            mx = np.max(data, axis=1, keepdims=True)
            mx[mx == 0] = 1
            data /= mx
        elif mode == "std":
            sd = np.std(data, axis=1, keepdims=True)
            sd[sd == 0] = 1
            data /= sd
        return data

    # ------------------------------------------------------------------
    # augmentations
    # ------------------------------------------------------------------

    def _generate_noise_data(self, data, ppks, spks):
        if len(ppks) > 0 and len(spks) > 0:
            for ppk, spk in zip(ppks, spks):
                coda_end = np.clip(int(spk + self.coda_ratio * (spk - ppk)),
                                   0, data.shape[-1], dtype=int)
                if ppk < coda_end:
                    data[:, ppk:coda_end] = np.random.randn(
                        data.shape[0], coda_end - ppk)
        return data, [], []

    def _add_event(self, data, ppks, spks, min_gap):
        target_idx = np.random.randint(0, len(ppks))
        ppk, spk = ppks[target_idx], spks[target_idx]
        coda_end = int(spk + self.coda_ratio * (spk - ppk))
        left = coda_end + min_gap
        right = data.shape[-1] - (spk - ppk) - min_gap
        if left < right:
            ppk_add = np.random.randint(left, right)
            spk_add = ppk_add + spk - ppk
            space = min(data.shape[-1] - ppk_add, coda_end - ppk)
            scale = np.random.random()
            data[:, ppk_add: ppk_add + space] += data[:, ppk: ppk + space] * scale
            ppks.append(ppk_add)
            spks.append(spk_add)
        ppks.sort()
        spks.sort()
        return data, ppks, spks

    def _shift_event(self, data, ppks, spks):
        shift = np.random.randint(0, data.shape[-1])
        data = np.concatenate((data[:, -shift:], data[:, :-shift]), axis=1)
        ppks = sorted((p + shift) % data.shape[-1] for p in ppks)
        spks = sorted((s + shift) % data.shape[-1] for s in spks)
        return data, ppks, spks

    def _drop_channel(self, data):
        if data.shape[0] < 2:
            return data
        drop_num = np.random.choice(range(1, data.shape[0]))
        candidates = list(range(data.shape[0]))
        for _ in range(drop_num):
            c = np.random.choice(candidates)
            candidates.remove(c)
            data[c, :] = 0.0
        return data

    def _adjust_amplitude(self, data):
        max_amp = np.max(np.abs(data), axis=1)
        if np.count_nonzero(max_amp) > 0:
            data *= data.shape[0] / np.count_nonzero(max_amp)
        return data

    def _scale_amplitude(self, data):
        if np.random.uniform(0, 1) < 0.5:
            data *= np.random.uniform(1, 3)
        else:
            data /= np.random.uniform(1, 3)
        return data

    def _pre_emphasis(self, data, pre_emphasis):
        for c in range(data.shape[0]):
            bpf = data[c, :]
            data[c, :] = np.append(bpf[0], bpf[1:] - pre_emphasis * bpf[:-1])
        return data

    def _add_noise(self, data):
        for c in range(data.shape[0]):
            x = data[c, :]
            snr = np.random.randint(10, 50)
            px = np.sum(x**2) / len(x)
            pn = px * 10 ** (-snr / 10.0)
            data[c, :] += np.random.randn(len(x)) * np.sqrt(pn)
        return data

    def _add_gaps(self, data, ppks, spks):
        phases = sorted(ppks + spks)
        if len(phases) > 0:
            phases.append(data.shape[-1] - 1)
            phases = sorted(set(phases))
            insert_pos = np.random.randint(0, len(phases) - 1)
            sgt = np.random.randint(phases[insert_pos], phases[insert_pos + 1])
            egt = np.random.randint(sgt, phases[insert_pos + 1])
        else:
            sgt = np.random.randint(0, data.shape[-1] - 1)
            egt = np.random.randint(sgt + 1, data.shape[-1])
        data[:, sgt:egt] = 0
        return data

    def _add_mask_windows(self, data, percent=50, window_size=20,
                          mask_value=1.0):
        p = np.clip(percent, 0, 100)
        num_windows = data.shape[-1] // window_size
        num_mask = num_windows * p // 100
        for i in np.random.choice(range(num_windows), num_mask, replace=False):
            data[:, i * window_size:(i + 1) * window_size] = mask_value
        return data

    def _add_noise_windows(self, data, percent=50, window_size=20):
        p = np.clip(percent, 0, 100)
        num_windows = data.shape[-1] // window_size
        num_block = num_windows * p // 100
        for i in np.random.choice(range(num_windows), num_block, replace=False):
            data[:, i * window_size:(i + 1) * window_size] = np.random.randn(
                data.shape[0], window_size)
        return data

    def _data_augmentation(self, event: dict) -> dict:
        data, ppks, spks = itemgetter("data", "ppks", "spks")(event)

        if np.random.random() < self.generate_noise_rate:
            data, ppks, spks = self._generate_noise_data(data, ppks, spks)
            self._clear_dict_except(event, "data")
            if np.random.random() < self.drop_channel_rate:
                data = self._adjust_amplitude(self._drop_channel(data))
            if np.random.random() < self.scale_amplitude_rate:
                data = self._scale_amplitude(data)
        else:
            for _ in range(self._max_event_num - len(ppks)):
                if np.random.random() < self.add_event_rate and ppks:
                    data, ppks, spks = self._add_event(
                        data, ppks, spks, self.min_event_gap)
            if np.random.random() < self.shift_event_rate:
                data, ppks, spks = self._shift_event(data, ppks, spks)
            if np.random.random() < self.drop_channel_rate:
                data = self._adjust_amplitude(self._drop_channel(data))
            if np.random.random() < self.scale_amplitude_rate:
                data = self._scale_amplitude(data)
            if np.random.random() < self.pre_emphasis_rate:
                data = self._pre_emphasis(data, self.pre_emphasis_ratio)
            if np.random.random() < self.add_noise_rate:
                data = self._add_noise(data)
            if np.random.random() < self.add_gap_rate:
                data = self._add_gaps(data, ppks, spks)

        if self.mask_percent > 0:
            data = self._add_mask_windows(
                data, percent=self.mask_percent,
                window_size=self.sampling_rate // 2)
        if self.noise_percent > 0:
            data = self._add_noise_windows(
                data, percent=self.noise_percent,
                window_size=self.sampling_rate // 2)

        event.update({"data": data, "ppks": ppks, "spks": spks})
        return event

    # ------------------------------------------------------------------
    # main entry
    # ------------------------------------------------------------------

    def _native_params(self):
        params = getattr(self, "_native_params_cache", None)
        if params is None:
            params = self._native_params_cache = dict(
                min_snr=float(self.min_snr), coda_ratio=float(self.coda_ratio),
                p_position_ratio=float(self.p_position_ratio),
                add_event_rate=float(self.add_event_rate),
                add_noise_rate=float(self.add_noise_rate),
                add_gap_rate=float(self.add_gap_rate),
                drop_channel_rate=float(self.drop_channel_rate),
                scale_amplitude_rate=float(self.scale_amplitude_rate),
                pre_emphasis_rate=float(self.pre_emphasis_rate),
                pre_emphasis_ratio=float(self.pre_emphasis_ratio),
                generate_noise_rate=float(self.generate_noise_rate),
                shift_event_rate=float(self.shift_event_rate),
                max_event_num=int(self._max_event_num),
                mask_percent=int(self.mask_percent),
                noise_percent=int(self.noise_percent),
                min_event_gap=int(self.min_event_gap),
                in_samples=int(self.in_samples),
                sampling_rate=int(self.sampling_rate),
                norm_mode={"": 0, "max": 1, "std": 2}[self.norm_mode])
        return params

    def process_native(self, event: dict, augmentation: bool, rng) -> dict:
        """Full pipeline in the C++ worker (data/_augment.cpp): bit-exact
        with :meth:`process` when ``rng`` continues np.random's stream
        (see SeismicDataset). Falls back to the numpy path — with the MT
        state synced both ways — outside the native envelope."""
        data = event["data"]
        fixed_p = 0.0 <= self.p_position_ratio <= 1.0
        ok = (isinstance(data, np.ndarray) and data.dtype == np.float32
              and data.ndim == 2 and data.flags["C_CONTIGUOUS"]
              and (fixed_p or data.shape[-1] >= self.in_samples))
        if not ok:
            keys, pos, hg, g = rng.get_state()
            np.random.set_state(("MT19937", np.array(keys, dtype=np.uint32),
                                 pos, int(hg), float(g)))
            event = self.process(event, augmentation)
            st = np.random.get_state()
            rng.set_state([int(k) for k in st[1]], int(st[2]), bool(st[3]),
                          float(st[4]))
            return event


        snr = np.atleast_1d(np.asarray(event["snr"], dtype=np.float64))
        out, ppks, spks, cleared = _native_data.process_event(
            data, [int(p) for p in event["ppks"]],
            [int(s) for s in event["spks"]], [float(v) for v in snr],
            bool(augmentation), self._native_params(), rng)
        if cleared:
            self._clear_dict_except(event, "data", "ppks", "spks")
        event["data"] = out
        event["ppks"] = list(ppks)
        event["spks"] = list(spks)
        return event

    def process(self, event: dict, augmentation: bool,
                inplace: bool = True) -> dict:
        if not inplace:
            event = copy.deepcopy(event)

        if self._is_noise(event["data"], event["ppks"], event["spks"],
                          event["snr"]):
            self._clear_dict_except(event, "data")

        event["ppks"], event["spks"] = _pad_phases(
            event["ppks"], event["spks"], self.min_event_gap, self.in_samples)

        if augmentation:
            event = self._data_augmentation(event)

        event["data"], event["ppks"], event["spks"] = self._cut_window(
            event["data"], event["ppks"], event["spks"], self.in_samples)

        event["data"] = self._normalize(event["data"], self.norm_mode)
        return event

    # ------------------------------------------------------------------
    # label generation
    # ------------------------------------------------------------------

    def _label_window(self, soft_label_width: int,
                      soft_label_shape: str) -> np.ndarray:
        left = int(soft_label_width / 2)
        right = soft_label_width - left
        if soft_label_shape == "gaussian":
            return np.exp(-(np.arange(-left, right + 1) ** 2) / (2 * 10**2))
        if soft_label_shape == "triangle":
            return 1 - np.abs(2 / soft_label_width * np.arange(-left, right + 1))
        if soft_label_shape == "box":
            return np.ones(soft_label_width + 1)
        if soft_label_shape == "sigmoid":
            l_l, l_r = -int(left / 2), left - int(left / 2)
            r_l, r_r = -int(right / 2), right - int(right / 2)
            x_l = -10 / left * np.arange(l_l, l_r)
            x_r = 10 / right * np.arange(r_l, r_r)
            sig = lambda x: 1 / (1 + np.exp(x))  # noqa: E731
            return np.concatenate((sig(x_l), [1.0], sig(x_r)), axis=0)
        raise NotImplementedError(
            f"Unsupported label shape: '{soft_label_shape}'")

    def _rasterize(self, idxs, length: int, soft_label_width: int,
                   soft_label_shape: str) -> np.ndarray:
        """Sum the label window at each index (clipped at the edges)."""
        if _native_data is not None and len(idxs) > 0:
            key = (soft_label_width, soft_label_shape)
            cache = getattr(self, "_window_cache", None)
            if cache is None:
                cache = self._window_cache = {}
            window = cache.get(key)
            if window is None:
                window = cache[key] = np.ascontiguousarray(
                    self._label_window(soft_label_width, soft_label_shape),
                    dtype=np.float64)
            return _native_data.rasterize(
                [int(i) for i in idxs], length, soft_label_width, window)
        slabel = np.zeros(length)
        if len(idxs) == 0:
            return slabel
        left = int(soft_label_width / 2)
        right = soft_label_width - left
        window = self._label_window(soft_label_width, soft_label_shape)
        for idx in idxs:
            if idx < 0:
                continue
            elif idx - left < 0:
                slabel[: idx + right + 1] += window[
                    soft_label_width + 1 - (idx + right + 1):]
            elif idx + right <= length - 1:
                slabel[idx - left: idx + right + 1] += window
            elif idx <= length - 1:
                slabel[-(length - (idx - left)):] += window[
                    : length - (idx - left)]
        return slabel

    def _generate_soft_label(self, name: str, event: dict,
                             soft_label_width: int,
                             soft_label_shape: str) -> np.ndarray:
        length = event["data"].shape[-1]

        def _clip(x):
            return min(max(x, 0), length)

        def _soft(idxs):
            return self._rasterize(idxs, length, soft_label_width,
                                   soft_label_shape)

        ppks, spks = _pad_phases(
            ppks=event["ppks"], spks=event["spks"],
            padding_idx=soft_label_width, num_samples=length)

        if name in ("ppk", "spk"):
            key = {"ppk": "ppks", "spk": "spks"}[name]
            label = _soft(event[key])
        elif name == "non":
            label = np.ones(length) - _soft(ppks) - _soft(spks)
            label[label < 0] = 0
        elif name == "det":
            label = np.zeros(length)
            assert len(ppks) == len(spks)
            for ppk, spk in zip(ppks, spks):
                det = int(spk + self.coda_ratio * (spk - ppk))
                label_i = _soft([ppk, det])
                label_i[_clip(ppk): _clip(det)] = 1.0
                label += label_i
            label[label > 1] = 1.0
        elif name in ("ppk+", "spk+"):
            key = {"ppk+": "ppks", "spk+": "spks"}[name]
            phases = event[key]
            label = np.zeros(length)
            for st in phases:
                label_i = _soft([st])
                label_i[_clip(st):] = 1.0
                label += label_i / len(phases)
        elif name in self.data_channels:
            label = event["data"][self.data_channels.index(name)]
        elif name in [f"d{c}" for c in self.data_channels]:
            ch = event["data"][self.data_channels.index(name[-1])]
            if (_native_data is not None and ch.dtype == np.float32
                    and ch.flags["C_CONTIGUOUS"]):
                label = _native_data.diff_label(ch)
            else:
                label = np.zeros_like(ch)
                label[1:] = np.diff(ch)
        else:
            raise NotImplementedError(f"Unsupported label name: '{name}'")
        return label.astype(self.dtype)

    def _get_io_item(self, name: Union[str, tuple, list], event: dict,
                     soft_label_width: int = None,
                     soft_label_shape: str = None):
        if isinstance(name, (tuple, list)):
            return np.array([self._get_io_item(sub, event) for sub in name])

        item_type = Config.get_type(name)
        if item_type == "soft":
            return self._generate_soft_label(
                name=name, event=event,
                soft_label_width=(soft_label_width or self.soft_label_width),
                soft_label_shape=(soft_label_shape or self.soft_label_shape))
        if item_type == "value":
            return np.array(event[name]).astype(self.dtype)
        if item_type == "onehot":
            cidx = event[name]
            if not len(cidx) > 0:
                raise ValueError(f"Item:{name}, Value:{cidx}")
            nc = Config.get_num_classes(name=name)
            return np.eye(nc)[cidx[0]].astype(np.int64)
        raise NotImplementedError(f"Unknown item: {name}")

    def get_inputs(self, event: dict, input_names: list):
        inputs = [self._get_io_item(name, event) for name in input_names]
        return tuple(inputs) if len(inputs) > 1 else inputs.pop()

    def get_targets_for_loss(self, event: dict, label_names: list) -> Any:
        targets = [self._get_io_item(name, event) for name in label_names]
        return tuple(targets) if len(targets) > 1 else targets.pop()

    def get_targets_for_metrics(self, event: dict, max_event_num: int,
                                task_names: list) -> dict:
        targets = {}
        for name in task_names:
            if name in ("ppk", "spk"):
                key = {"ppk": "ppks", "spk": "spks"}[name]
                tgt = self._get_io_item(name=key, event=event)
                tgt = _pad_array(tgt, length=max_event_num,
                                 padding_value=int(-1e7)).astype(np.int64)
            elif name == "det":
                padded_ppks, padded_spks = _pad_phases(
                    event["ppks"], event["spks"], self.soft_label_width,
                    self.in_samples)
                detections = []
                for ppk, spk in zip(padded_ppks, padded_spks):
                    st = int(np.clip(ppk, 0, self.in_samples))
                    et = int(spk + self.coda_ratio * (spk - ppk))
                    detections.extend([st, et])
                expected_num = (self._max_event_num
                                + int(bool(self.add_event_rate))
                                + int(bool(self.shift_event_rate))
                                + int(0 <= self.p_position_ratio <= 1))
                if len(detections) // 2 < expected_num:
                    detections += [1, 0] * (expected_num
                                            - len(detections) // 2)
                tgt = np.array(detections).astype(np.int64)
            else:
                tgt = self._get_io_item(name=name, event=event)
            targets[name] = tgt
        return targets


class SeismicDataset(Dataset):
    """torch Dataset: raw reader -> preprocess/augment -> labels.

    With augmentation on, the logical length doubles and the second half is
    the augmented copy of the first (reference preprocess.py:918-938).
    """

    def __init__(self, args: argparse.Namespace, input_names: list,
                 label_names: list, task_names: list, mode: str):
        self._seed = int(args.seed)
        self._mode = mode.lower()
        self._input_names = input_names
        self._label_names = label_names
        self._task_names = task_names
        self._max_event_num = args.max_event_num

        self._augmentation = args.augmentation and self._mode == "train"
        if self._augmentation != args.augmentation:
            logger.warning(f"[{self._mode}]Augmentation -> {self._augmentation}")

        self._dataset = build_dataset(
            dataset_name=args.dataset_name, seed=self._seed, mode=self._mode,
            data_dir=args.data, shuffle=args.shuffle,
            data_split=args.data_split, train_size=args.train_size,
            val_size=args.val_size,
            **getattr(args, "dataset_kwargs", {}) or {})
        logger.info(str(self._dataset))
        self._dataset_size = len(self._dataset)
        if self._augmentation:
            logger.info(f"Data augmentation: dataset size -> "
                        f"{self._dataset_size * 2}")

        def _flat(names):
            for n in names:
                if isinstance(n, (tuple, list)):
                    yield from _flat(n)
                else:
                    yield n

        # Regression/classification labels ("value"/"onehot" io items) have
        # no target on a noise-replaced window: the reference crashes at
        # collate ([] vs [1]) unless the user remembers
        # --generate-noise-rate 0. Force it off here, mirroring the
        # reference's own p_position_ratio -> generate_noise_rate=0 rule
        # (reference training/preprocess.py:125-131).
        generate_noise_rate = args.generate_noise_rate
        scalar_labels = [n for n in _flat(self._label_names)
                         if Config.get_type(n) in ("value", "onehot")]
        if scalar_labels and generate_noise_rate > 0:
            logger.warning(
                f"labels {scalar_labels} are undefined on generated noise; "
                f"`generate_noise_rate` -> 0.0")
            generate_noise_rate = 0.0

        self._preprocessor = DataPreprocessor(
            data_channels=self._dataset.channels(),
            sampling_rate=self._dataset.sampling_rate(),
            in_samples=args.in_samples,
            min_snr=args.min_snr,
            coda_ratio=args.coda_ratio,
            norm_mode=args.norm_mode,
            p_position_ratio=args.p_position_ratio,
            add_event_rate=args.add_event_rate,
            add_noise_rate=args.add_noise_rate,
            add_gap_rate=args.add_gap_rate,
            drop_channel_rate=args.drop_channel_rate,
            scale_amplitude_rate=args.scale_amplitude_rate,
            pre_emphasis_rate=args.pre_emphasis_rate,
            pre_emphasis_ratio=args.pre_emphasis_ratio,
            max_event_num=args.max_event_num,
            generate_noise_rate=generate_noise_rate,
            shift_event_rate=args.shift_event_rate,
            mask_percent=args.mask_percent,
            noise_percent=args.noise_percent,
            min_event_gap_sec=args.min_event_gap,
            soft_label_shape=args.label_shape,
            soft_label_width=int(args.label_width
                                 * self._dataset.sampling_rate()),
            dtype=np.float32,
        )

    def sampling_rate(self):
        return self._dataset.sampling_rate()

    def data_channels(self):
        return self._dataset.channels()

    def name(self):
        return f"{self._dataset.name()}_{self._mode}"

    def __len__(self):
        return 2 * self._dataset_size if self._augmentation \
            else self._dataset_size

    def __getstate__(self):
        # the pybind RandomState is not picklable (spawn-mode workers);
        # each process re-adopts np.random's state lazily
        state = dict(self.__dict__)
        state.pop("_native_rng_obj", None)
        return state

    def _native_rng(self):
        """Per-process C++ RandomState continuing np.random's exact MT
        stream (lazily adopted, so a DataLoader worker that forked with a
        copied numpy state behaves identically to the Python path)."""
        rng = getattr(self, "_native_rng_obj", None)
        if rng is None:
            st = np.random.get_state()
            rng = _native_data.RandomState(0)
            rng.set_state([int(k) for k in st[1]], int(st[2]), bool(st[3]),
                          float(st[4]))
            self._native_rng_obj = rng
        return rng

    def __getitem__(self, idx: int):
        event, meta_data = self._dataset[idx % self._dataset_size]
        augment = self._augmentation and idx >= self._dataset_size
        if _native_data is not None:
            event = self._preprocessor.process_native(
                event=event, augmentation=augment, rng=self._native_rng())
        else:
            event = self._preprocessor.process(
                event=event, augmentation=augment)
        inputs = self._preprocessor.get_inputs(
            event=event, input_names=self._input_names)
        loss_targets = self._preprocessor.get_targets_for_loss(
            event=event, label_names=self._label_names)
        metrics_targets = self._preprocessor.get_targets_for_metrics(
            event=event, task_names=self._task_names,
            max_event_num=self._max_event_num)
        meta_data_json = json.dumps(meta_data, default=str)
        return inputs, loss_targets, metrics_targets, meta_data_json
