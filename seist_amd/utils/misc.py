"""Seeding, timing, argument formatting, and SNR estimation.

Capability parity with /root/reference/utils/misc.py (seeding :14-21, time
helpers :24-38, safe paths :41-52, strfargs :206-221, count_parameters
:224-225, cal_snr :228-274). Distributed helpers live in
``seist_amd.parallel.dist`` in this framework.
"""

import datetime
import os
import random

import numpy as np
import torch


def setup_seed(seed: int) -> None:
    """Seed torch / numpy / random and force deterministic conv algorithms."""
    torch.manual_seed(seed)
    torch.cuda.manual_seed_all(seed)
    np.random.seed(seed)
    random.seed(seed)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False


def get_time_str() -> str:
    return datetime.datetime.now().strftime("%Y-%m-%d_%H-%M-%S")


def strftimedelta(td: datetime.timedelta) -> str:
    """Format a timedelta as ``{h}h {m}min {s}s``."""
    total = int(td.days * 86400 + td.seconds + td.microseconds // 1e6)
    return f"{total // 3600}h {(total % 3600) // 60}min {total % 60}s"


def get_safe_path(path: str, tag: str = "new") -> str:
    """Return a non-existing variant of ``path`` (appends ``_<tag>`` while taken)."""
    d = os.path.dirname(path)
    if d and not os.path.exists(d):
        os.makedirs(d, exist_ok=True)
    while os.path.exists(path):
        root, ext = os.path.splitext(path)
        path = f"{root}_{str(tag).replace(' ', '_')}{ext}"
    return path


def strfargs(args, configs) -> str:
    """Render CLI args and the Config class as a readable string."""
    lines = ["", "Arguments:"]
    for k, v in vars(args).items():
        lines.append(f"{k}: {v}")
    lines.append("")
    lines.append("Configs:")
    for k in dir(configs):
        if k.startswith("__"):
            continue
        v = getattr(configs, k)
        if callable(v):
            continue
        lines.append(f"{k}: {v}")
    lines.append("")
    return "\n".join(lines)


def count_parameters(module: torch.nn.Module) -> int:
    return sum(p.numel() for p in module.parameters())


def cal_snr(data: np.ndarray, pat: int, window: int = 500, method: str = "power") -> float:
    """Estimate SNR (dB) around a phase arrival.

    Signal window is the ``window`` samples after ``pat``, noise window the
    ``window`` samples before; both shrink symmetrically near the trace edges
    (reference semantics, utils/misc.py:228-274).
    """
    pat = int(pat)
    assert window < data.shape[-1] / 2, f"window = {window}, data.shape = {data.shape}"
    assert 0 < pat < data.shape[-1], f"pat = {pat}"

    if pat + window > data.shape[-1]:
        window = data.shape[-1] - pat
    elif pat < window:
        window = pat
    nw = data[:, pat - window : pat]
    sw = data[:, pat : pat + window]

    if method == "power":
        snr = np.mean(sw**2) / (np.mean(nw**2) + 1e-6)
    elif method == "std":
        snr = np.std(sw) / (np.std(nw) + 1e-6)
    else:
        raise ValueError(f"Unknown method: {method}")
    return round(10 * np.log10(snr), 2)
