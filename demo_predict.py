#!/usr/bin/env python3
"""Standalone inference demo — the deployment recipe the README points to
(capability parity with /root/reference/demo_predict.py).

Loads a checkpoint, normalizes one waveform window, runs the model on the
MI355X kernel path, and renders the phase-picking figure. Without a real
DiTing HDF5 archive it falls back to a synthetic trace so the demo runs
anywhere.
"""

import argparse

import numpy as np
import torch

from seist_amd.models import create_model, load_checkpoint
from seist_amd.utils.visualization import vis_phase_picking


def normalize(data: np.ndarray, mode: str = "std"):
    data = data - np.mean(data, axis=1, keepdims=True)
    if mode == "max":
        mx = np.max(data, axis=1, keepdims=True)
        mx[mx == 0] = 1
        data /= mx
    elif mode == "std":
        sd = np.std(data, axis=1, keepdims=True)
        sd[sd == 0] = 1
        data /= sd
    elif mode != "":
        raise ValueError(f"Supported mode: 'max','std', got '{mode}'")
    return data


def load_data(data_path: str, trace_name: str) -> np.ndarray:
    """Read one trace (C, L) from a DiTing-format HDF5 archive."""
    import h5py
    with h5py.File(data_path, "r") as f:
        return np.array(f.get(f"earthquake/{trace_name}")).astype(np.float32).T


def synthetic_data(num_samples: int = 8192) -> np.ndarray:
    from seist_amd.data.synthetic import Synthetic
    ds = Synthetic(seed=0, mode="test", data_dir="", size=4,
                   num_samples=num_samples)
    return ds[0][0]["data"]


def load_model(model_name: str, ckpt_path: str, device: torch.device,
               in_channels: int = 3, in_samples: int = 8192):
    model = create_model(model_name=model_name, in_channels=in_channels,
                         in_samples=in_samples)
    if ckpt_path:
        ckpt = load_checkpoint(ckpt_path, device=device)
        model.load_state_dict(ckpt["model_dict"])
    return model.to(device).eval()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-name", default="seist_m_dpk")
    p.add_argument("--checkpoint", default="",
                   help="path to a .pth checkpoint (ours or the reference's)")
    p.add_argument("--data", default="",
                   help="DiTing HDF5 part file; omit for a synthetic trace")
    p.add_argument("--trace-name", default="000159.0004")
    p.add_argument("--in-samples", type=int, default=8192)
    p.add_argument("--save-dir", default="./")
    args = p.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model = load_model(args.model_name, args.checkpoint, device,
                       in_channels=3, in_samples=args.in_samples)

    if args.data:
        waveform = load_data(args.data, args.trace_name)
    else:
        waveform = synthetic_data(args.in_samples + 512)
    waveform = normalize(waveform[:, : args.in_samples], mode="std")
    x = torch.from_numpy(waveform).reshape(1, 3, -1).to(device)

    with torch.no_grad():
        preds = model(x)
    preds = preds.float().cpu().numpy().reshape(3, -1)

    paths = vis_phase_picking(
        waveforms=waveform,
        waveforms_labels=["Z", "N", "E"],
        preds=preds,
        true_phase_idxs=None,
        true_phase_labels=None,
        pred_phase_labels=[r"$\hat{D}$", r"$\hat{P}$", r"$\hat{S}$"],
        sampling_rate=None,
        save_name="demo_prediction",
        save_dir=args.save_dir,
        formats=["png"],
    )
    print(f"saved: {paths}")


if __name__ == "__main__":
    main()
