"""Matplotlib visualization: waveform/pred/target panels and the phase-
picking figure used by the inference demo.

Capability parity with /root/reference/utils/visualization.py:18-186.
matplotlib is imported lazily so headless installs without it still run
training/testing.
"""

import datetime
import os

import numpy as np


def _plt():
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    matplotlib.rcParams.update({
        "font.family": "sans-serif",
        "font.size": 8,
        "mathtext.fontset": "stix",
        "axes.unicode_minus": False,
    })
    return plt


def _timestamp():
    return datetime.datetime.now().strftime("%Y-%m-%d_%H-%M-%S")


def vis_waves_preds_targets(waveforms: np.ndarray, preds: np.ndarray,
                            targets: np.ndarray, sampling_rate=None,
                            save_dir="./", format="png"):
    """Stacked panels: input channels, prediction traces, target traces."""
    plt = _plt()
    plt.figure()
    groups = (("Channel", waveforms), ("Pred", preds), ("Target", targets))
    num_row = sum(g.shape[0] for _, g in groups)
    row = 0
    for label, group in groups:
        for idx, trace in enumerate(group):
            row += 1
            plt.subplot(num_row, 1, row)
            x = (np.arange(len(trace)) / sampling_rate
                 if sampling_rate else np.arange(len(trace)))
            plt.plot(x, trace, "-", color="k", linewidth=0.15, alpha=0.8)
            plt.text(0.001, 0.95, f"{label}-{idx}",
                     horizontalalignment="left", verticalalignment="top",
                     transform=plt.gca().transAxes, fontsize="small")
            if label == "Channel":
                plt.ylim(-1, 1)
            plt.yticks([])
    plt.xlabel("Time (s)" if sampling_rate else "Sample points")
    plt.tight_layout()
    plt.subplots_adjust(wspace=0.1, hspace=0.0)
    os.makedirs(save_dir, exist_ok=True)
    path = os.path.join(save_dir, f"{_timestamp()}.{format}")
    plt.savefig(path, dpi=400)
    plt.close()
    return path


def vis_phase_picking(waveforms: np.ndarray, waveforms_labels: list,
                      preds: np.ndarray, true_phase_idxs, true_phase_labels,
                      pred_phase_labels: list, sampling_rate: int = None,
                      save_name="", save_dir="./", formats=("png",)):
    """Waveform channels with true-phase markers + probability traces."""
    plt = _plt()
    plt.figure(figsize=(10 / 2.54, 10 / 2.54))
    L = len(waveforms[0])
    x = (np.arange(L) / sampling_rate if sampling_rate else np.arange(L))
    num_row = waveforms.shape[0] + 1
    lo, hi = float(np.min(waveforms)), float(np.max(waveforms))
    panel = "abcdefgh"

    for idx, wave in enumerate(waveforms):
        plt.subplot(num_row, 1, idx + 1)
        plt.plot(x, wave, "-", color="k", linewidth=1, alpha=0.8,
                 label=waveforms_labels[idx])
        if idx == 0 and true_phase_idxs:
            for pi, (pidx, plabel, color) in enumerate(zip(
                    true_phase_idxs, true_phase_labels, ("C1", "C5"))):
                plt.vlines(x=[pidx], ymin=lo * 1.1, ymax=hi * 1.1,
                           colors=[color], linestyles="solid", label=plabel)
        plt.ylim(lo * 1.2, hi * 1.2)
        plt.ylabel("Amplitude")
        plt.yticks([])
        plt.xticks([])
        plt.text(0.05, 0.78, f"({panel[idx]})",
                 horizontalalignment="center",
                 transform=plt.gca().transAxes, fontsize=8)
        plt.legend(loc="upper right", fontsize=8, ncol=1)

    plt.subplot(num_row, 1, num_row)
    plt.text(0.05, 0.78, f"({panel[num_row - 1]})",
             horizontalalignment="center",
             transform=plt.gca().transAxes, fontsize=8)
    styles = ("-.C0", "--C1", "--C5")
    for trace, style, label in zip(preds, styles, pred_phase_labels):
        plt.plot(x, trace, style, linewidth=1, alpha=0.8, label=label)
    plt.ylabel("Probability")
    plt.xlabel("Time (s)" if sampling_rate else "Samples")
    plt.legend(loc="upper right", fontsize=8, ncol=1)
    plt.tight_layout()
    plt.gcf().align_labels()
    plt.subplots_adjust(wspace=0.1, hspace=0.05)

    os.makedirs(save_dir, exist_ok=True)
    if not isinstance(formats, (list, tuple)):
        formats = [formats]
    paths = []
    for fmt in formats:
        path = os.path.join(save_dir, f"{_timestamp()}{save_name}.{fmt}")
        plt.savefig(path, dpi=400)
        paths.append(path)
    plt.close()
    return paths
