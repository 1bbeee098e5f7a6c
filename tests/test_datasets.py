"""Dataset reader read-path coverage with synthetic on-disk layouts.

DiTing/PNW need h5py (absent in this image) and real archives; their
registration/config surface is pinned in test_inventory. The SOS reader
is npz-based and runs fully — including the read path that is dead code
in the reference (its sos.py:71 reads `self.data_dir`/`self.mode`, which
do not exist -> AttributeError; fixed here)."""

import os

import numpy as np
import pandas as pd
import pytest

from seist_amd.data import build_dataset


@pytest.fixture()
def sos_root(tmp_path):
    for mode in ("train", "val", "test"):
        d = tmp_path / mode
        d.mkdir()
        rows = []
        rng = np.random.default_rng(7)
        for i in range(3):
            fname = f"ev{i}.npz"
            np.savez(d / fname,
                     data=rng.standard_normal((2000, 1)).astype(np.float32))
            rows.append({"fname": fname, "itp": 800 + i, "its": 1200 + i})
        np.savez(d / "noise.npz",
                 data=rng.standard_normal((2000, 1)).astype(np.float32))
        rows.append({"fname": "noise.npz", "itp": -1, "its": -1})
        pd.DataFrame(rows).to_csv(d / "_all_label.csv", index=False)
    return str(tmp_path)


def test_sos_reader_end_to_end(sos_root):
    for mode, n in (("train", 4), ("val", 4), ("test", 4)):
        ds = build_dataset("sos", seed=0, mode=mode, data_dir=sos_root,
                           shuffle=False)
        assert len(ds) == n
        ev, meta = ds[0]
        assert ev["data"].shape == (1, 2000)          # channels-first
        assert ev["data"].dtype == np.float32
        assert ev["ppks"] == [800] and ev["spks"] == [1200]
        assert np.isfinite(ev["snr"]).all()
        assert meta["fname"] == "ev0.npz"
        # noise row: no picks, snr 0
        evn, _ = ds[3]
        assert evn["ppks"] == [] and evn["spks"] == []
        assert float(np.asarray(evn["snr"]).reshape(-1)[0]) == 0.0


def test_sos_reader_feeds_preprocessor(sos_root):
    """SOS events flow through the full SeismicDataset pipeline (window
    cut, normalize, soft labels) without augmentation."""
    from seist_amd.data.preprocess import SeismicDataset
    from types import SimpleNamespace
    args = SimpleNamespace(
        seed=0, data=sos_root, dataset_name="sos", shuffle=False,
        data_split=False, train_size=0.8, val_size=0.1, in_samples=1024,
        augmentation=False, min_snr=float("-inf"), p_position_ratio=-1,
        coda_ratio=1.4, norm_mode="std", label_shape="gaussian",
        label_width=0.5, add_event_rate=0, add_noise_rate=0,
        add_gap_rate=0, drop_channel_rate=0, scale_amplitude_rate=0,
        pre_emphasis_rate=0, pre_emphasis_ratio=0.97, max_event_num=1,
        generate_noise_rate=0, shift_event_rate=0, mask_percent=0,
        noise_percent=0, min_event_gap=0.5, workers=0)
    ds = SeismicDataset(args=args, input_names=[["z"]],
                        label_names=[["non", "ppk", "spk"]],
                        task_names=["ppk", "spk"], mode="train")
    inputs, loss_targets, metric_targets, meta_json = ds[0]
    x = inputs[0] if isinstance(inputs, (list, tuple)) else inputs
    assert x.shape[-1] == 1024
    assert np.isfinite(np.asarray(x)).all()
    t = loss_targets[0] if isinstance(loss_targets, (list, tuple)) \
        else loss_targets
    assert t.shape[-1] == 1024  # soft labels rasterized to the window
    import json
    assert json.loads(meta_json)["fname"] == "ev0.npz"
