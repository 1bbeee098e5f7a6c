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


def test_diting_light_reader_with_fake_h5(tmp_path, monkeypatch):
    """DiTing reader logic without real archives: a fake h5py exercises
    the key zero-padding quirk ("123.45" -> "000123.4500", reference
    datasets/diting.py:137-138), channel transpose, polarity/clarity
    encoding, baz wrap and ms->ml magnitude conversion."""
    import seist_amd.data.diting as dt

    cols = {
        "part": [0, 0], "key": ["123.45", "7.1"], "ev_id": [1, 2],
        "evmag": ["3.0", "2.0"], "mag_type": ["ms", "ml"],
        "p_pick": [4000, 5000], "p_clarity": ["I", "E"],
        "p_motion": ["U", "D"], "s_pick": [6000, 7000],
        "net": ["AA", "BB"], "sta_id": [1, 2], "dis": [10.0, 20.0],
        "st_mag": ["3.0", "2.0"], "baz": ["370.0", "45.0"],
        "P_residual": ["0", "0"], "S_residual": ["0", "0"],
    }
    for c in ("Z_P", "Z_S", "N_P", "N_S", "E_P", "E_S"):
        cols[f"{c}_amplitude_snr"] = [5.0, 6.0]
        cols[f"{c}_power_snr"] = [5.0, 6.0]
    pd.DataFrame(cols).to_csv(tmp_path / "DiTing330km_light.csv")

    requested = []

    class FakeFile:
        def __init__(self, path, mode):
            assert path.endswith("DiTing330km_part_0.hdf5")

        def __enter__(self):
            return self

        def __exit__(self, *a):
            return False

        def get(self, key):
            requested.append(key)
            rng = np.random.default_rng(0)
            return rng.standard_normal((8192, 3)).astype(np.float64)

    import types
    fake = types.SimpleNamespace(File=FakeFile)
    monkeypatch.setattr(dt, "_h5py", lambda: fake)

    ds = build_dataset("diting_light", seed=0, mode="train",
                       data_dir=str(tmp_path), shuffle=False,
                       data_split=False)
    assert len(ds) == 2
    ev, meta = ds[0]
    assert requested[-1] == "earthquake/000123.4500"   # padding quirk
    assert ev["data"].shape == (3, 8192)               # transposed
    assert ev["ppks"] == [4000] and ev["spks"] == [6000]
    assert ev["pmp"] == [0] and ev["clr"] == [0]       # U->0, I->0
    assert ev["baz"] == [10.0]                         # 370 % 360
    # ms->ml magnitude conversion: (3.0 + 1.08) / 1.13
    assert ev["emg"][0] == pytest.approx((3.0 + 1.08) / 1.13, rel=1e-6)

    ev2, _ = ds[1]
    assert requested[-1] == "earthquake/000007.1000"
    assert ev2["pmp"] == [1] and ev2["clr"] == [1]     # D->1, E->1
    assert ev2["emg"][0] == pytest.approx(2.0)         # ml unchanged


def test_pnw_reader_with_fake_h5(tmp_path, monkeypatch):
    """PNW reader logic without the archive: seisbench trace_name parsing
    ("bucket$n,:c,:l" -> data/<bucket>[n]), NaN zeroing, 4-class polarity
    map, and the '|'-joined snr string parse (reference
    datasets/pnw.py:102-138)."""
    import seist_amd.data.pnw as pw

    df = pd.DataFrame({
        "trace_name": ["bucket3$7,:3,:6000", "bucket1$0,:3,:6000"],
        "trace_P_polarity": ["positive", "undecidable"],
        "preferred_source_magnitude_type": ["ml", "ml"],
        "preferred_source_magnitude": [9.5, 2.0],   # clipped to 8
        "trace_snr_db": ["10.0|nan|12.5", "3.0|4.0|5.0"],
        "trace_P_arrival_sample": [1000, 2000],
        "trace_S_arrival_sample": [1500, 2500],
    })
    df.to_csv(tmp_path / "comcat_metadata.csv", index=False)

    class FakeDset:
        def __init__(self, bucket):
            self.bucket = bucket

        def __getitem__(self, n):
            rng = np.random.default_rng(n)
            arr = rng.standard_normal((3, 6000)).astype(np.float64)
            arr[0, 0] = np.nan          # reader must nan_to_num
            return arr

    class FakeFile:
        def __init__(self, path, mode):
            assert path.endswith("comcat_waveforms.hdf5")

        def __enter__(self):
            return self

        def __exit__(self, *a):
            return False

        def get(self, key):
            assert key.startswith("data/bucket")
            return FakeDset(key)

    import types
    monkeypatch.setattr(pw, "_h5py",
                        lambda: types.SimpleNamespace(File=FakeFile))

    ds = build_dataset("pnw", seed=0, mode="train", data_dir=str(tmp_path),
                       shuffle=False, data_split=False)
    assert len(ds) == 2
    ev, meta = ds[0]
    assert ev["data"].shape == (3, 6000)
    assert np.isfinite(ev["data"]).all()         # NaN zeroed
    assert ev["pmp"] == [0]                      # positive -> 0
    assert float(ev["emg"][0]) == 8.0            # clipped
    assert ev["snr"].tolist() == [10.0, 0.0, 12.5]
    ev2, _ = ds[1]
    assert ev2["pmp"] == [2]                     # undecidable -> 2


def test_diting_multipart_concat_and_split(tmp_path, monkeypatch):
    """Full DiTing variant: multi-part CSV concat, deterministic
    seed-shuffle and train/val/test row split."""
    import seist_amd.data.diting as dt

    def part_df(part, n):
        cols = {
            "part": [part] * n, "key": [f"{100 + 50 * part + i}.0" for i in range(n)],
            "ev_id": list(range(n)), "evmag": ["1.0"] * n,
            "mag_type": ["ml"] * n, "p_pick": [4000] * n,
            "p_clarity": ["I"] * n, "p_motion": ["U"] * n,
            "s_pick": [6000] * n, "net": ["AA"] * n,
            "sta_id": list(range(n)), "dis": [1.0] * n,
            "st_mag": ["1.0"] * n, "baz": ["0.0"] * n,
            "P_residual": ["0"] * n, "S_residual": ["0"] * n,
        }
        for c in ("Z_P", "Z_S", "N_P", "N_S", "E_P", "E_S"):
            cols[f"{c}_amplitude_snr"] = [5.0] * n
            cols[f"{c}_power_snr"] = [5.0] * n
        return pd.DataFrame(cols)

    part_df(0, 6).to_csv(tmp_path / "DiTing330km_part_0.csv")
    part_df(1, 4).to_csv(tmp_path / "DiTing330km_part_1.csv")
    monkeypatch.setattr(dt.DiTing, "_part_range", (0, 2))

    tr = build_dataset("diting", seed=3, mode="train",
                       data_dir=str(tmp_path), shuffle=True, data_split=True,
                       train_size=0.8, val_size=0.1)
    va = build_dataset("diting", seed=3, mode="val", data_dir=str(tmp_path),
                       shuffle=True, data_split=True, train_size=0.8,
                       val_size=0.1)
    te = build_dataset("diting", seed=3, mode="test", data_dir=str(tmp_path),
                       shuffle=True, data_split=True, train_size=0.8,
                       val_size=0.1)
    assert len(tr) + len(va) + len(te) == 10
    assert len(tr) == 8 and len(va) == 1 and len(te) == 1
    # deterministic: same seed -> identical split membership
    tr2 = build_dataset("diting", seed=3, mode="train",
                        data_dir=str(tmp_path), shuffle=True,
                        data_split=True, train_size=0.8, val_size=0.1)
    keys = lambda ds: [ds._meta_data.iloc[i]["key"] for i in range(len(ds))]
    assert keys(tr) == keys(tr2)
    # disjointness across modes
    all_keys = keys(tr) + keys(va) + keys(te)
    assert len(set(all_keys)) == 10
