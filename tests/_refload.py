"""Helpers to load the upstream reference implementation (read-only at
/root/reference) for golden-parity tests. The reference needs timm, GPUtil
and obspy which are absent in this image — stub them before import.

All parity tests must skip cleanly when the reference tree is absent
(e.g. on the GPU box, where only the repo snapshot is copied).
"""

import importlib.util
import os
import sys
import types

REF_ROOT = "/root/reference"


def reference_available() -> bool:
    return os.path.isdir(REF_ROOT)


def _install_stubs():
    if "timm" not in sys.modules:
        timm = types.ModuleType("timm")
        m1 = types.ModuleType("timm.models")
        m2 = types.ModuleType("timm.models.layers")
        from seist_amd.models.seist import DropPath
        m2.DropPath = DropPath
        timm.models = m1
        m1.layers = m2
        sys.modules.update({"timm": timm, "timm.models": m1,
                            "timm.models.layers": m2})
    if "GPUtil" not in sys.modules:
        gputil = types.ModuleType("GPUtil")
        gputil.getGPUs = lambda: []
        sys.modules["GPUtil"] = gputil
    if "h5py" not in sys.modules:
        try:
            import h5py  # noqa: F401
        except ImportError:
            h5py = types.ModuleType("h5py")
            h5py.File = None
            sys.modules["h5py"] = h5py
    if "obspy" not in sys.modules:
        obspy = types.ModuleType("obspy")
        sig = types.ModuleType("obspy.signal")
        trg = types.ModuleType("obspy.signal.trigger")
        trg.trigger_onset = None  # set by tests if needed
        obspy.signal = sig
        sig.trigger = trg
        sys.modules.update({"obspy": obspy, "obspy.signal": sig,
                            "obspy.signal.trigger": trg})


def load_ref_module(relpath: str, name: str):
    """Load a single reference module file standalone (its package
    ``__init__`` is NOT executed, so heavy deps are avoided)."""
    _install_stubs()
    if REF_ROOT not in sys.path:
        sys.path.insert(0, REF_ROOT)
    spec = importlib.util.spec_from_file_location(
        name, os.path.join(REF_ROOT, relpath))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def load_ref_utils():
    """Import the reference ``utils`` package (Metrics, misc, ...)."""
    _install_stubs()
    if REF_ROOT not in sys.path:
        sys.path.insert(0, REF_ROOT)
    import utils as ref_utils  # noqa
    return ref_utils


def load_ref_models():
    """Import the reference ``models`` package (registry + all models)."""
    _install_stubs()
    if REF_ROOT not in sys.path:
        sys.path.insert(0, REF_ROOT)
    import models as ref_models  # noqa
    return ref_models
