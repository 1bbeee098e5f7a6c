"""torch.jit traceability of the model zoo (eval mode).

The reference keeps its models export-friendly on purpose (explicit
padding instead of padding='same', reference models/seist.py:24); the op
wrappers here route through plain composites while the tracer runs
(`_trace_eager`), so traced CPU graphs contain only standard ATen ops.
"""

import warnings

import pytest
import torch

from seist_amd.models import create_model

MODELS = ["seist_s_dpk", "seist_m_dpk", "seist_l_dpk", "seist_m_pmp",
          "seist_m_emg", "phasenet", "eqtransformer", "magnet",
          "ditingmotion", "baz_network", "distpt_network"]


def _maxdiff(a, b):
    if isinstance(a, (list, tuple)):
        return max(_maxdiff(ai, bi) for ai, bi in zip(a, b))
    return (a - b).abs().max().item()


@pytest.mark.parametrize("name", MODELS)
def test_jit_trace_eval(name):
    torch.manual_seed(0)
    m = create_model(name, in_channels=3, in_samples=8192).eval()
    x = torch.randn(1, 3, 8192)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        with torch.no_grad():
            ref = m(x)
            traced = torch.jit.trace(m, x, check_trace=False)
            out = traced(x)
    assert _maxdiff(ref, out) < 1e-5, name
