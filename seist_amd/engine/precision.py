"""Mixed-precision policy for MI355X training.

The reference trains fp32 with TF32 matmul (main.py:224-225). gfx950 has no
TF32/xf32 path, so the MI355X-native regime is bf16 compute with fp32
BatchNorm statistics/parameters and fp32 LSTM cells, plus fp32 master
weights inside the fused Adam (ops/adam.py). ``convert_to_bf16`` applies
that policy module-wise; fp32 stays the default for parity runs.
"""

import torch
import torch.nn as nn

_BF16_TYPES = (nn.Conv1d, nn.ConvTranspose1d, nn.Linear)
_FP32_TYPES = (nn.BatchNorm1d, nn.LayerNorm, nn.LSTM, nn.GRU,
               nn.InstanceNorm1d, nn.GroupNorm)


def convert_to_bf16(model: nn.Module) -> nn.Module:
    """Cast conv/linear parameters to bf16; keep norms and recurrent cells
    in fp32 (their params are small, their numerics are touchy).

    Models containing LSTM/LayerNorm (EQTransformer, MagNet) get the
    SELECTIVE policy: only Conv1d/ConvTranspose1d go bf16 — the big
    encoder/decoder tensors run on the bf16 matrix cores while the tiny
    recurrent/attention stage (L=64) keeps the reference's fp32 numerics.
    The conv path casts activations to the weight dtype and the models
    cast back to fp32 at their LSTM/LayerNorm boundaries."""
    selective = any(isinstance(m, (nn.LSTM, nn.GRU, nn.LayerNorm))
                    for m in model.modules())
    for m in model.modules():
        if isinstance(m, (nn.Conv1d, nn.ConvTranspose1d)):
            m.to(torch.bfloat16)
        elif not selective and isinstance(m, nn.Linear):
            m.to(torch.bfloat16)
        elif isinstance(m, _FP32_TYPES):
            m.to(torch.float32)
        elif not selective:
            for p in m.parameters(recurse=False):
                p.data = p.data.to(torch.bfloat16)
    return model


def cast_inputs(x, dtype):
    if isinstance(x, (list, tuple)):
        return type(x)(cast_inputs(xi, dtype) for xi in x)
    if torch.is_tensor(x) and x.is_floating_point():
        return x.to(dtype)
    return x
