"""EQTransformer (Mousavi et al. 2020) — detection + picking with
conv encoder, BiLSTM stack, additive-attention transformers and three
upsampling decoders.

Parity with /root/reference/models/eqtransformer.py: 7 conv+maxpool encoder
blocks (8192 -> 64), 5 residual conv blocks, 3 BiLSTM blocks, 2 global
single-head additive-attention transformer layers (L=64), decoders with
optional LSTM + width-3 banded local attention, 7 upsample(x2)+conv blocks
and sigmoid outputs concatenated to (N,3,8192). Optional L1-on-gradient
hooks on the encoder convs. Module names match for checkpoint interop.
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ._blocks import run_bn, run_conv, run_conv_bn
from ._registry import register_model

_EPS = 1e-6


def _same_pad(k: int):
    return ((k - 1) // 2, k - 1 - (k - 1) // 2)


class ConvBlock(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size,
                 kernel_l1_alpha, bias_l1_alpha):
        super().__init__()
        assert kernel_l1_alpha >= 0.0 and bias_l1_alpha >= 0.0
        self.conv_padding_same = _same_pad(kernel_size)
        self.conv = nn.Conv1d(in_channels, out_channels, kernel_size)
        self.relu = nn.ReLU()
        self.pool = nn.MaxPool1d(kernel_size=2, padding=0)
        if kernel_l1_alpha > 0.0:
            self.conv.weight.register_hook(
                lambda g: g.data + kernel_l1_alpha * torch.sign(self.conv.weight.data))
        if bias_l1_alpha > 0.0:
            self.conv.bias.register_hook(
                lambda g: g.data + bias_l1_alpha * torch.sign(self.conv.bias.data))

    def forward(self, x):
        x = run_conv(self.conv, x, *self.conv_padding_same).relu()
        x = F.pad(x, (0, x.size(-1) % 2), "constant", -1 / _EPS)
        return ops.max_pool1d(x, 2)


class ResConvBlock(nn.Module):
    def __init__(self, io_channels, kernel_size, drop_rate):
        super().__init__()
        self.conv_padding_same = _same_pad(kernel_size)
        self.bn0 = nn.BatchNorm1d(io_channels)
        self.relu0 = nn.ReLU()
        self.dropout0 = nn.Dropout1d(drop_rate)
        self.conv0 = nn.Conv1d(io_channels, io_channels, kernel_size)
        self.bn1 = nn.BatchNorm1d(io_channels)
        self.relu1 = nn.ReLU()
        self.dropout1 = nn.Dropout1d(drop_rate)
        self.conv1 = nn.Conv1d(io_channels, io_channels, kernel_size)

    def forward(self, x):
        x1 = self.dropout0(run_bn(self.bn0, x, act="relu"))
        x1 = run_conv_bn(self.conv0, self.bn1, x1, act="relu",
                         padl=self.conv_padding_same[0],
                         padr=self.conv_padding_same[1])
        x1 = self.dropout1(x1)
        x1 = run_conv(self.conv1, x1, *self.conv_padding_same)
        return x + x1


class BiLSTMBlock(nn.Module):
    def __init__(self, in_channels, out_channels, drop_rate):
        super().__init__()
        self.bilstm = nn.LSTM(in_channels, out_channels, batch_first=True,
                              bidirectional=True)
        self.dropout = nn.Dropout(drop_rate)
        self.conv = nn.Conv1d(2 * out_channels, out_channels, kernel_size=1)
        self.bn = nn.BatchNorm1d(out_channels)

    def forward(self, x):
        # LSTM weights stay fp32; cast bf16 encoder activations up.
        # On GPU the recurrence runs as the persistent K11 kernel (input
        # projections = one GEMM), not MIOpen's per-step launches.
        # ops.lstm owns the fp32 upcast (bf16 pre-projection GEMMs
        # measured slower same-box — see the note in ops/eqt.py)
        x = ops.lstm(x.permute(0, 2, 1).contiguous(), self.bilstm)
        x = self.dropout(x).permute(0, 2, 1)
        x = run_conv(self.conv, x)
        return run_bn(self.bn, x)


class AttentionLayer(nn.Module):
    """Single-head additive attention with optional banded mask
    (reference eqtransformer.py:135-198; K10 of SURVEY §2.4)."""

    def __init__(self, in_channels, d_model, attn_width=None):
        super().__init__()
        self.attn_width = attn_width
        self.Wx = nn.Parameter(torch.empty((in_channels, d_model)))
        self.Wt = nn.Parameter(torch.empty((in_channels, d_model)))
        self.bh = nn.Parameter(torch.empty(d_model))
        self.Wa = nn.Parameter(torch.empty((d_model, 1)))
        self.ba = nn.Parameter(torch.empty(1))
        nn.init.xavier_uniform_(self.Wx)
        nn.init.xavier_uniform_(self.Wt)
        nn.init.xavier_uniform_(self.Wa)
        nn.init.zeros_(self.bh)
        nn.init.zeros_(self.ba)

    def forward(self, x):
        x = x.permute(0, 2, 1)                       # (N,L,C)
        q = torch.matmul(x, self.Wt)                 # (N,L,d)
        k = torch.matmul(x, self.Wx)                 # (N,L,d)
        # fused K10 kernel on GPU: the (N,L,L,d) tanh tensor never exists
        a = ops.additive_attention_weights(q, k, self.bh, self.Wa, self.ba,
                                           self.attn_width)
        v = torch.matmul(a, x).permute(0, 2, 1)
        return v, a


class FeedForward(nn.Module):
    def __init__(self, io_channels, feedforward_dim, drop_rate):
        super().__init__()
        self.lin0 = nn.Linear(io_channels, feedforward_dim)
        self.relu = nn.ReLU()
        self.dropout = nn.Dropout(drop_rate)
        self.lin1 = nn.Linear(feedforward_dim, io_channels)
        nn.init.xavier_uniform_(self.lin0.weight)
        nn.init.zeros_(self.lin0.bias)
        nn.init.xavier_uniform_(self.lin1.weight)
        nn.init.zeros_(self.lin1.bias)

    def forward(self, x):
        return self.lin1(self.dropout(self.relu(self.lin0(x))))


class TransformerLayer(nn.Module):
    def __init__(self, io_channels, d_model, feedforward_dim, drop_rate,
                 attn_width=None):
        super().__init__()
        self.attn = AttentionLayer(io_channels, d_model, attn_width)
        self.ln0 = nn.LayerNorm(io_channels)
        self.ff = FeedForward(io_channels, feedforward_dim, drop_rate)
        self.ln1 = nn.LayerNorm(io_channels)

    def forward(self, x):
        x = x.float()  # attention/LayerNorm stage stays fp32 (tiny, L=64)
        x1, w = self.attn(x)
        x2 = (x1 + x).permute(0, 2, 1)
        x2 = ops.layer_norm(x2, self.ln0.weight, self.ln0.bias, self.ln0.eps)
        x4 = ops.layer_norm(self.ff(x2) + x2, self.ln1.weight, self.ln1.bias,
                            self.ln1.eps).permute(0, 2, 1)
        return x4, w


class Encoder(nn.Module):
    def __init__(self, in_channels, conv_channels, conv_kernels,
                 resconv_kernels, num_lstm_blocks, num_transformer_layers,
                 transformer_io_channels, transformer_d_model,
                 feedforward_dim, drop_rate,
                 conv_kernel_l1_regularization=0.0,
                 conv_bias_l1_regularization=0.0):
        super().__init__()
        self.convs = nn.Sequential(*[
            ConvBlock(inc, outc, kers, conv_kernel_l1_regularization,
                      conv_bias_l1_regularization)
            for inc, outc, kers in zip([in_channels] + conv_channels[:-1],
                                       conv_channels, conv_kernels)
        ])
        self.res_convs = nn.Sequential(*[
            ResConvBlock(conv_channels[-1], kers, drop_rate)
            for kers in resconv_kernels
        ])
        self.bilstms = nn.Sequential(*[
            BiLSTMBlock(inc, outc, drop_rate)
            for inc, outc in zip(
                [conv_channels[-1]] + [transformer_io_channels] * (num_lstm_blocks - 1),
                [transformer_io_channels] * num_lstm_blocks)
        ])
        self.transformers = nn.ModuleList([
            TransformerLayer(transformer_io_channels, transformer_d_model,
                             feedforward_dim, drop_rate)
            for _ in range(num_transformer_layers)
        ])

    def forward(self, x):
        x = self.bilstms(self.res_convs(self.convs(x)))
        for tr in self.transformers:
            x, w = tr(x)
        return x, w


class UpSamplingBlock(nn.Module):
    def __init__(self, in_channels, out_channels, out_samples, kernel_size,
                 kernel_l1_alpha, bias_l1_alpha):
        super().__init__()
        assert kernel_l1_alpha >= 0.0 and bias_l1_alpha >= 0.0
        self.out_samples = out_samples
        self.conv_padding_same = _same_pad(kernel_size)
        self.upsampling = nn.Upsample(scale_factor=2)
        self.conv = nn.Conv1d(in_channels, out_channels, kernel_size)
        self.relu = nn.ReLU()
        if kernel_l1_alpha > 0.0:
            self.conv.weight.register_hook(
                lambda g: g.data + kernel_l1_alpha * torch.sign(self.conv.weight.data))
        if bias_l1_alpha > 0.0:
            self.conv.bias.register_hook(
                lambda g: g.data + bias_l1_alpha * torch.sign(self.conv.bias.data))

    def forward(self, x):
        x = ops.upsample2x(x)[:, :, : self.out_samples]
        return run_conv(self.conv, x, *self.conv_padding_same).relu()


class IdentityNTuple(nn.Identity):
    def __init__(self, *args, ntuple: int = 1, **kwargs):
        super().__init__(*args, **kwargs)
        assert ntuple >= 1
        self.ntuple = ntuple

    def forward(self, input: torch.Tensor):
        if self.ntuple > 1:
            return (input,) * self.ntuple
        return input


class Decoder(nn.Module):
    def __init__(self, conv_channels, conv_kernels, transformer_io_channels,
                 transformer_d_model, feedforward_dim, drop_rate, out_samples,
                 has_lstm=True, has_local_attn=True, local_attn_width=3,
                 conv_kernel_l1_regularization=0.0,
                 conv_bias_l1_regularization=0.0):
        super().__init__()
        self.lstm = (nn.LSTM(transformer_io_channels, transformer_io_channels,
                             batch_first=True, bidirectional=False)
                     if has_lstm else IdentityNTuple(ntuple=2))
        self.lstm_dropout = nn.Dropout(drop_rate) if has_lstm else nn.Identity()
        self.transformer = (
            TransformerLayer(transformer_io_channels, transformer_d_model,
                             feedforward_dim, drop_rate,
                             attn_width=local_attn_width)
            if has_local_attn else IdentityNTuple(ntuple=2))

        crop_sizes = [out_samples]
        for _ in range(len(conv_kernels) - 1):
            crop_sizes.insert(0, math.ceil(crop_sizes[0] / 2))
        self.upsamplings = nn.Sequential(*[
            UpSamplingBlock(inc, outc, crop, kers,
                            conv_kernel_l1_regularization,
                            conv_bias_l1_regularization)
            for inc, outc, crop, kers in zip(
                [transformer_io_channels] + conv_channels[:-1],
                conv_channels, crop_sizes, conv_kernels)
        ])
        self.conv_out = nn.Conv1d(conv_channels[-1], 1, kernel_size=11,
                                  padding=5)

    def forward(self, x):
        x = x.permute(0, 2, 1).float().contiguous()  # decoder LSTM stays fp32
        if isinstance(self.lstm, nn.LSTM):
            x = ops.lstm(x, self.lstm)
        else:
            x, _ = self.lstm(x)
        x = self.lstm_dropout(x).permute(0, 2, 1)
        x, _ = self.transformer(x)
        x = self.upsamplings(x)
        x = run_conv(self.conv_out, x)
        return x.sigmoid()


class EQTransformer(nn.Module):
    def __init__(self, in_channels=3, in_samples=8192,
                 conv_channels=[8, 16, 16, 32, 32, 64, 64],
                 conv_kernels=[11, 9, 7, 7, 5, 5, 3],
                 resconv_kernels=[3, 3, 3, 2, 2],
                 num_lstm_blocks=3, num_transformer_layers=2,
                 transformer_io_channels=16, transformer_d_model=32,
                 feedforward_dim=128, local_attention_width=3, drop_rate=0.1,
                 decoder_with_attn_lstm=[False, True, True],
                 conv_kernel_l1_regularization=0.0,
                 conv_bias_l1_regularization=0.0, **kwargs):
        super().__init__()
        assert len(conv_channels) == len(conv_kernels)
        self.encoder = Encoder(
            in_channels=in_channels, conv_channels=conv_channels,
            conv_kernels=conv_kernels, resconv_kernels=resconv_kernels,
            num_lstm_blocks=num_lstm_blocks,
            num_transformer_layers=num_transformer_layers,
            transformer_io_channels=transformer_io_channels,
            transformer_d_model=transformer_d_model,
            feedforward_dim=feedforward_dim, drop_rate=drop_rate,
            conv_kernel_l1_regularization=conv_kernel_l1_regularization,
            conv_bias_l1_regularization=conv_bias_l1_regularization)
        self.decoders = nn.ModuleList([
            Decoder(conv_channels=conv_channels[::-1],
                    conv_kernels=conv_kernels[::-1],
                    transformer_io_channels=transformer_io_channels,
                    transformer_d_model=transformer_d_model,
                    feedforward_dim=feedforward_dim, drop_rate=drop_rate,
                    out_samples=in_samples, has_lstm=has_al,
                    has_local_attn=has_al,
                    local_attn_width=local_attention_width,
                    conv_kernel_l1_regularization=conv_kernel_l1_regularization,
                    conv_bias_l1_regularization=conv_bias_l1_regularization)
            for has_al in decoder_with_attn_lstm
        ])

    def forward(self, x):
        feature, _ = self.encoder(x)
        return torch.cat([decoder(feature) for decoder in self.decoders],
                         dim=1)


@register_model
def eqtransformer(**kwargs):
    return EQTransformer(**kwargs)
