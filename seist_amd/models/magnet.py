"""MagNet (Mousavi & Beroza 2020) — conv + BiLSTM magnitude estimator.

Parity with /root/reference/models/magnet.py: two conv+maxpool blocks,
one bidirectional LSTM (hidden 100), linear head emitting (mag, log-var)
for MousaviLoss.
"""

import torch
import torch.nn as nn

from .. import ops
from ..ops.functional import auto_pad_lr
from ._blocks import run_conv
from ._registry import register_model


class ConvBlock(nn.Module):
    def __init__(self, in_channels, out_channels, conv_kernel_size,
                 pool_kernel_size, drop_rate):
        super().__init__()
        self.conv = nn.Conv1d(in_channels, out_channels, conv_kernel_size)
        self.dropout = nn.Dropout(drop_rate)
        self.pool = nn.MaxPool1d(pool_kernel_size, ceil_mode=True)

    def forward(self, x):
        pl, pr = auto_pad_lr(x.size(-1), self.conv.kernel_size[0])
        x = run_conv(self.conv, x, pl, pr)
        x = self.dropout(x)
        return ops.max_pool1d(x, self.pool.kernel_size, ceil_mode=True)


class MagNet(nn.Module):
    def __init__(self, in_channels: int, conv_channels: list = [64, 32],
                 lstm_dim: int = 100, drop_rate: float = 0.2, **kwargs):
        super().__init__()
        self.conv_layers = nn.Sequential(*[
            ConvBlock(inc, outc, conv_kernel_size=3, pool_kernel_size=4,
                      drop_rate=drop_rate)
            for inc, outc in zip([in_channels] + conv_channels[:-1],
                                 conv_channels)
        ])
        self.lstm = nn.LSTM(conv_channels[-1], lstm_dim, num_layers=1,
                            batch_first=True, bidirectional=True)
        self.lin = nn.Linear(lstm_dim * 2, 2)

    def forward(self, x):
        x = self.conv_layers(x)
        x = x.transpose(-1, -2).float().contiguous()
        if x.is_cuda:
            # K11 persistent recurrence; h_n per direction is the last
            # processed step: t = L-1 forward, t = 0 reverse
            y = ops.lstm(x, self.lstm)
            H = self.lstm.hidden_size
            h = torch.cat([y[:, -1, :H], y[:, 0, H:]], dim=-1)
        else:
            hs, (h, c) = self.lstm(x)
            h = h.transpose(0, 1).flatten(1)
        return self.lin(h)


@register_model
def magnet(**kwargs):
    return MagNet(**kwargs)
