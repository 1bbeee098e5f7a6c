"""dist-PT network (Mousavi & Beroza 2020) — dilated-causal TCN with dual
(distance, p-travel-time) heads.

Parity with /root/reference/models/distpt_network.py: ResBlocks of two
causal dilated convs (dilations 2^0..2^10) + 1x1 residual, sum of block
outputs, last-timestep readout.
"""

import torch.nn as nn

from .. import ops
from ._blocks import run_bn, run_conv
from ._registry import register_model


class ResBlock(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, dilation,
                 drop_rate):
        super().__init__()
        self.conv0 = nn.Conv1d(in_channels, out_channels, kernel_size,
                               dilation=dilation)
        self.bn0 = nn.BatchNorm1d(out_channels)
        self.relu0 = nn.ReLU()
        self.dropout0 = nn.Dropout1d(drop_rate)
        self.conv1 = nn.Conv1d(out_channels, out_channels, kernel_size,
                               dilation=dilation)
        self.bn1 = nn.BatchNorm1d(out_channels)
        self.relu1 = nn.ReLU()
        self.dropout1 = nn.Dropout1d(drop_rate)
        self.conv_out = nn.Conv1d(out_channels, out_channels, kernel_size=1)

    def forward(self, x):
        causal_pad = (self.conv0.kernel_size[0] - 1) * self.conv0.dilation[0]
        x = run_conv(self.conv0, x, causal_pad, 0)
        x = run_bn(self.bn0, x, act="relu")
        x = self.dropout0(x)
        x = run_conv(self.conv1, x, causal_pad, 0)
        x = run_bn(self.bn1, x, act="relu")
        x = self.dropout1(x)
        x1 = x + ops.pointwise_conv(x, self.conv_out.weight,
                                    self.conv_out.bias)
        return x1, x


class TemporalConvLayer(nn.Module):
    def __init__(self, in_channels, out_channels=64, kernel_size=2,
                 num_conv_blocks=1, dilations=[1, 2, 4, 8, 16, 32],
                 drop_rate=0.0, return_sequences=False):
        super().__init__()
        self.conv_in = nn.Conv1d(in_channels, out_channels, kernel_size=1)
        self.conv_blocks = nn.ModuleList([
            ResBlock(out_channels, out_channels, kernel_size, dilation,
                     drop_rate)
            for dilation in dilations * num_conv_blocks
        ])
        self.return_sequences = return_sequences

    def forward(self, x):
        x = ops.pointwise_conv(x, self.conv_in.weight, self.conv_in.bias)
        shortcuts = []
        for conv in self.conv_blocks:
            x, sc = conv(x)
            shortcuts.append(sc)
        x = sum(shortcuts)
        if not self.return_sequences:
            x = x[:, :, -1]
        return x


class DistPT_Network(nn.Module):
    def __init__(self, in_channels, tcn_channels=20, kernel_size=6,
                 num_conv_blocks=1, dilations=[2**i for i in range(11)],
                 drop_rate=0.1, **kwargs):
        super().__init__()
        self.tcn = TemporalConvLayer(
            in_channels=in_channels, out_channels=tcn_channels,
            kernel_size=kernel_size, num_conv_blocks=num_conv_blocks,
            dilations=dilations, drop_rate=drop_rate)
        self.lin_dist = nn.Linear(tcn_channels, 2)
        self.lin_ptrvl = nn.Linear(tcn_channels, 2)

    def forward(self, x):
        x = self.tcn(x)
        return self.lin_dist(x), self.lin_ptrvl(x)


@register_model
def distpt_network(**kwargs):
    return DistPT_Network(**kwargs)
