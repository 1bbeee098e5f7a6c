"""PhaseNet (Zhu & Beroza 2019) — 1D U-Net for phase picking.

Architecture parity with /root/reference/models/phasenet.py: stride-4
conv encoder over 5 channel stages (8192 -> 32), ConvTranspose1d decoder
with crop+concat skips, 3-class softmax output. Module/parameter names
match the reference for checkpoint interop.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F  # noqa: F401

from .. import ops
from ._blocks import run_bn, run_conv, run_conv_bn
from ._registry import register_model


class ConvBlock(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride,
                 drop_rate, has_stride_conv=True):
        super().__init__()
        self.stride = stride if has_stride_conv else 1
        self.kernel_padding = kernel_size - stride if has_stride_conv else 0
        self.conv0 = (nn.Conv1d(in_channels, in_channels, kernel_size,
                                stride=stride, bias=False)
                      if has_stride_conv else nn.Identity())
        self.bn0 = (nn.BatchNorm1d(in_channels) if has_stride_conv
                    else nn.Identity())
        self.relu0 = nn.ReLU() if has_stride_conv else nn.Identity()
        self.drop0 = nn.Dropout(drop_rate) if has_stride_conv else nn.Identity()

        self.conv_padding_same = (
            (kernel_size - 1) // 2,
            kernel_size - 1 - (kernel_size - 1) // 2,
        )
        self.conv1 = nn.Conv1d(in_channels, out_channels, kernel_size,
                               bias=False)
        self.bn1 = nn.BatchNorm1d(out_channels)
        self.relu1 = nn.ReLU()
        self.drop1 = nn.Dropout(drop_rate)

    def forward(self, x):
        p = (self.stride - (x.size(-1) % self.stride)) % self.stride \
            + self.kernel_padding
        if isinstance(self.conv0, nn.Conv1d):
            x = run_conv_bn(self.conv0, self.bn0, x, act="relu",
                            padl=p // 2, padr=p - p // 2)
            x = self.drop0(x)
        x = run_conv_bn(self.conv1, self.bn1, x, act="relu",
                        padl=self.conv_padding_same[0],
                        padr=self.conv_padding_same[1])
        return self.drop1(x)


class ConvTransBlock(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride,
                 drop_rate, has_conv_same=True, has_conv_trans=True):
        super().__init__()
        self.conv_padding_same = (
            ((kernel_size - 1) // 2, kernel_size - 1 - (kernel_size - 1) // 2)
            if has_conv_same else (0, 0)
        )
        self.conv0 = (nn.Conv1d(2 * in_channels, in_channels, kernel_size,
                                bias=False)
                      if has_conv_same else nn.Identity())
        self.bn0 = nn.BatchNorm1d(in_channels) if has_conv_same else nn.Identity()
        self.relu0 = nn.ReLU() if has_conv_same else nn.Identity()
        self.drop0 = nn.Dropout(drop_rate) if has_conv_trans else nn.Identity()
        self.convt = (nn.ConvTranspose1d(in_channels, out_channels,
                                         kernel_size, stride=stride, bias=False)
                      if has_conv_trans else nn.Identity())
        self.bn1 = (nn.BatchNorm1d(out_channels) if has_conv_trans
                    else nn.Identity())
        self.relu1 = nn.ReLU() if has_conv_trans else nn.Identity()
        self.drop1 = nn.Dropout(drop_rate) if has_conv_same else nn.Identity()

    def forward(self, x):
        if isinstance(self.conv0, nn.Conv1d):
            x = run_conv_bn(self.conv0, self.bn0, x, act="relu",
                            padl=self.conv_padding_same[0],
                            padr=self.conv_padding_same[1])
        x = self.drop0(x)
        if isinstance(self.convt, nn.ConvTranspose1d):
            x = ops.conv_transpose1d(x, self.convt.weight, self.convt.bias,
                                     stride=self.convt.stride[0])
            x = run_bn(self.bn1, x, act="relu")
        return self.drop1(x)


class PhaseNet(nn.Module):
    def __init__(self, in_channels=3, kernel_size=7, stride=4,
                 conv_channels=[8, 16, 32, 64, 128], drop_rate=0.1, **kwargs):
        super().__init__()
        self.in_channels = in_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.conv_channels = conv_channels
        self.depth = len(conv_channels)

        self.conv_padding_same = (
            (kernel_size - 1) // 2,
            kernel_size - 1 - (kernel_size - 1) // 2,
        )
        self.conv_in = nn.Conv1d(in_channels, conv_channels[0], kernel_size)
        self.bn_in = nn.BatchNorm1d(conv_channels[0])
        self.relu_in = nn.ReLU()
        self.drop_in = nn.Dropout(drop_rate)

        self.down_convs = nn.ModuleList([
            ConvBlock(inc, outc, kernel_size, stride, drop_rate,
                      has_stride_conv=(i != 0))
            for i, inc, outc in zip(
                range(self.depth),
                conv_channels[:1] + conv_channels[:-1],
                conv_channels)
        ])
        self.up_convs = nn.ModuleList([
            ConvTransBlock(inc, outc, kernel_size, stride, drop_rate,
                           has_conv_same=(i < self.depth - 1),
                           has_conv_trans=(i > 0))
            for i, inc, outc in zip(
                range(self.depth)[::-1],
                conv_channels[::-1],
                conv_channels[-2::-1] + [None])
        ])
        self.conv_out = nn.Conv1d(conv_channels[0], 3, 1)
        self.softmax = nn.Softmax(dim=1)

    def forward(self, x):
        x = run_conv(self.conv_in, x, *self.conv_padding_same)
        x = run_bn(self.bn_in, x, act="relu")
        x = self.drop_in(x)

        shortcuts = []
        for conv in self.down_convs[:-1]:
            x = conv(x)
            shortcuts.append(x)
        x = self.down_convs[-1](x)

        for convt, shortcut in zip(self.up_convs[:-1], shortcuts[::-1]):
            x = convt(x)
            p = ((self.stride - (shortcut.size(-1) % self.stride)) % self.stride
                 + self.kernel_size - self.stride)
            lp = p // 2
            rp = p - lp
            x = torch.cat([shortcut, x[:, :, lp:-rp]], dim=1)

        x = self.up_convs[-1](x)
        x = ops.pointwise_conv(x, self.conv_out.weight, self.conv_out.bias)
        return self.softmax(x)


@register_model
def phasenet(**kwargs):
    return PhaseNet(**kwargs)
