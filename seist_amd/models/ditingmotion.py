"""DiTingMotion (Zhao et al. 2023) — first-motion polarity classifier.

Parity with /root/reference/models/ditingmotion.py: dense multi-kernel
CombConvLayers, 5 blocks with concat shortcut + maxpool, dual side-layer
heads (clarity + polarity) on blocks 3-5, fuse MLPs; the final outputs are
the averages of the side and fused sigmoid outputs. Input is 2 channels
[z, dz] (config).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.functional import auto_pad_lr
from ._blocks import run_conv
from .. import ops
from ._registry import register_model


class CombConvLayer(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_sizes,
                 out_kernel_size, drop_rate):
        super().__init__()
        self.convs = nn.ModuleList([
            nn.Sequential(
                nn.Conv1d(in_channels, out_channels, kernel_size=kers),
                nn.ReLU(),
            )
            for kers in kernel_sizes
        ])
        self.dropout = nn.Dropout(drop_rate)
        self.out_conv = nn.Conv1d(
            in_channels + len(kernel_sizes) * out_channels,
            out_channels, kernel_size=out_kernel_size)
        self.out_relu = nn.ReLU()

    def forward(self, x):
        outs = [x]
        for conv_relu in self.convs:
            conv = conv_relu[0]
            pl, pr = auto_pad_lr(x.size(-1), conv.kernel_size[0])
            outs.append(run_conv(conv, x, pl, pr).relu())
        x = self.dropout(torch.cat(outs, dim=1))
        pl, pr = auto_pad_lr(x.size(-1), self.out_conv.kernel_size[0])
        return run_conv(self.out_conv, x, pl, pr).relu()


class BasicBlock(nn.Module):
    def __init__(self, in_channels, layer_channels, comb_kernel_sizes,
                 comb_out_kernel_size, drop_rate, pool_size):
        super().__init__()
        self.conv_layers = nn.Sequential(*[
            CombConvLayer(inc, outc, comb_kernel_sizes, comb_out_kernel_size,
                          drop_rate)
            for inc, outc in zip([in_channels] + layer_channels[:-1],
                                 layer_channels)
        ])
        self.pool = nn.MaxPool1d(pool_size)

    def forward(self, x):
        x1 = self.conv_layers(x)
        return ops.max_pool1d(torch.cat([x, x1], dim=1),
                              self.pool.kernel_size[0] if isinstance(self.pool.kernel_size, tuple) else self.pool.kernel_size)


class SideLayer(nn.Module):
    def __init__(self, in_channels, out_channels, comb_kernel_sizes,
                 comb_out_kernel_size, drop_rate, linear_in_dim,
                 linear_hidden_dim, linear_out_dim):
        super().__init__()
        self.conv_layer = CombConvLayer(in_channels, out_channels,
                                        comb_kernel_sizes,
                                        comb_out_kernel_size, drop_rate)
        self.flatten = nn.Flatten(1)
        self.lin0 = nn.Linear(linear_in_dim, linear_hidden_dim)
        self.relu = nn.ReLU()
        self.lin1 = nn.Linear(linear_hidden_dim, linear_out_dim)
        self.sigmoid = nn.Sigmoid()
        self.conv_out_channels = out_channels
        self.linear_in_dim = linear_in_dim

    def forward(self, x):
        x = self.conv_layer(x)
        N, C, L = x.size()
        if C * L != self.linear_in_dim:
            # official model expects (2, 128); interpolate to fit other shapes
            x = ops.nearest_resize(x,
                                   self.linear_in_dim // self.conv_out_channels)
        x1 = self.flatten(x)
        x2 = self.relu(self.lin0(x1))
        x3 = self.sigmoid(self.lin1(x2))
        return x1, x2, x3


class DiTingMotion(nn.Module):
    def __init__(
        self,
        in_channels: int,
        blocks_layer_channels: list = [[8, 8], [8, 8], [8, 8, 8], [8, 8, 8],
                                       [8, 8, 8]],
        side_layer_conv_channels: int = 2,
        blocks_sidelayer_linear_in_dims: list = [None, None, 32, 16, 16],
        blocks_sidelayer_linear_hidden_dims: list = [None, None, 8, 8, 8],
        comb_kernel_sizes: list = [3, 3, 5, 5],
        comb_out_kernel_size: int = 3,
        pool_size: int = 2,
        drop_rate: float = 0.2,
        fuse_hidden_dim: int = 8,
        num_polarity_classes: int = 2,
        num_clarity_classes: int = 2,
        **kwargs,
    ):
        super().__init__()
        self.blocks = nn.ModuleList()
        self.clarity_side_layers = nn.ModuleList()
        self.polarity_side_layers = nn.ModuleList()

        blocks_in_channels = [in_channels]
        for blc in blocks_layer_channels[:-1]:
            blocks_in_channels.append(blc[-1] + blocks_in_channels[-1])

        fuse_polarity_in_dim = fuse_clarity_in_dim = 0
        for inc, layer_channels, lin_in, lin_hidden in zip(
                blocks_in_channels, blocks_layer_channels,
                blocks_sidelayer_linear_in_dims,
                blocks_sidelayer_linear_hidden_dims):
            self.blocks.append(BasicBlock(
                inc, layer_channels, comb_kernel_sizes, comb_out_kernel_size,
                drop_rate, pool_size))
            if lin_in is not None:
                self.clarity_side_layers.append(SideLayer(
                    layer_channels[-1] + inc, side_layer_conv_channels,
                    comb_kernel_sizes, comb_out_kernel_size, drop_rate,
                    lin_in, lin_hidden, num_clarity_classes))
                self.polarity_side_layers.append(SideLayer(
                    layer_channels[-1] + inc, side_layer_conv_channels,
                    comb_kernel_sizes, comb_out_kernel_size, drop_rate,
                    lin_in, lin_hidden, num_polarity_classes))
                fuse_clarity_in_dim += lin_in
                fuse_polarity_in_dim += lin_hidden
            else:
                self.clarity_side_layers.append(None)
                self.polarity_side_layers.append(None)

        self.fuse_polarity = nn.Sequential(
            nn.Linear(fuse_polarity_in_dim, fuse_hidden_dim),
            nn.Linear(fuse_hidden_dim, num_polarity_classes),
            nn.Sigmoid(),
        )
        self.fuse_clarity = nn.Sequential(
            nn.Linear(fuse_clarity_in_dim, fuse_hidden_dim),
            nn.Linear(fuse_hidden_dim, num_clarity_classes),
            nn.Sigmoid(),
        )

    def forward(self, x):
        clarity_to_fuse, polarity_to_fuse = [], []
        clarity_outs, polarity_outs = [], []
        for block, c_side, p_side in zip(self.blocks,
                                         self.clarity_side_layers,
                                         self.polarity_side_layers):
            x = block(x)
            if c_side is not None and p_side is not None:
                c0, _, c2 = c_side(x)
                clarity_to_fuse.append(c0)
                clarity_outs.append(c2)
                _, p1, p2 = p_side(x)
                polarity_to_fuse.append(p1)
                polarity_outs.append(p2)

        clarity_outs.append(self.fuse_clarity(torch.cat(clarity_to_fuse, -1)))
        polarity_outs.append(self.fuse_polarity(torch.cat(polarity_to_fuse, -1)))

        final_clarity = sum(clarity_outs) / len(clarity_outs)
        final_polarity = sum(polarity_outs) / len(polarity_outs)
        return final_clarity, final_polarity


@register_model
def ditingmotion(**kwargs):
    kwargs.setdefault("num_polarity_classes", 2)
    kwargs.setdefault("num_clarity_classes", 2)
    return DiTingMotion(**kwargs)
