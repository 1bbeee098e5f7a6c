"""Loss library.

Numerical parity with /root/reference/models/loss.py:8-210 (CELoss, BCELoss,
FocalLoss, BinaryFocalLoss, MSELoss, CombinationLoss, MousaviLoss, plus the
``HuberLoss`` re-export). All operate on probabilities (post-sigmoid/softmax
model outputs) like the reference; epsilon = 1e-6.

BCE and CE (the flagship/phasenet losses) dispatch to the fused K15 kernel
on GPU for fp32 scalar-weight inputs — ONE pass forward (grid-stride sum),
one elementwise pass backward (`ops/hip/loss.hip`) — instead of the ~10
eager elementwise+reduce kernels. The eager composites below remain the
semantics ground truth and run on CPU / per-channel-weight / non-fp32
inputs. The remaining losses stay eager on purpose: they are <1% of any
step they appear in (rocprof, profiles/step_profile_r01.md).
"""

from typing import Tuple

import torch
import torch.nn as nn
from torch.nn import HuberLoss  # noqa: F401  (re-export, parity with reference)

from ..ops.functional import LOSS_BCE, LOSS_CE, fused_prob_loss

_EPS = 1e-6


def _register_weight(module: nn.Module, weight) -> None:
    if weight is not None:
        w = torch.tensor(weight, dtype=torch.float32)
    else:
        w = torch.tensor(1.0, dtype=torch.float32)
    module.register_buffer("weight", w)


class CELoss(nn.Module):
    """Cross entropy over probability inputs: mean over batch of
    sum_c -w_c * t * log(p + eps). Shapes (N,C,L) or (N,Classes)."""

    def __init__(self, weight=None) -> None:
        super().__init__()
        _register_weight(self, weight)

    def forward(self, preds, targets):
        fused = fused_prob_loss(preds, targets, self.weight, LOSS_CE)
        if fused is not None:
            return fused
        loss = -targets * torch.log(preds + _EPS) * self.weight
        return loss.sum(1).mean()


class BCELoss(nn.Module):
    """Binary cross entropy with per-channel weights; mean over all elems."""

    def __init__(self, weight=None) -> None:
        super().__init__()
        _register_weight(self, weight)

    def forward(self, preds, targets):
        fused = fused_prob_loss(preds, targets, self.weight, LOSS_BCE)
        if fused is not None:
            return fused
        loss = -(
            targets * torch.log(preds + _EPS)
            + (1.0 - targets) * torch.log(1.0 - preds + _EPS)
        )
        return (loss * self.weight).mean()


class FocalLoss(nn.Module):
    """Multi-class focal loss; optionally applies softmax internally."""

    def __init__(self, gamma=2, weight=None, has_softmax=True):
        super().__init__()
        self.gamma = gamma
        self.has_softmax = has_softmax
        _register_weight(self, weight)

    def forward(self, preds, targets):
        if self.has_softmax:
            preds = torch.softmax(preds, dim=1)
        loss = -targets * torch.log(preds + _EPS)
        loss = loss * torch.pow(1.0 - preds, self.gamma) * self.weight
        return loss.sum(1).mean()


class BinaryFocalLoss(nn.Module):
    """Binary focal loss on sigmoid outputs."""

    def __init__(self, gamma=2, alpha=1, weight=None):
        super().__init__()
        self.gamma = gamma
        self.alpha = alpha
        _register_weight(self, weight)

    def forward(self, preds, targets):
        loss = -(
            self.alpha
            * torch.pow(1.0 - preds, self.gamma)
            * targets
            * torch.log(preds + _EPS)
            + (1.0 - self.alpha)
            * torch.pow(preds, self.gamma)
            * (1.0 - targets)
            * torch.log(1.0 - preds + _EPS)
        )
        return (loss * self.weight).mean()


class MSELoss(nn.Module):
    def __init__(self, weight=None) -> None:
        super().__init__()
        _register_weight(self, weight)

    def forward(self, preds, targets):
        return (((preds - targets) ** 2) * self.weight).mean()


class CombinationLoss(nn.Module):
    """Weighted sum of sub-losses for multi-output models."""

    def __init__(self, losses: list, losses_weights: list = None) -> None:
        super().__init__()
        assert len(losses) >= 2, (
            "`CombinationLoss` is for multi-task training and needs >= 2 losses"
        )
        if losses_weights is not None:
            assert len(losses) == len(losses_weights)
            self.losses_weights = losses_weights
        else:
            self.losses_weights = [1.0] * len(losses)
        self.losses = nn.ModuleList([L() for L in losses])

    def forward(self, preds: Tuple[torch.Tensor], targets: Tuple[torch.Tensor]):
        total = 0.0
        for pred, target, fn, w in zip(preds, targets, self.losses, self.losses_weights):
            total = total + fn(pred, target) * w
        return total


class MousaviLoss(nn.Module):
    """Heteroscedastic regression loss (MagNet / dist-PT):
    sum( 0.5*exp(-s)*(t - y)^2 + 0.5*s ) where preds[:,0]=y, preds[:,1]=s."""

    def forward(self, preds, targets):
        y_hat = preds[:, 0].reshape(-1, 1)
        s = preds[:, 1].reshape(-1, 1)
        return torch.sum(
            0.5 * torch.exp(-s) * torch.square(torch.abs(targets - y_hat)) + 0.5 * s
        )
