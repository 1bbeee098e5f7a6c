"""Loss numerics vs the reference loss module (golden parity) and vs
hand-computed values."""

import pytest
import torch

from seist_amd.models import (BCELoss, BinaryFocalLoss, CELoss,
                              CombinationLoss, FocalLoss, MousaviLoss,
                              MSELoss)

from _refload import load_ref_module, reference_available


def test_celoss_hand_value():
    p = torch.tensor([[[0.5], [0.5]]])
    t = torch.tensor([[[1.0], [0.0]]])
    loss = CELoss()(p, t)
    assert torch.allclose(loss, -torch.log(torch.tensor(0.5 + 1e-6)))


def test_mousavi_hand_value():
    preds = torch.tensor([[2.0, 0.0]])
    targets = torch.tensor([[3.0]])
    # 0.5*exp(0)*1 + 0 = 0.5
    assert torch.allclose(MousaviLoss()(preds, targets), torch.tensor(0.5))


@pytest.mark.skipif(not reference_available(), reason="reference absent")
def test_all_losses_match_reference():
    ref = load_ref_module("models/loss.py", "ref_loss")
    torch.manual_seed(0)
    p = torch.rand(4, 3, 64).clamp(1e-4, 1 - 1e-4)
    t = torch.rand(4, 3, 64)
    pairs = [
        (CELoss(weight=[[1], [2], [3]]), ref.CELoss(weight=[[1], [2], [3]])),
        (BCELoss(weight=[[0.5], [1], [1]]),
         ref.BCELoss(weight=[[0.5], [1], [1]])),
        (FocalLoss(), ref.FocalLoss()),
        (BinaryFocalLoss(), ref.BinaryFocalLoss()),
        (MSELoss(), ref.MSELoss()),
    ]
    for ours, theirs in pairs:
        a = ours(p, t)
        b = theirs(p, t)
        assert torch.allclose(a, b, atol=1e-6), type(ours).__name__

    # regression losses on (N,2)/(N,1)
    preds = torch.randn(8, 2)
    targets = torch.randn(8, 1)
    assert torch.allclose(MousaviLoss()(preds, targets),
                          ref.MousaviLoss()(preds, targets))

    comb_o = CombinationLoss(losses=[MSELoss, MSELoss])
    comb_r = ref.CombinationLoss(losses=[ref.MSELoss, ref.MSELoss])
    po = (torch.randn(8, 1), torch.randn(8, 1))
    to = (torch.randn(8, 1), torch.randn(8, 1))
    assert torch.allclose(comb_o(po, to), comb_r(po, to))
