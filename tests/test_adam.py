"""FusedAdam CPU path vs torch.optim.Adam / AdamW."""

import pytest
import torch

from seist_amd.ops import FusedAdam


@pytest.mark.parametrize("adamw,wd", [(False, 0.0), (False, 0.01),
                                      (True, 0.01)])
def test_fused_adam_matches_torch(adamw, wd):
    torch.manual_seed(0)
    net_a = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 1))
    net_b = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 1))
    net_b.load_state_dict(net_a.state_dict())

    opt_a = FusedAdam(net_a.parameters(), lr=1e-2, weight_decay=wd,
                      adamw=adamw)
    cls = torch.optim.AdamW if adamw else torch.optim.Adam
    opt_b = cls(net_b.parameters(), lr=1e-2, weight_decay=wd)

    x = torch.randn(32, 8)
    y = torch.randn(32, 1)
    for _ in range(5):
        for net, opt in ((net_a, opt_a), (net_b, opt_b)):
            opt.zero_grad()
            loss = ((net(x) - y) ** 2).mean()
            loss.backward()
            opt.step()
    for pa, pb in zip(net_a.parameters(), net_b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), (pa - pb).abs().max()


def test_fused_adam_bf16_master_weights():
    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.randn(64, dtype=torch.bfloat16))
    opt = FusedAdam([p], lr=1e-2)
    for _ in range(3):
        opt.zero_grad()
        (p.float() ** 2).sum().backward()
        opt.step()
    state = opt.state[p]
    assert state["master"].dtype == torch.float32
    # master tracks more precisely than the bf16 copy
    assert torch.allclose(p.float(), state["master"],
                          atol=0.01 * state["master"].abs().max().item() + 1e-2)
