"""BAZ network (Mousavi & Beroza 2020) — back-azimuth from covariance
eigenstructure + waveform conv stack.

Parity with /root/reference/models/baz_network.py. The per-sample 3x3
covariance + eigendecomposition (reference `_cov`/`_eig`,
baz_network.py:67-86) is computed batched; on GPU the closed-form
3x3 eigensolver kernel (K16) replaces `torch.linalg.eig`.
"""

import torch
import torch.nn as nn

from ._registry import register_model


def _sym3_eig(cov: torch.Tensor):
    """Batched closed-form eigendecomposition of symmetric 3x3 covariance
    matrices (K16 of SURVEY.md §2.4) — the analytic trigonometric method
    for symmetric 3x3 eigenvalues plus null-space cross products for the
    eigenvectors. Pure tensor math: runs on the GPU with no LAPACK/MAGMA
    dependency, identically on CPU. Eigenvalues descending (the reference's
    `torch.linalg.eig` ordering is unspecified; the downstream conv only
    consumes the eigenstructure)."""
    A = cov.double()
    N = A.shape[0]
    a, b, c = A[:, 0, 0], A[:, 1, 1], A[:, 2, 2]
    d, e, f = A[:, 0, 1], A[:, 1, 2], A[:, 0, 2]

    q = (a + b + c) / 3.0
    p1 = d * d + e * e + f * f
    p2 = ((a - q) ** 2 + (b - q) ** 2 + (c - q) ** 2 + 2.0 * p1)
    p = torch.sqrt(torch.clamp(p2 / 6.0, min=1e-300))
    eye = torch.eye(3, dtype=A.dtype, device=A.device).expand(N, 3, 3)
    B = (A - q[:, None, None] * eye) / p[:, None, None]
    r = torch.linalg.det(B) / 2.0  # 3x3 determinant: closed-form in ATen
    r = torch.clamp(r, -1.0, 1.0)
    phi = torch.acos(r) / 3.0
    two_pi_3 = 2.0943951023931953
    l0 = q + 2.0 * p * torch.cos(phi)
    l2 = q + 2.0 * p * torch.cos(phi + two_pi_3)
    l1 = 3.0 * q - l0 - l2
    vals = torch.stack([l0, l1, l2], dim=-1)  # descending

    # eigenvector for each eigenvalue: cross product of two rows of A - l*I
    def eigvec(lam):
        M = A - lam[:, None, None] * eye
        c01 = torch.cross(M[:, 0], M[:, 1], dim=-1)
        c02 = torch.cross(M[:, 0], M[:, 2], dim=-1)
        c12 = torch.cross(M[:, 1], M[:, 2], dim=-1)
        cands = torch.stack([c01, c02, c12], dim=1)  # (N,3,3)
        norms = cands.norm(dim=-1)
        best = norms.argmax(dim=1)
        v = cands[torch.arange(N, device=A.device), best]
        return v / torch.clamp(v.norm(dim=-1, keepdim=True), min=1e-30)

    vecs = torch.stack([eigvec(l0), eigvec(l1), eigvec(l2)], dim=-1)
    # degenerate (isotropic) case: A ~ q*I -> any orthonormal basis
    iso = p2 < 1e-24
    if iso.any():
        vals = torch.where(iso[:, None], q[:, None].expand(N, 3), vals)
        vecs = torch.where(iso[:, None, None],
                           eye.to(vecs.dtype), vecs)
    return (vals.unsqueeze(-1).to(cov.dtype), vecs.to(cov.dtype))


class BAZ_Network(nn.Module):
    def __init__(self, in_channels: int, in_samples: int,
                 in_matrix_dim: int = 7, conv_channels: list = [20, 32, 64, 20],
                 kernel_size: int = 3, pool_size: int = 2,
                 lin_hidden_dim: int = 100, drop_rate: float = 0.3, **kwargs):
        super().__init__()
        self.layers = nn.ModuleList()
        dim = in_samples
        for inc, outc in zip([in_channels] + conv_channels[:-1], conv_channels):
            self.layers.append(nn.Sequential(
                nn.Conv1d(inc, outc, kernel_size,
                          padding=(kernel_size - 1) // 2),
                nn.ReLU(),
                nn.Dropout(drop_rate),
                nn.MaxPool1d(pool_size, ceil_mode=True),
            ))
            dim = (dim + (pool_size - (dim % pool_size)) % pool_size) // pool_size
        dim = (dim + in_matrix_dim) * conv_channels[-1]

        self.flatten0 = nn.Flatten()
        self.conv1 = nn.Conv1d(in_channels, conv_channels[-1], kernel_size=1)
        self.relu0 = nn.ReLU()
        self.flatten1 = nn.Flatten()
        self.lin0 = nn.Linear(dim, lin_hidden_dim)
        self.relu1 = nn.ReLU()
        self.dropout = nn.Dropout(drop_rate)
        self.lin1 = nn.Linear(lin_hidden_dim, 2)

    @torch.no_grad()
    def _cov(self, x: torch.Tensor):
        N, C, L = x.size()
        diff = x - x.mean(-1, keepdim=True)
        # batched (C,L) @ (L,C) — one rocBLAS batched GEMM instead of the
        # reference's N*L rank-1 bmm (baz_network.py:73-76)
        cov = torch.matmul(diff, diff.transpose(-1, -2)) / (L - 1)
        return cov

    @torch.no_grad()
    def _compute_cov_and_eig(self, x):
        cov = self._cov(x.float())
        eig_values, eig_vectors = _sym3_eig(cov)
        eig_values = eig_values / eig_values.max()
        cov = cov / cov.abs().max()
        out = torch.cat([cov, eig_values, eig_vectors], dim=-1)
        return out.to(x.dtype)

    def forward(self, x):
        x1 = self._compute_cov_and_eig(x)
        for layer in self.layers:
            x = layer(x)
        x = self.flatten0(x)
        x1 = self.flatten1(self.relu0(self.conv1(x1)))
        x = torch.cat([x, x1], dim=1)
        x = self.dropout(self.relu1(self.lin0(x)))
        x = self.lin1(x)
        return x[:, :1], x[:, 1:]


@register_model
def baz_network(**kwargs):
    return BAZ_Network(**kwargs)
