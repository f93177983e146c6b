"""Deep shrinkage autoencoder prototype (reference experiments/deep_ae_testing.py).

A small multi-layer encoder with learned soft-shrinkage between layers and a
normalized linear decoder; exploratory, not wired into the sweep engine
(mirrors the reference's status as a standalone prototype).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from sparse_coding_amd.models.learned_dict import normalize_rows


class ShrinkLayer(nn.Module):
    def __init__(self, d_in: int, d_out: int):
        super().__init__()
        self.lin = nn.Linear(d_in, d_out)
        # softplus(theta) is the shrinkage threshold; start it near 0.01
        # (softplus^-1(0.01) ≈ -4.6) so codes are alive at init.
        self.theta = nn.Parameter(torch.full((d_out,), -4.6))

    def forward(self, x):
        z = self.lin(x)
        return torch.sign(z) * F.relu(z.abs() - F.softplus(self.theta))


class DeepShrinkageAE(nn.Module):
    def __init__(self, activation_size: int, n_dict: int, depth: int = 3, l1_alpha: float = 1e-3):
        super().__init__()
        dims = [activation_size] + [n_dict] * depth
        self.layers = nn.ModuleList(ShrinkLayer(a, b) for a, b in zip(dims[:-1], dims[1:]))
        dec = torch.empty(n_dict, activation_size)
        nn.init.xavier_uniform_(dec)
        self.decoder = nn.Parameter(dec)
        self.l1_alpha = l1_alpha
        self.n_feats = n_dict
        self.activation_size = activation_size

    def encode(self, x):
        c = x
        for layer in self.layers:
            c = layer(c)
        return F.relu(c)

    def forward(self, x):
        c = self.encode(x)
        x_hat = c @ normalize_rows(self.decoder)
        mse = (x_hat - x).pow(2).mean()
        l1 = self.l1_alpha * torch.norm(c, 1, dim=-1).mean()
        return mse + l1, mse, l1, c

    def train_on(self, batches, lr: float = 1e-3, log_every: int = 0):
        opt = torch.optim.Adam(self.parameters(), lr=lr)
        for i, x in enumerate(batches):
            opt.zero_grad()
            loss, mse, l1, _ = self(x)
            loss.backward()
            opt.step()
            if log_every and i % log_every == 0:
                print(f"step {i}: loss={loss.item():.5f} mse={mse.item():.5f}")
        return self
