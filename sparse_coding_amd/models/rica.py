"""Reconstruction ICA (reference autoencoders/rica.py:9-60; orphan module —
not wired into the ensemble engine there either)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class RICA(nn.Module):
    def __init__(self, activation_size, n_dict_components, sparsity_coef=0.0, sparsity_loss="smooth_l1"):
        super().__init__()
        self.n_dict_components = n_dict_components
        self.activation_size = activation_size
        self.weights = nn.Parameter(torch.empty(n_dict_components, activation_size))
        nn.init.xavier_uniform_(self.weights)
        self.sparsity_loss = sparsity_loss
        self.sparsity_coef = sparsity_coef

    def forward(self, x):
        c = x @ self.weights.T
        x_hat = c @ self.weights
        return x_hat, c

    def loss(self, x, x_hat, c):
        l_rec = F.mse_loss(x, x_hat)
        if self.sparsity_loss == "smooth_l1":
            l_sp = F.smooth_l1_loss(c, torch.zeros_like(c))
        elif self.sparsity_loss == "l1":
            l_sp = F.l1_loss(c, torch.zeros_like(c))
        else:
            raise ValueError(self.sparsity_loss)
        return l_rec + self.sparsity_coef * l_sp, l_rec, l_sp

    def train_batch(self, batch, optimizer=None):
        if optimizer is None:
            raise ValueError("optimizer must be specified for RICA")
        optimizer.zero_grad()
        x_hat, c = self(batch)
        loss, l_rec, l_sp = self.loss(batch, x_hat, c)
        loss.backward()
        optimizer.step()
        return loss.detach(), l_rec.detach(), l_sp.detach()

    def get_dict(self):
        return self.weights

    def configure_optimizers(self, **kwargs):
        return torch.optim.Adam(self.parameters(), **kwargs)


RICA.__module__ = "autoencoders.rica"
