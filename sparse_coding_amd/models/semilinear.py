"""Semilinear SAE: 2-layer MLP encoder + normalized linear decoder.

Parity with reference ``autoencoders/semilinear_autoencoder.py:14-83``.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from sparse_coding_amd.models.learned_dict import normalize_rows
from sparse_coding_amd.models.sae_signatures import DictSignature


class FFLayer:
    @staticmethod
    def init(input_size, output_size, device=None, dtype=None):
        w = torch.empty(output_size, input_size, device=device, dtype=dtype)
        nn.init.xavier_uniform_(w)
        return {"weight": w, "bias": torch.zeros(output_size, device=device, dtype=dtype)}

    @staticmethod
    def forward(params, x):
        return torch.clamp(x @ params["weight"].T + params["bias"], min=0.0)


class SemiLinearSAE(DictSignature):
    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, device=None, dtype=None, hidden_size=None):
        if hidden_size is None:
            hidden_size = n_dict_components
        dec = torch.empty(n_dict_components, activation_size, device=device, dtype=dtype)
        nn.init.xavier_uniform_(dec)
        params = {
            "encoder_layers": [
                FFLayer.init(activation_size, hidden_size, device=device, dtype=dtype),
                FFLayer.init(hidden_size, n_dict_components, device=device, dtype=dtype),
            ],
            "decoder": dec,
        }
        buffers = {"l1_alpha": torch.tensor(l1_alpha, device=device, dtype=dtype)}
        return params, buffers

    @staticmethod
    def loss(params, buffers, batch):
        c = batch
        for layer in params["encoder_layers"]:
            c = FFLayer.forward(layer, c)
        normed = normalize_rows(params["decoder"])
        x_hat = c @ normed
        l_rec = (x_hat - batch).pow(2).mean()
        l_l1 = buffers["l1_alpha"] * torch.norm(c, 1, dim=-1).mean()
        total = l_rec + l_l1
        return total, ({"loss": total, "l_reconstruction": l_rec, "l_l1": l_l1}, {"c": c})


for _cls in (FFLayer, SemiLinearSAE):
    _cls.__module__ = "autoencoders.semilinear_autoencoder"
