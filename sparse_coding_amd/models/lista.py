"""LISTA / residual-denoising unrolled sparse encoders.

Parity with reference ``autoencoders/residual_denoising_autoencoder.py``
(LISTALayer :15-36, FunctionalLISTADenoisingSAE :39-103,
ResidualDenoising* :125-201; learned-ISTA per arXiv 2008.02683).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from sparse_coding_amd.models.learned_dict import LearnedDict, normalize_rows
from sparse_coding_amd.models.sae_signatures import DictSignature
from sparse_coding_amd.utils.tree import tree_map


def shrinkage(r: torch.Tensor, theta: torch.Tensor) -> torch.Tensor:
    return torch.sign(r) * F.relu(torch.abs(r) - theta)


class LISTALayer:
    @staticmethod
    def init(d_activation, n_features, dtype=torch.float32):
        w = torch.empty(n_features, d_activation, dtype=dtype)
        torch.nn.init.orthogonal_(w)
        return {
            "W": w,
            "theta": torch.randn(n_features, dtype=dtype) * 0.02,
            "rho": torch.tensor(0.1, dtype=dtype),
        }

    @staticmethod
    def forward(params, y, b, x, A):
        # one learned-ISTA iteration solving A^T c ~= b
        m = torch.clamp(params["rho"], min=0.0, max=1.0)
        Ay = y @ A  # einsum("ij,bi->bj", A, y)
        r = y + (b - Ay) @ params["W"].T
        x_ = shrinkage(r, params["theta"])
        y_ = x_ + m * (x_ - x)
        return y_, x_


class FunctionalLISTADenoisingSAE(DictSignature):
    @staticmethod
    def init(d_activation, n_features, n_hidden_layers, l1_alpha, dtype=torch.float32):
        dec = torch.empty(n_features, d_activation, dtype=dtype)
        torch.nn.init.orthogonal_(dec)
        params = {
            "decoder": dec,
            "encoder_layers": [LISTALayer.init(d_activation, n_features, dtype=dtype) for _ in range(n_hidden_layers)],
        }
        buffers = {"l1_alpha": torch.tensor(l1_alpha, dtype=dtype)}
        return params, buffers

    @staticmethod
    def encode(params, b, learned_dict):
        y = b @ learned_dict.T
        x = y
        for layer in params["encoder_layers"]:
            y, x = LISTALayer.forward(layer, y, b, x, learned_dict)
        return y

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["decoder"])
        c = FunctionalLISTADenoisingSAE.encode(params, batch, learned_dict)
        x_hat = c @ learned_dict
        l_rec = (x_hat - batch).pow(2).mean()
        l_sp = buffers["l1_alpha"] * torch.norm(c, 1, dim=-1).mean()
        total = l_rec + l_sp
        return total, ({"loss": total, "l_reconstruction": l_rec, "l_l1": l_sp}, {"c": c})

    @staticmethod
    def to_learned_dict(params, buffers):
        return LISTADenoisingSAE(params)

    @staticmethod
    def init_lr(n_hidden_layers, lr, lr_encoder=None):
        if lr_encoder is None:
            lr_encoder = lr
        lrs = {
            "decoder": lr,
            "encoder_embedding": lr_encoder,
            "encoder_bias": lr_encoder,
            "encoder_layers": [{"weight": lr, "bias": lr} for _ in range(n_hidden_layers)],
        }
        return lrs


class LISTADenoisingSAE(LearnedDict):
    def __init__(self, params):
        self.params = params
        self.n_feats, self.activation_size = params["decoder"].shape

    def encode(self, x):
        return FunctionalLISTADenoisingSAE.encode(self.params, x, self.get_learned_dict())

    def to_device(self, device):
        self.params = tree_map(lambda t: t.to(device=device), self.params)

    def get_learned_dict(self):
        return normalize_rows(self.params["decoder"])


class ResidualDenoisingLayer:
    @staticmethod
    def init(d_activation, n_features, dtype=torch.float32):
        w = torch.empty(n_features, n_features, dtype=dtype)
        torch.nn.init.orthogonal_(w)
        return {"W": w, "theta": torch.randn(n_features, dtype=dtype) * 0.02}

    @staticmethod
    def forward(params, x):
        x_ = F.relu(x + params["theta"])
        return x_ @ params["W"].T + x


class FunctionalResidualDenoisingSAE(DictSignature):
    @staticmethod
    def init(d_activation, n_features, n_hidden_layers, l1_alpha, dtype=torch.float32):
        dec = torch.empty(n_features, d_activation, dtype=dtype)
        torch.nn.init.orthogonal_(dec)
        params = {
            "decoder": dec,
            "encoder_layers": [ResidualDenoisingLayer.init(d_activation, n_features, dtype=dtype) for _ in range(n_hidden_layers)],
            "encoder_bias": torch.randn(n_features, dtype=dtype) * 0.02,
        }
        buffers = {"l1_alpha": torch.tensor(l1_alpha, dtype=dtype)}
        return params, buffers

    @staticmethod
    def encode(params, b, learned_dict):
        x = b @ learned_dict.T
        for layer in params["encoder_layers"]:
            x = ResidualDenoisingLayer.forward(layer, x)
        return F.relu(x + params["encoder_bias"])

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["decoder"])
        c = FunctionalResidualDenoisingSAE.encode(params, batch, learned_dict)
        x_hat = c @ learned_dict
        l_rec = (x_hat - batch).pow(2).mean()
        l_sp = buffers["l1_alpha"] * torch.norm(c, 1, dim=-1).mean()
        total = l_rec + l_sp
        return total, ({"loss": total, "l_reconstruction": l_rec, "l_l1": l_sp}, {"c": c})

    @staticmethod
    def to_learned_dict(params, buffers):
        return ResidualDenoisingSAE(params)


class ResidualDenoisingSAE(LearnedDict):
    def __init__(self, params):
        self.params = params
        self.n_feats, self.activation_size = params["decoder"].shape

    def encode(self, x):
        return FunctionalResidualDenoisingSAE.encode(self.params, x, self.get_learned_dict())

    def to_device(self, device):
        self.params = tree_map(lambda t: t.to(device=device), self.params)

    def get_learned_dict(self):
        return normalize_rows(self.params["decoder"])


for _cls in (
    LISTALayer,
    FunctionalLISTADenoisingSAE,
    LISTADenoisingSAE,
    ResidualDenoisingLayer,
    FunctionalResidualDenoisingSAE,
    ResidualDenoisingSAE,
):
    _cls.__module__ = "autoencoders.residual_denoising_autoencoder"
