"""Nonnegative-encoder (positive) tied SAEs used for MLP-space experiments.

Parity with reference ``autoencoders/mlp_tests.py`` (TiedPositiveSAE:8,
UntiedPositiveSAE:38, FunctionalPositiveTiedSAE:68-125, including the
hard-coded +0.18 input shift at :104,110).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from sparse_coding_amd.models.learned_dict import LearnedDict, TiedSAE, normalize_rows, relu_encode
from sparse_coding_amd.models.sae_signatures import DictSignature

INPUT_SHIFT = 0.18  # reference mlp_tests.py:104,110


class TiedPositiveSAE(LearnedDict):
    def __init__(self, encoder, encoder_bias, norm_encoder=False):
        self.encoder = torch.abs(encoder)
        self.encoder_bias = encoder_bias
        self.norm_encoder = norm_encoder
        self.n_feats, self.activation_size = self.encoder.shape

    def get_learned_dict(self):
        return normalize_rows(self.encoder)

    def to_device(self, device):
        self.encoder = self.encoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)

    def encode(self, batch):
        enc = normalize_rows(self.encoder) if self.norm_encoder else self.encoder
        return relu_encode(enc, self.encoder_bias, batch)


class UntiedPositiveSAE(LearnedDict):
    def __init__(self, encoder, encoder_bias, decoder, norm_encoder=False):
        self.encoder = torch.abs(encoder)
        self.decoder = decoder
        self.encoder_bias = encoder_bias
        self.norm_encoder = norm_encoder
        self.n_feats, self.activation_size = self.encoder.shape

    def get_learned_dict(self):
        return normalize_rows(self.encoder)

    def to_device(self, device):
        self.encoder = self.encoder.to(device)
        self.decoder = self.decoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)

    def encode(self, batch):
        return relu_encode(self.encoder, self.encoder_bias, batch)


class FunctionalPositiveTiedSAE(DictSignature):
    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, bias_decay=0.0, device=None, dtype=None):
        enc = torch.empty(n_dict_components, activation_size, device=device, dtype=dtype)
        nn.init.xavier_uniform_(enc)
        params = {
            "encoder": torch.abs(enc),
            "encoder_bias": torch.full((n_dict_components,), -1.0, device=device, dtype=dtype),
        }
        buffers = {
            "l1_alpha": torch.tensor(l1_alpha, device=device, dtype=dtype),
            "bias_decay": torch.tensor(bias_decay, device=device, dtype=dtype),
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        return TiedSAE(params["encoder"], params["encoder_bias"], norm_encoder=True)

    @staticmethod
    def loss(params, buffers, batch):
        enc = torch.clamp(params["encoder"], min=0.0)
        learned_dict = normalize_rows(enc)
        c = relu_encode(learned_dict, params["encoder_bias"], batch + INPUT_SHIFT)
        x_hat = c @ learned_dict
        l_rec = ((x_hat - INPUT_SHIFT) - batch).pow(2).mean()
        l_l1 = buffers["l1_alpha"] * torch.norm(c, 1, dim=-1).mean()
        l_bd = buffers["bias_decay"] * torch.norm(params["encoder_bias"], 2)
        total = l_rec + l_l1 + l_bd
        return total, (
            {"loss": total, "l_reconstruction": l_rec, "l_l1": l_l1, "l_bias_decay": l_bd},
            {"c": c},
        )


for _cls in (TiedPositiveSAE, UntiedPositiveSAE, FunctionalPositiveTiedSAE):
    _cls.__module__ = "autoencoders.mlp_tests"
