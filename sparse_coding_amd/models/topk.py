"""TopK encoder signature + learned dict (reference autoencoders/topk_encoder.py).

The training path has a dedicated HIP kernel (K8 in SURVEY.md §2.4): GEMM +
per-row top-k selection + scatter + ReLU fused; this module is the eager
oracle and the eval-time wrapper.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from sparse_coding_amd.models.learned_dict import LearnedDict, normalize_rows
from sparse_coding_amd.models.sae_signatures import DictSignature


class TopKEncoder(DictSignature):
    @staticmethod
    def init(d_activation, n_features, sparsity, dtype=torch.float32):
        params = {"dict": torch.randn(n_features, d_activation, dtype=dtype)}
        buffers = {"sparsity": torch.tensor(sparsity, dtype=torch.long)}
        return params, buffers

    @staticmethod
    def encode(b, sparsity, normed_dict):
        scores = b @ normed_dict.T
        top_idx = torch.topk(scores, sparsity, dim=-1).indices
        code = torch.zeros_like(scores)
        code.scatter_(-1, top_idx, scores.gather(-1, top_idx))
        return F.relu(code)

    @staticmethod
    def loss(params, buffers, batch):
        # NOTE: reference topk_encoder.py:31 normalizes WITHOUT the 1e-8 clamp
        normed = params["dict"] / torch.norm(params["dict"], dim=-1)[:, None]
        code = TopKEncoder.encode(batch, buffers["sparsity"], normed)
        x_hat = code @ normed
        loss = F.mse_loss(batch, x_hat)
        return loss, ({"loss": loss}, {"c": code})

    @staticmethod
    def to_learned_dict(params, buffers):
        normed = params["dict"] / torch.norm(params["dict"], dim=-1)[:, None]
        return TopKLearnedDict(normed, buffers["sparsity"].item())


class TopKLearnedDict(LearnedDict):
    def __init__(self, dict, sparsity):
        self.dict = dict
        self.sparsity = sparsity
        self.n_feats, self.activation_size = self.dict.shape

    def to_device(self, device):
        self.dict = self.dict.to(device)

    def encode(self, x):
        return TopKEncoder.encode(x, self.sparsity, self.dict)

    def get_learned_dict(self):
        return self.dict


for _cls in (TopKEncoder, TopKLearnedDict):
    _cls.__module__ = "autoencoders.topk_encoder"
