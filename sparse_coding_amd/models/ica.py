"""FastICA baseline (CPU/sklearn, as in the reference autoencoders/ica.py).

sklearn's FastICA is inherently CPU float64; the reference quotes ~15 min/GB
(ica.py:43).  Kept CPU-side here too (SURVEY.md §7 stage 7) — the GPU budget
goes to SAE training, not sklearn baselines.
"""

from __future__ import annotations

from datetime import datetime

import numpy as np
import torch
from sklearn.decomposition import FastICA
from sklearn.preprocessing import StandardScaler

from sparse_coding_amd.models.learned_dict import LearnedDict
from sparse_coding_amd.models.topk import TopKLearnedDict


class ICAEncoder(LearnedDict):
    def __init__(self, activation_size: int, n_components: int = 0):
        self.activation_size = activation_size
        self.n_feats = n_components if n_components else activation_size
        self.ica = FastICA()
        self.scaler = StandardScaler()

    def to_device(self, device):
        pass

    def encode(self, x):
        assert x.shape[1] == self.activation_size
        x_std = self.scaler.transform(x.cpu().numpy().astype(np.float64))
        return torch.tensor(self.ica.transform(x_std), device=x.device)

    def train(self, dataset: torch.Tensor):
        assert dataset.shape[1] == self.activation_size
        print(f"Fitting ICA on {dataset.shape[0]} activations")
        rescaled = self.scaler.fit_transform(dataset.cpu().numpy().astype(np.float64))
        t0 = datetime.now()
        out = self.ica.fit_transform(rescaled)
        print(f"ICA fit in {datetime.now() - t0}")
        return out

    def get_learned_dict(self):
        comps = torch.tensor(self.ica.components_, dtype=torch.float32)
        return comps / torch.norm(comps, dim=-1, keepdim=True)

    def to_topk_dict(self, sparsity: int) -> TopKLearnedDict:
        pos = self.ica.components_.copy()
        comps = np.concatenate([pos, -pos], axis=0)
        return TopKLearnedDict(torch.tensor(comps, dtype=torch.float32), sparsity)

    def to_nneg_dict(self) -> "NNegICAEncoder":
        return NNegICAEncoder(self.activation_size, self.ica, scaler=self.scaler)


class NNegICAEncoder(LearnedDict):
    """Split-sign (nonnegative) ICA code (reference ica.py:61-81; the
    reference version references an unset ``self.scaler`` and np.clamp —
    fixed here)."""

    def __init__(self, activation_size: int, ica, scaler=None):
        self.activation_size = activation_size
        self.n_feats = 2 * ica.components_.shape[0]
        self.ica = ica
        self.scaler = scaler

    def to_device(self, device):
        pass

    def encode(self, x):
        assert x.shape[1] == self.activation_size
        x_np = x.cpu().numpy().astype(np.float64)
        if self.scaler is not None:
            x_np = self.scaler.transform(x_np)
        c = self.ica.transform(x_np)
        c_pos = np.clip(c, 0, None)
        c_neg = np.clip(-c, 0, None)
        return torch.cat(
            [torch.tensor(c_pos, device=x.device), torch.tensor(c_neg, device=x.device)],
            dim=-1,
        )

    def get_learned_dict(self):
        comps = torch.tensor(self.ica.components_, dtype=torch.float32)
        comps = torch.cat([comps, -comps], dim=0)
        return comps / torch.norm(comps, dim=-1, keepdim=True)


for _cls in (ICAEncoder, NNegICAEncoder):
    _cls.__module__ = "autoencoders.ica"
