"""NMF baseline (sklearn CPU), reference autoencoders/nmf.py:30-66."""

from __future__ import annotations

from datetime import datetime

import numpy as np
import torch
from sklearn.decomposition import NMF

from sparse_coding_amd.models.learned_dict import LearnedDict
from sparse_coding_amd.models.topk import TopKLearnedDict


class NMFEncoder(LearnedDict):
    def __init__(self, activation_size: int, n_components: int = 0, shift: float = 0.0):
        self.activation_size = activation_size
        self.n_feats = n_components if n_components else activation_size
        self.nmf = NMF()
        self.shift = shift

    def to_device(self, device):
        pass

    def encode(self, x):
        if torch.min(x) < self.shift:
            print("Warning: data has values below expected minimum for NMF. This may cause errors.")
        x = torch.clamp(x - self.shift, min=0.0)
        c = self.nmf.transform(x.cpu().numpy().astype(np.float64))
        return torch.tensor(c, device=x.device)

    def train(self, dataset: torch.Tensor):
        if torch.min(dataset) < self.shift:
            self.shift = torch.min(dataset).item()
        dataset = dataset - self.shift
        assert dataset.shape[1] == self.activation_size
        print(f"Fitting NMF on {dataset.shape[0]} activations")
        t0 = datetime.now()
        self.nmf.fit(dataset.cpu().numpy())
        print(f"NMF fit in {datetime.now() - t0}")

    def get_learned_dict(self):
        return torch.tensor(self.nmf.components_, dtype=torch.float32)

    def to_topk_dict(self, sparsity: int) -> TopKLearnedDict:
        return TopKLearnedDict(self.get_learned_dict(), sparsity)


NMFEncoder.__module__ = "autoencoders.nmf"
