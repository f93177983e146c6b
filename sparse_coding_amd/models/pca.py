"""Streaming (batched) PCA baseline + PCA-based dictionary exports.

Semantics parity with reference ``autoencoders/pca.py`` (BatchedPCA streaming
covariance/mean update :54-69, exports :71-110, PCAEncoder :113-135).  The
covariance accumulation runs on-GPU; the streaming rank-update kernel target
is K10 in SURVEY.md §2.4 (the einsum outer-product is replaced by a single
``addmm`` on the centered batch, which rocBLAS maps to MFMA).
"""

from __future__ import annotations

from typing import Optional

import torch

from sparse_coding_amd.models.learned_dict import LearnedDict, Rotation, TiedSAE
from sparse_coding_amd.models.topk import TopKLearnedDict


class BatchedMean:
    def __init__(self, n_dims: int, device):
        self.n_dims = n_dims
        self.device = device
        self.mean = torch.zeros(n_dims, device=device)
        self.n_samples = 0

    def train_batch(self, activations: torch.Tensor) -> None:
        b = activations.shape[0]
        total = self.n_samples + b
        self.mean *= self.n_samples / total
        self.mean += activations.sum(dim=0) / total
        self.n_samples = total

    def get_mean(self) -> torch.Tensor:
        return self.mean


class BatchedPCA:
    """Streaming mean + covariance with the exact same rank-update recurrence
    as the reference (pca.py:54-64), but computed with one GEMM per batch
    (``corrected.T @ (x - new_mean) / B``) instead of materializing the
    [B, d, d] outer-product tensor — O(B d) memory instead of O(B d^2)."""

    def __init__(self, n_dims: int, device):
        self.n_dims = n_dims
        self.device = device
        self.cov = torch.zeros(n_dims, n_dims, device=device)
        self.mean = torch.zeros(n_dims, device=device)
        self.n_samples = 0

    def get_mean(self) -> torch.Tensor:
        return self.mean

    def train_batch(self, activations: torch.Tensor) -> None:
        b = activations.shape[0]
        total = self.n_samples + b
        corrected = activations - self.mean
        new_mean = self.mean + corrected.mean(dim=0) * b / total
        # mean_b outer(corrected_b, x_b - new_mean) == corrected.T @ (x - new_mean) / b
        cov_update = corrected.T @ (activations - new_mean) / b
        self.cov = self.cov * (self.n_samples / total) + cov_update * (b / total)
        self.mean = new_mean
        self.n_samples = total

    def get_pca(self):
        cov_symm = (self.cov + self.cov.T) / 2
        return torch.linalg.eigh(cov_symm)

    def get_centering_transform(self):
        eigvals, eigvecs = self.get_pca()
        eigvals = torch.clamp(eigvals, min=1e-6)
        scaling = 1.0 / torch.sqrt(eigvals)
        assert torch.all(~torch.isnan(scaling)), "Scaling has NaNs"
        return self.get_mean(), eigvecs, scaling

    def get_dict(self) -> torch.Tensor:
        eigvals, eigvecs = self.get_pca()
        return eigvecs[:, torch.argsort(eigvals, descending=True)].T

    def to_learned_dict(self, sparsity: int) -> "PCAEncoder":
        return PCAEncoder(self.get_dict(), sparsity)

    def to_topk_dict(self, sparsity: int) -> TopKLearnedDict:
        dirs = self.get_dict()
        return TopKLearnedDict(torch.cat([dirs, -dirs], dim=0), sparsity)

    def to_rotation_dict(self, n_components: Optional[int] = None) -> Rotation:
        if n_components is None:
            n_components = self.n_dims
        return Rotation(self.get_dict()[:n_components])

    def to_pve_rotation_dict(self, n_components: Optional[int] = None) -> TiedSAE:
        if n_components is None:
            n_components = self.n_dims
        dirs = self.get_dict()[:n_components]
        dirs_ = torch.cat([dirs, -dirs], dim=0)
        return TiedSAE(
            dirs_,
            torch.zeros(2 * n_components),
            centering=(self.get_mean(), None, None),
            norm_encoder=True,
        )


def calc_pca(activations: torch.Tensor, batch_size: int = 512, device="cuda:0") -> BatchedPCA:
    pca = BatchedPCA(activations.shape[1], device)
    for i in range(0, activations.shape[0], batch_size):
        pca.train_batch(activations[i : i + batch_size].to(device))
    return pca


def calc_mean(activations: torch.Tensor, batch_size: int = 512, device="cuda:0") -> torch.Tensor:
    m = BatchedMean(activations.shape[1], device)
    for i in range(0, activations.shape[0], batch_size):
        m.train_batch(activations[i : i + batch_size].to(device))
    return m.get_mean()


class PCAEncoder(LearnedDict):
    """Top-|k| PCA inference dict (reference pca.py:113-135)."""

    def __init__(self, pca_dict: torch.Tensor, sparsity: int):
        self.pca_dict = pca_dict / torch.norm(pca_dict, dim=-1)[:, None]
        self.sparsity = sparsity
        self.n_feats, self.activation_size = self.pca_dict.shape

    def to_device(self, device):
        self.pca_dict = self.pca_dict.to(device)

    def encode(self, x):
        scores = x @ self.pca_dict.T
        top_idx = torch.topk(scores.abs(), self.sparsity, dim=-1).indices
        code = torch.zeros_like(scores)
        code.scatter_(-1, top_idx, scores.gather(-1, top_idx))
        return code

    def get_learned_dict(self):
        return self.pca_dict


for _cls in (BatchedMean, BatchedPCA, PCAEncoder):
    _cls.__module__ = "autoencoders.pca"
