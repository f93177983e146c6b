"""Synthetic ground-truth sparse-dictionary activation generators.

Parity with reference ``sc_datasets/random_dataset.py`` (RandomDatasetGenerator
:17-73, SparseMixDataset :77-142, generate_* :160-279): decaying per-feature
inclusion probability, optional MVN-CDF-correlated mask, uniform strengths,
codes @ unit-norm feature dictionary, optional MVN noise.  All batch
generation stays on-device (pure GPU tensor ops).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Generator, Optional, Tuple, Union

import numpy as np
import torch

DeviceLike = Union[torch.device, str]


def generate_rand_feats(feat_dim: int, num_feats: int, device: DeviceLike) -> torch.Tensor:
    """Unit-norm random feature directions [num_feats, feat_dim]
    (reference random_dataset.py:248-261)."""
    feats = np.random.multivariate_normal(np.zeros(feat_dim), np.eye(feat_dim), size=num_feats)
    feats = feats / np.linalg.norm(feats, axis=1, keepdims=True)
    return torch.from_numpy(feats).to(device).float()


def generate_corr_matrix(num_feats: int, device: DeviceLike) -> torch.Tensor:
    """Random symmetric PSD-ified correlation matrix (reference :264-279)."""
    corr = np.random.rand(num_feats, num_feats)
    corr = (corr + corr.T) / 2
    min_eig = np.min(np.real(np.linalg.eigvals(corr)))
    if min_eig < 0:
        corr -= 1.001 * min_eig * np.eye(num_feats)
    return torch.from_numpy(corr).to(device).float()


def generate_rand_dataset(
    n_ground_truth_components: int,
    dataset_size: int,
    feature_probs: torch.Tensor,
    feats: torch.Tensor,
    device: DeviceLike,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Sparse uniform codes with independent per-feature probabilities
    (reference :160-188)."""
    thresh = torch.rand(dataset_size, n_ground_truth_components, device=device)
    values = torch.rand(dataset_size, n_ground_truth_components, device=device)
    codes = torch.where(thresh <= feature_probs, values, torch.zeros_like(thresh))
    strengths = torch.rand(dataset_size, n_ground_truth_components, device=device)
    data = (codes * strengths) @ feats
    return feats, codes, data


def generate_correlated_dataset(
    n_ground_truth_components: int,
    dataset_size: int,
    corr_matrix: torch.Tensor,
    feats: torch.Tensor,
    frac_nonzero: float,
    decay: torch.Tensor,
    device: DeviceLike,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Correlated sparse codes via MVN sample → normal CDF → decayed and
    rescaled inclusion probabilities (reference :191-245)."""
    mvn = torch.distributions.MultivariateNormal(
        loc=torch.zeros(n_ground_truth_components, device=device),
        covariance_matrix=corr_matrix,
    )
    corr_thresh = mvn.sample()
    normal = torch.distributions.Normal(
        torch.tensor([0.0], device=device), torch.tensor([1.0], device=device)
    )
    cdf = normal.cdf(corr_thresh.squeeze())
    component_probs = cdf * decay
    component_probs = component_probs * (frac_nonzero / torch.mean(component_probs))

    thresh = torch.rand(dataset_size, n_ground_truth_components, device=device)
    values = torch.rand(dataset_size, n_ground_truth_components, device=device)
    codes = torch.where(thresh <= component_probs, values, torch.zeros_like(thresh))

    # no all-zero rows: give such rows one random active feature
    zero_rows = (codes.count_nonzero(dim=1) == 0).nonzero()[:, 0]
    rand_feat = torch.randint(0, n_ground_truth_components, (zero_rows.shape[0],), device=codes.device)
    codes[zero_rows, rand_feat] = 1.0

    strengths = torch.rand(dataset_size, n_ground_truth_components, device=device)
    data = (codes * strengths) @ feats
    return feats, codes, data


def generate_noise_dataset(
    dataset_size: int,
    noise_covariance: torch.Tensor,
    noise_magnitude_scale: float,
    device: DeviceLike,
) -> torch.Tensor:
    noise = torch.distributions.MultivariateNormal(
        loc=torch.zeros(noise_covariance.shape[0], device=device),
        covariance_matrix=noise_covariance,
    ).sample(torch.Size([dataset_size]))
    return noise * noise_magnitude_scale


@dataclass
class RandomDatasetGenerator(Generator):
    activation_dim: int
    n_ground_truth_components: int
    batch_size: int
    feature_num_nonzero: int
    feature_prob_decay: float
    correlated: bool
    device: DeviceLike

    frac_nonzero: float = field(init=False)
    decay: torch.Tensor = field(init=False)
    feats: torch.Tensor = field(init=False)
    corr_matrix: Optional[torch.Tensor] = field(init=False, default=None)
    component_probs: Optional[torch.Tensor] = field(init=False, default=None)

    def __post_init__(self):
        self.frac_nonzero = self.feature_num_nonzero / self.n_ground_truth_components
        self.decay = torch.tensor(
            [self.feature_prob_decay**i for i in range(self.n_ground_truth_components)]
        ).to(self.device)
        if self.correlated:
            self.corr_matrix = generate_corr_matrix(self.n_ground_truth_components, device=self.device)
        else:
            self.component_probs = self.decay * self.frac_nonzero
        self.feats = generate_rand_feats(self.activation_dim, self.n_ground_truth_components, device=self.device)
        self.t_type = torch.float32

    def send(self, ignored_arg: Any) -> torch.Tensor:
        if self.correlated:
            _, _, data = generate_correlated_dataset(
                self.n_ground_truth_components, self.batch_size, self.corr_matrix,
                self.feats, self.frac_nonzero, self.decay, self.device,
            )
        else:
            _, _, data = generate_rand_dataset(
                self.n_ground_truth_components, self.batch_size, self.component_probs,
                self.feats, self.device,
            )
        return data.to(self.t_type)

    def throw(self, type: Any = None, value: Any = None, traceback: Any = None) -> None:
        raise StopIteration


@dataclass
class SparseMixDataset(Generator):
    """Correlated sparse codes + MVN noise (reference :77-142)."""

    activation_dim: int
    n_sparse_components: int
    batch_size: int
    feature_num_nonzero: int
    feature_prob_decay: float
    noise_magnitude_scale: float
    device: DeviceLike

    sparse_component_dict: Optional[torch.Tensor] = None
    sparse_component_covariance: Optional[torch.Tensor] = None
    noise_covariance: Optional[torch.Tensor] = None
    t_type: Optional[torch.dtype] = None

    sparse_component_probs: Optional[torch.Tensor] = field(init=False, default=None)

    def __post_init__(self):
        self.frac_nonzero = self.feature_num_nonzero / self.n_sparse_components
        if self.sparse_component_dict is None:
            self.sparse_component_dict = generate_rand_feats(
                self.activation_dim, self.n_sparse_components, device=self.device
            )
        if self.sparse_component_covariance is None:
            self.sparse_component_covariance = generate_corr_matrix(self.n_sparse_components, device=self.device)
        if self.noise_covariance is None:
            self.noise_covariance = torch.eye(self.activation_dim, device=self.device)
        self.sparse_component_probs = torch.tensor(
            [self.feature_prob_decay**i for i in range(self.n_sparse_components)],
            dtype=torch.float32,
        ).to(self.device)
        if self.t_type is None:
            self.t_type = torch.float32

    def send(self, batch_size: Optional[int]) -> torch.Tensor:
        bs = self.batch_size if batch_size is None else batch_size
        _, _, sparse_data = generate_correlated_dataset(
            self.n_sparse_components, bs, self.sparse_component_covariance,
            self.sparse_component_dict, self.frac_nonzero,
            self.sparse_component_probs, self.device,
        )
        noise_data = generate_noise_dataset(bs, self.noise_covariance, self.noise_magnitude_scale, self.device)
        return (sparse_data + noise_data).to(self.t_type)

    def throw(self, type: Any = None, value: Any = None, traceback: Any = None) -> None:
        raise StopIteration


for _cls in (RandomDatasetGenerator, SparseMixDataset):
    _cls.__module__ = "sc_datasets.random_dataset"
