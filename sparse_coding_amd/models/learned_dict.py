"""LearnedDict: the dictionary-of-directions model interface.

API parity with reference ``autoencoders/learned_dict.py:16-293`` (LearnedDict
ABC + the evaluation wrapper classes).  Classes are re-exported (and their
``__module__`` pinned) under ``autoencoders.learned_dict`` so that pickled
checkpoints round-trip between this framework and the reference
(``learned_dicts.pt`` stores pickled class paths, reference
``big_sweep.py:368-381``).

Implementation notes (MI355X build): ``encode``/``decode`` are expressed with
``F.linear`` / matmul so they dispatch to rocBLAS on the eval path; the
training hot loop does NOT go through these classes — it runs in the fused
HIP kernels of ``sparse_coding_amd.ops`` via the ensemble engine.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

EPS_NORM = 1e-8


def normalize_rows(w: torch.Tensor, eps: float = EPS_NORM) -> torch.Tensor:
    """Rows scaled to unit L2 norm; norms clamped at ``eps`` (reference
    ``learned_dict.py:137-138``: ``decoder / clamp(norm, 1e-8)``)."""
    norms = torch.norm(w, 2, dim=-1, keepdim=True)
    return w / torch.clamp(norms, min=eps)


def relu_encode(weight: torch.Tensor, bias: torch.Tensor, batch: torch.Tensor) -> torch.Tensor:
    """c = relu(batch @ weight^T + bias) — the canonical SAE encoder."""
    return F.relu(F.linear(batch, weight, bias))


class LearnedDict(ABC):
    """A learned dictionary: encode to a (sparse) code, decode against a
    unit-norm dictionary.  Reference: ``autoencoders/learned_dict.py:16-53``."""

    n_feats: int
    activation_size: int

    @abstractmethod
    def get_learned_dict(self) -> torch.Tensor:  # [n_feats, activation_size]
        ...

    @abstractmethod
    def encode(self, batch: torch.Tensor) -> torch.Tensor:  # [B, d] -> [B, n]
        ...

    @abstractmethod
    def to_device(self, device) -> None:
        ...

    def decode(self, code: torch.Tensor) -> torch.Tensor:
        # einsum("nd,bn->bd", dict, code) == code @ dict
        return code @ self.get_learned_dict()

    def center(self, batch: torch.Tensor) -> torch.Tensor:
        return batch

    def uncenter(self, batch: torch.Tensor) -> torch.Tensor:
        return batch

    def predict(self, batch: torch.Tensor) -> torch.Tensor:
        return self.uncenter(self.decode(self.encode(self.center(batch))))

    def n_dict_components(self) -> int:
        return self.get_learned_dict().shape[0]


class Identity(LearnedDict):
    def __init__(self, activation_size: int, device=None):
        self.n_feats = activation_size
        self.activation_size = activation_size
        self.device = device if device is not None else "cpu"

    def get_learned_dict(self):
        return torch.eye(self.n_feats, device=self.device)

    def encode(self, batch):
        return batch

    def to_device(self, device):
        self.device = device


class IdentityPositive(LearnedDict):
    """Identity split into +/- halves so codes are nonnegative."""

    def __init__(self, activation_size: int, device=None):
        self.n_feats = activation_size
        self.activation_size = activation_size
        self.device = device if device is not None else "cpu"

    def get_learned_dict(self):
        eye = torch.eye(self.n_feats, device=self.device)
        return torch.cat([eye, -eye], dim=0)

    def encode(self, batch):
        return F.relu(torch.cat([batch, -batch], dim=-1))

    def to_device(self, device):
        self.device = device


class IdentityReLU(LearnedDict):
    def __init__(self, activation_size: int, bias: Optional[torch.Tensor] = None):
        self.n_feats = activation_size
        self.activation_size = activation_size
        self.bias = bias if bias is not None else torch.zeros(activation_size)
        assert self.bias.shape == (activation_size,)

    def get_learned_dict(self):
        return torch.eye(self.n_feats)

    def encode(self, batch):
        return F.relu(batch + self.bias)

    def to_device(self, device):
        self.bias = self.bias.to(device)


class RandomDict(LearnedDict):
    def __init__(self, activation_size: int, n_feats: Optional[int] = None):
        self.n_feats = n_feats if n_feats else activation_size
        self.activation_size = activation_size
        self.encoder = torch.randn(self.n_feats, activation_size)
        self.encoder_bias = torch.zeros(self.n_feats)

    def get_learned_dict(self):
        return self.encoder

    def encode(self, batch):
        return relu_encode(self.encoder, self.encoder_bias, batch)

    def to_device(self, device):
        self.encoder = self.encoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)


class UntiedSAE(LearnedDict):
    """Untied SAE: raw encoder, row-normalized decoder as the dictionary.
    Reference: ``autoencoders/learned_dict.py:129-149``."""

    def __init__(self, encoder, decoder, encoder_bias):
        self.encoder = encoder
        self.decoder = decoder
        self.encoder_bias = encoder_bias
        self.n_feats, self.activation_size = self.encoder.shape

    def get_learned_dict(self):
        return normalize_rows(self.decoder)

    def encode(self, batch):
        return relu_encode(self.encoder, self.encoder_bias, batch)

    def to_device(self, device):
        self.encoder = self.encoder.to(device)
        self.decoder = self.decoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)


class TiedSAE(LearnedDict):
    """Tied SAE with optional affine whitening-centering (trans/rot/scale).
    Reference: ``autoencoders/learned_dict.py:152-215``."""

    def __init__(self, encoder, encoder_bias, centering: Tuple = (None, None, None), norm_encoder: bool = True):
        self.encoder = encoder
        self.encoder_bias = encoder_bias
        self.norm_encoder = norm_encoder
        self.n_feats, self.activation_size = self.encoder.shape

        trans, rot, scale = centering
        self.center_trans = trans if trans is not None else torch.zeros(self.activation_size)
        self.center_rot = rot if rot is not None else torch.eye(self.activation_size)
        self.center_scale = scale if scale is not None else torch.ones(self.activation_size)

    def initialize_missing(self):
        dev = self.encoder.device
        if not hasattr(self, "center_trans"):
            self.center_trans = torch.zeros(self.activation_size, device=dev)
        if not hasattr(self, "center_rot"):
            self.center_rot = torch.eye(self.activation_size, device=dev)
        if not hasattr(self, "center_scale"):
            self.center_scale = torch.ones(self.activation_size, device=dev)

    def center(self, batch):
        # einsum("cu,bu->bc", rot, x - t) * s
        return (batch - self.center_trans) @ self.center_rot.T * self.center_scale

    def uncenter(self, batch):
        return (batch / self.center_scale) @ self.center_rot + self.center_trans

    def get_learned_dict(self):
        return normalize_rows(self.encoder)

    def encode(self, batch):
        enc = normalize_rows(self.encoder) if self.norm_encoder else self.encoder
        return relu_encode(enc, self.encoder_bias, batch)

    def to_device(self, device):
        self.initialize_missing()
        self.encoder = self.encoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)
        self.center_trans = self.center_trans.to(device)
        self.center_rot = self.center_rot.to(device)
        self.center_scale = self.center_scale.to(device)


class ReverseSAE(LearnedDict):
    """Tied SAE that subtracts the bias from active features before decoding.
    Reference: ``autoencoders/learned_dict.py:218-257``."""

    def __init__(self, encoder, encoder_bias, norm_encoder: bool = False):
        self.encoder = encoder
        self.encoder_bias = encoder_bias
        self.norm_encoder = norm_encoder
        self.n_feats, self.activation_size = self.encoder.shape

    def get_learned_dict(self):
        return normalize_rows(self.encoder)

    def _weights(self):
        return normalize_rows(self.encoder) if self.norm_encoder else self.encoder

    def encode(self, batch):
        return relu_encode(self._weights(), self.encoder_bias, batch)

    def decode(self, c):
        on = c > 0.0
        c = torch.where(on, c - self.encoder_bias, c)
        # NOTE: reference learned_dict.py:256 writes einsum("dn,bn->bd") which
        # only type-checks for square dictionaries; we implement the intended
        # [n,d] contraction (matches FunctionalReverseSAE.loss, sae_ensemble.py:486).
        return c @ self._weights()

    def to_device(self, device):
        self.encoder = self.encoder.to(device)
        self.encoder_bias = self.encoder_bias.to(device)


class AddedNoise(LearnedDict):
    def __init__(self, noise_mag: float, activation_size: int, device=None):
        self.noise_mag = noise_mag
        self.activation_size = activation_size
        self.n_feats = activation_size
        self.device = device if device is not None else "cpu"

    def get_learned_dict(self):
        return torch.eye(self.activation_size, device=self.device)

    def encode(self, batch):
        return batch + torch.randn_like(batch) * self.noise_mag

    def to_device(self, device):
        self.device = device


class Rotation(LearnedDict):
    def __init__(self, matrix, device=None):
        self.matrix = matrix
        self.activation_size = matrix.shape[0]
        self.n_feats = matrix.shape[0]
        self.device = device if device is not None else "cpu"
        self.matrix = self.matrix.to(self.device)

    def get_learned_dict(self):
        return self.matrix

    def encode(self, batch):
        return F.linear(batch, self.matrix)

    def to_device(self, device):
        self.matrix = self.matrix.to(device)
        self.device = device


# Pickle compatibility: reference checkpoints reference these classes as
# ``autoencoders.learned_dict.<Name>``; pin __module__ so
# checkpoints written by THIS framework load in the reference too.
_PICKLE_PUBLIC = (
    Identity,
    IdentityPositive,
    IdentityReLU,
    RandomDict,
    UntiedSAE,
    TiedSAE,
    ReverseSAE,
    AddedNoise,
    Rotation,
    LearnedDict,
)
for _cls in _PICKLE_PUBLIC:
    _cls.__module__ = "autoencoders.learned_dict"
