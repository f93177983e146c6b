"""Self-parsing argparse dataclasses (reference config.py:7-140).

Every field of a BaseArgs subclass auto-becomes ``--field``.  Unlike the
reference, CLI parsing happens only when requested (``from_cli()`` or
``parse=True``) so that constructing a config inside pytest or a library
call never touches ``sys.argv`` (the reference parses in ``__post_init__``,
config.py:15-21, which breaks embedding).
"""

from __future__ import annotations

import argparse
from dataclasses import dataclass, field
from typing import Any, List, Optional

import torch


def _str2bool(v: str) -> bool:
    if isinstance(v, bool):
        return v
    return v.lower() in ("1", "true", "yes", "y", "t")


@dataclass
class BaseArgs:
    def parse_args(self, argv: Optional[List[str]] = None) -> argparse.Namespace:
        parser = argparse.ArgumentParser()
        for key, value in vars(self).items():
            ty = type(value)
            if ty is bool:
                parser.add_argument(f"--{key}", type=_str2bool, default=None)
            elif value is None:
                parser.add_argument(f"--{key}", default=None)
            else:
                parser.add_argument(f"--{key}", type=ty, default=None)
        return parser.parse_args(argv)

    def update(self, args: Any) -> None:
        for key, value in vars(args).items():
            if value is not None:
                setattr(self, key, value)

    @classmethod
    def from_cli(cls, argv: Optional[List[str]] = None, **overrides):
        self = cls(**overrides)
        self.update(self.parse_args(argv))
        return self

    def as_dict(self):
        d = {}
        for k, v in vars(self).items():
            if isinstance(v, torch.dtype):
                v = str(v)
            d[k] = v
        return d

    # reference cfg objects support dict(cfg) via keys/__getitem__
    def keys(self):
        return vars(self).keys()

    def __getitem__(self, k):
        return getattr(self, k)


def _default_device() -> str:
    return "cuda:0" if torch.cuda.is_available() else "cpu"


@dataclass
class TrainArgs(BaseArgs):
    layer: int = 2
    layer_loc: str = "residual"
    model_name: str = "pythia-70m-deduped"
    dataset_name: str = "openwebtext"
    dataset_folder: str = ""
    device: str = field(default_factory=_default_device)
    tied_ae: bool = False
    seed: int = 0
    learned_dict_ratio: float = 1.0
    output_folder: str = "outputs"
    dtype: torch.dtype = torch.float32
    epochs: int = 1
    center_dataset: bool = False
    n_chunks: int = 30
    chunk_size_gb: float = 2.0
    batch_size: int = 256
    use_wandb: bool = False
    wandb_images: bool = False
    lr: float = 1e-3
    l1_alpha: float = 1e-3
    save_every: int = 5
    n_epochs: int = 1
    # fields the reference reads but never declares (SURVEY.md known-danglers)
    n_repetitions: Optional[int] = None
    center_activations: bool = False


@dataclass
class EnsembleArgs(TrainArgs):
    activation_width: int = 512
    use_synthetic_dataset: bool = False
    bias_decay: float = 0.0
    # in-sweep dead-feature resampling (engine/resample.py); 0 = off
    resample_every_chunks: int = 0
    resample_protocol: str = "anthropic"
    resample_n_track: int = 512
    resample_warmup_steps: int = 1000


@dataclass
class SyntheticEnsembleArgs(EnsembleArgs):
    noise_magnitude_scale: float = 0.0
    feature_prob_decay: float = 0.99
    feature_num_nonzero: int = 10
    gen_batch_size: int = 4096
    dataset_folder: str = "activation_data"
    n_ground_truth_components: int = 512
    correlated_components: bool = False


@dataclass
class ErasureArgs(BaseArgs):
    model_name: str = "EleutherAI/pythia-70m-deduped"
    device: str = field(default_factory=_default_device)
    layer: Optional[int] = None
    count_cutoff: int = 10000
    output_folder: str = "output_erasure_pca"
    activation_filename: str = "activation_data_erasure.pt"
    dict_filename: str = ""


@dataclass
class ToyArgs(BaseArgs):
    layer: int = 2
    layer_loc: str = "residual"
    model_name: str = "pythia-70m-deduped"
    dataset_name: str = "openwebtext"
    device: str = field(default_factory=_default_device)
    tied_ae: bool = False
    seed: int = 0
    learned_dict_ratio: float = 1.0
    output_folder: str = "outputs"
    dtype: torch.dtype = torch.float32
    activation_dim: int = 256
    feature_prob_decay: float = 0.99
    feature_num_nonzero: int = 5
    correlated_components: bool = False
    n_ground_truth_components: int = 512
    noise_std: float = 0.1
    l1_exp_low: int = -12
    l1_exp_high: int = -11
    l1_exp_base: float = 10 ** (1 / 4)
    dict_ratio_exp_low: int = 1
    dict_ratio_exp_high: int = 7
    dict_ratio_exp_base: float = 2
    batch_size: int = 4096
    lr: float = 1e-3
    epochs: int = 1
    noise_level: float = 0.0
    n_components_dictionary: int = 512
    l1_alpha: float = 1e-3


@dataclass
class InterpArgs(BaseArgs):
    layer: int = 2
    model_name: str = "EleutherAI/pythia-70m-deduped"
    layer_loc: str = "residual"
    device: str = field(default_factory=_default_device)
    n_feats_explain: int = 10
    load_interpret_autoencoder: str = ""
    tied_ae: bool = False
    interp_name: str = ""
    sort_mode: str = "max"
    use_decoder: bool = True
    df_n_feats: int = 200
    top_k: int = 50
    save_loc: str = ""


@dataclass
class InterpGraphArgs(BaseArgs):
    layer: int = 1
    model_name: str = "EleutherAI/pythia-70m-deduped"
    layer_loc: str = "mlp"
    score_mode: str = "all"
    run_all: bool = False


@dataclass
class InvestigateArgs(BaseArgs):
    threshold: float = 0.9
    layer: int = 2
    device: str = field(default_factory=_default_device)
