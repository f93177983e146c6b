"""Experiment catalogue: named ensemble builders + sweep entry points.

Covers the reference's ``big_sweep_experiments.py`` catalogue (C18): tied vs
untied grids (:42-229), topk (:232), synthetic ranges (:265), dense L1 range
(:294), thresholding (:403), masked dict-ratio (:543), zero-L1 baselines
(:497,910), across-layers/locations (:643-769), Pythia-1.4B (:851-907) and
GPT-2-small (:1174-1269) — expressed through one parametrized builder
instead of ~20 near-copies.

An ensemble-init function returns
``(ensembles, ensemble_hyperparams, buffer_hyperparams, hyperparam_ranges)``
where ensembles is a list of (FunctionalEnsemble, args, name)
(reference big_sweep.py:328-336).
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import numpy as np
import torch

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.models.sae_signatures import (
    FunctionalMaskedTiedSAE,
    FunctionalSAE,
    FunctionalThresholdingSAE,
    FunctionalTiedSAE,
)
from sparse_coding_amd.models.topk import TopKEncoder
from sparse_coding_amd.sweep.big_sweep import sweep


def default_devices(n: Optional[int] = None) -> List[str]:
    if not torch.cuda.is_available():
        return ["cpu"]
    count = torch.cuda.device_count()
    if n is not None:
        count = min(count, n)
    return [f"cuda:{i}" for i in range(count)]


def make_grid_ensembles(
    cfg,
    sig=FunctionalTiedSAE,
    l1_values: Optional[Sequence[float]] = None,
    dict_ratios: Optional[Sequence[float]] = None,
    devices: Optional[List[str]] = None,
    init_kwargs: Optional[dict] = None,
):
    """One ensemble per (device, dict_ratio): models within an ensemble share
    shapes and vary l1_alpha (the stacking dimension), mirroring the
    reference's placement (big_sweep_experiments.py:42-229)."""
    if l1_values is None:
        l1_values = np.logspace(-4, -2, 8)
    if dict_ratios is None:
        dict_ratios = [cfg.learned_dict_ratio]
    if devices is None:
        devices = default_devices()
    init_kwargs = init_kwargs or {}

    d = cfg.activation_width
    ensembles = []
    for gi, ratio in enumerate(dict_ratios):
        device = devices[gi % len(devices)]
        n_dict = int(d * ratio)
        models = [sig.init(d, n_dict, float(l1), device=device, **init_kwargs) for l1 in l1_values]
        ens = FunctionalEnsemble(models, sig, adam, {"lr": cfg.lr}, device=device)
        args = {"batch_size": cfg.batch_size, "device": device, "dict_size": n_dict}
        ensembles.append((ens, args, f"dict_ratio_{ratio}"))

    ensemble_hyperparams = ["dict_size"]
    buffer_hyperparams = ["l1_alpha"]
    hyperparam_ranges = {
        "l1_alpha": [float(x) for x in l1_values],
        "dict_size": [int(d * r) for r in dict_ratios],
    }
    return ensembles, ensemble_hyperparams, buffer_hyperparams, hyperparam_ranges


# -- named experiment builders ----------------------------------------------

def tied_vs_untied_init(cfg):
    """Both families side by side (reference :42-229)."""
    l1_values = np.logspace(-4, -2, 8)
    devices = default_devices()
    tied = make_grid_ensembles(cfg, FunctionalTiedSAE, l1_values, [cfg.learned_dict_ratio], devices[: max(1, len(devices) // 2)])
    untied = make_grid_ensembles(cfg, FunctionalSAE, l1_values, [cfg.learned_dict_ratio], devices[max(1, len(devices) // 2):] or devices)
    ensembles = tied[0] + untied[0]
    return ensembles, ["dict_size"], ["l1_alpha"], tied[3]


def dense_l1_range_init(cfg, n_points: int = 16):
    """Dense log-spaced l1 grid, the canonical operating-point sweep
    (reference :294-338; l1≈8.5e-4 is index 7 of logspace(-4,-2,16))."""
    return make_grid_ensembles(cfg, FunctionalTiedSAE if cfg.tied_ae else FunctionalSAE, np.logspace(-4, -2, n_points))


def zero_l1_baseline_init(cfg):
    return make_grid_ensembles(cfg, FunctionalTiedSAE, [0.0], [cfg.learned_dict_ratio])


def dict_ratio_range_init(cfg, ratios: Sequence[float] = (0.5, 1, 2, 4, 8, 16, 32)):
    """Across dict sizes, one ensemble per ratio (reference :543-640).
    Shape-homogeneous per ensemble, so no masking is needed; the masked
    variant below stacks them into ONE ensemble instead."""
    return make_grid_ensembles(cfg, FunctionalTiedSAE, np.logspace(-4, -2, 4), ratios)


def masked_dict_ratio_init(cfg, ratios: Sequence[float] = (0.5, 1, 2, 4), l1: float = 8.5e-4):
    """Different dict sizes stacked into one ensemble with coef masks
    (reference :543; sae_ensemble.py:309)."""
    d = cfg.activation_width
    devices = default_devices()
    n_stack = int(d * max(ratios))
    models = [
        FunctionalMaskedTiedSAE.init(d, int(d * r), n_stack, l1, device=devices[0]) for r in ratios
    ]
    ens = FunctionalEnsemble(models, FunctionalMaskedTiedSAE, adam, {"lr": cfg.lr}, device=devices[0])
    args = {"batch_size": cfg.batch_size, "device": devices[0], "dict_size": n_stack}
    return (
        [(ens, args, "masked_ratios")],
        ["dict_size"],
        ["l1_alpha", "dict_size"],
        {"l1_alpha": [l1], "dict_size": [int(d * r) for r in ratios]},
    )


def topk_init(cfg, ks: Sequence[int] = (4, 8, 16, 32, 64, 128)):
    """TopK encoders across sparsity levels (reference :232-262).  TopK
    stacks are shape-homogeneous but k varies per model; k lives in buffers
    so vmap batches it."""
    d = cfg.activation_width
    devices = default_devices()
    n_dict = int(d * cfg.learned_dict_ratio)
    models = [TopKEncoder.init(d, n_dict, int(k)) for k in ks]
    ens = FunctionalEnsemble(models, TopKEncoder, adam, {"lr": cfg.lr}, device=devices[0], no_stacking=True)
    args = {"batch_size": cfg.batch_size, "device": devices[0], "dict_size": n_dict}
    return (
        [(ens, args, "topk")],
        ["dict_size"],
        ["sparsity"],
        {"sparsity": list(ks), "dict_size": [n_dict]},
    )


def thresholding_init(cfg):
    """Soft-thresholding SAEs (reference :403-494)."""
    return make_grid_ensembles(cfg, FunctionalThresholdingSAE, np.logspace(-4, -2, 8))


def synthetic_sweep_init(cfg):
    """Synthetic ground-truth sweep (reference :265-291); pair with
    cfg.use_synthetic_dataset=True."""
    return make_grid_ensembles(cfg, FunctionalTiedSAE, np.logspace(-5, -2, 8))


# -- runnable experiments ----------------------------------------------------

def run_dense_l1_range(cfg):
    return sweep(dense_l1_range_init, cfg)


def run_topk(cfg):
    return sweep(topk_init, cfg)


def run_across_layers(cfg, layers: Sequence[int], init: Callable = dense_l1_range_init):
    """Sweep each layer's dataset in turn (reference :643-769)."""
    base_dataset_folder = cfg.dataset_folder
    base_output_folder = cfg.output_folder
    results = {}
    for layer in layers:
        cfg.layer = layer
        cfg.dataset_folder = f"{base_dataset_folder}_l{layer}"
        cfg.output_folder = f"{base_output_folder}_l{layer}"
        results[layer] = sweep(init, cfg)
    return results


def run_pythia_1_4_b_sweep(cfg):
    cfg.model_name = "pythia-1.4b-deduped"
    cfg.layer_loc = "residual"
    return sweep(dense_l1_range_init, cfg)


def run_gpt2_small_mlp_sweep(cfg):
    cfg.model_name = "gpt2"
    cfg.layer_loc = "mlpout"
    return sweep(lambda c: make_grid_ensembles(c, FunctionalTiedSAE, np.logspace(-4, -2, 8), (1, 2, 4, 8)), cfg)


def lista_init(cfg, n_hidden_layers: int = 3):
    """LISTA denoising SAEs across l1 (reference :341-400)."""
    from sparse_coding_amd.models.lista import FunctionalLISTADenoisingSAE

    d = cfg.activation_width
    devices = default_devices()
    n_dict = int(d * cfg.learned_dict_ratio)
    l1s = np.logspace(-4, -2, 8)
    models = [FunctionalLISTADenoisingSAE.init(d, n_dict, n_hidden_layers, float(l1)) for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalLISTADenoisingSAE, adam, {"lr": cfg.lr}, device=devices[0])
    args = {"batch_size": cfg.batch_size, "device": devices[0], "dict_size": n_dict}
    return ([(ens, args, "lista")], ["dict_size"], ["l1_alpha"],
            {"l1_alpha": [float(x) for x in l1s], "dict_size": [n_dict]})


def residual_denoising_init(cfg, n_hidden_layers: int = 2):
    """Residual-denoising SAEs (reference residual_denoising_experiment /
    run_resid_denoise :1280-area)."""
    from sparse_coding_amd.models.lista import FunctionalResidualDenoisingSAE

    d = cfg.activation_width
    devices = default_devices()
    n_dict = int(d * cfg.learned_dict_ratio)
    l1s = np.logspace(-4, -2, 8)
    models = [FunctionalResidualDenoisingSAE.init(d, n_dict, n_hidden_layers, float(l1)) for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalResidualDenoisingSAE, adam, {"lr": cfg.lr}, device=devices[0])
    args = {"batch_size": cfg.batch_size, "device": devices[0], "dict_size": n_dict}
    return ([(ens, args, "resid_denoise")], ["dict_size"], ["l1_alpha"],
            {"l1_alpha": [float(x) for x in l1s], "dict_size": [n_dict]})


def positive_init(cfg):
    """Non-negative-encoder tied SAEs with the reference's +0.18 input shift
    (reference run_positive :1034-1171; mlp_tests.py)."""
    from sparse_coding_amd.models.positive import FunctionalPositiveTiedSAE

    return make_grid_ensembles(cfg, FunctionalPositiveTiedSAE, np.logspace(-4, -2, 8))


def run_thresholding(cfg):
    return sweep(thresholding_init, cfg)


def run_lista(cfg):
    return sweep(lista_init, cfg)


def run_resid_denoise(cfg):
    return sweep(residual_denoising_init, cfg)


def run_positive(cfg):
    return sweep(positive_init, cfg)


def run_zero_l1_baseline(cfg):
    return sweep(zero_l1_baseline_init, cfg)


def run_dict_ratio(cfg):
    return sweep(dict_ratio_range_init, cfg)


def run_synthetic(cfg):
    cfg.use_synthetic_dataset = True
    return sweep(synthetic_sweep_init, cfg)


def run_across_layers_attn(cfg, layers: Sequence[int]):
    cfg.layer_loc = "attn"
    return run_across_layers(cfg, layers)


def run_across_layers_mlp_out(cfg, layers: Sequence[int]):
    cfg.layer_loc = "mlpout"
    return run_across_layers(cfg, layers)


def run_across_layers_mlp_untied(cfg, layers: Sequence[int]):
    cfg.layer_loc = "mlp"
    cfg.tied_ae = False
    return run_across_layers(cfg, layers)


def run_across_layers_mlp_long(cfg, layers: Sequence[int], n_chunks: int = 60):
    """Long MLP sweep (reference long_mlp_sweep :956-1031): more chunks,
    mlp location."""
    cfg.layer_loc = "mlp"
    cfg.n_chunks = n_chunks
    return run_across_layers(cfg, layers)


def run_single_layer(cfg, layer: int = 2):
    """One layer end to end on the default model (reference run_single_layer)."""
    cfg.layer = layer
    return sweep(dense_l1_range_init, cfg)


def run_single_layer_gpt2(cfg, layer: int = 5):
    cfg.model_name = "gpt2"
    cfg.layer = layer
    return sweep(dense_l1_range_init, cfg)
