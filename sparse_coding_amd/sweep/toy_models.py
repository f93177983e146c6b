"""Superposition toy-model replication (reference replicate_toy_models.py, C23).

Trains small SAEs on synthetic superposition data across an (l1, dict-ratio)
grid and reports MMCS-to-ground-truth and dead-neuron counts.  Uses the
framework's own ensemble engine rather than the reference's frozen
standalone nn.Module (replicate_toy_models.py:208-229) — the decoder-renorm-
in-forward semantics are identical (FunctionalTiedSAE).
"""

from __future__ import annotations

import os
import pickle
from typing import Dict, List, Tuple

import numpy as np
import torch

from sparse_coding_amd.data.random_dataset import RandomDatasetGenerator
from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.metrics import standard_metrics as sm
from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE


def run_single_go(
    activation_dim: int = 256,
    n_ground_truth: int = 512,
    feature_num_nonzero: int = 5,
    l1_values=None,
    dict_ratios=(1, 2, 4),
    n_steps: int = 2000,
    batch_size: int = 1024,
    lr: float = 1e-3,
    device: str = "cpu",
    correlated: bool = False,
    backend: str = "auto",
) -> List[Dict]:
    """One grid sweep on one synthetic generator; returns per-setting results
    (reference run_single_go :279)."""
    gen = RandomDatasetGenerator(
        activation_dim=activation_dim,
        n_ground_truth_components=n_ground_truth,
        batch_size=batch_size,
        feature_num_nonzero=feature_num_nonzero,
        feature_prob_decay=0.99,
        correlated=correlated,
        device=device,
    )
    if l1_values is None:
        l1_values = np.logspace(-4, -2, 4)

    results = []
    for ratio in dict_ratios:
        n_dict = int(activation_dim * ratio)
        models = [FunctionalTiedSAE.init(activation_dim, n_dict, float(l1), device=device) for l1 in l1_values]
        ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": lr}, device=device, backend=backend)
        for _ in range(n_steps):
            ens.step_batch(gen.send(None))
        for ld, l1 in zip(ens.to_learned_dicts(), l1_values):
            sample = gen.send(None).cpu()
            results.append({
                "dict_ratio": ratio,
                "l1_alpha": float(l1),
                "mmcs_to_ground_truth": sm.mmcs_to_fixed(ld, gen.feats.cpu()).item(),
                "representedness": sm.representedness(gen.feats.cpu(), ld).mean().item(),
                "fvu": sm.fraction_variance_unexplained(ld, sample).item(),
                "dead_count": int(sm.dead_feature_fraction(ld, sample) * n_dict),
                "mean_l0": sm.mean_l0(ld, sample).item(),
            })
    return results


def main(output_folder: str = "outputs_toy", device: str = "cpu", **kwargs):
    """Grid + pickle outputs (reference main :446, pickles :530-535)."""
    os.makedirs(output_folder, exist_ok=True)
    results = run_single_go(device=device, **kwargs)
    with open(os.path.join(output_folder, "toy_results.pkl"), "wb") as f:
        pickle.dump(results, f)
    # mmcs grid plot
    ratios = sorted({r["dict_ratio"] for r in results})
    l1s = sorted({r["l1_alpha"] for r in results})
    grid = np.zeros((len(l1s), len(ratios)))
    for r in results:
        grid[l1s.index(r["l1_alpha"]), ratios.index(r["dict_ratio"])] = r["mmcs_to_ground_truth"]
    fig = sm.plot_grid(grid, ratios, l1s, "dict_ratio", "l1_alpha", cmap="viridis")
    fig.savefig(os.path.join(output_folder, "mmcs_grid.png"), dpi=120)
    return results


if __name__ == "__main__":
    main()
