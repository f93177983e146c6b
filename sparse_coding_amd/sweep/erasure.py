"""Concept-erasure evaluation: dictionary-feature ablation vs LEACE-style
linear erasure (reference plotting/erasure_plot.py + ErasureArgs, C27).

Pipeline: collect activations + binary concept labels (e.g. the gender
prompt set), then compare probe AUROC after
  (a) ablating the k dictionary features most correlated with the label,
  (b) rank-k LEACE-style linear concept erasure,
as k grows (the "bottleneck" curves of plotting/bottleneck_plot.py).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from sparse_coding_amd.metrics.standard_metrics import logistic_regression_auroc


def feature_label_correlation(code: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """|corr| of each feature activation with the binary label."""
    c = code - code.mean(dim=0)
    l = (labels.float() - labels.float().mean()).unsqueeze(1)
    num = (c * l).mean(dim=0)
    denom = c.std(dim=0) * l.std() + 1e-8
    return (num / denom).abs()


def ablate_top_features(learned_dict, acts: torch.Tensor, labels: torch.Tensor, k: int) -> torch.Tensor:
    """Remove the k most label-correlated dictionary features from the
    activations: x' = x - sum_i c_i * d_i over the selected features."""
    code = learned_dict.encode(learned_dict.center(acts))
    corr = feature_label_correlation(code, labels)
    top = torch.argsort(corr, descending=True)[:k]
    d = learned_dict.get_learned_dict()
    removed = code[:, top] @ d[top]
    return acts - learned_dict.uncenter(removed) + learned_dict.uncenter(torch.zeros_like(removed))


def leace_erase(acts: torch.Tensor, labels: torch.Tensor, rank: int = 1) -> torch.Tensor:
    """Rank-k least-squares concept erasure: project out the top directions
    of the class-mean difference in whitened space (LEACE, Belrose et al.)."""
    x = acts - acts.mean(dim=0)
    y = labels.float() - labels.float().mean()
    # cross-covariance direction(s)
    sigma = (x.T @ x) / x.shape[0] + 1e-4 * torch.eye(x.shape[1])
    cross = (x * y[:, None]).mean(dim=0, keepdim=True).T  # [d, 1]
    # whitened projection directions
    evals, evecs = torch.linalg.eigh(sigma)
    w_inv_half = evecs @ torch.diag(evals.clamp_min(1e-6).rsqrt()) @ evecs.T
    z = w_inv_half @ cross
    q, _ = torch.linalg.qr(z)
    q = q[:, :rank]
    # erase in whitened space, unwhiten
    w_half = evecs @ torch.diag(evals.clamp_min(1e-6).sqrt()) @ evecs.T
    proj = w_inv_half @ q @ q.T @ w_half
    return acts - x @ proj.T


def erasure_curves(
    learned_dict,
    acts: torch.Tensor,
    labels: torch.Tensor,
    ks: Optional[List[int]] = None,
) -> Dict[str, List[float]]:
    """Probe AUROC after each erasure method at increasing k."""
    if ks is None:
        ks = [0, 1, 2, 4, 8, 16, 32]
    out: Dict[str, List[float]] = {"k": [], "dict_ablation": [], "leace": []}
    base = logistic_regression_auroc(acts, labels, max_iter=200)
    for k in ks:
        out["k"].append(k)
        if k == 0:
            out["dict_ablation"].append(base)
            out["leace"].append(base)
            continue
        erased = ablate_top_features(learned_dict, acts, labels, k)
        out["dict_ablation"].append(logistic_regression_auroc(erased, labels, max_iter=200))
        out["leace"].append(logistic_regression_auroc(leace_erase(acts, labels, rank=min(k, acts.shape[1])), labels, max_iter=200))
    return out


def plot_erasure(curves: Dict[str, List[float]], save_path: str = "erasure.png"):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots()
    ks = curves["k"]
    ax.plot(ks, curves["dict_ablation"], "o-", label="dict-feature ablation")
    ax.plot(ks, curves["leace"], "s-", label="LEACE rank-k")
    ax.axhline(0.5, color="gray", linestyle=":")
    ax.set_xlabel("k (features / rank erased)")
    ax.set_ylabel("probe AUROC")
    ax.legend()
    fig.tight_layout()
    fig.savefig(save_path, dpi=120)
    return fig
