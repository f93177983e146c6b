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
    """One-shot LEACE erasure (fit + apply on the same data); see
    :class:`LeaceEraser` for the estimator."""
    return LeaceEraser.fit(acts, labels, rank=rank)(acts)


class LeaceEraser:
    """The full LEACE estimator (Belrose et al. 2023, arXiv:2306.03819):
    the least-squares-optimal affine eraser
        r(x) = x - Sigma^{1/2} P Sigma^{-1/2} (x - mu)
    with P the orthogonal projection onto the whitened cross-covariance
    colspace W @ Sigma_xz.  Fit once, apply anywhere (train/eval split —
    the reference's erasure studies fit on one prompt set and score
    transfer on another, plotting/erasure_plot.py:199-215).

    ``labels`` may be binary {0,1} (rank-1 eraser) or integer classes
    (one-hot z, rank up to C-1)."""

    def __init__(self, eraser: torch.Tensor, mu: torch.Tensor):
        self.eraser = eraser  # [d, d]
        self.mu = mu

    @staticmethod
    def fit(acts: torch.Tensor, labels: torch.Tensor, rank: Optional[int] = None) -> "LeaceEraser":
        x = acts - acts.mean(dim=0)
        classes = labels.unique()
        if len(classes) <= 2:
            z = (labels.float() - labels.float().mean()).unsqueeze(1)
        else:
            z = torch.nn.functional.one_hot(labels.long()).float()
            z = z - z.mean(dim=0)
        sigma = (x.T @ x) / x.shape[0] + 1e-4 * torch.eye(x.shape[1])
        cross = (x.T @ z) / x.shape[0]  # [d, C]
        evals, evecs = torch.linalg.eigh(sigma)
        w_inv_half = evecs @ torch.diag(evals.clamp_min(1e-6).rsqrt()) @ evecs.T
        w_half = evecs @ torch.diag(evals.clamp_min(1e-6).sqrt()) @ evecs.T
        q, _ = torch.linalg.qr(w_inv_half @ cross)
        if rank is not None:
            q = q[:, :rank]
        eraser = w_half @ q @ q.T @ w_inv_half  # Sigma^{1/2} P Sigma^{-1/2}
        return LeaceEraser(eraser, acts.mean(dim=0))

    def __call__(self, acts: torch.Tensor) -> torch.Tensor:
        return acts - (acts - self.mu) @ self.eraser.T

    def mean_edit(self, acts: torch.Tensor) -> float:
        return (self(acts) - acts).norm(dim=-1).mean().item()


def mean_erase(acts: torch.Tensor, labels: torch.Tensor,
               fit_acts: Optional[torch.Tensor] = None,
               fit_labels: Optional[torch.Tensor] = None,
               affine: bool = False) -> torch.Tensor:
    """The reference study's "Mean" erasers (plotting/erasure_plot.py:140-152):
    project out the class-mean-difference direction; ``affine=True`` instead
    translates every class's mean onto the global mean (an affine edit)."""
    fa = acts if fit_acts is None else fit_acts
    fl = labels if fit_labels is None else fit_labels
    mu0 = fa[fl == 0].mean(dim=0)
    mu1 = fa[fl == 1].mean(dim=0)
    if affine:
        mu = fa.mean(dim=0)
        out = acts.clone()
        out[labels == 0] += mu - mu0
        out[labels == 1] += mu - mu1
        return out
    v = mu1 - mu0
    v = v / v.norm().clamp_min(1e-8)
    return acts - (acts @ v)[:, None] * v[None, :]


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
