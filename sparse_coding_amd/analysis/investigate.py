"""Converged-feature analyses (reference experiments/ misc, C29).

- investigate_converged: which features two dicts agree on (investigate.py)
- dict_vs_embedding_cosines: dictionary directions vs token-embedding rows
  (experiments/check_l0_tokens.py)
- moment_interp_correlation: activation-moment statistics vs autointerp
  scores (experiments/interp_moment_corrs.py)
- pca_perplexity: perplexity under top-k PCA reconstruction across k
  (experiments/pca_perplexity.py)
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm
from sparse_coding_amd.models.learned_dict import LearnedDict


def investigate_converged(ld_a: LearnedDict, ld_b: LearnedDict, threshold: float = 0.9):
    """Indices of a's features with a close (cosine > threshold) partner in b,
    plus the match matrix statistics (reference investigate.py)."""
    cos = ld_a.get_learned_dict() @ ld_b.get_learned_dict().T
    best, idx = cos.max(dim=-1)
    converged = torch.where(best > threshold)[0]
    return {
        "converged_idx": converged,
        "n_converged": int(converged.numel()),
        "frac_converged": float((best > threshold).float().mean()),
        "best_match": best,
        "match_idx": idx,
    }


def dict_vs_embedding_cosines(ld: LearnedDict, embedding: torch.Tensor, top_k: int = 5):
    """Max cosine sims between dictionary directions and (unit-normalized)
    token-embedding rows (reference experiments/check_l0_tokens.py)."""
    emb = embedding / torch.clamp(torch.norm(embedding, dim=-1, keepdim=True), 1e-8)
    cos = ld.get_learned_dict() @ emb.T
    top = torch.topk(cos, top_k, dim=-1)
    return top.values, top.indices


def moment_interp_correlation(
    ld: LearnedDict,
    activations: torch.Tensor,
    interp_scores: Dict[int, float],
) -> Dict[str, float]:
    """Correlation of per-feature activation moments (skew/kurtosis/variance)
    with autointerp scores (reference experiments/interp_moment_corrs.py)."""
    times_active, mean, var, skew, kurt, m4 = sm.calc_moments_streaming(ld, activations)
    feats = sorted(interp_scores.keys())
    scores = np.array([interp_scores[f] for f in feats])
    out = {}
    for name, stat in [("mean", mean), ("var", var), ("skew", skew), ("kurtosis", kurt)]:
        vals = stat[feats].cpu().numpy()
        ok = np.isfinite(vals) & np.isfinite(scores)
        if ok.sum() > 2 and vals[ok].std() > 1e-9 and scores[ok].std() > 1e-9:
            out[name] = float(np.corrcoef(vals[ok], scores[ok])[0, 1])
        else:
            out[name] = float("nan")
    return out


def pca_perplexity_curve(
    model,
    tokenizer,
    pca,
    layer: int,
    layer_loc: str,
    token_ids: torch.Tensor,
    ks: Sequence[int] = (1, 2, 4, 8, 16, 32, 64),
    device: str = "cuda:0",
) -> List[Tuple[int, float]]:
    """Perplexity with the activation replaced by its top-k PCA
    reconstruction (reference experiments/pca_perplexity.py)."""
    out = []
    for k in ks:
        ld = pca.to_learned_dict(k)
        ld.to_device(device)
        ppl = sm.calculate_perplexity(model, tokenizer, ld, layer, layer_loc, token_ids, device=device)
        out.append((k, ppl))
    return out
