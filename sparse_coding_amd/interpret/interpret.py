"""Auto-interpretation: per-feature activation fragments + scoring protocol.

Parity with reference ``interpret.py`` (C22): build a per-feature activation
table from 64-token text fragments (make_feature_activation_dataset :82-212),
cache it (:215-262), select top-k + random activation records per feature
(:282-316), drive an explainer/simulator (:265-386), and read/plot scores
(:691-761).

The reference hard-depends on the OpenAI ``neuron_explainer`` package and an
API key at import time (:30-32).  Here the LLM calls are injected: pass any
callable ``explain_fn(records) -> str`` / ``simulate_fn(explanation,
records) -> score``; without one, the fragment/record machinery still runs
(it is what the quality metrics need) and explanation calls raise cleanly.
"""

from __future__ import annotations

import json
import os
from dataclasses import asdict, dataclass, field
from typing import Callable, Dict, List, Optional, Sequence

import numpy as np
import torch

FRAGMENT_LEN = 64  # reference interpret.py uses 64-token fragments
TOP_K_RECORDS = 20
RANDOM_RECORDS = 20


@dataclass
class ActivationRecord:
    """One fragment's per-token activations for one feature (mirrors the
    neuron-explainer protocol's ActivationRecord)."""

    tokens: List[str]
    activations: List[float]


@dataclass
class FeatureInterpretation:
    feature: int
    explanation: str = ""
    score: float = float("nan")
    top_records: List[ActivationRecord] = field(default_factory=list)
    random_records: List[ActivationRecord] = field(default_factory=list)

    def save(self, folder: str) -> None:
        os.makedirs(folder, exist_ok=True)
        with open(os.path.join(folder, f"feature_{self.feature}.json"), "w") as f:
            json.dump(asdict(self), f)


def make_feature_activation_dataset(
    model,
    learned_dict,
    layer: int,
    layer_loc: str,
    token_batches: Sequence[torch.Tensor],
    tokenizer=None,
    device: str = "cuda:0",
    feature_indices: Optional[Sequence[int]] = None,
    max_features: int = 200,
):
    """Run the LM over fragments, encode hooked activations with the learned
    dict, and return a dense [n_fragments, frag_len, n_feats_kept] activation
    array + the token id array (reference :82-212, DF-free: a tensor table is
    smaller and GPU-friendly; pandas conversion is one call away).
    """
    from sparse_coding_amd.data.activation_dataset import capture_activation_hook

    if feature_indices is None:
        feature_indices = list(range(min(learned_dict.n_feats, max_features)))
    feature_indices = list(feature_indices)

    all_acts = []
    all_tokens = []
    model.eval()
    with torch.no_grad():
        for batch in token_batches:
            batch = batch.to(device)
            store: List[torch.Tensor] = []
            with capture_activation_hook(model, layer, layer_loc, store):
                model(input_ids=batch)
            acts = store[0].to(torch.float32)  # [b*l, d_act]
            code = learned_dict.encode(learned_dict.center(acts))  # [b*l, n]
            code = code[:, feature_indices]
            b, l = batch.shape
            all_acts.append(code.reshape(b, l, -1).cpu())
            all_tokens.append(batch.cpu())
    return torch.cat(all_acts), torch.cat(all_tokens), feature_indices


def get_df(cache_path: str, builder: Callable, rebuild: bool = False):
    """Tensor-table cache (reference caches an HDF DataFrame, :215-262)."""
    if os.path.exists(cache_path) and not rebuild:
        return torch.load(cache_path, weights_only=False)
    data = builder()
    os.makedirs(os.path.dirname(cache_path) or ".", exist_ok=True)
    torch.save(data, cache_path)
    return data


def _decode_tokens(tokenizer, ids: torch.Tensor) -> List[str]:
    if tokenizer is None:
        return [f"<{int(t)}>" for t in ids]
    return [tokenizer.decode([int(t)]) for t in ids]


def select_activation_records(
    acts: torch.Tensor,      # [n_frag, frag_len, n_feats]
    tokens: torch.Tensor,    # [n_frag, frag_len]
    feature: int,
    tokenizer=None,
    top_k: int = TOP_K_RECORDS,
    n_random: int = RANDOM_RECORDS,
    sort_mode: str = "max",
    seed: int = 0,
):
    """Top-k fragments by max (or mean) activation + random fragments
    (reference :282-316)."""
    feat_acts = acts[:, :, feature]
    if sort_mode == "max":
        scores = feat_acts.max(dim=-1).values
    else:
        scores = feat_acts.mean(dim=-1)
    order = torch.argsort(scores, descending=True)
    top_idx = order[:top_k]
    rng = np.random.default_rng(seed)
    rest = order[top_k:].numpy()
    rand_idx = rng.choice(rest, size=min(n_random, len(rest)), replace=False) if len(rest) else []

    def record(i):
        return ActivationRecord(
            tokens=_decode_tokens(tokenizer, tokens[i]),
            activations=[float(a) for a in feat_acts[i]],
        )

    return [record(int(i)) for i in top_idx], [record(int(i)) for i in rand_idx]


def interpret_features(
    acts: torch.Tensor,
    tokens: torch.Tensor,
    feature_indices: Sequence[int],
    tokenizer=None,
    explain_fn: Optional[Callable] = None,
    simulate_fn: Optional[Callable] = None,
    output_folder: Optional[str] = None,
    sort_mode: str = "max",
) -> List[FeatureInterpretation]:
    """The explain+score loop (reference async interpret() :265-386).

    explain_fn(top_records) -> explanation string
    simulate_fn(explanation, records) -> float score (explained-variance of
    simulated vs real activations, the neuron-explainer convention)
    """
    results = []
    for fi, feature in enumerate(feature_indices):
        top, rand = select_activation_records(acts, tokens, fi, tokenizer, sort_mode=sort_mode)
        interp = FeatureInterpretation(feature=int(feature), top_records=top, random_records=rand)
        if explain_fn is not None:
            interp.explanation = explain_fn(top)
            if simulate_fn is not None:
                interp.score = float(simulate_fn(interp.explanation, top + rand))
        if output_folder:
            interp.save(output_folder)
        results.append(interp)
    return results


def correlation_score_simulator(predict_fn: Callable) -> Callable:
    """Build a simulate_fn that scores by correlation between predicted and
    real activations — the scoring rule of the neuron-explainer protocol."""

    def simulate(explanation: str, records: List[ActivationRecord]) -> float:
        real, pred = [], []
        for rec in records:
            p = predict_fn(explanation, rec.tokens)
            real.extend(rec.activations)
            pred.extend(p)
        real_a, pred_a = np.asarray(real), np.asarray(pred)
        if real_a.std() < 1e-9 or pred_a.std() < 1e-9:
            return 0.0
        return float(np.corrcoef(real_a, pred_a)[0, 1])

    return simulate


def read_results(folder: str) -> Dict[int, Dict]:
    """Load saved per-feature interpretations (reference read_results :691)."""
    out = {}
    for fname in sorted(os.listdir(folder)):
        if fname.startswith("feature_") and fname.endswith(".json"):
            with open(os.path.join(folder, fname)) as f:
                rec = json.load(f)
            out[rec["feature"]] = rec
    return out


def plot_scores(results_by_name: Dict[str, List[float]], save_path: Optional[str] = None):
    """Violin plot of autointerp scores per dict family, fixed -0.2..0.6 axis
    (reference :702-761, interpret.py:50-51)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots()
    names = list(results_by_name.keys())
    data = [results_by_name[n] for n in names]
    if any(len(d) for d in data):
        ax.violinplot([d if len(d) else [0.0] for d in data], showmeans=True)
    ax.set_xticks(range(1, len(names) + 1))
    ax.set_xticklabels(names, rotation=45, ha="right")
    ax.set_ylim(-0.2, 0.6)
    ax.set_ylabel("autointerp score")
    fig.tight_layout()
    if save_path:
        fig.savefig(save_path)
    return fig


def fragment_table_to_dataframe(acts, tokens, feature_indices, tokenizer=None):
    """The reference's activation-DataFrame schema (interpret.py:131-212 /
    read at :271-283): one row per fragment with ``fragment_token_ids``,
    ``fragment_token_strs`` and, per kept feature i,
    ``feature_{i}_activation_{j}`` (j < fragment length) + ``feature_{i}_max``
    columns.  The tensor table stays the native format; this converter gives
    reference-analysis code the exact columns it indexes."""
    import pandas as pd

    n_frag, frag_len, n_feats = acts.shape
    cols = {
        "fragment_token_ids": [tokens[i].tolist() for i in range(n_frag)],
        "fragment_token_strs": [_decode_tokens(tokenizer, tokens[i]) for i in range(n_frag)],
    }
    acts_np = acts.cpu().numpy()
    for fi, feature in enumerate(feature_indices):
        for j in range(frag_len):
            cols[f"feature_{feature}_activation_{j}"] = acts_np[:, j, fi]
        cols[f"feature_{feature}_max"] = acts_np[:, :, fi].max(axis=1)
    return pd.DataFrame(cols)
