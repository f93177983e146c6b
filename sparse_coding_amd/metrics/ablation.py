"""Feature-ablation graphs: causal influence between dictionary features at
different layer locations.

Parity with reference ``standard_metrics.py:110-222``
(build_ablation_graph / _non_positional + the ablate interventions), on HF
forward hooks instead of TransformerLens run_with_hooks.

Location = (layer, layer_loc) e.g. (2, "residual").
"""

from __future__ import annotations

import contextlib
from typing import Dict, List, Tuple

import torch

from sparse_coding_amd.data.activation_dataset import (
    capture_activation_hook,
    resolve_hook_point,
)

Location = Tuple[int, str]


@contextlib.contextmanager
def _intervention_hook(model, layer: int, layer_loc: str, fn):
    """Apply fn to the flattened [B*L, d] activation at (layer, layer_loc)."""
    module, kind = resolve_hook_point(model, layer, layer_loc)
    if kind != "output0":
        raise NotImplementedError("interventions support output locations only")

    def hook(mod, inputs, output):
        is_tuple = isinstance(output, tuple)
        out = output[0] if is_tuple else output
        shape = out.shape
        new = fn(out.reshape(-1, shape[-1])).reshape(shape)
        return (new,) + tuple(output[1:]) if is_tuple else new

    handle = module.register_forward_hook(hook)
    try:
        yield
    finally:
        handle.remove()


def ablate_feature_intervention(learned_dict, feature_idx: int):
    """Subtract one feature's contribution from the activation
    (reference ablate_feature_intervention_non_positional :163-177)."""

    def go(flat: torch.Tensor) -> torch.Tensor:
        f32 = flat.to(torch.float32)
        c = learned_dict.encode(learned_dict.center(f32))
        contribution = torch.outer(c[:, feature_idx], learned_dict.get_learned_dict()[feature_idx])
        return (f32 - contribution).to(flat.dtype)

    return go


@torch.no_grad()
def cache_all_feature_activations(
    model,
    dicts: Dict[Location, "LearnedDict"],
    tokens: torch.Tensor,
    device: str = "cuda:0",
    extra_hooks=(),
) -> Dict[Location, torch.Tensor]:
    """Per-location [B, L, n_feats] feature activations for one forward."""
    stores: Dict[Location, list] = {loc: [] for loc in dicts}
    B, L = tokens.shape
    with contextlib.ExitStack() as stack:
        for loc in dicts:
            stack.enter_context(capture_activation_hook(model, loc[0], loc[1], stores[loc]))
        for ctx in extra_hooks:
            stack.enter_context(ctx)
        model(input_ids=tokens.to(device))

    out = {}
    for loc, ld in dicts.items():
        acts = stores[loc][0].to(torch.float32)
        code = ld.encode(ld.center(acts))
        out[loc] = code.reshape(B, L, -1)
    return out


@torch.no_grad()
def build_ablation_graph(
    model,
    dicts: Dict[Location, "LearnedDict"],
    tokens: torch.Tensor,
    features_to_ablate: Dict[Location, List[int]],
    target_features: Dict[Location, List[int]] = None,
    device: str = "cuda:0",
) -> Dict[Tuple[Tuple[Location, int], Tuple[Location, int]], float]:
    """Edge weight = mean |Δ target-feature activation| when a source feature
    is ablated (reference build_ablation_graph_non_positional :180-222)."""
    if target_features is None:
        target_features = {loc: list(range(d.n_feats)) for loc, d in dicts.items()}
    all_targets = [(loc, f) for loc, feats in target_features.items() for f in feats]

    base = cache_all_feature_activations(model, dicts, tokens, device)

    graph = {}
    for loc, ld in dicts.items():
        for feature in features_to_ablate.get(loc, []):
            hook = _intervention_hook(model, loc[0], loc[1], ablate_feature_intervention(ld, feature))
            ablated = cache_all_feature_activations(model, dicts, tokens, device, extra_hooks=[hook])
            for loc_t, f_t in all_targets:
                if loc_t == loc and f_t == feature:
                    continue
                delta = base[loc_t][:, :, f_t] - ablated[loc_t][:, :, f_t]
                graph[((loc, feature), (loc_t, f_t))] = torch.norm(delta, dim=-1).mean().item()
    return graph
