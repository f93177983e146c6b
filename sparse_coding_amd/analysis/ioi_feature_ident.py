"""IOI feature identification: which dictionary features carry the
indirect-object signal?

Fills the role of the reference's MISSING ``ioi_feature_ident.py`` (its
``do_ioi_multiple_layers.sh:4`` calls it but the file does not exist —
SURVEY.md §2 "known dangling references") using the full IOIDataset
semantics (data/ioi_counterfact.py) and the ablation machinery:

1. build clean prompts + IO-flipped counterfacts (the Redwood flip);
2. capture layer activations at the END position (where the model must
   prefer the IO over the S name);
3. encode with the learned dict; rank features by the mean activation
   difference between clean and counterfact runs;
4. verify by ablating the top features and measuring the change in the host
   LM's IO-vs-S logit difference.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch


def capture_end_activations(model, dataset, layer: int, layer_loc: str,
                            device: str = "cpu") -> torch.Tensor:
    """[N, d] activations at each prompt's end position."""
    from sparse_coding_amd.data.activation_dataset import capture_activation_hook

    store: List[torch.Tensor] = []
    toks = dataset.toks.to(device)
    with torch.no_grad(), capture_activation_hook(model, layer, layer_loc, store):
        model(input_ids=toks)
    acts = store[0].float().reshape(toks.shape[0], toks.shape[1], -1)
    end = dataset.word_idx["end"].to(device)
    return acts[torch.arange(acts.shape[0], device=device), end]


def rank_ioi_features(learned_dict, clean_acts: torch.Tensor,
                      flipped_acts: torch.Tensor, top_k: int = 16):
    """Features ranked by |mean clean - mean flipped| code activation."""
    c_clean = learned_dict.encode(learned_dict.center(clean_acts))
    c_flip = learned_dict.encode(learned_dict.center(flipped_acts))
    diff = (c_clean.mean(dim=0) - c_flip.mean(dim=0)).abs()
    order = torch.argsort(diff, descending=True)[:top_k]
    return order, diff[order], c_clean, c_flip


def logit_diff(model, dataset, device: str = "cpu",
               edit_layer: Optional[int] = None, layer_loc: str = "residual",
               edit_fn=None) -> float:
    """Mean (IO logit - S logit) at the end position, optionally with an
    activation edit patched in at one layer (the ablation eval hook)."""

    from sparse_coding_amd.data.activation_dataset import resolve_hook_point

    toks = dataset.toks.to(device)
    handle = None
    if edit_fn is not None:
        module, kind = resolve_hook_point(model, edit_layer, layer_loc)

        def hook(mod, inputs, output):
            if isinstance(output, tuple):
                return (edit_fn(output[0]),) + output[1:]
            return edit_fn(output)

        handle = module.register_forward_hook(hook)
    try:
        with torch.no_grad():
            logits = model(input_ids=toks).logits
    finally:
        if handle is not None:
            handle.remove()
    end = dataset.word_idx["end"].to(device)
    final = logits[torch.arange(logits.shape[0], device=device), end]
    io = dataset.io_token_ids().to(device)
    s = dataset.s_token_ids().to(device)
    rows = torch.arange(final.shape[0], device=device)
    return (final[rows, io] - final[rows, s]).mean().item()


def ablate_features_edit(learned_dict, features: torch.Tensor, device="cpu"):
    """edit_fn that removes the chosen dictionary features from a layer's
    activations (used with logit_diff's patch hook)."""
    def edit(h):
        shape = h.shape
        flat = h.reshape(-1, shape[-1]).float()
        code = learned_dict.encode(learned_dict.center(flat))
        removed = code[:, features] @ learned_dict.get_learned_dict()[features]
        out = flat - learned_dict.uncenter(removed) + learned_dict.uncenter(
            torch.zeros_like(removed))
        return out.reshape(shape).to(h.dtype)

    return edit


def run_ioi_feature_ident(learned_dict, model, layer: int = 2,
                          layer_loc: str = "residual", n_prompts: int = 64,
                          top_k: int = 8, device: str = "cpu",
                          tokenizer=None) -> Dict:
    """End-to-end study; returns ranked features + ablation verification."""
    from sparse_coding_amd.data.ioi_counterfact import IOIDataset

    ds = IOIDataset("mixed", N=n_prompts, tokenizer=tokenizer, seed=0)
    ds_flip = ds.gen_flipped_prompts("IO", seed=1)
    clean = capture_end_activations(model, ds, layer, layer_loc, device)
    flipped = capture_end_activations(model, ds_flip, layer, layer_loc, device)
    feats, scores, _, _ = rank_ioi_features(learned_dict, clean, flipped, top_k=top_k)

    base = logit_diff(model, ds, device)
    ablated = logit_diff(model, ds, device, edit_layer=layer, layer_loc=layer_loc,
                         edit_fn=ablate_features_edit(learned_dict, feats, device))
    return {
        "features": feats.tolist(),
        "diff_scores": scores.tolist(),
        "base_logit_diff": base,
        "ablated_logit_diff": ablated,
    }


if __name__ == "__main__":
    import argparse

    from sparse_coding_amd.data.activation_dataset import load_model

    p = argparse.ArgumentParser()
    p.add_argument("--learned-dict", required=True)
    p.add_argument("--model-name", default="pythia-70m-deduped")
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--n-prompts", type=int, default=64)
    p.add_argument("--top-k", type=int, default=8)
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()

    ld = torch.load(args.learned_dict, map_location="cpu", weights_only=False)
    if isinstance(ld, list):
        ld = ld[0][0]
    ld.to_device(args.device)
    model = load_model(args.model_name, device=args.device)
    out = run_ioi_feature_ident(ld, model, args.layer, args.layer_loc,
                                args.n_prompts, args.top_k, args.device)
    print(out)
