"""KL divergence of LM predictions under activation replacement.

Covers reference ``plotting/plot_kl_div.py``: for each learned dict, replace
the hooked activation with its reconstruction and measure the KL divergence
of the output distribution against the clean model, plotted against L0.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse
from typing import Dict, List, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from sparse_coding_amd.data.activation_dataset import replace_activation_hook
from sparse_coding_amd.metrics import standard_metrics as sm


@torch.no_grad()
def kl_under_reconstruction(
    model, learned_dict, layer: int, layer_loc: str, token_ids: torch.Tensor,
    device: str = "cuda:0", batch_size: int = 8,
) -> float:
    """mean KL( p_clean || p_replaced ) over all token positions."""
    model.eval()
    total_kl, total_tok = 0.0, 0
    for i in range(0, token_ids.shape[0], batch_size):
        ids = token_ids[i : i + batch_size].to(device)
        clean = model(input_ids=ids).logits.float()
        with replace_activation_hook(model, layer, layer_loc, learned_dict):
            repl = model(input_ids=ids).logits.float()
        logp_c = torch.log_softmax(clean, dim=-1)
        logp_r = torch.log_softmax(repl, dim=-1)
        kl = (logp_c.exp() * (logp_c - logp_r)).sum(dim=-1)
        total_kl += kl.sum().item()
        total_tok += kl.numel()
    return total_kl / max(total_tok, 1)


def score_dicts_kl(
    learned_dicts_path: str, model, layer: int, layer_loc: str,
    token_ids: torch.Tensor, sample: torch.Tensor, device: str = "cuda:0",
) -> List[Tuple[float, float, dict]]:
    dicts = torch.load(learned_dicts_path, map_location="cpu", weights_only=False)
    out = []
    for ld, hp in dicts:
        ld.to_device(device)
        kl = kl_under_reconstruction(model, ld, layer, layer_loc, token_ids, device)
        l0 = sm.mean_l0(ld, sample.to(device)).item()
        out.append((l0, kl, hp))
    return out


def plot_kl(points: List[Tuple[float, float, dict]], save_path: str = "kl_div.png"):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots()
    pts = sorted((p[0], p[1]) for p in points)
    ax.plot([p[0] for p in pts], [p[1] for p in pts], "o-")
    ax.set_xlabel("mean L0")
    ax.set_ylabel("KL(clean || reconstructed)")
    ax.set_yscale("log")
    fig.tight_layout()
    fig.savefig(save_path, dpi=120)
    return fig


def main():
    from sparse_coding_amd.data.activation_dataset import load_model, synthetic_token_batches

    p = argparse.ArgumentParser()
    p.add_argument("--learned-dicts", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--model", default="pythia-70m-deduped")
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    p.add_argument("--out", default="kl_div.png")
    args = p.parse_args()

    model = load_model(args.model, device=args.device)
    token_ids = torch.cat(list(synthetic_token_batches(model.config.vocab_size, 4, 128, 4)))
    chunk = torch.load(args.chunk, map_location="cpu").float()
    sample = chunk[np.random.choice(len(chunk), size=min(10000, len(chunk)), replace=False)]
    points = score_dicts_kl(args.learned_dicts, model, args.layer, args.layer_loc, token_ids, sample, args.device)
    plot_kl(points, args.out)


if __name__ == "__main__":
    main()
