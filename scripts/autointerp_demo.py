"""End-to-end autointerp artifact (VERDICT item 3 'Done' criterion): train a
flagship-config dict on Pythia-70m-shape activations, run the full
explain/simulate protocol offline, and emit scored explanations + the violin
plot.

GPU box:  python scripts/autointerp_demo.py --steps 2000
CPU test scale:  --model-name tiny-gptneox --d-model 64 --steps 50
"""

from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-name", default="pythia-70m-deduped")
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--dict-ratio", type=int, default=8)
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--l1", type=float, default=8.5e-4)
    p.add_argument("--n-feats-explain", type=int, default=40)
    p.add_argument("--n-fragments", type=int, default=256)
    p.add_argument("--out-dir", default="docs/autointerp_r02")
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    np.random.seed(0)
    os.makedirs(args.out_dir, exist_ok=True)

    from sparse_coding_amd.config import InterpArgs
    from sparse_coding_amd.data.activation_dataset import (
        capture_activation_hook,
        load_model,
        synthetic_token_batches,
    )
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.interpret import drivers
    from sparse_coding_amd.interpret.interpret import plot_scores
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    # 1. train a dict on the host LM's activations
    model = load_model(args.model_name, device=device)
    d = args.d_model
    n_dict = d * args.dict_ratio

    def act_batches(n):
        bsz = max(1, args.batch // 256)
        for toks in synthetic_token_batches(model.config.vocab_size, bsz, 256, n):
            store = []
            with torch.no_grad(), capture_activation_hook(model, args.layer, "residual", store):
                model(input_ids=toks.to(device))
            yield store[0].float()

    models = [FunctionalTiedSAE.init(d, n_dict, args.l1, device=device)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=device)
    backend = type(ens._hip_step).__name__ if ens._hip_step else "torch"
    pool = list(act_batches(8))
    for step in range(args.steps):
        if step % 64 == 0 and step:
            pool = list(act_batches(8))
        losses, _ = ens.step_batch(pool[step % len(pool)])
    ld = ens.to_learned_dicts()[0]
    dict_path = os.path.join(args.out_dir, "learned_dict.pt")
    torch.save(ld, dict_path)
    print(f"trained ({backend}): final loss {losses['loss'].item():.4f}")

    # 2. the full protocol (offline mock client), reference result layout
    cfg = InterpArgs()
    cfg.model_name = args.model_name
    cfg.layer = args.layer
    cfg.device = device
    cfg.df_n_feats = max(args.n_feats_explain, 40)
    cfg.n_feats_explain = args.n_feats_explain
    cfg.save_loc = os.path.join(args.out_dir, "sparse_coding")
    scores = drivers.run(ld, cfg, model=model, n_fragments=args.n_fragments)
    print(f"scored {len(scores)} features; mean={np.mean(list(scores.values())) if scores else float('nan'):.3f}")

    # 3. readers + violin plot over the result tree
    by_mode = {}
    for mode in ("top", "random", "top_random"):
        sc = drivers.read_scores(args.out_dir, mode)
        by_mode[mode] = {k: v[1] for k, v in sc.items()}
    plot_scores({k: v for k, v in by_mode["top_random"].items()},
                save_path=os.path.join(args.out_dir, "autointerp_violin.png"))
    with open(os.path.join(args.out_dir, "summary.json"), "w") as f:
        json.dump({
            "config": {"model": args.model_name, "d_model": d, "dict_size": n_dict,
                       "l1": args.l1, "steps": args.steps, "backend": backend,
                       "client": "MockLLMClient (offline; swap in HFLocalClient/OpenAI "
                                 "via drivers.run(client=...))"},
            "n_scored": len(scores),
            "mean_score": float(np.mean(list(scores.values()))) if scores else None,
            "scores": {str(k): round(v, 4) for k, v in scores.items()},
        }, f, indent=1)
    print(f"wrote {args.out_dir}")


if __name__ == "__main__":
    main()
