"""Kernel-variant sweep for the fused SAE training step (GPU box).

Times ensemble.step_batch over the (staging, bk, prio) kernel-config grid on
the flagship bench shape and prints one JSON line per combo plus the winner.
Run on a GPU box:  python scripts/ktune.py [--steps 30] [--batch 2048]

The measured winner is then baked into sparse_coding_amd/ops/kconfig.DEFAULTS.
"""

from __future__ import annotations

import argparse
import itertools
import json
import time

import os
import sys

import numpy as np
import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def bench_combo(args, staging, bk, prio, bk_gw=None, bk_dec=None, bn=128):
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.ops.kconfig import set_kernel_config

    set_kernel_config(staging=staging, bk=bk, prio=prio, bk_grad_w=bk_gw, bk_dec=bk_dec, bn=bn)
    device = "cuda:0"
    d, n_dict, M, B = args.d_model, args.d_model * args.dict_ratio, args.n_models, args.batch
    torch.manual_seed(0)
    models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=device)
              for l1 in np.logspace(-4, -2, M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                             device=device, backend="hip")
    x = torch.randn(B, d, device=device)
    for _ in range(args.warmup):
        ens.step_batch(x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        ens.step_batch(x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    return dt * 1e3, B / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--dict-ratio", type=int, default=8)
    p.add_argument("--n-models", type=int, default=8)
    args = p.parse_args()

    results = []
    for staging, bk, prio in itertools.product(["t", "pre"], [32, 16], [False, True]):
        ms, acts = bench_combo(args, staging, bk, prio)
        rec = {"staging": staging, "bk": bk, "prio": prio,
               "ms_per_step": round(ms, 4), "acts_per_sec": round(acts)}
        results.append(rec)
        print(json.dumps(rec), flush=True)

    best = min(results, key=lambda r: r["ms_per_step"])
    print(json.dumps({"winner": best}), flush=True)

    # mixed: winner's staging/prio with per-kernel bk for the long-K
    # kernels (grad_w K=B, dec K=n): deep tiles can win there even when
    # occupancy wins the short-K kernels
    # wide-tile probe (128x256): 4 accumulators/wave, 2 blocks/CU
    ms, acts = bench_combo(args, best["staging"], 16, best["prio"], bn=256)
    print(json.dumps({"staging": best["staging"], "bn": 256,
                      "ms_per_step": round(ms, 4), "acts_per_sec": round(acts)}), flush=True)

    for key in ("bk_grad_w", "bk_dec"):
        for bk_k in (16, 32):
            if bk_k == best["bk"]:
                continue
            ms, acts = bench_combo(args, best["staging"], best["bk"], best["prio"],
                                   **({"bk_gw": bk_k} if key == "bk_grad_w" else {"bk_dec": bk_k}))
            rec = {"staging": best["staging"], "bk": best["bk"], "prio": best["prio"],
                   key: bk_k, "ms_per_step": round(ms, 4), "acts_per_sec": round(acts)}
            print(json.dumps(rec), flush=True)


if __name__ == "__main__":
    main()
