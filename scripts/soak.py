"""Long-horizon numeric soak of the fused steps (GPU box).

Runs tens of thousands of fused steps per signature, asserting finite
losses/params throughout — drift/overflow insurance beyond the short
numerics tests.  python scripts/soak.py [--tied 50000 --thresh 20000 --topk 20000]
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import numpy as np
import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def soak(name, sig, models, steps, B, d, no_stack=False):
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam

    ens = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, device="cuda:0",
                             backend="hip", no_stacking=no_stack)
    pool = [torch.randn(B, d, device="cuda:0") for _ in range(4)]
    t0 = time.perf_counter()
    losses = None
    for i in range(steps):
        losses, _ = ens.step_batch(pool[i % 4])
        if i % 5000 == 0:
            assert torch.isfinite(losses["loss"]).all(), (name, i)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert torch.isfinite(losses["loss"]).all(), name
    for k, v in ens.params.items():
        if torch.is_tensor(v):
            assert torch.isfinite(v).all(), (name, k)
    print(f"[soak {name}] {steps} steps, {steps * B / dt:,.0f} acts/s, "
          f"final mean loss {float(losses['loss'].mean()):.5f}", flush=True)
    del ens, pool
    torch.cuda.empty_cache()


def main():
    from sparse_coding_amd.models.sae_signatures import (
        FunctionalThresholdingSAE,
        FunctionalTiedSAE,
    )
    from sparse_coding_amd.models.topk import TopKEncoder

    p = argparse.ArgumentParser()
    p.add_argument("--tied", type=int, default=50000)
    p.add_argument("--thresh", type=int, default=20000)
    p.add_argument("--topk", type=int, default=20000)
    args = p.parse_args()

    torch.manual_seed(0)
    np.random.seed(0)
    d, n, M, B = 512, 4096, 8, 2048
    l1s = np.logspace(-4, -2, M)
    if args.tied:
        soak("tied", FunctionalTiedSAE,
             [FunctionalTiedSAE.init(d, n, float(l1), device="cuda:0") for l1 in l1s],
             args.tied, B, d)
    if args.thresh:
        soak("thresholding", FunctionalThresholdingSAE,
             [FunctionalThresholdingSAE.init(d, n, float(l1), device="cuda:0") for l1 in l1s],
             args.thresh, B, d)
    if args.topk:
        soak("topk", TopKEncoder, [TopKEncoder.init(d, n, 32) for _ in range(M)],
             args.topk, B, d, no_stack=True)


if __name__ == "__main__":
    main()
