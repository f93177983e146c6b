"""Scaled-down end-to-end runs of the BASELINE.json configs on ONE GPU.

Each smoke runs the real code path of its config — model widths and dict
ratios are the named ones; batch counts are shrunk so the whole file runs in
a couple of GPU-minutes — and prints one JSON line with throughput + sanity
metrics.  (Configs 4-5 name 8xMI355X; their smokes run the same ensembles
and kernels on one GPU — the DP layer itself is exercised by bench.py under
torchrun and by tests/test_full_stack_gpu.py's RCCL-path test.)

  config 2: Pythia-70m layer-2 residual (d=512), 8x dict, 8-way L1 ensemble
  config 3: GPT-2-small MLP-out (d=768), 32-SAE (l1 x dict_size) grid —
            ragged dict sizes in ONE stacked ensemble via the masked step
  config 4: Pythia-410m residual (d=1024), 8x dict, L1 ensemble
  config 5: TopK (k=32) + dead-neuron resampling, Pythia-1.4b residual
            (d=2048), 8x dict

Run: python scripts/config_smokes.py [--steps N]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def _train(ens, gen, steps, resampler=None):
    # pre-generate a batch pool so the timing is the training step, not the
    # synthetic generator (same protocol as bench.py)
    pool = [gen.send(None).contiguous() for _ in range(8)]
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        batch = pool[i % len(pool)]
        losses, aux = ens.step_batch(batch)
        if resampler is not None:
            resampler.observe(batch, aux)
            if (i + 1) % max(steps // 2, 1) == 0 and i < steps - 5:
                resampler.resample()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return losses, dt


def smoke(name, ens, gen, steps, B, extra=None, resampler=None):
    losses, dt = _train(ens, gen, steps, resampler=resampler)
    loss = losses["loss"]
    rec = {
        "config": name,
        "backend": type(ens._hip_step).__name__ if ens._hip_step else "torch",
        "n_models": ens.n_models,
        "steps": steps,
        "acts_per_sec": round(B * steps / dt),
        "ms_per_step": round(dt / steps * 1e3, 3),
        "final_loss_finite": bool(torch.isfinite(loss).all()),
        "final_loss_mean": round(float(loss.mean()), 5),
    }
    if extra:
        rec.update(extra)
    print(json.dumps(rec), flush=True)
    return rec


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--batch", type=int, default=1024)
    args = p.parse_args()

    from sparse_coding_amd.data.random_dataset import RandomDatasetGenerator
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.engine.resample import EnsembleResampler
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import (
        FunctionalMaskedTiedSAE,
        FunctionalTiedSAE,
    )
    from sparse_coding_amd.models.topk import TopKEncoder

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    B, steps = args.batch, args.steps
    torch.manual_seed(0)
    np.random.seed(0)

    def generator(d):
        return RandomDatasetGenerator(
            activation_dim=d, n_ground_truth_components=2 * d, batch_size=B,
            feature_num_nonzero=32, feature_prob_decay=0.999,
            correlated=False, device=device)

    # config 2: Pythia-70m resid l2, 8x dict, 8-way L1 ensemble
    d = 512
    models = [FunctionalTiedSAE.init(d, 8 * d, float(l1), device=device)
              for l1 in np.logspace(-4, -2, 8)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                             device=device, backend="auto")
    smoke("2:pythia-70m-resid-8x-8L1", ens, generator(d), steps, B,
          extra={"d_model": d, "dict_size": 8 * d})
    del ens, models

    # config 3: GPT-2-small MLP-out, 32-SAE (l1 x dict_size) grid in one
    # ragged (masked) stacked ensemble
    d = 768
    ratios = (1, 2, 4, 8)
    l1s = np.logspace(-4, -2, 8)
    n_stack = int(max(ratios) * d)
    models = [FunctionalMaskedTiedSAE.init(d, int(r * d), n_stack, float(l1), device=device)
              for r in ratios for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalMaskedTiedSAE, adam, {"lr": 1e-3},
                             device=device, backend="auto")
    assert ens.n_models == 32
    smoke("3:gpt2sm-mlpout-32grid-masked", ens, generator(d), max(steps // 2, 10), B,
          extra={"d_model": d, "dict_sizes": sorted({int(r * d) for r in ratios}),
                 "n_stack": n_stack})
    del ens, models

    # config 4: Pythia-410m resid, 8x dict (the 8-GPU DP axis runs in bench.py)
    d = 1024
    models = [FunctionalTiedSAE.init(d, 8 * d, float(l1), device=device)
              for l1 in np.logspace(-4, -2, 8)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                             device=device, backend="auto")
    smoke("4:pythia-410m-resid-8x", ens, generator(d), max(steps // 2, 10), B,
          extra={"d_model": d, "dict_size": 8 * d})
    del ens, models

    # config 5: TopK k=32 + dead-neuron resampling at Pythia-1.4b width
    d = 2048
    models = [TopKEncoder.init(d, 8 * d, 32) for _ in range(2)]
    # no_stacking: torch.topk's k is per-model data-dependent, so the vmap
    # oracle loops models; the fused HipTopKStep (GPU) stacks fine
    ens = FunctionalEnsemble(models, TopKEncoder, adam, {"lr": 1e-3},
                             device=device, backend="auto", no_stacking=True)
    rs = EnsembleResampler(ens, n_track=256)
    smoke("5:pythia-1.4b-topk32-resample", ens, generator(d), max(steps // 3, 8), B,
          extra={"d_model": d, "dict_size": 8 * d, "k": 32}, resampler=rs)


if __name__ == "__main__":
    main()
