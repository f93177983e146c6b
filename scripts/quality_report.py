"""Quality-curve report: FVU-vs-L0 pareto, MMCS-to-ground-truth, dead
fraction for the flagship ensemble config on synthetic activations.

The reference publishes quality curves, not throughput; its
training semantics are reproduced exactly (tests/test_hip_numerics.py shows
the fused step tracks the vmap oracle), so this report measures the same
curves this framework produces on known ground truth.

Run on an MI355X:  python scripts/quality_report.py --steps 4000
Outputs: docs/quality/quality_report.json + fvu_l0_pareto.png
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=4000)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--dict-ratio", type=int, default=8)
    p.add_argument("--n-true", type=int, default=1024)
    p.add_argument("--nonzero", type=int, default=20)
    p.add_argument("--n-models", type=int, default=8)
    p.add_argument("--out-dir", default="docs/quality")
    p.add_argument("--backend", default="auto")
    p.add_argument("--resample-every", type=int, default=0,
                   help="resample dead features every N steps (0 = off)")
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    np.random.seed(0)

    from sparse_coding_amd.data.random_dataset import RandomDatasetGenerator
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.metrics import standard_metrics as sm
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    d = args.d_model
    n_dict = d * args.dict_ratio
    gen = RandomDatasetGenerator(
        activation_dim=d, n_ground_truth_components=args.n_true,
        batch_size=args.batch, feature_num_nonzero=args.nonzero,
        feature_prob_decay=1.0, correlated=False, device=device,
    )
    l1s = np.logspace(-4.5, -3.2, args.n_models)
    models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=device) for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=device, backend=args.backend)
    backend = type(ens._hip_step).__name__ if ens._hip_step else "torch"
    print(f"training {args.n_models} models, backend={backend}")

    resampler = None
    if args.resample_every:
        from sparse_coding_amd.engine.resample import EnsembleResampler

        resampler = EnsembleResampler(ens, n_track=512)

    t0 = time.time()
    for step in range(args.steps):
        batch = gen.send(None)
        losses, aux = ens.step_batch(batch)
        if resampler is not None:
            resampler.observe(batch, aux)
            # no resampling in the final quarter: freshly re-initialized
            # features need training time before they count as alive
            if (step + 1) % args.resample_every == 0 and step < args.steps * 3 // 4:
                n_dead = resampler.resample()
                print(f"step {step}: resampled {n_dead.tolist() if hasattr(n_dead, 'tolist') else n_dead}")
        if step % 500 == 0:
            print(f"step {step}: loss={[round(v, 4) for v in losses['loss'].tolist()]}")
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    train_s = time.time() - t0

    # evaluate on fresh data
    sample = torch.cat([gen.send(None) for _ in range(4)]).cpu()
    feats_cpu = gen.feats.cpu()
    rows = []
    for ld, l1 in zip(ens.to_learned_dicts(), l1s):
        rows.append({
            "l1_alpha": float(l1),
            "fvu": sm.fraction_variance_unexplained(ld, sample).item(),
            "mean_l0": sm.mean_l0(ld, sample).item(),
            "dead_frac": sm.dead_feature_fraction(ld, sample),
            "mmcs_to_truth": sm.mmcs_to_fixed(ld, feats_cpu).item(),
            "representedness": sm.representedness(feats_cpu, ld).mean().item(),
        })
        print(rows[-1])

    os.makedirs(args.out_dir, exist_ok=True)
    report = {
        "config": {
            "d_model": d, "dict_size": n_dict, "n_models": args.n_models,
            "batch": args.batch, "steps": args.steps, "backend": backend,
            "n_ground_truth": args.n_true, "feature_num_nonzero": args.nonzero,
            "dtype": "fp32", "data": "synthetic ground-truth sparse dict",
            "resample_every": args.resample_every,
        },
        "train_seconds": train_s,
        "acts_per_sec": args.batch * args.steps / train_s,
        "models": rows,
    }
    with open(os.path.join(args.out_dir, "quality_report.json"), "w") as f:
        json.dump(report, f, indent=1)

    from plotting.fvu_sparsity_plot import plot_fvu_sparsity

    curves = {"tied_8x": [(r["mean_l0"], r["fvu"], {"l1_alpha": r["l1_alpha"]}) for r in rows]}
    plot_fvu_sparsity(curves, save_path=os.path.join(args.out_dir, "fvu_l0_pareto.png"),
                      title=f"FVU vs L0 (d={d}, dict={n_dict}, {args.steps} steps, fp32)")
    print(f"wrote {args.out_dir}")


if __name__ == "__main__":
    main()
