"""Quality-curve report: FVU-vs-L0 pareto, MMCS-to-ground-truth, dead
fraction for an ensemble config on synthetic ground-truth activations.

The reference publishes quality curves, not throughput; its
training semantics are reproduced exactly (tests/test_hip_numerics.py shows
the fused step tracks the vmap oracle), so this report measures the same
curves this framework produces on known ground truth.

Round-2 additions:
* --protocol anthropic: loss^2-weighted resampling + alive-norm scaling +
  post-resample lr warmup (engine/resample.py), the retention protocol.
* --compare-oracle: trains a second, identically-seeded ensemble on the
  torch/vmap oracle backend and overlays both pareto curves — the
  fused-vs-oracle quality-parity evidence.
* dead-fraction trajectory sampled during training (not just at the end).

Canonical operating point (reference sweep_baselines.py:46-54): tied,
ratio 1, l1 = 8.5e-4:
  python scripts/quality_report.py --dict-ratio 1 --steps 30000 \
      --protocol anthropic --resample-every 1000 --out-dir docs/quality_r02
Flagship 8x config: --dict-ratio 8 (default).
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


def build_ensemble(args, l1s, device, backend):
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(0)
    d = args.d_model
    n_dict = d * args.dict_ratio
    models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=device) for l1 in l1s]
    return FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                              device=device, backend=backend)


def evaluate(ens, l1s, gen, feats_cpu):
    from sparse_coding_amd.metrics import standard_metrics as sm

    sample = torch.cat([gen.send(None) for _ in range(4)]).cpu()
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
    return rows


def train(ens, gen, args, label, log_dead_every=0, l1s=None, feats_cpu=None):
    resampler = None
    if args.resample_every:
        from sparse_coding_amd.engine.resample import EnsembleResampler

        resampler = EnsembleResampler(
            ens, n_track=args.n_track, protocol=args.protocol,
            warmup_steps=args.warmup_steps)

    resample_until = int(args.steps * args.resample_until)
    dead_traj = []
    t0 = time.time()
    for step in range(args.steps):
        batch = gen.send(None)
        losses, aux = ens.step_batch(batch)
        if resampler is not None:
            resampler.observe(batch, aux)
            if (step + 1) % args.resample_every == 0 and step < resample_until:
                n_dead = resampler.resample()
                lst = n_dead.tolist() if hasattr(n_dead, "tolist") else n_dead
                print(f"[{label}] step {step}: resampled {lst}")
        if log_dead_every and (step + 1) % log_dead_every == 0:
            from sparse_coding_amd.metrics import standard_metrics as sm

            sample = gen.send(None).cpu()
            dead = [sm.dead_feature_fraction(ld, sample) for ld in ens.to_learned_dicts()]
            dead_traj.append({"step": step + 1, "dead_frac": dead})
            print(f"[{label}] step {step + 1}: dead_frac={[round(v, 3) for v in dead]}")
        elif step % 2000 == 0:
            print(f"[{label}] step {step}: loss={[round(v, 4) for v in losses['loss'].tolist()]}")
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return time.time() - t0, dead_traj


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=4000)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--dict-ratio", type=int, default=8)
    p.add_argument("--n-true", type=int, default=1024)
    p.add_argument("--nonzero", type=int, default=20)
    p.add_argument("--n-models", type=int, default=8)
    p.add_argument("--l1-lo", type=float, default=-4.5, help="log10 of the l1 grid lower end")
    p.add_argument("--l1-hi", type=float, default=-3.2)
    p.add_argument("--out-dir", default="docs/quality")
    p.add_argument("--backend", default="auto")
    p.add_argument("--resample-every", type=int, default=0,
                   help="resample dead features every N steps (0 = off)")
    p.add_argument("--protocol", default="anthropic", choices=["worst", "anthropic"])
    p.add_argument("--warmup-steps", type=int, default=1000)
    p.add_argument("--n-track", type=int, default=512)
    p.add_argument("--resample-until", type=float, default=0.875,
                   help="stop resampling after this fraction of training")
    p.add_argument("--dead-log-every", type=int, default=0,
                   help="record a dead-fraction trajectory point every N steps")
    p.add_argument("--compare-oracle", action="store_true",
                   help="also train the torch/vmap oracle from the same seed and overlay")
    p.add_argument("--unit-scale", action="store_true",
                   help="rescale synthetic activations to ~unit per-dim variance "
                        "(matches real residual-stream scale, so the reference's "
                        "l1 grid lands in the same FVU/L0 regime)")
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    np.random.seed(0)

    from sparse_coding_amd.data.random_dataset import RandomDatasetGenerator

    d = args.d_model
    n_dict = d * args.dict_ratio
    l1s = np.logspace(args.l1_lo, args.l1_hi, args.n_models)

    class _ScaledGen:
        """Wraps the generator, scaling outputs to ~unit per-dim variance."""

        def __init__(self, gen, scale):
            self.gen = gen
            self.scale = scale
            self.feats = gen.feats

        def send(self, arg):
            return self.gen.send(arg) * self.scale

    def make_gen():
        # generate_rand_feats draws from NUMPY's RNG — seed both so every
        # make_gen() call rebuilds the SAME ground-truth dictionary
        torch.manual_seed(100)
        np.random.seed(100)
        g = RandomDatasetGenerator(
            activation_dim=d, n_ground_truth_components=args.n_true,
            batch_size=args.batch, feature_num_nonzero=args.nonzero,
            feature_prob_decay=1.0, correlated=False, device=device,
        )
        if args.unit_scale:
            scale = float(1.0 / g.send(None).std())
            return _ScaledGen(g, scale)
        return g

    gen = make_gen()
    feats_cpu = gen.feats.cpu()

    ens = build_ensemble(args, l1s, device, args.backend)
    backend = type(ens._hip_step).__name__ if ens._hip_step else "torch"
    print(f"training {args.n_models} models, backend={backend}")
    train_s, dead_traj = train(ens, gen, args, backend,
                               log_dead_every=args.dead_log_every)
    rows = evaluate(ens, l1s, make_gen(), feats_cpu)
    for r in rows:
        print(r)

    oracle_rows = None
    if args.compare_oracle:
        gen2 = make_gen()
        ens2 = build_ensemble(args, l1s, device, "torch")
        oracle_s, _ = train(ens2, gen2, args, "oracle")
        oracle_rows = evaluate(ens2, l1s, make_gen(), feats_cpu)
        print("oracle:")
        for r in oracle_rows:
            print(r)

    os.makedirs(args.out_dir, exist_ok=True)
    report = {
        "config": {
            "d_model": d, "dict_size": n_dict, "n_models": args.n_models,
            "batch": args.batch, "steps": args.steps, "backend": backend,
            "n_ground_truth": args.n_true, "feature_num_nonzero": args.nonzero,
            "dtype": "fp32", "data": "synthetic ground-truth sparse dict",
            "resample_every": args.resample_every, "protocol": args.protocol,
            "warmup_steps": args.warmup_steps, "n_track": args.n_track,
            "resample_until": args.resample_until,
            "l1_grid": [float(v) for v in l1s],
        },
        "train_seconds": train_s,
        "acts_per_sec": args.batch * args.steps / train_s,
        "models": rows,
        "dead_trajectory": dead_traj,
    }
    if oracle_rows is not None:
        report["oracle_models"] = oracle_rows
    with open(os.path.join(args.out_dir, "quality_report.json"), "w") as f:
        json.dump(report, f, indent=1)

    from plotting.fvu_sparsity_plot import plot_fvu_sparsity

    curves = {f"tied_{args.dict_ratio}x_{backend}":
              [(r["mean_l0"], r["fvu"], {"l1_alpha": r["l1_alpha"]}) for r in rows]}
    if oracle_rows is not None:
        curves["tied_oracle"] = [(r["mean_l0"], r["fvu"], {"l1_alpha": r["l1_alpha"]})
                                 for r in oracle_rows]
    plot_fvu_sparsity(curves, save_path=os.path.join(args.out_dir, "fvu_l0_pareto.png"),
                      title=f"FVU vs L0 (d={d}, dict={n_dict}, {args.steps} steps, fp32)")
    print(f"wrote {args.out_dir}")


if __name__ == "__main__":
    main()
