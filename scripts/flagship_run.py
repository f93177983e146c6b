"""A real end-to-end training run of the sweep() driver at flagship scale.

Synthetic ground-truth activations (no network), flagship ensemble shape
(d=512, 8x dict, 8-way L1 grid), the actual dispatch/checkpoint/resume
machinery, and a final quality report against the generating dictionary.
Artifacts (metrics JSONL, config.yaml, quality summary) are small and meant
to be committed; the activation chunks and checkpoints stay in the run dir.

GPU box:  python scripts/flagship_run.py --out-dir gpurun_out/flagship_run
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


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out-dir", default="gpurun_out/flagship_run")
    p.add_argument("--n-chunks", type=int, default=4)
    p.add_argument("--chunk-gb", type=float, default=0.5)
    p.add_argument("--n-repetitions", type=int, default=2)
    p.add_argument("--persistent", action="store_true",
                   help="persistent dispatch workers (one spawn per ensemble)")
    args = p.parse_args()

    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.metrics import standard_metrics as sm
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = 512
    cfg.n_ground_truth_components = 1024
    cfg.gen_batch_size = 4096
    cfg.feature_num_nonzero = 32
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = args.chunk_gb
    cfg.n_chunks = args.n_chunks
    cfg.n_repetitions = args.n_repetitions
    cfg.batch_size = 2048
    cfg.device = device
    cfg.dataset_folder = os.path.join(args.out_dir, "chunks")
    cfg.output_folder = os.path.join(args.out_dir, "out")
    cfg.use_wandb = False
    cfg.wandb_images = False
    cfg.persistent_workers = args.persistent

    l1s = np.logspace(-4.5, -3.2, 8)

    def init_func(c):
        return make_grid_ensembles(c, FunctionalTiedSAE, l1s, [8.0], devices=[device])

    t0 = time.time()
    dicts = big_sweep.sweep(init_func, cfg)
    wall = time.time() - t0

    # quality vs the generating dictionary
    gen = torch.load(os.path.join(cfg.output_folder, "generator.pt"), weights_only=False)
    feats = gen.sparse_component_dict.cpu().float()
    sample = torch.load(os.path.join(cfg.dataset_folder, "0.pt"), weights_only=False).float()[:65536]
    rows = []
    for ld, hp in dicts:
        rows.append({
            "l1_alpha": hp["l1_alpha"],
            "fvu": sm.fraction_variance_unexplained(ld, sample).item(),
            "mean_l0": sm.mean_l0(ld, sample).item(),
            "dead_frac": sm.dead_feature_fraction(ld, sample),
            "representedness": sm.representedness(feats, ld).mean().item(),
        })
        print(rows[-1], flush=True)

    n_acts = cfg.n_chunks * args.n_repetitions * sample.shape[0] * 0  # computed below
    total_acts = sum(torch.load(os.path.join(cfg.dataset_folder, f"{i}.pt"),
                                weights_only=False).shape[0]
                     for i in range(cfg.n_chunks)) * args.n_repetitions
    summary = {
        "persistent_workers": args.persistent,
        "wall_seconds": round(wall, 1),
        "total_activations_seen": int(total_acts),
        "end_to_end_acts_per_sec": round(total_acts / wall),
        "checkpoints": sorted(os.listdir(cfg.output_folder)),
        "resume_state": os.path.exists(os.path.join(cfg.output_folder, "resume_state.pt")),
        "models": rows,
    }
    with open(os.path.join(args.out_dir, "run_summary.json"), "w") as f:
        json.dump(summary, f, indent=1)
    print(json.dumps({k: v for k, v in summary.items() if k != "models"}), flush=True)


if __name__ == "__main__":
    main()
