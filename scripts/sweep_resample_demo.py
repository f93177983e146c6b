"""Long sweep()-driver run with in-sweep anthropic resampling (GPU evidence
for cfg.resample_every_chunks at flagship scale: dispatched children, fused
step, k_resample + lr_mult, checkpoints).

GPU box: python scripts/sweep_resample_demo.py --n-repetitions 8
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
    p.add_argument("--n-chunks", type=int, default=4)
    p.add_argument("--n-repetitions", type=int, default=8)
    p.add_argument("--chunk-gb", type=float, default=0.25)
    p.add_argument("--resample-every-chunks", type=int, default=4)
    p.add_argument("--data-dir", default="/tmp/sweeprs_data")
    p.add_argument("--out-dir", default="/tmp/sweeprs_out")
    p.add_argument("--summary", default="gpurun_out/sweeprs_summary.json")
    args = p.parse_args()

    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.metrics import standard_metrics as sm
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = 512
    cfg.n_ground_truth_components = 8192
    cfg.gen_batch_size = 4096
    cfg.feature_num_nonzero = 40
    # flat inclusion probability: the default 0.99 decay over 8192
    # components renormalizes early-feature probs past 1 (degenerate data)
    cfg.feature_prob_decay = 1.0
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = args.chunk_gb
    cfg.n_chunks = args.n_chunks
    cfg.n_repetitions = args.n_repetitions
    cfg.batch_size = 2048
    cfg.device = "cuda:0" if torch.cuda.is_available() else "cpu"
    cfg.dataset_folder = args.data_dir
    cfg.output_folder = args.out_dir
    cfg.use_wandb = False
    cfg.wandb_images = False
    cfg.resample_every_chunks = args.resample_every_chunks
    cfg.resample_n_track = 512

    def init_func(c):
        return make_grid_ensembles(c, FunctionalTiedSAE,
                                   list(np.logspace(-4, -2.3, 8)), [8.0],
                                   devices=[cfg.device])

    t0 = time.time()
    dicts = big_sweep.sweep(init_func, cfg)
    dt = time.time() - t0

    chunk0 = torch.load(os.path.join(args.data_dir, "0.pt"), weights_only=False).float()
    sample = chunk0[:8192]
    rows = []
    for ld, hp in dicts:
        rows.append({
            "l1": round(hp["l1_alpha"], 6),
            "fvu": round(sm.fraction_variance_unexplained(ld, sample).item(), 4),
            "l0": round(sm.mean_l0(ld, sample).item(), 1),
            "dead": round(sm.dead_feature_fraction(ld, sample), 4),
        })
    n_resampled = 0
    metrics_path = os.path.join(args.out_dir, "metrics.jsonl")
    if os.path.exists(metrics_path):
        for line in open(metrics_path):
            rec = json.loads(line)
            n_resampled += sum(v for k, v in rec.items() if k.endswith("_resampled"))
    summary = {"wall_s": round(dt, 1), "n_resampled_total": int(n_resampled), "rows": rows}
    print(json.dumps(summary))
    os.makedirs(os.path.dirname(args.summary) or ".", exist_ok=True)
    with open(args.summary, "w") as f:
        json.dump(summary, f, indent=1)


if __name__ == "__main__":
    main()
