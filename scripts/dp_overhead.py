"""World-1 cost of the DP choreography (VERDICT weak #2 evidence): the
force_dp_path step runs the exact event/stream/collective call sequence the
8-GPU run uses (single-member communicator, real RCCL calls); comparing
against the plain fused step isolates the machinery's overhead from any
communication time.  GPU box: python scripts/dp_overhead.py"""

from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
import torch.distributed as dist


def timed(fn, steps, warmup):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3


def main():
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29621")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        d, n, M, B = 512, 4096, 8, 2048
        x = torch.randn(B, d, device="cuda:0")
        rows = {}
        for mode in ("plain", "allreduce", "rs_ag"):
            torch.manual_seed(0)
            models = [FunctionalTiedSAE.init(d, n, float(l1), device="cuda:0")
                      for l1 in np.logspace(-4, -2, M)]
            ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                                     device="cuda:0", backend="hip")
            if mode == "plain":
                fn = lambda: ens.step_batch(x)
            else:
                tr = DataParallelEnsembleTrainer(ens, force_dp_path=True, dp_mode=mode)
                fn = lambda: tr.step(x)
            rows[mode] = round(timed(fn, 300, 20), 4)
            del ens
            torch.cuda.empty_cache()
        rows["allreduce_overhead_us"] = round((rows["allreduce"] - rows["plain"]) * 1e3, 1)
        rows["rs_ag_overhead_us"] = round((rows["rs_ag"] - rows["plain"]) * 1e3, 1)
        print(json.dumps(rows))
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
