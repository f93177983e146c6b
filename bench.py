"""Flagship benchmark: SAE-ensemble training throughput on MI355X.

Measures the BASELINE.json headline metric — activations/sec through the SAE
ensemble on the Pythia-70m layer-2 residual config (d_model=512, 8x dict =
4096 features, 8-way L1 ensemble) — on synthetic activations with
random-init weights (no network in this environment; BASELINE.md: the
reference publishes no throughput number, so vs_baseline is null).

Single GPU:   python bench.py --steps 50 --warmup 10
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...
Weak scaling: each rank trains on its own batch of `--batch` activations
(global batch = N x batch); gradients are averaged over RCCL/xGMI each step.

Timing contract: W untimed warmup steps; barrier + torch.cuda.synchronize on
both sides of exactly K timed steps; elapsed is MAX over ranks; rank 0 prints
ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    # defaults give a multi-second timed region (~6 s at the flagship
    # ~3 ms/step) so driver-observed wall time corroborates the printed value
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--batch", type=int, default=2048, help="per-GPU batch of activation vectors")
    p.add_argument("--d-model", type=int, default=512, help="activation width (Pythia-70m residual)")
    p.add_argument("--dict-ratio", type=int, default=8, help="dict size multiple")
    p.add_argument("--n-models", type=int, default=8, help="L1-ensemble size")
    p.add_argument("--tied", action="store_true", default=True)
    p.add_argument("--untied", dest="tied", action="store_false")
    # --family is the torchrun-safe alias: torchrun's own parser abbrev-
    # matches "--sig" to --signals-to-handle and steals it from script args
    p.add_argument("--sig", "--family", dest="sig",
                   choices=["sae", "topk", "thresholding"], default="sae",
                   help="model family: sae (tied/untied per --tied; the BASELINE default), "
                        "topk (k=32 + dead-neuron resampling, config 5), thresholding")
    p.add_argument("--backend", choices=["auto", "hip", "torch"], default="auto")
    p.add_argument("--dp-mode", choices=["allreduce", "rs_ag"], default="allreduce",
                   help="multi-GPU gradient exchange: chunked overlapped all-reduce, or "
                        "reduce-scatter + sharded Adam + all-gather (ZeRO-style)")
    p.add_argument("--dtype", choices=["fp32"], default="fp32",
                   help="compute dtype; fp32 matches the reference (fp32 params, BASELINE.md)")
    p.add_argument("--profile-tag", default="", help="extra tag echoed in the JSON config")
    return p.parse_args()


def main():
    args = parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world_size > 1

    use_cuda = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)

    if distributed:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl" if use_cuda else "gloo", rank=rank, world_size=world_size)

    torch.manual_seed(1234 + rank)
    np.random.seed(1234 + rank)

    from sparse_coding_amd.data.random_dataset import RandomDatasetGenerator
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalSAE, FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    d = args.d_model
    n_dict = d * args.dict_ratio
    M = args.n_models
    B = args.batch

    # synthetic activations of the flagship shape: sparse ground-truth dict data
    gen = RandomDatasetGenerator(
        activation_dim=d,
        n_ground_truth_components=n_dict,
        batch_size=B,
        feature_num_nonzero=32,
        feature_prob_decay=0.997,
        correlated=False,
        device=device,
    )
    n_pool = 8
    data_pool = [gen.send(None).contiguous() for _ in range(n_pool)]

    l1_values = np.logspace(-4, -2, M)
    # build on the target device directly (288 GB HBM: everything stays resident)
    torch.manual_seed(1234)  # identical replicas on every rank
    no_stacking = False
    resampler = None
    if args.sig == "topk":
        from sparse_coding_amd.models.topk import TopKEncoder

        sig = TopKEncoder
        models = [TopKEncoder.init(d, n_dict, 32) for _ in range(M)]
        no_stacking = not use_cuda  # vmap cannot stack data-dependent topk
    elif args.sig == "thresholding":
        from sparse_coding_amd.models.sae_signatures import FunctionalThresholdingSAE

        sig = FunctionalThresholdingSAE
        models = [sig.init(d, n_dict, float(l1), device=device) for l1 in l1_values]
    else:
        sig = FunctionalTiedSAE if args.tied else FunctionalSAE
        models = [sig.init(d, n_dict, float(l1), device=device) for l1 in l1_values]
    ensemble = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, device=device,
                                  backend=args.backend, no_stacking=no_stacking)
    trainer = DataParallelEnsembleTrainer(ensemble, dp_mode=args.dp_mode)
    if args.sig == "topk":
        from sparse_coding_amd.engine.resample import EnsembleResampler

        resampler = EnsembleResampler(ensemble, n_track=256)

    def barrier_sync():
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        losses, aux = trainer.step(data_pool[i % n_pool])
        if resampler is not None:
            resampler.observe(data_pool[i % n_pool], aux)

    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        losses, aux = trainer.step(data_pool[i % n_pool])
        if resampler is not None:
            # config 5 names resampling: tracking runs every step, the
            # (rare) resample itself once mid-run
            resampler.observe(data_pool[i % n_pool], aux)
            if i == args.steps // 2:
                trainer.resample(resampler)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], device=device if use_cuda else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    acts_per_sec = world_size * B * args.steps / elapsed

    if rank == 0:
        result = {
            "metric": "activations/sec through SAE ensemble (Pythia-70m resid, 8x dict)",
            "value": acts_per_sec,
            "unit": "activations/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "pythia-70m-resid-l2",
                "sig": args.sig,
                "d_model": d,
                "dict_size": n_dict,
                "n_models": M,
                "tied": bool(args.tied),
                "global_batch": world_size * B,
                "seq_len": None,
                "parallelism": f"dp{world_size}" + ("-rsag" if args.dp_mode == "rs_ag" else ""),
                "backend": ensemble._hip_step.__class__.__name__ if ensemble._hip_step else "torch-eager",
                "optimizer": "adam",
                "tag": args.profile_tag,
            },
        }
        print(json.dumps(result))

    if distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
