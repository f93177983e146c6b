"""torchrun-able sweep: ensemble-per-rank over an RCCL chunk broadcast.

The single-process ``sweep()`` driver parents its workers (per-chunk spawn
or the persistent pool).  This variant is the fully RCCL-native P1: launch
one rank per GPU with torchrun; rank 0 generates/loads each chunk and
broadcasts it over xGMI; every rank trains ITS OWN ensemble (a slice of the
hyperparameter grid) against its HBM-resident copy; checkpoints gather to
rank 0 in the reference ``_{i}/learned_dicts.pt`` layout.

    torchrun --standalone --nproc-per-node 8 -m sparse_coding_amd.sweep.sharded_sweep

Library use: ``sharded_sweep(init_func_for_rank, cfg)`` where
``init_func_for_rank(cfg, rank, world) -> (ensemble, args, name)``.
"""

from __future__ import annotations

import os
from typing import Callable

import numpy as np
import torch
import torch.distributed as dist
import yaml

from sparse_coding_amd.parallel.chunk_feed import ShardedEnsembleRunner
from sparse_coding_amd.parallel.dp import init_distributed
from sparse_coding_amd.sweep.big_sweep import (
    ensemble_train_loop,
    init_synthetic_dataset,
    unstacked_to_learned_dicts,
)
from sparse_coding_amd.utils.logging import RunLogger


def sharded_sweep(init_func_for_rank: Callable, cfg):
    """Chunk-epoch loop, one ensemble per rank, chunks broadcast from rank 0."""
    rank, local_rank, world = init_distributed()
    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
    cfg.device = device
    torch.manual_seed(0)
    np.random.seed(0)

    os.makedirs(cfg.dataset_folder, exist_ok=True)
    os.makedirs(cfg.output_folder, exist_ok=True)
    if rank == 0:
        cfg.logger = RunLogger(cfg.output_folder, name="sharded_sweep",
                               use_wandb=getattr(cfg, "use_wandb", False))
        if cfg.use_synthetic_dataset:
            init_synthetic_dataset(cfg)
    else:
        cfg.logger = None
    if dist.is_initialized():
        dist.barrier()

    ensemble, args, name = init_func_for_rank(cfg, rank, world)
    runner = ShardedEnsembleRunner(ensemble, cfg, args, name,
                                   ensemble_train_loop, device)

    n_chunks = len([f for f in os.listdir(cfg.dataset_folder) if f.endswith(".pt")])
    chunk_order = np.random.permutation(n_chunks)
    if getattr(cfg, "n_repetitions", None) is not None:
        chunk_order = np.tile(chunk_order, cfg.n_repetitions)

    learned_dicts = None
    for i, chunk_idx in enumerate(chunk_order):
        chunk = None
        if rank == 0:
            chunk = torch.load(os.path.join(cfg.dataset_folder, f"{chunk_idx}.pt"),
                               weights_only=False).to(torch.float32)
        runner.run_chunk(chunk)

        if i == len(chunk_order) - 1 or (i + 1) in [2**j for j in range(3, 10)]:
            learned_dicts = runner.gather_learned_dicts(
                cfg.ensemble_hyperparams, cfg.buffer_hyperparams)
            if rank == 0:
                iter_folder = os.path.join(cfg.output_folder, f"_{i}")
                os.makedirs(iter_folder, exist_ok=True)
                torch.save(learned_dicts, os.path.join(iter_folder, "learned_dicts.pt"))
                with open(os.path.join(iter_folder, "config.yaml"), "w") as f:
                    cfg_dict = cfg.as_dict() if hasattr(cfg, "as_dict") else dict(cfg)
                    cfg_dict.pop("logger", None)
                    yaml.dump({k: v for k, v in cfg_dict.items()
                               if isinstance(v, (int, float, str, bool, list, type(None)))}, f)

    if rank == 0 and cfg.logger is not None:
        cfg.logger.close()
    return learned_dicts


def _demo_init_for_rank(cfg, rank: int, world: int):
    """Default grid slice: rank r trains an 8-way L1 ensemble at dict ratio
    2^r (so 8 ranks cover ratios 1..128)."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(1000 + rank)
    d = cfg.activation_width
    ratio = 2 ** rank
    n_dict = int(d * ratio)
    l1s = np.logspace(-4, -2, 8)
    models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=cfg.device) for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": cfg.lr}, device=cfg.device)
    args = {"batch_size": cfg.batch_size, "device": cfg.device, "dict_size": n_dict}
    return ens, args, f"ratio_{ratio}"


def main():
    """Synthetic sharded-sweep demo:

    torchrun --standalone --nproc-per-node N -m sparse_coding_amd.sweep.sharded_sweep
    """
    import argparse

    from sparse_coding_amd.config import SyntheticEnsembleArgs

    p = argparse.ArgumentParser()
    p.add_argument("--dataset-folder", default="activation_data_sharded")
    p.add_argument("--output-folder", default="output_sharded")
    p.add_argument("--n-chunks", type=int, default=4)
    p.add_argument("--chunk-gb", type=float, default=0.5)
    p.add_argument("--activation-width", type=int, default=512)
    args = p.parse_args()

    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = args.activation_width
    cfg.n_ground_truth_components = 2 * args.activation_width
    cfg.gen_batch_size = 4096
    cfg.feature_num_nonzero = 32
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = args.chunk_gb
    cfg.n_chunks = args.n_chunks
    cfg.batch_size = 2048
    cfg.dataset_folder = args.dataset_folder
    cfg.output_folder = args.output_folder
    cfg.use_wandb = False
    cfg.ensemble_hyperparams = ["dict_size"]
    cfg.buffer_hyperparams = ["l1_alpha"]

    dicts = sharded_sweep(_demo_init_for_rank, cfg)
    if dicts is not None:
        print(f"rank 0: gathered {len(dicts)} learned dicts")


if __name__ == "__main__":
    main()
