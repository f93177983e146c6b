"""torchrun-able sweep: ensemble-per-rank over an RCCL chunk broadcast.

The single-process ``sweep()`` driver parents its workers (per-chunk spawn
or the persistent pool).  This variant is the fully RCCL-native P1: launch
one rank per GPU with torchrun; rank 0 generates/loads each chunk and
broadcasts it over xGMI; every rank trains ITS OWN ensemble (a slice of the
hyperparameter grid) against its HBM-resident copy; checkpoints gather to
rank 0 in the reference ``_{i}/learned_dicts.pt`` layout.

    torchrun --standalone --nproc-per-node 8 -m sparse_coding_amd.sweep.sharded_sweep_demo

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
