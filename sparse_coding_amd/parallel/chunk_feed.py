"""Chunk broadcast over RCCL/xGMI + ensemble-per-rank sharded training.

The reference's primary multi-GPU mode pins one
activation chunk in host shared memory and lets 8 worker processes each
pull the batches they need over PCIe.  The MI355X-native replacement
(SURVEY.md's own design note for P1): ONE rank reads/generates the chunk,
stages it into its HBM, and RCCL-broadcasts it to every rank over xGMI
(7 p2p links, ~10x the host link) — after which each rank trains its OWN
ensemble against its GPU-resident copy (ensemble sharding, zero gradient
traffic; contrast parallel/dp.py which replicates ONE ensemble and
all-reduces gradients).

Launch:  torchrun --standalone --nproc-per-node 8 ... and drive
`ShardedEnsembleRunner` (see tests/test_parallel.py for the gloo-on-CPU
plumbing test; the collective path is backend-agnostic).
"""

from __future__ import annotations

from typing import Callable, List, Optional, Tuple

import torch
import torch.distributed as dist


class BroadcastChunkFeeder:
    """Rank `src` supplies each chunk; every rank receives a device-resident
    copy via one broadcast."""

    def __init__(self, device, src: int = 0, group=None):
        self.device = torch.device(device)
        self.src = src
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0

    def feed(self, chunk: Optional[torch.Tensor]) -> torch.Tensor:
        """On rank `src`, pass the chunk (any device); on other ranks pass
        None.  Returns the chunk on this rank's device."""
        if not dist.is_initialized() or dist.get_world_size(self.group) == 1:
            assert chunk is not None
            return chunk.to(self.device)

        if self.rank == self.src:
            assert chunk is not None
            if chunk.dim() != 2:
                # the receiver-side shape buffer is fixed at 2 entries; a
                # higher-rank chunk would corrupt the shape exchange — fail
                # loudly on the source instead
                raise ValueError(f"BroadcastChunkFeeder.feed wants [N, d] chunks, got shape {tuple(chunk.shape)}")
            chunk = chunk.to(self.device, torch.float32)
            shape = torch.tensor(list(chunk.shape), device=self.device, dtype=torch.long)
        else:
            shape = torch.zeros(2, device=self.device, dtype=torch.long)
        dist.broadcast(shape, src=self.src, group=self.group)
        if self.rank != self.src:
            chunk = torch.empty(int(shape[0]), int(shape[1]), device=self.device)
        dist.broadcast(chunk, src=self.src, group=self.group)
        return chunk


class ShardedEnsembleRunner:
    """P1, RCCL-native: rank r owns ensemble r (its own hyperparameter
    slice); chunks are broadcast once and every rank trains locally with no
    further communication."""

    def __init__(self, ensemble, cfg, args, name, job: Callable, device, src: int = 0):
        self.ensemble = ensemble
        self.cfg = cfg
        self.args = args
        self.name = name
        self.job = job
        self.feeder = BroadcastChunkFeeder(device, src=src)

    def run_chunk(self, chunk: Optional[torch.Tensor]) -> None:
        from torch.utils.data import BatchSampler, RandomSampler

        local = self.feeder.feed(chunk)
        batch_size = self.args.get("batch_size", getattr(self.cfg, "batch_size", 256))
        sampler = BatchSampler(RandomSampler(range(local.shape[0])),
                               batch_size=batch_size, drop_last=False)

        class _Counter:  # mp.Value-compatible progress stub
            value = 0

        self.job(self.ensemble, self.cfg, self.args, self.name, sampler, local, _Counter())

    def gather_learned_dicts(self, ensemble_hyperparams, buffer_hyperparams,
                             dst: int = 0) -> Optional[List[Tuple]]:
        """Collect every rank's unstacked LearnedDicts on `dst` (object
        gather; dictionaries are small next to activation chunks)."""
        from sparse_coding_amd.sweep.big_sweep import unstacked_to_learned_dicts

        local = unstacked_to_learned_dicts(self.ensemble, self.args,
                                           ensemble_hyperparams, buffer_hyperparams)
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return local
        gathered: Optional[List] = [None] * dist.get_world_size() if dist.get_rank() == dst else None
        dist.gather_object(local, gathered, dst=dst)
        if gathered is None:
            return None
        return [ld for part in gathered for ld in part]
