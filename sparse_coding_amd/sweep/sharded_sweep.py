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
        cfg._chunk_i = i  # resample schedule (big_sweep.ensemble_train_loop)
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


def lm_chunk_generator(cfg, device):
    """Infinite stream of [chunk_activations, d] fp32 activation chunks from
    the host LM, entirely in the generator rank's HBM (no disk round-trip).
    The LM runs bf16 on GPU (ROADMAP item 7: ~2x generation throughput;
    activations are cast fp32 for training as the reference trains fp32)."""
    from sparse_coding_amd.data.activation_dataset import (
        capture_activation_hook,
        load_model,
        synthetic_token_batches,
    )

    dtype = torch.bfloat16 if str(device).startswith("cuda") else None
    model = load_model(cfg.model_name, device=device, dtype=dtype)
    vocab = model.config.vocab_size
    bsz = getattr(cfg, "model_batch_size", 4)
    max_len = getattr(cfg, "max_length", 256)
    per_batch = bsz * max_len
    chunk_acts = getattr(cfg, "chunk_activations", 1 << 18)
    batches_per_chunk = max(1, (chunk_acts + per_batch - 1) // per_batch)
    layer, layer_loc = cfg.layer, getattr(cfg, "layer_loc", "residual")

    while True:
        parts = []
        for toks in synthetic_token_batches(vocab, bsz, max_len, batches_per_chunk):
            store = []
            with torch.no_grad(), capture_activation_hook(model, layer, layer_loc, store):
                model(input_ids=toks.to(device))
            parts.append(store[0].float())
        yield torch.cat(parts)[:chunk_acts].contiguous()


def generator_trainer_sweep(init_func_for_rank: Callable, cfg,
                            make_generator: Callable = None):
    """Config-4/5 data plane: RANK 0 IS A DEDICATED
    GENERATOR — it runs the host LM and streams each activation chunk
    straight from its HBM to every trainer rank over one RCCL broadcast
    (xGMI p2p; the chunk never touches disk or host RAM) — while ranks
    >= 1 train their own ensembles on the previous chunk.  The pipeline
    overlap is the natural one: rank 0 generates chunk i+1 while the
    trainers are still stepping through chunk i; the broadcast is the only
    sync point.

    make_generator(cfg, device) -> iterator of [N, d] fp32 chunks
    (default: lm_chunk_generator — the host-LM stream).
    Returns the gathered learned_dicts on rank 0, None elsewhere.
    """
    from torch.utils.data import BatchSampler, RandomSampler

    from sparse_coding_amd.parallel.chunk_feed import BroadcastChunkFeeder

    rank, local_rank, world = init_distributed()
    if world < 2:
        raise RuntimeError("generator_trainer_sweep needs world_size >= 2 "
                           "(one generator rank + trainers)")
    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
    cfg.device = device
    torch.manual_seed(0)
    np.random.seed(0)
    os.makedirs(cfg.output_folder, exist_ok=True)

    feeder = BroadcastChunkFeeder(device, src=0)
    logger = RunLogger(cfg.output_folder, name="generator_trainer_sweep",
                       use_wandb=getattr(cfg, "use_wandb", False)) if rank == 0 else None

    ensemble, args, name = (None, None, None)
    if rank == 0:
        gen = (make_generator or lm_chunk_generator)(cfg, device)
    else:
        ensemble, args, name = init_func_for_rank(cfg, rank, world)

    class _Counter:  # mp.Value-compatible progress stub
        value = 0

    n_steps = int(getattr(cfg, "n_chunks", 4)) * int(getattr(cfg, "n_repetitions", 1) or 1)
    learned_dicts = None
    for i in range(n_steps):
        cfg._chunk_i = i  # resample schedule (big_sweep.ensemble_train_loop)
        if rank == 0:
            chunk = next(gen)
            feeder.feed(chunk)
            if logger is not None:
                logger.log({"chunk": i, "chunk_rows": int(chunk.shape[0])})
        else:
            local = feeder.feed(None)
            sampler = BatchSampler(RandomSampler(range(local.shape[0])),
                                   batch_size=args.get("batch_size", 256), drop_last=False)
            ensemble_train_loop(ensemble, cfg, args, name, sampler, local, _Counter())

        if i == n_steps - 1 or (i + 1) in [2**j for j in range(3, 10)]:
            local_dicts = [] if rank == 0 else unstacked_to_learned_dicts(
                ensemble, args, cfg.ensemble_hyperparams, cfg.buffer_hyperparams)
            gathered = [None] * world if rank == 0 else None
            dist.gather_object(local_dicts, gathered, dst=0)
            if rank == 0:
                learned_dicts = [ld for part in gathered for ld in part]
                iter_folder = os.path.join(cfg.output_folder, f"_{i}")
                os.makedirs(iter_folder, exist_ok=True)
                torch.save(learned_dicts, os.path.join(iter_folder, "learned_dicts.pt"))

    if logger is not None:
        logger.close()
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
