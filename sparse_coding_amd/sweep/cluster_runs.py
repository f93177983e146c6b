"""Per-GPU process dispatch: one OS process per ensemble, one shared chunk.

Parity with reference ``cluster_runs.py`` (dispatch_job_on_chunk :100-157,
job_wrapper :15-36, dispatch_lite :50, collect_lite :89): the chunk is pinned
and moved to shared memory once, each ensemble's stacked tensors are shared,
and each child attaches via ``FunctionalEnsemble.from_state`` and runs the
job with a BatchSampler over the shared chunk (a DataLoader would copy it).

On an MI355X node each ensemble is pinned to its own GPU; the preferred
multi-GPU path for a SINGLE ensemble is the RCCL data-parallel trainer in
``sparse_coding_amd.parallel`` — this module exists for reference-API parity
and for many-ensemble sweeps.
"""

from __future__ import annotations

import time
from typing import Any, Callable, List, Tuple

import torch
import torch.multiprocessing as mp
from torch.utils.data import BatchSampler, RandomSampler

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble

# Always fork-free: a fork while the parent's OpenMP/BLAS pools hold locks
# deadlocks the child inside its first torch op (observed on CPU runs where
# a vmap-heavy step preceded dispatch).  The reference forces spawn globally
# (big_sweep.py:302); here the context is explicit so library users are safe
# no matter what the global start method is.
_ctx = mp.get_context("spawn")


def job_wrapper(job, ensemble_state_dict, cfg, args, tag, dataset, done_flag, progress_counter):
    torch.manual_seed(0)
    ensemble = FunctionalEnsemble.from_state(ensemble_state_dict)
    batch_size = args.get("batch_size", getattr(cfg, "batch_size", 256))
    sampler = BatchSampler(
        RandomSampler(range(dataset.shape[0])),
        batch_size=batch_size,
        drop_last=False,
    )
    job(ensemble, cfg, args, tag, sampler, dataset, progress_counter)
    done_flag.value = 1


def dispatch_job_on_chunk(ensembles: List[Tuple[Any, dict, str]], cfg, chunk: torch.Tensor, job: Callable, poll_s: float = 0.1):
    """Run `job` on every (ensemble, args, name) against one shared chunk.

    NOTE the reference "pins" the chunk with a discarded `chunk.pin_memory()`
    (cluster_runs.py:101 — pin_memory returns a copy, so it is a no-op) and
    then streams every batch over unpinned H2D.  Here the chunk goes to
    shared CPU memory for the spawn children, and each child stages it into
    its GPU's HBM ONCE (big_sweep.ensemble_train_loop) — with 288 GB per GPU,
    whole-chunk residency is the right call.
    """
    chunk.share_memory_()

    processes = []
    done_flags = []
    progress_counters = []
    n_batches = []

    for ensemble, args, name in ensembles:
        ensemble.to_shared_memory()
        done = _ctx.Value("i", 0)
        counter = _ctx.Value("i", 0)
        batch_size = args.get("batch_size", getattr(cfg, "batch_size", 256))
        n_batches.append((chunk.shape[0] + batch_size - 1) // batch_size)
        proc = _ctx.Process(
            target=job_wrapper,
            args=(job, ensemble.state_dict(), cfg, args, name, chunk, done, counter),
        )
        proc.start()
        processes.append(proc)
        done_flags.append(done)
        progress_counters.append(counter)

    import sys as _sys

    # carriage-return progress only on a real terminal; plain logs get one
    # line per chunk instead of thousands of \r frames
    show_progress = bool(getattr(cfg, "show_progress", True)) and _sys.stdout.isatty()
    total = sum(n_batches)
    while not all(f.value == 1 for f in done_flags):
        # a crashed child must not hang the sweep (reference busy-polls
        # forever, cluster_runs.py:145-154)
        for proc, flag in zip(processes, done_flags):
            if not proc.is_alive() and flag.value != 1:
                for p in processes:
                    if p.is_alive():
                        p.terminate()
                raise RuntimeError(f"ensemble worker pid={proc.pid} died (exitcode={proc.exitcode})")
        if show_progress:
            done_batches = sum(c.value for c in progress_counters)
            print(f"\r[dispatch] {done_batches}/{total} batches", end="", flush=True)
        time.sleep(poll_s)
    if show_progress:
        print()

    for proc in processes:
        proc.join()


def dispatch_lite(cfg, chunk: torch.Tensor, ensemble, name: str, job: Callable):
    """Single-ensemble async dispatch (reference cluster_runs.py:50-86)."""
    chunk.share_memory_()
    ensemble.to_shared_memory()
    done = _ctx.Value("i", 0)
    counter = _ctx.Value("i", 0)
    args = {"batch_size": getattr(cfg, "batch_size", 256), "device": ensemble.device}
    proc = _ctx.Process(
        target=job_wrapper,
        args=(job, ensemble.state_dict(), cfg, args, name, chunk, done, counter),
    )
    proc.start()
    return proc, done, counter


def collect_lite(handles):
    """Join the processes started by dispatch_lite (reference :89-97)."""
    for proc, done, _ in handles:
        proc.join()
        if proc.exitcode != 0:
            raise RuntimeError(f"worker pid={proc.pid} exited with {proc.exitcode}")
