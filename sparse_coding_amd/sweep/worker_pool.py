"""Persistent dispatch workers: spawn once, stream chunks through queues.

The reference forks one process per ensemble PER CHUNK
(cluster_runs.py:100-157): a 60-chunk sweep over 8 ensembles pays ~480
process spawns, each a fresh CUDA context + module import (~3-5 s) — the
dominant cost of small sweeps (docs/runs/flagship_synthetic).  Here each
ensemble gets ONE long-lived spawn worker:

  * the ensemble's stacked tensors are moved to shared memory once and the
    worker attaches once — parameter updates are visible to the parent
    continuously, exactly as in the per-chunk design;
  * chunks travel by shared-memory handle through an mp.Queue;
  * the parent detects dead workers (raises instead of hanging) and tears
    the pool down with sentinels (context-manager or close()).

`sweep(cfg)` uses the pool when ``cfg.persistent_workers = True``; results
are bit-identical to the per-chunk dispatcher (the train loop reseeds per
chunk) — tests/test_parallel.py::test_persistent_pool_matches_dispatch.
"""

from __future__ import annotations

import queue as pyqueue
from typing import Any, Callable, List, Tuple

import torch
import torch.multiprocessing as mp
from torch.utils.data import BatchSampler, RandomSampler

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble

_ctx = mp.get_context("spawn")


def _worker_main(job, ensemble_state, cfg, args, name, task_q, done_q, progress_counter):
    torch.manual_seed(0)
    ensemble = FunctionalEnsemble.from_state(ensemble_state)
    batch_size = args.get("batch_size", getattr(cfg, "batch_size", 256))
    while True:
        task = task_q.get()
        if task is None:
            done_q.put(("exit", name))
            return
        chunk = task
        try:
            sampler = BatchSampler(
                RandomSampler(range(chunk.shape[0])), batch_size=batch_size, drop_last=False
            )
            progress_counter.value = 0
            job(ensemble, cfg, args, name, sampler, chunk, progress_counter)
            done_q.put(("ok", name))
        except Exception as e:  # noqa: BLE001 - surfaced in the parent
            done_q.put(("error", f"{name}: {type(e).__name__}: {e}"))
            raise
        finally:
            del chunk


class PersistentWorkerPool:
    """One long-lived spawn worker per (ensemble, args, name)."""

    def __init__(self, ensembles: List[Tuple[Any, dict, str]], cfg, job: Callable):
        self.entries = []
        self.cfg = cfg
        for ensemble, args, name in ensembles:
            ensemble.to_shared_memory()
            task_q = _ctx.Queue()
            done_q = _ctx.Queue()
            counter = _ctx.Value("i", 0)
            proc = _ctx.Process(
                target=_worker_main,
                args=(job, ensemble.state_dict(), cfg, args, name, task_q, done_q, counter),
                daemon=True,
            )
            proc.start()
            self.entries.append(dict(proc=proc, task_q=task_q, done_q=done_q,
                                     counter=counter, name=name, args=args))

    def run_chunk(self, chunk: torch.Tensor, poll_s: float = 0.1) -> None:
        """Train every ensemble on `chunk`; returns when all workers finish."""
        chunk.share_memory_()
        for e in self.entries:
            e["task_q"].put(chunk)
        pending = {e["name"]: e for e in self.entries}
        while pending:
            for name, e in list(pending.items()):
                try:
                    status, payload = e["done_q"].get(timeout=poll_s)
                except pyqueue.Empty:
                    if not e["proc"].is_alive():
                        self.close(force=True)
                        raise RuntimeError(
                            f"persistent worker {name} died (exitcode={e['proc'].exitcode})"
                        )
                    continue
                if status == "error":
                    self.close(force=True)
                    raise RuntimeError(f"persistent worker failed: {payload}")
                del pending[name]

    def close(self, force: bool = False) -> None:
        for e in self.entries:
            if e["proc"].is_alive():
                if force:
                    e["proc"].terminate()
                else:
                    e["task_q"].put(None)
        for e in self.entries:
            e["proc"].join(timeout=30)
            if e["proc"].is_alive():
                e["proc"].terminate()
        self.entries = []

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
