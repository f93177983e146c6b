"""Data parallelism over activation shards with RCCL over xGMI.

Replaces the reference's two multi-GPU paths:
  P1 cluster process-per-ensemble  → kept in sweep/cluster_runs.py
  P3 DDP-over-gloo big-SAE trainer → THIS module: one process per GPU,
     torch.distributed backend "nccl" (= RCCL on ROCm), gradient all-reduce
     bucketed and overlapped with the grad kernels on a side stream.

Topology note: xGMI is 7 p2p links × ≈153 GB/s per GPU; the
[M,n,d] fp32 gradient set is tens-to-hundreds of MB, so we bucket to ~64 MB
and launch all-reduces as soon as a bucket's producers complete, letting RCCL
schedule rings across links while later grad GEMMs still run.
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from sparse_coding_amd.utils.tree import tree_flatten


def init_distributed(backend: Optional[str] = None) -> Tuple[int, int, int]:
    """Initialize torch.distributed from torchrun env; returns
    (rank, local_rank, world_size).  Safe to call when WORLD_SIZE is unset
    (returns a single-process layout without init)."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(backend=backend, world_size=world_size, rank=rank)
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
    return rank, local_rank, world_size


class GradBucketAllReducer:
    """Flattens a fixed tree of gradient tensors into persistent buckets and
    all-reduces them (avg) on a dedicated stream.

    The bucket layout is computed once (sorted-key tree flatten order is
    deterministic across ranks — utils/tree.py), so every step is: copy
    grads into buckets (device-side, async) → all_reduce per bucket →
    copy back.  With the HIP step the copies are fused away by reducing the
    grad workspaces directly when they are bucket-contiguous.
    """

    def __init__(self, grads_template, bucket_bytes: int = 64 << 20, group=None):
        self.group = group
        leaves, _ = tree_flatten(grads_template)
        self.shapes = [g.shape for g in leaves]
        self.numels = [g.numel() for g in leaves]
        device = leaves[0].device
        self.buckets: List[torch.Tensor] = []
        self.assignments: List[List[Tuple[int, int, int]]] = []  # per bucket: (leaf_idx, offset, numel)
        cur: List[Tuple[int, int, int]] = []
        cur_size = 0
        for i, n in enumerate(self.numels):
            nb = n * 4
            if cur and cur_size + nb > bucket_bytes:
                self._seal(cur, device)
                cur, cur_size = [], 0
            cur.append((i, cur_size // 4, n))
            cur_size += nb
        if cur:
            self._seal(cur, device)
        self.comm_stream = torch.cuda.Stream() if device.type == "cuda" else None

    def _seal(self, assignment, device):
        total = sum(n for _, _, n in assignment)
        self.buckets.append(torch.empty(total, device=device, dtype=torch.float32))
        self.assignments.append(list(assignment))

    def all_reduce_(self, grads) -> None:
        """In-place average of `grads` across ranks."""
        if not dist.is_initialized() or dist.get_world_size(self.group) == 1:
            return
        leaves, _ = tree_flatten(grads)
        world = dist.get_world_size(self.group)

        works = []
        for bucket, assignment in zip(self.buckets, self.assignments):
            for li, off, n in assignment:
                bucket[off : off + n].copy_(leaves[li].reshape(-1), non_blocking=True)
            if self.comm_stream is not None:
                # per-bucket event between the (current-stream) pack copies
                # and the (comm-stream) all_reduce — a single wait_stream at
                # the top would not order copies issued after it
                ev = torch.cuda.Event()
                ev.record()
                self.comm_stream.wait_event(ev)
                with torch.cuda.stream(self.comm_stream):
                    works.append((dist.all_reduce(bucket, async_op=True), bucket, assignment))
            else:
                works.append((dist.all_reduce(bucket, async_op=True), bucket, assignment))

        for work, bucket, assignment in works:
            work.wait()
            bucket.div_(world)
            for li, off, n in assignment:
                leaves[li].reshape(-1).copy_(bucket[off : off + n], non_blocking=True)
        if self.comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self.comm_stream)


def _reduce_scatter_rows(flat: torch.Tensor, out: torch.Tensor, group) -> "dist.Work":
    """reduce_scatter_tensor(sum) of a [world*k, ...] tensor into a [k, ...]
    shard; falls back to all_reduce + local narrow on backends without a
    native reduce-scatter (gloo) — numerically identical (sum then slice)."""
    try:
        return dist.reduce_scatter_tensor(out, flat, op=dist.ReduceOp.SUM,
                                          group=group, async_op=True)
    except (RuntimeError, ValueError):
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=group, async_op=True)

        class _SliceWork:
            def wait(self_inner):
                work.wait()
                rank = dist.get_rank(group)
                k = out.shape[0]
                out.copy_(flat[rank * k : (rank + 1) * k])

        return _SliceWork()


def _all_gather_rows(out_flat: torch.Tensor, shard: torch.Tensor, group) -> None:
    """all_gather of row shards into the full [world*k, ...] tensor."""
    try:
        dist.all_gather_into_tensor(out_flat, shard.contiguous(), group=group)
    except (RuntimeError, ValueError):
        world = dist.get_world_size(group)
        k = shard.shape[0]
        parts = [out_flat[r * k : (r + 1) * k] for r in range(world)]
        dist.all_gather(parts, shard.contiguous(), group=group)


class DataParallelEnsembleTrainer:
    """DP over activation shards: every rank holds a replica of the SAME
    ensemble; each step consumes a per-rank shard of the global batch.

    dp_mode:
      "allreduce" (default) — all-reduce the [M, n, d] gradients (chunked
        over model halves, overlapped with the remaining grad GEMMs on a
        side stream) and apply the full, deterministic Adam update locally;
        replicas stay bit-identical without a broadcast.
      "rs_ag" — ZeRO-style: reduce-scatter each weight-gradient chunk into
        per-rank row shards (half the exposed bytes of an all-reduce on a
        per-link-bound xGMI ring), run the renorm-projected Adam on the
        LOCAL shard only (optimizer compute and moments sharded 1/world),
        then all-gather the updated parameter rows.  Optimizer state is
        authoritative only on the owning rank — call
        consolidate_optim_state() before unstack()/checkpointing.
        Supported for the plain tied/untied HipSAEStep; other fused steps
        and the torch backend fall back to "allreduce"."""

    def __init__(self, ensemble, bucket_bytes: int = 64 << 20, group=None,
                 force_dp_path: bool = False, graph_capture: Optional[bool] = None,
                 dp_mode: str = "allreduce"):
        if dp_mode not in ("allreduce", "rs_ag"):
            raise ValueError(f"unknown dp_mode {dp_mode!r}")
        self.ensemble = ensemble
        self.group = group
        self.dp_mode = dp_mode
        self._reducer: Optional[GradBucketAllReducer] = None
        self.bucket_bytes = bucket_bytes
        self.world_size = dist.get_world_size(group) if dist.is_initialized() else 1
        # force_dp_path: run the overlapped all-reduce path even at world
        # size 1 — lets a single-GPU box exercise the exact RCCL call
        # sequence the 8-GPU scaling run uses (tests/test_full_stack_gpu.py)
        self.force_dp_path = force_dp_path
        # opt-in hipGraph capture of the WHOLE multi-GPU step including the
        # RCCL collectives (RCCL supports capture); default off until the
        # 8-GPU behavior is measured (docs/ROADMAP.md item 2).  Falls back
        # to eager on capture failure.
        if graph_capture is None:
            graph_capture = os.environ.get("SC_AMD_DP_GRAPH") == "1"
        self.graph_capture = graph_capture
        self._graph = None
        self._graph_B = None
        self._graph_ws = None
        self._dp_steps = 0

    def broadcast_state(self) -> None:
        """One-time parameter/optimizer broadcast from rank 0 (use when
        replicas were not constructed from the same seed)."""
        if not dist.is_initialized():
            return
        for tree in (self.ensemble.params, self.ensemble.buffers, self.ensemble.optim_states):
            for leaf in tree_flatten(tree)[0]:
                if leaf.dtype.is_floating_point or leaf.dtype in (torch.int32, torch.int64):
                    dist.broadcast(leaf, src=0, group=self.group)

    def _rs_ag_supported(self, hs) -> bool:
        from sparse_coding_amd.engine.hip_step import HipSAEStep

        return (type(hs) is HipSAEStep and not hs.masked and not hs.reverse
                and hs.n_dict % max(self.world_size, 1) == 0)

    def step(self, local_batch: torch.Tensor):
        if self.world_size == 1 and not self.force_dp_path:
            # single-process: take the ensemble's own step (hipGraph-captured
            # on the fused path)
            return self.ensemble.step_batch(local_batch)
        hs = getattr(self.ensemble, "_hip_step", None)
        if self.dp_mode == "rs_ag":
            if hs is not None and self._rs_ag_supported(hs):
                return self._step_rs_ag(hs, local_batch)
            if hs is None and getattr(self.ensemble.optimizer_func, "__name__", "") == "adam":
                return self._step_rs_ag_torch(local_batch)
            if not getattr(self, "_rs_ag_warned", False):
                print(f"[dp] rs_ag unsupported for {type(hs).__name__}; using allreduce")
                self._rs_ag_warned = True
        if hs is not None:
            # fused path: gradient tensors are all-reduced as they become
            # final (g_bias after k_gc; the [M,n,d] weight grads in model
            # halves) on a side stream, overlapping the remaining grad GEMMs
            works = []
            comm_stream = None
            if torch.cuda.is_available():
                if not hasattr(self, "_comm_stream"):
                    self._comm_stream = torch.cuda.Stream()
                comm_stream = self._comm_stream

            def on_grads(tensors):
                if comm_stream is not None:
                    ev = torch.cuda.Event()
                    ev.record()
                    comm_stream.wait_event(ev)
                    with torch.cuda.stream(comm_stream):
                        for t in tensors:
                            works.append((dist.all_reduce(t, async_op=True, group=self.group), t))
                else:
                    for t in tensors:
                        works.append((dist.all_reduce(t, async_op=True, group=self.group), t))

            use_dp = self.world_size > 1 or self.force_dp_path

            def dp_step(x):
                works.clear()
                B = hs.grads_phase(x, on_grads=on_grads if use_dp else None)
                if use_dp:
                    for w, t in works:
                        w.wait()
                    if comm_stream is not None:
                        torch.cuda.current_stream().wait_stream(comm_stream)
                    for _, t in works:
                        t.div_(self.world_size)
                hs.update_phase(B)
                return B

            if self.graph_capture and torch.cuda.is_available():
                B = local_batch.shape[0]
                # a graph is valid only while the step's workspaces are the
                # ones it captured: an eager step at a different B reallocates
                # them (HipSAEStep._alloc), after which replay would write
                # into freed tensors — identity-check the workspace object
                # (strong ref held at capture, so the id cannot be reused)
                if self._graph is not None and getattr(hs, "c", None) is not self._graph_ws:
                    self._graph = None
                    self._graph_B = None
                if self._graph is not None and self._graph_B == B:
                    self._x_static.copy_(local_batch)
                    self._graph.replay()
                    return hs._loss_data(B), {"c": hs.c}
                if self._dp_steps >= 2 and self._graph is None and self._graph_B != B:
                    try:
                        self._x_static = local_batch.contiguous().clone()
                        torch.cuda.synchronize()
                        g = torch.cuda.CUDAGraph()
                        with torch.cuda.graph(g):
                            dp_step(self._x_static)
                        self._graph = g
                        self._graph_B = B
                        self._graph_ws = hs.c
                        self._graph.replay()  # capture does not execute
                        return hs._loss_data(B), {"c": hs.c}
                    except Exception as e:  # noqa: BLE001 - optimization only
                        print(f"[dp] graph capture failed ({e}); staying eager")
                        self.graph_capture = False
                        self._graph = None

            B = dp_step(local_batch)
            self._dp_steps += 1
            return hs._loss_data(B), {"c": hs.c}

        grads, (loss_data, aux) = self.ensemble.compute_grads(local_batch)
        if self.world_size > 1:
            if self._reducer is None:
                self._reducer = GradBucketAllReducer(grads, self.bucket_bytes, self.group)
            self._reducer.all_reduce_(grads)
        self.ensemble.apply_grads(grads)
        return loss_data, aux

    # -- rs_ag (ZeRO-style) ---------------------------------------------------
    def _step_rs_ag(self, hs, x: torch.Tensor):
        """Fused-step rs_ag: weight-grad chunks are reduce-scattered into row
        shards from the on_grads callback (launched on the comm stream so
        they overlap the remaining grad GEMMs), the renorm-projected Adam
        (k_project_adam) runs on the LOCAL rows only, and the updated
        parameter rows are all-gathered.  Bias-sized grads ([M, n] and
        smaller) are all-reduced and updated everywhere — they are <0.1% of
        the bytes and the whole-bias L2-decay term does not shard."""
        import contextlib

        from sparse_coding_amd.engine.hip_step import EPS_NORM

        ens = self.ensemble
        world = max(self.world_size, 1)
        rank = dist.get_rank(self.group) if (dist.is_initialized() and world > 1) else 0
        d = hs.d_act
        R = hs.n_models * hs.n_dict

        comm_stream = None
        if torch.cuda.is_available():
            if not hasattr(self, "_comm_stream"):
                self._comm_stream = torch.cuda.Stream()
            comm_stream = self._comm_stream
        if not hasattr(self, "_rs_shards"):
            self._rs_shards = {}

        w_pend = []  # (work|None, grad shard, kind, global row0, chunk rows)
        b_pend = []  # (work, tensor)

        def locate(t):
            for kind, base in (("gw", hs.gw), ("gw_enc", getattr(hs, "gw_enc", None))):
                if base is None:
                    continue
                off_b = t.data_ptr() - base.data_ptr()
                if 0 <= off_b < base.numel() * 4:
                    return kind, off_b // (4 * d)
            raise RuntimeError("rs_ag: gradient tensor not in a known workspace")

        def on_grads(tensors):
            if comm_stream is not None:
                ev = torch.cuda.Event()
                ev.record()
                comm_stream.wait_event(ev)
            ctx = torch.cuda.stream(comm_stream) if comm_stream is not None else contextlib.nullcontext()
            with ctx:
                for t in tensors:
                    if t.dim() == 2:  # bias-sized
                        if world > 1:
                            b_pend.append((dist.all_reduce(t, async_op=True, group=self.group), t))
                    else:
                        kind, row0 = locate(t)
                        rows = t.shape[0] * t.shape[1]
                        k = rows // world
                        shard = self._rs_shards.get((kind, row0))
                        if shard is None or shard.shape[0] != k:
                            shard = torch.empty(k, d, device=t.device, dtype=torch.float32)
                            self._rs_shards[(kind, row0)] = shard
                        flat = t.reshape(rows, d)
                        if world > 1:
                            work = _reduce_scatter_rows(flat, shard, self.group)
                        else:  # world-1 force path: same code shape, local copy
                            shard.copy_(flat)
                            work = None
                        w_pend.append((work, shard, kind, row0, rows))

        B = hs.grads_phase(x, on_grads=on_grads)

        for work, *_ in w_pend:
            if work is not None:
                work.wait()
        for work, _ in b_pend:
            work.wait()
        if comm_stream is not None:
            torch.cuda.current_stream().wait_stream(comm_stream)
        for _, t in b_pend:
            t.div_(world)

        st = ens.optim_states
        st["step"] += 1.0
        step0 = st["step"].reshape(-1)[0:1]
        p = ens.params
        self._rs_consolidate = []
        norms_flat = hs.norms.reshape(R)
        lrm_flat = hs.lr_mult.reshape(R)
        for work, shard, kind, row0, rows in w_pend:
            if world > 1:
                shard.div_(world)
            k = rows // world
            g0 = row0 + rank * k
            if hs.tied:
                pname, project = "encoder", True
            elif kind == "gw":
                pname, project = "decoder", True
            else:
                pname, project = "encoder", False
            Wflat = p[pname].reshape(R, d)
            mu = st["mu"][pname].reshape(R, d)
            nu = st["nu"][pname].reshape(R, d)
            sl = slice(g0, g0 + k)
            hs.ext.project_adam(Wflat[sl], shard, norms_flat[sl], mu[sl], nu[sl],
                                step0, k, hs.lr, hs.beta1, hs.beta2, hs.eps,
                                EPS_NORM, project, lr_mult=lrm_flat[sl])
            if world > 1:
                _all_gather_rows(Wflat[row0 : row0 + rows], Wflat[sl], self.group)
            self._rs_consolidate.append((st["mu"][pname], row0, rows))
            self._rs_consolidate.append((st["nu"][pname], row0, rows))
        # bias params: identical full update on every rank
        hs.ext.bias_adam(p["encoder_bias"], hs.g_bias, hs.bias_decay,
                         st["mu"]["encoder_bias"], st["nu"]["encoder_bias"],
                         st["step"], hs.lr, hs.beta1, hs.beta2, hs.eps,
                         lr_mult=hs.lr_mult)
        return hs._loss_data(B), {"c": hs.c}

    def _step_rs_ag_torch(self, x: torch.Tensor):
        """Generic-tree rs_ag for the torch backend (and the CPU/gloo tests):
        weight leaves ([M, n, d] and deeper) are reduce-scattered by row and
        Adam-updated on the local shard only; small leaves are all-reduced
        and updated everywhere.  Matches functional.optim.adam's exact fp32
        operation order so an rs_ag run is bit-identical to an allreduce
        run (the gloo reduce-scatter fallback sums identically)."""
        ens = self.ensemble
        okw = ens.optimizer_kwargs
        lr = float(okw.get("lr", 1e-3))
        b1, b2 = okw.get("betas", (0.9, 0.999))
        eps = float(okw.get("eps", 1e-8))
        world = max(self.world_size, 1)
        rank = dist.get_rank(self.group) if (dist.is_initialized() and world > 1) else 0

        grads, (loss_data, aux) = ens.compute_grads(x)
        st = ens.optim_states
        st["step"] += 1.0
        step_t = st["step"].reshape(-1)[0]
        bc1 = 1.0 - b1 ** step_t
        bc2 = 1.0 - b2 ** step_t

        g_leaves, _ = tree_flatten(grads)
        p_leaves, _ = tree_flatten(ens.params)
        mu_leaves, _ = tree_flatten(st["mu"])
        nu_leaves, _ = tree_flatten(st["nu"])

        def adam_(pp, gg, mu, nu):
            # same expressions as functional.optim.adam (bitwise-matching)
            new_mu = b1 * mu + (1.0 - b1) * gg
            new_nu = b2 * nu + (1.0 - b2) * gg * gg
            mu.copy_(new_mu)
            nu.copy_(new_nu)
            pp.add_(-lr * (new_mu / bc1) / (torch.sqrt(new_nu / bc2) + eps))

        self._rs_consolidate = []
        for pp, gg, mu, nu in zip(p_leaves, g_leaves, mu_leaves, nu_leaves):
            rows = pp.numel() // pp.shape[-1] if pp.dim() >= 1 else 1
            if pp.dim() >= 3 and world > 1 and rows % world == 0:
                dl = pp.shape[-1]
                gflat = gg.reshape(rows, dl).contiguous()
                k = rows // world
                shard = torch.empty(k, dl, device=pp.device, dtype=pp.dtype)
                _reduce_scatter_rows(gflat, shard, self.group).wait()
                shard.div_(world)
                sl = slice(rank * k, (rank + 1) * k)
                pf = pp.reshape(rows, dl)
                adam_(pf[sl], shard, mu.reshape(rows, dl)[sl], nu.reshape(rows, dl)[sl])
                _all_gather_rows(pf, pf[sl].clone(), self.group)
                self._rs_consolidate.append((mu, 0, rows))
                self._rs_consolidate.append((nu, 0, rows))
            else:
                if world > 1:
                    dist.all_reduce(gg, group=self.group)
                    gg.div_(world)
                adam_(pp, gg, mu, nu)
        return loss_data, aux

    def resample(self, resampler):
        """DP-correct dead-feature resample: fired counts are summed across
        ranks (each rank only saw its own batch shard) and the replacement
        pool is taken from rank 0, so every replica applies the IDENTICAL
        deterministic rewrite and stays bit-identical — no param broadcast
        needed.  examples_seen is scaled to the global count for the
        rate-threshold dead rule."""
        if dist.is_initialized() and self.world_size > 1:
            dist.all_reduce(resampler.fired, group=self.group)
            dist.broadcast(resampler.pool_examples, src=0, group=self.group)
            resampler.examples_seen *= self.world_size
        return resampler.resample()

    def consolidate_optim_state(self) -> None:
        """rs_ag mode shards the Adam moments of the weight rows: each rank's
        mu/nu are authoritative only for its own rows.  Call this before
        unstack()/state_dict()/checkpointing to all-gather the full moments
        onto every rank.  No-op in allreduce mode or at world 1."""
        if self.dp_mode != "rs_ag" or self.world_size <= 1:
            return
        ops = getattr(self, "_rs_consolidate", None)
        if not ops:
            return
        rank = dist.get_rank(self.group)
        world = self.world_size
        for mom, row0, rows in ops:
            k = rows // world
            d_last = mom.shape[-1]
            flat = mom.reshape(-1, d_last)
            sl = slice(row0 + rank * k, row0 + rank * k + k)
            _all_gather_rows(flat[row0 : row0 + rows], flat[sl].clone(), self.group)


def shard_batch(batch: torch.Tensor, rank: int, world_size: int) -> torch.Tensor:
    """Contiguous equal shard of a global batch."""
    n = batch.shape[0] // world_size
    return batch[rank * n : (rank + 1) * n]
