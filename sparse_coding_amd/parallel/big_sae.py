"""Single big-dictionary SAE trained data-parallel, with dead-neuron
resampling — the MI355X replacement for the reference's DDP experiment
(``experiments/huge_batch_size.py``, C24: plain nn.Module SAE :25-101, DDP
over gloo :259-345, the repo's only resampling implementation :224-254 +
WorstIndices :120-146).

Here: one process per GPU over RCCL ("nccl" on ROCm), gradient all-reduce by
DDP, and the resampling rule kept semantically identical: a feature is dead
if it never fired over the chunk; dead encoder rows are re-initialized from
the worst-reconstructed examples (scaled to 0.2× the mean encoder row norm)
and their Adam state is zeroed.  The fused-kernel ensemble path exposes the
same rule on-device (engine.resample / ops k_resample).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from sparse_coding_amd.parallel.dp import init_distributed


class BigSAE(nn.Module):
    """Untied SAE with a learnable centering vector (reference
    huge_batch_size.py:25-101)."""

    def __init__(self, activation_size: int, n_features: int, l1_alpha: float):
        super().__init__()
        self.activation_size = activation_size
        self.n_features = n_features
        self.l1_alpha = l1_alpha

        enc = torch.empty(n_features, activation_size)
        nn.init.xavier_uniform_(enc)
        dec = torch.empty(n_features, activation_size)
        nn.init.xavier_uniform_(dec)
        self.encoder = nn.Parameter(enc)
        self.decoder = nn.Parameter(dec)
        self.encoder_bias = nn.Parameter(torch.zeros(n_features))
        self.centering = nn.Parameter(torch.zeros(activation_size))

    def normed_decoder(self) -> torch.Tensor:
        norms = torch.norm(self.decoder, 2, dim=-1, keepdim=True)
        return self.decoder / torch.clamp(norms, 1e-8)

    def forward(self, x: torch.Tensor):
        x_c = x - self.centering
        c = F.relu(F.linear(x_c, self.encoder, self.encoder_bias))
        x_hat = c @ self.normed_decoder()
        per_example_mse = (x_hat - x_c).pow(2).mean(dim=-1)
        mse = per_example_mse.mean()
        l1 = self.l1_alpha * torch.norm(c, 1, dim=-1).mean()
        loss = mse + l1
        return loss, mse, l1, c, per_example_mse


class WorstExampleTracker:
    """Device-side top-K worst-reconstructed example tracker.

    Replaces the reference's Python-loop WorstIndices (:120-146) with a
    batched topk merge: O(B log K) on device, no host sync per batch.
    """

    def __init__(self, k: int, activation_size: int, device):
        self.k = k
        self.losses = torch.full((k,), -float("inf"), device=device)
        self.examples = torch.zeros(k, activation_size, device=device)

    def update(self, batch: torch.Tensor, per_example_loss: torch.Tensor) -> None:
        losses = torch.cat([self.losses, per_example_loss.detach()])
        examples = torch.cat([self.examples, batch.detach()])
        top = torch.topk(losses, self.k)
        self.losses = top.values
        self.examples = examples[top.indices]

    def get_worst(self, n: int) -> torch.Tensor:
        order = torch.argsort(self.losses, descending=True)
        return self.examples[order[:n]]


@torch.no_grad()
def resample_dead_features(
    model: BigSAE,
    optimizer: torch.optim.Optimizer,
    c_totals: torch.Tensor,
    tracker: WorstExampleTracker,
    encoder_norm_ratio: float = 0.2,
) -> int:
    """Reference resampling rule (huge_batch_size.py:224-254) on [n,d] rows."""
    dead = torch.where(c_totals == 0)[0]
    n_replace = int(dead.numel())
    if n_replace == 0:
        return 0
    worst = tracker.get_worst(n_replace)
    if worst.shape[0] < n_replace:
        dead = dead[: worst.shape[0]]
        n_replace = worst.shape[0]

    avg_norm = torch.norm(model.encoder, dim=-1).mean()
    worst_unit = worst / torch.clamp(torch.norm(worst, dim=-1, keepdim=True), 1e-8)
    model.encoder[dead] = worst_unit * encoder_norm_ratio * avg_norm
    model.decoder[dead] = worst_unit
    model.encoder_bias[dead] = 0.0

    for param in (model.encoder, model.decoder, model.encoder_bias):
        state = optimizer.state.get(param)
        if state and "exp_avg" in state:
            state["exp_avg"][dead] = 0
            state["exp_avg_sq"][dead] = 0
    return n_replace


def train_big_sae(
    chunk_paths: List[str],
    activation_size: int = 1024,
    n_features: int = 16384,
    l1_alpha: float = 1e-3,
    lr: float = 1e-3,
    batch_size: int = 4096,
    reinit_every_chunks: int = 10,
    device: Optional[str] = None,
    log_fn=print,
) -> BigSAE:
    """DDP training loop (reference process_main/process_reinit :150-345).

    Launch with torchrun (one rank per GPU); RCCL handles the gradient
    all-reduce inside DDP backward, bucketed and overlapped.
    """
    rank, local_rank, world_size = init_distributed()
    if device is None:
        device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"

    torch.manual_seed(0)
    model = BigSAE(activation_size, n_features, l1_alpha).to(device)
    if world_size > 1:
        model_ddp = nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if torch.cuda.is_available() else None
        )
    else:
        model_ddp = model
    optimizer = torch.optim.Adam(model_ddp.parameters(), lr=lr)

    n_samples = 0
    for chunk_i, path in enumerate(chunk_paths):
        data = torch.load(path, map_location="cpu").float()
        sampler_idx = torch.randperm(data.shape[0])
        # shard across ranks (DistributedSampler equivalent for a tensor)
        sampler_idx = sampler_idx[rank::world_size]

        c_totals = torch.zeros(n_features, device=device)
        tracker = WorstExampleTracker(n_features, activation_size, device)

        for s in range(0, sampler_idx.shape[0] - batch_size + 1, batch_size):
            x = data[sampler_idx[s : s + batch_size]].to(device, non_blocking=True)
            optimizer.zero_grad(set_to_none=True)
            loss, mse, l1, c, per_ex = model_ddp(x)
            loss.backward()  # RCCL all-reduce overlapped here
            optimizer.step()

            c_totals += (c > 0).float().sum(dim=0)
            tracker.update(x, per_ex)
            n_samples += x.shape[0] * world_size

        if rank == 0:
            log_fn(f"[big_sae] chunk {chunk_i}: loss={loss.item():.5f} mse={mse.item():.5f} "
                   f"n_samples={n_samples}")

        if reinit_every_chunks and (chunk_i + 1) % reinit_every_chunks == 0:
            # identical decision on every rank: all-reduce the fired counts
            if world_size > 1:
                dist.all_reduce(c_totals)
            n_dead = resample_dead_features(model, optimizer, c_totals, tracker)
            if world_size > 1:
                # keep replicas bit-identical after the (rank-local) worst-example refill
                for p in model.parameters():
                    dist.broadcast(p.data, src=0)
                for p, state in optimizer.state.items():
                    for key in ("exp_avg", "exp_avg_sq"):
                        if key in state:
                            dist.broadcast(state[key], src=0)
            if rank == 0:
                log_fn(f"[big_sae] resampled {n_dead} dead features")

    return model


def main():
    """torchrun entry point (reference huge_batch_size.py main :358-366):

    torchrun --standalone --nproc-per-node N -m sparse_coding_amd.parallel.big_sae \\
        --chunk-dir activation_data [--n-features 16384] [...]

    Falls back to synthetic chunks when --chunk-dir has none (no network).
    """
    import argparse
    import glob

    p = argparse.ArgumentParser()
    p.add_argument("--chunk-dir", default="activation_data")
    p.add_argument("--activation-size", type=int, default=1024)
    p.add_argument("--n-features", type=int, default=16384)
    p.add_argument("--l1-alpha", type=float, default=1e-3)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--batch-size", type=int, default=4096)
    p.add_argument("--reinit-every-chunks", type=int, default=10)
    p.add_argument("--synthetic-chunks", type=int, default=0,
                   help="generate N synthetic chunks into --chunk-dir first")
    p.add_argument("--save", default="big_sae_state.pt")
    args = p.parse_args()

    if args.synthetic_chunks:
        os.makedirs(args.chunk_dir, exist_ok=True)
        for i in range(args.synthetic_chunks):
            torch.save(torch.randn(65536, args.activation_size, dtype=torch.float16),
                       os.path.join(args.chunk_dir, f"{i}.pt"))

    paths = sorted(glob.glob(os.path.join(args.chunk_dir, "*.pt")))
    if not paths:
        raise SystemExit(f"no chunks in {args.chunk_dir} (use --synthetic-chunks N)")

    model = train_big_sae(
        paths,
        activation_size=args.activation_size,
        n_features=args.n_features,
        l1_alpha=args.l1_alpha,
        lr=args.lr,
        batch_size=args.batch_size,
        reinit_every_chunks=args.reinit_every_chunks,
    )
    rank = int(os.environ.get("RANK", "0"))
    if rank == 0 and args.save:
        torch.save(model.state_dict(), args.save)
        print(f"saved {args.save}")
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
