"""Minimal single-device L1 sweep without process dispatch.

Parity with reference ``basic_l1_sweep.py:46-145``: build an M-model tied or
untied ensemble over a log-spaced l1 grid, train through the chunk files on
one device, save learned_dicts per epoch/chunk.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from sparse_coding_amd.config import BaseArgs
from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.models.sae_signatures import FunctionalSAE, FunctionalTiedSAE
from sparse_coding_amd.sweep.big_sweep import unstacked_to_learned_dicts


@dataclass
class SweepArgs(BaseArgs):
    dataset_dir: str = "activation_data"
    output_dir: str = "outputs_basic_sweep"
    device: str = "cuda:0"
    tied: bool = True
    n_models: int = 16
    l1_exp_low: float = -4.0
    l1_exp_high: float = -2.0
    dict_ratio: float = 4.0
    batch_size: int = 256
    lr: float = 1e-3
    n_epochs: int = 1
    backend: str = "auto"


def basic_l1_sweep(cfg: SweepArgs):
    os.makedirs(cfg.output_dir, exist_ok=True)
    chunk_files = sorted(f for f in os.listdir(cfg.dataset_dir) if f.endswith(".pt"))
    assert chunk_files, f"no chunks in {cfg.dataset_dir}"

    d = torch.load(os.path.join(cfg.dataset_dir, chunk_files[0]), map_location="cpu").shape[1]
    n_dict = int(d * cfg.dict_ratio)
    l1_values = np.logspace(cfg.l1_exp_low, cfg.l1_exp_high, cfg.n_models)

    sig = FunctionalTiedSAE if cfg.tied else FunctionalSAE
    models = [sig.init(d, n_dict, float(l1), device=cfg.device) for l1 in l1_values]
    ensemble = FunctionalEnsemble(models, sig, adam, {"lr": cfg.lr}, device=cfg.device, backend=cfg.backend)

    args = {"batch_size": cfg.batch_size, "device": cfg.device, "dict_size": n_dict}

    for epoch in range(cfg.n_epochs):
        for ci, fname in enumerate(np.random.permutation(chunk_files)):
            chunk = torch.load(os.path.join(cfg.dataset_dir, fname), map_location="cpu").float()
            if torch.cuda.is_available():
                chunk = chunk.pin_memory()
            perm = torch.randperm(chunk.shape[0])
            losses_acc = []
            for s in range(0, chunk.shape[0] - cfg.batch_size + 1, cfg.batch_size):
                batch = chunk[perm[s : s + cfg.batch_size]].to(cfg.device, non_blocking=True)
                losses, _ = ensemble.step_batch(batch)
                losses_acc.append(losses["loss"])
            mean_loss = torch.stack(losses_acc).mean(dim=0) if losses_acc else None
            print(f"epoch {epoch} chunk {ci}: loss={None if mean_loss is None else mean_loss.tolist()}")

            lds = unstacked_to_learned_dicts(ensemble, args, ["dict_size"], ["l1_alpha"])
            out = os.path.join(cfg.output_dir, f"epoch_{epoch}_chunk_{ci}")
            os.makedirs(out, exist_ok=True)
            torch.save(lds, os.path.join(out, "learned_dicts.pt"))
    return ensemble


if __name__ == "__main__":
    basic_l1_sweep(SweepArgs.from_cli())
