"""Classical-baseline suite: PCA / PCA-topk / ICA / ICA-topk / random /
identity-ReLU dictionaries per layer, sparsity-matched to the trained SAE.

Parity with reference ``sweep_baselines.py`` (run_layer_baselines :27-114,
run_ica :16, resave_change_sparsity :117, mp.Pool device fan-out :158-172).
"""

from __future__ import annotations

import multiprocessing as mp
import os
from typing import List, Optional

import torch

from sparse_coding_amd.models.ica import ICAEncoder
from sparse_coding_amd.models.learned_dict import IdentityReLU, RandomDict
from sparse_coding_amd.models.nmf import NMFEncoder
from sparse_coding_amd.models.pca import BatchedPCA


def run_ica(activations: torch.Tensor, n_components: int = 0) -> ICAEncoder:
    ica = ICAEncoder(activations.shape[1], n_components)
    ica.train(activations)
    return ica


def run_layer_baselines(
    layer: int,
    chunk_path: str,
    output_folder: str,
    device: str = "cuda:0",
    sparsity: int = 50,
    do_ica: bool = True,
    do_nmf: bool = False,
    pca_batch_size: int = 2048,
    max_ica_samples: int = 200_000,
) -> dict:
    """Train/save the baseline dict family for one layer's chunk 0."""
    os.makedirs(output_folder, exist_ok=True)
    activations = torch.load(chunk_path, map_location="cpu").float()
    d = activations.shape[1]

    out = {}

    pca = BatchedPCA(d, device)
    for i in range(0, activations.shape[0], pca_batch_size):
        pca.train_batch(activations[i : i + pca_batch_size].to(device))

    pca_dict = pca.to_learned_dict(sparsity)
    pca_dict.to_device("cpu")
    torch.save(pca_dict, os.path.join(output_folder, f"pca_l{layer}.pt"))
    out["pca"] = pca_dict

    pca_topk = pca.to_topk_dict(sparsity)
    pca_topk.to_device("cpu")
    torch.save(pca_topk, os.path.join(output_folder, f"pca_topk_l{layer}.pt"))
    out["pca_topk"] = pca_topk

    rot = pca.to_pve_rotation_dict()
    rot.to_device("cpu")
    torch.save(rot, os.path.join(output_folder, f"pca_rot_l{layer}.pt"))
    out["pca_rot"] = rot

    if do_ica:
        ica = run_ica(activations[:max_ica_samples])
        torch.save(ica, os.path.join(output_folder, f"ica_l{layer}.pt"))
        torch.save(ica.to_topk_dict(sparsity), os.path.join(output_folder, f"ica_topk_l{layer}.pt"))
        out["ica"] = ica

    if do_nmf:
        nmf = NMFEncoder(d)
        nmf.train(activations[:max_ica_samples].clone())
        torch.save(nmf, os.path.join(output_folder, f"nmf_l{layer}.pt"))
        out["nmf"] = nmf

    rand = RandomDict(d)
    torch.save(rand, os.path.join(output_folder, f"random_l{layer}.pt"))
    out["random"] = rand

    ident = IdentityReLU(d)
    torch.save(ident, os.path.join(output_folder, f"identity_relu_l{layer}.pt"))
    out["identity_relu"] = ident

    return out


def resave_change_sparsity(dict_path: str, new_sparsity: int) -> None:
    """Adjust the k of a saved topk-style dict in place (reference :117-155)."""
    d = torch.load(dict_path, map_location="cpu")
    if hasattr(d, "sparsity"):
        d.sparsity = new_sparsity
        torch.save(d, dict_path)
    else:
        raise ValueError(f"{dict_path} has no sparsity attribute")


def run_all(
    layers: List[int],
    chunk_paths: List[str],
    output_folder: str,
    devices: Optional[List[str]] = None,
    sparsity: int = 50,
    n_procs: int = 6,
) -> None:
    """Fan the per-layer jobs over a process pool pinned to devices
    (reference :158-172)."""
    if devices is None:
        n = torch.cuda.device_count() or 1
        devices = [f"cuda:{i}" for i in range(n)] if torch.cuda.is_available() else ["cpu"]
    jobs = [
        (layer, chunk, output_folder, devices[i % len(devices)], sparsity)
        for i, (layer, chunk) in enumerate(zip(layers, chunk_paths))
    ]
    with mp.get_context("spawn").Pool(min(n_procs, len(jobs))) as pool:
        pool.starmap(run_layer_baselines, jobs)
