"""Activation-dataset generation CLI (reference generate_test_data.py, C25)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

import torch

from sparse_coding_amd.config import BaseArgs
from sparse_coding_amd.data.activation_dataset import load_model, setup_data


@dataclass
class GenTestArgs(BaseArgs):
    model_name: str = "pythia-70m-deduped"
    dataset_name: str = "synthetic"
    dataset_folder: str = "activation_data"
    layer_loc: str = "residual"
    layers: str = "2"  # comma-separated
    n_chunks: int = 1
    chunk_size_gb: float = 0.1
    device: str = "cuda:0" if torch.cuda.is_available() else "cpu"
    center_dataset: bool = False


def main():
    cfg = GenTestArgs.from_cli()
    layers = [int(x) for x in str(cfg.layers).split(",")]
    model = load_model(cfg.model_name, device=cfg.device)
    tokenizer = None
    try:
        from transformers import AutoTokenizer

        tokenizer = AutoTokenizer.from_pretrained(cfg.model_name)
    except Exception:  # noqa: BLE001 (offline)
        pass
    n = setup_data(
        tokenizer,
        model,
        dataset_name=cfg.dataset_name,
        dataset_folder=cfg.dataset_folder,
        layer=layers if len(layers) > 1 else layers[0],
        layer_loc=cfg.layer_loc,
        n_chunks=cfg.n_chunks,
        chunk_size_gb=cfg.chunk_size_gb,
        device=cfg.device,
        center_dataset=cfg.center_dataset,
        model_name=cfg.model_name,
    )
    print(f"generated {n} activations into {cfg.dataset_folder}")


if __name__ == "__main__":
    main()
