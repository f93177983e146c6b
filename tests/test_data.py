"""Data plane: synthetic generators, PCA/baselines, activation capture from a
tiny random-init HF model on CPU."""

import os

import numpy as np
import pytest
import torch

from sparse_coding_amd.data.random_dataset import (
    RandomDatasetGenerator,
    SparseMixDataset,
    generate_corr_matrix,
    generate_rand_feats,
)
from sparse_coding_amd.models.pca import BatchedMean, BatchedPCA


def test_rand_feats_unit_norm():
    feats = generate_rand_feats(16, 32, "cpu")
    assert feats.shape == (32, 16)
    assert torch.allclose(torch.norm(feats, dim=-1), torch.ones(32), atol=1e-5)


def test_corr_matrix_psd():
    corr = generate_corr_matrix(24, "cpu")
    eigvals = torch.linalg.eigvalsh(corr)
    assert (eigvals > -1e-5).all()


def test_random_generator_sparsity():
    torch.manual_seed(0)
    np.random.seed(0)
    gen = RandomDatasetGenerator(
        activation_dim=32, n_ground_truth_components=64, batch_size=256,
        feature_num_nonzero=8, feature_prob_decay=1.0, correlated=False, device="cpu",
    )
    batch = gen.send(None)
    assert batch.shape == (256, 32)
    assert batch.dtype == torch.float32


def test_sparse_mix_dataset():
    torch.manual_seed(0)
    np.random.seed(0)
    gen = SparseMixDataset(
        activation_dim=16, n_sparse_components=32, batch_size=128,
        feature_num_nonzero=4, feature_prob_decay=0.99, noise_magnitude_scale=0.1,
        device="cpu", sparse_component_covariance=torch.eye(32),
    )
    b = gen.send(None)
    assert b.shape == (128, 16)
    b2 = gen.send(64)
    assert b2.shape == (64, 16)


def test_batched_pca_matches_exact():
    torch.manual_seed(0)
    x = torch.randn(2000, 12) @ torch.randn(12, 12) + torch.randn(12)
    pca = BatchedPCA(12, "cpu")
    for i in range(0, 2000, 128):
        pca.train_batch(x[i : i + 128])
    assert torch.allclose(pca.get_mean(), x.mean(dim=0), atol=1e-4)
    exact_cov = ((x - x.mean(0)).T @ (x - x.mean(0))) / x.shape[0]
    assert torch.allclose(pca.cov, exact_cov, atol=1e-3)
    eigvals, _ = pca.get_pca()
    exact_eigvals = torch.linalg.eigvalsh((exact_cov + exact_cov.T) / 2)
    assert torch.allclose(eigvals, exact_eigvals, atol=1e-3)


def test_batched_mean_streaming():
    x = torch.randn(1000, 5)
    bm = BatchedMean(5, "cpu")
    for i in range(0, 1000, 100):
        bm.train_batch(x[i : i + 100])
    assert torch.allclose(bm.get_mean(), x.mean(dim=0), atol=1e-5)


def test_pca_topk_dict():
    x = torch.randn(500, 8)
    pca = BatchedPCA(8, "cpu")
    pca.train_batch(x)
    topk = pca.to_topk_dict(3)
    code = topk.encode(torch.randn(10, 8))
    assert (code != 0).sum(dim=-1).max() <= 3
    penc = pca.to_learned_dict(2)
    c = penc.encode(torch.randn(10, 8))
    assert ((c != 0).sum(dim=-1) <= 2).all()


# ---------------------------------------------------------------------------
# activation capture (tiny LM, CPU)
# ---------------------------------------------------------------------------

def _tiny_model():
    from transformers import GPTNeoXConfig, GPTNeoXForCausalLM

    cfg = GPTNeoXConfig(
        hidden_size=32, num_hidden_layers=2, num_attention_heads=4,
        intermediate_size=64, vocab_size=128, max_position_embeddings=64,
    )
    return GPTNeoXForCausalLM(cfg).eval()


def test_activation_capture_residual_and_mlp(tmp_path):
    from sparse_coding_amd.data.activation_dataset import (
        capture_activation_hook,
        make_activation_dataset_hf,
        synthetic_token_batches,
    )

    model = _tiny_model()

    # direct capture shape check
    store = []
    with capture_activation_hook(model, 1, "residual", store):
        model(input_ids=torch.randint(0, 128, (2, 16)))
    assert store and store[0].shape == (32, 32)

    store = []
    with capture_activation_hook(model, 0, "mlp", store):
        model(input_ids=torch.randint(0, 128, (2, 16)))
    assert store[0].shape == (32, 64)

    # chunked dataset writer
    total = make_activation_dataset_hf(
        synthetic_token_batches(128, 2, 16, 12),
        model, [1], "residual",
        chunk_size=128, n_chunks=2,
        output_folder=str(tmp_path), device="cpu", model_name="pythia-70m",
    )
    files = sorted(os.listdir(tmp_path))
    assert "0.pt" in files and "1.pt" in files
    chunk = torch.load(tmp_path / "0.pt")
    assert chunk.shape == (128, 32) and chunk.dtype == torch.float16


def test_replace_activation_hook_changes_logits():
    from sparse_coding_amd.data.activation_dataset import replace_activation_hook
    from sparse_coding_amd.models.learned_dict import UntiedSAE

    model = _tiny_model()
    ids = torch.randint(0, 128, (1, 8))
    with torch.no_grad():
        clean = model(input_ids=ids).logits
        zero_dict = UntiedSAE(torch.zeros(4, 32), torch.zeros(4, 32), torch.full((4,), -1e9))
        with replace_activation_hook(model, 0, "residual", zero_dict):
            ablated = model(input_ids=ids).logits
    assert not torch.allclose(clean, ablated)


def test_perplexity_under_reconstruction():
    from sparse_coding_amd.metrics.standard_metrics import calculate_perplexity
    from sparse_coding_amd.models.learned_dict import Identity

    model = _tiny_model()
    ids = torch.randint(0, 128, (4, 16))
    clean = calculate_perplexity(model, None, None, 1, "residual", ids, device="cpu", batch_size=2)
    ident = calculate_perplexity(model, None, Identity(32), 1, "residual", ids, device="cpu", batch_size=2)
    assert clean > 0
    assert abs(clean - ident) / clean < 1e-3  # identity replacement ≈ clean


def test_get_activation_size_and_tensor_names():
    from sparse_coding_amd.data.activation_dataset import get_activation_size, make_tensor_name

    assert get_activation_size("pythia-70m", "residual") == 512
    assert get_activation_size("pythia-70m", "mlp") == 2048
    assert get_activation_size("gpt2", "residual") == 768
    assert get_activation_size("pythia-1.4b", "residual") == 2048
    assert make_tensor_name(2, "residual", "pythia-70m") == "blocks.2.hook_resid_post"
    assert make_tensor_name(5, "mlpout", "gpt2") == "blocks.5.hook_mlp_out"


@pytest.mark.timeout(300)
def test_generate_test_data_cli(tmp_path):
    """The reference-named generate_test_data.py CLI (C25) writes the chunk
    layout end to end."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(root, "generate_test_data.py"),
         "--model_name", "tiny-gptneox", "--layers", "1",
         "--layer_loc", "residual", "--dataset_folder", str(tmp_path / "gtd"),
         "--n_chunks", "1", "--chunk_size_gb", "0.0002", "--device", "cpu"],
        capture_output=True, text=True, cwd=root, timeout=280)
    assert r.returncode == 0, r.stderr[-500:]
    chunk = torch.load(tmp_path / "gtd" / "0.pt", weights_only=False)
    assert chunk.dtype == torch.float16 and chunk.shape[1] == 64
