"""Interpretation machinery, eval prompt sets, erasure curves, plotting."""

import os

import numpy as np
import pytest
import torch

from sparse_coding_amd.models.learned_dict import Identity, TiedSAE


def _tiny_model():
    from transformers import GPTNeoXConfig, GPTNeoXForCausalLM

    cfg = GPTNeoXConfig(
        hidden_size=32, num_hidden_layers=2, num_attention_heads=4,
        intermediate_size=64, vocab_size=128, max_position_embeddings=128,
    )
    return GPTNeoXForCausalLM(cfg).eval()


def test_fragment_dataset_and_records(tmp_path):
    from sparse_coding_amd.interpret.interpret import (
        get_df,
        interpret_features,
        make_feature_activation_dataset,
        select_activation_records,
    )

    model = _tiny_model()
    ld = TiedSAE(torch.randn(16, 32), torch.zeros(16))
    batches = [torch.randint(0, 128, (2, 64)) for _ in range(2)]
    acts, tokens, feats = make_feature_activation_dataset(
        model, ld, 1, "residual", batches, device="cpu", max_features=8
    )
    assert acts.shape == (4, 64, 8)
    assert tokens.shape == (4, 64)

    # cache round-trip
    cache = str(tmp_path / "df.pt")
    data = get_df(cache, lambda: (acts, tokens, feats))
    data2 = get_df(cache, lambda: (_ for _ in ()).throw(RuntimeError("must not rebuild")))
    assert torch.equal(data2[0], acts)

    top, rand = select_activation_records(acts, tokens, 0, top_k=2, n_random=1)
    assert len(top) == 2 and len(top[0].tokens) == 64

    # explain/simulate injection
    results = interpret_features(
        acts, tokens, feats[:3],
        explain_fn=lambda recs: "fires on everything",
        simulate_fn=lambda expl, recs: 0.25,
        output_folder=str(tmp_path / "interp"),
    )
    assert len(results) == 3
    assert results[0].score == 0.25
    from sparse_coding_amd.interpret.interpret import read_results

    loaded = read_results(str(tmp_path / "interp"))
    assert loaded[feats[0]]["explanation"] == "fires on everything"


def test_correlation_simulator():
    from sparse_coding_amd.interpret.interpret import ActivationRecord, correlation_score_simulator

    rec = ActivationRecord(tokens=["a", "b", "c"], activations=[0.0, 1.0, 2.0])
    sim_perfect = correlation_score_simulator(lambda expl, toks: [0.0, 0.5, 1.0])
    assert abs(sim_perfect("x", [rec]) - 1.0) < 1e-6
    sim_anti = correlation_score_simulator(lambda expl, toks: [1.0, 0.5, 0.0])
    assert sim_anti("x", [rec]) < -0.99


def test_plot_scores(tmp_path):
    from sparse_coding_amd.interpret.interpret import plot_scores

    fig = plot_scores({"sae": [0.2, 0.3], "pca": [0.05]}, save_path=str(tmp_path / "v.png"))
    assert os.path.exists(tmp_path / "v.png")


def test_ioi_dataset():
    from transformers import AutoTokenizer

    from sparse_coding_amd.data.eval_prompts import generate_induction_dataset, generate_ioi_dataset

    try:
        tok = AutoTokenizer.from_pretrained("gpt2")
    except Exception:
        pytest.skip("no tokenizer available offline")
    clean, corr = generate_ioi_dataset(tok, 4, 4)
    assert clean.shape == corr.shape
    assert clean.shape[0] == 8
    ind = generate_induction_dataset(tok, 8, 16)
    assert torch.equal(ind[:, :8], ind[:, 8:])


def test_erasure_curves():
    from sparse_coding_amd.sweep.erasure import erasure_curves, leace_erase

    torch.manual_seed(0)
    d = 16
    labels = torch.cat([torch.zeros(100), torch.ones(100)]).long()
    concept = torch.randn(d)
    acts = torch.randn(200, d) * 0.5 + labels[:, None].float() * concept
    ld = Identity(d)
    curves = erasure_curves(ld, acts, labels, ks=[0, 4])
    assert curves["dict_ablation"][0] > 0.9  # probe finds the concept
    erased = leace_erase(acts, labels, rank=1)
    from sparse_coding_amd.metrics.standard_metrics import logistic_regression_auroc

    assert logistic_regression_auroc(erased, labels, max_iter=200) < curves["dict_ablation"][0]


def test_toy_models_quick():
    from sparse_coding_amd.sweep.toy_models import run_single_go

    res = run_single_go(
        activation_dim=16, n_ground_truth=24, feature_num_nonzero=3,
        l1_values=[1e-3], dict_ratios=(2,), n_steps=150, batch_size=256,
        device="cpu", backend="torch",
    )
    assert len(res) == 1
    assert res[0]["fvu"] < 1.0
    assert 0 <= res[0]["mmcs_to_ground_truth"] <= 1.0


def test_fvu_sparsity_plot_script(tmp_path):
    import sys

    sys.path.insert(0, "plotting")
    from plotting.fvu_sparsity_plot import pareto_auc, plot_fvu_sparsity, score_learned_dicts

    dicts = [(TiedSAE(torch.randn(8, 8), torch.zeros(8)), {"l1_alpha": 1e-3, "dict_size": 8}),
             (TiedSAE(torch.randn(8, 8), torch.zeros(8)), {"l1_alpha": 1e-2, "dict_size": 8})]
    path = tmp_path / "learned_dicts.pt"
    torch.save(dicts, path)
    sample = torch.randn(256, 8)
    curves = score_learned_dicts(str(path), sample)
    assert len(curves) == 1
    plot_fvu_sparsity(curves, save_path=str(tmp_path / "p.png"))
    assert os.path.exists(tmp_path / "p.png")
    assert np.isnan(pareto_auc([(1.0, 0.5)]))


def test_ablation_graph_tiny():
    from sparse_coding_amd.metrics.ablation import build_ablation_graph

    model = _tiny_model()
    ld0 = TiedSAE(torch.randn(8, 32), torch.zeros(8))
    ld1 = TiedSAE(torch.randn(8, 32), torch.zeros(8))
    tokens = torch.randint(0, 128, (2, 8))
    dicts = {(0, "residual"): ld0, (1, "residual"): ld1}
    graph = build_ablation_graph(
        model, dicts, tokens,
        features_to_ablate={(0, "residual"): [0, 1]},
        target_features={(1, "residual"): [0, 1, 2]},
        device="cpu",
    )
    assert len(graph) == 6
    # ablating an upstream residual feature must perturb downstream features
    assert any(v > 0 for v in graph.values())


def test_scan_layer_moments(tmp_path):
    from sparse_coding_amd.metrics.standard_metrics import scan_layer_moments

    dicts = [(TiedSAE(torch.randn(8, 8), torch.zeros(8)), {"l1_alpha": 1e-3})]
    torch.save(dicts, tmp_path / "ld.pt")
    torch.save(torch.randn(500, 8, dtype=torch.float16), tmp_path / "chunk.pt")
    res = scan_layer_moments([str(tmp_path / "ld.pt")], [str(tmp_path / "chunk.pt")], devices=["cpu"], n_procs=1)
    assert len(res) == 1 and "prop_active" in res[0][0]


def test_interpret_cli(tmp_path):
    import interpret as interp_cli
    from sparse_coding_amd.models.learned_dict import TiedSAE

    ld = TiedSAE(torch.randn(16, 512), torch.zeros(16))  # pythia-70m d_model
    ld_path = tmp_path / "ld.pt"
    torch.save(ld, ld_path)
    cache = str(tmp_path / "cache.pt")
    interp_cli.main([
        "make_fragments", "--learned-dict", str(ld_path), "--model-name", "pythia-70m",
        "--layer", "1", "--n-fragments", "8", "--df-n-feats", "8",
        "--cache", cache, "--device", "cpu",
    ])
    assert os.path.exists(cache)
    out = str(tmp_path / "res")
    interp_cli.main(["interpret", "--cache", cache, "--output-folder", out, "--n-feats-explain", "3"])
    assert len(os.listdir(out)) == 3


def test_deep_ae():
    from sparse_coding_amd.analysis.deep_ae import DeepShrinkageAE

    torch.manual_seed(0)
    ae = DeepShrinkageAE(16, 32, depth=2)
    x = torch.randn(64, 16)
    loss0, *_ = ae(x)
    ae.train_on([x] * 200, lr=5e-3)
    loss1, *_ = ae(x)
    assert loss1 < loss0


@pytest.mark.timeout(600)
def test_example_notebooks_execute(tmp_path, monkeypatch):
    """The example notebooks run top to bottom on CPU (code cells exec'd in
    order; plots land in tmp)."""
    import json

    monkeypatch.chdir(tmp_path)
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for name in ("dict_across_time", "dict_compare", "interpreting_sparse_dictionaries",
                 "case_studies_loop", "inter_layer_comparison",
                 "inter_dict_connections", "feature_interp"):
        nb = json.load(open(os.path.join(root, "examples", f"{name}.ipynb")))
        src = "\n".join("".join(c["source"]) for c in nb["cells"] if c["cell_type"] == "code")
        exec(compile(src, name, "exec"), {})  # noqa: S102 - our own notebooks
