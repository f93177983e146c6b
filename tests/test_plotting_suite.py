"""Every named plotting script runs end-to-end on CPU over
toy artifacts: sweep checkpoint layouts, protocol score folders, erasure
score files.  Each test invokes the script's main(argv) and checks the PNG
lands on disk."""

import os
import sys

import numpy as np
import pytest
import torch

PLOT_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "plotting")
sys.path.insert(0, PLOT_DIR)

D, N = 32, 64


def _ld():
    from sparse_coding_amd.models.learned_dict import TiedSAE

    return TiedSAE(torch.randn(N, D), torch.zeros(N))


def _sweep_layout(root, folders, epochs=(0, 9), l1s=(1e-4, 1e-3)):
    for folder in folders:
        for e in epochs:
            d = root / folder / f"_{e}"
            os.makedirs(d, exist_ok=True)
            torch.save([(_ld(), {"l1_alpha": l1, "dict_size": N}) for l1 in l1s],
                       d / "learned_dicts.pt")


@pytest.fixture()
def chunk_file(tmp_path):
    path = tmp_path / "chunk.pt"
    torch.save(torch.randn(2048, D).half(), path)
    return str(path)


def test_plot_n_active_over_time(tmp_path, chunk_file):
    import plot_n_active_over_time as m

    _sweep_layout(tmp_path, ["tied_residual_l2_r1.0"])
    out = str(tmp_path / "o.png")
    m.main(["--load-dir", str(tmp_path), "--chunk", chunk_file,
            "--epochs", "0,9", "--out", out])
    assert os.path.exists(out)


def test_plot_n_active_long(tmp_path, chunk_file):
    import plot_n_active_long as m

    _sweep_layout(tmp_path, ["tied_residual_l2_r1.0_long"])
    out = str(tmp_path / "o.png")
    m.main(["--load-dir", str(tmp_path), "--chunk", chunk_file,
            "--epochs", "0,9", "--out", out])
    assert os.path.exists(out)


def test_plot_n_active_big_70m(tmp_path, chunk_file):
    import plot_n_active_big_70m as m

    _sweep_layout(tmp_path, ["tied_residual_l2_r1.0", "tied_residual_l2_r2.0"])
    out = str(tmp_path / "o.png")
    m.main(["--load-dir", str(tmp_path), "--chunk", chunk_file,
            "--ratios", "1.0,2.0", "--epoch", "9", "--out", out])
    assert os.path.exists(out)


def test_plot_n_active_gpt2sm(tmp_path, chunk_file):
    import plot_n_active_gpt2sm as m

    _sweep_layout(tmp_path, ["gpt2sm_mlpout_l5_r1.0"])
    out = str(tmp_path / "o.png")
    m.main(["--load-dir", str(tmp_path), "--chunk", chunk_file,
            "--ratios", "1.0", "--epoch", "9", "--out", out])
    assert os.path.exists(out)


def test_plot_n_active_summary(tmp_path, chunk_file):
    import plot_n_active_summary as m

    _sweep_layout(tmp_path, ["tied_residual_l0_r2.0", "tied_residual_l1_r2.0"])
    out = str(tmp_path / "o.png")
    m.main(["--load-dir", str(tmp_path), "--chunk-template", chunk_file,
            "--layers", "0,1", "--epoch", "9", "--out", out])
    assert os.path.exists(out)


def test_num_dead_plot(tmp_path, chunk_file):
    import num_dead_plot as m

    _sweep_layout(tmp_path, ["out_r1"])
    out = str(tmp_path / "o.png")
    m.main(["--dict-files", f"1:{tmp_path}/out_r1/_9/learned_dicts.pt",
            "--chunk", chunk_file, "--n-samples", "512", "--out", out])
    assert os.path.exists(out)


def _protocol_scores(root, transforms, n_feats=6):
    """Write reference-layout explanation.txt trees with known scores."""
    rng = np.random.default_rng(0)
    for t in transforms:
        for f in range(n_feats):
            d = root / t / f"feature_{f}"
            os.makedirs(d, exist_ok=True)
            s = rng.uniform(0, 0.4)
            with open(d / "explanation.txt", "w") as fh:
                fh.write(f"tokens: 'x'\nScore: {s:.2f}\nExplainer model: local\n"
                         f"Simulator model: local\nTop only score: {s:.2f}\n"
                         f"Random only score: {s / 2:.2f}\n")


def test_plot_autointerp_violins(tmp_path):
    import plot_autointerp_violins as m

    _protocol_scores(tmp_path / "l2_residual", ["sparse_coding", "pca"])
    m.main(["--base-path", str(tmp_path), "--score-modes", "top,top_random"])
    assert os.path.exists(tmp_path / "l2_residual" / "top_means_and_violin.png")


def test_plot_autointerp_vs_baselines(tmp_path):
    import plot_autointerp_vs_baselines as m

    for layer in (0, 1):
        _protocol_scores(tmp_path / f"l{layer}_residual",
                         ["tied_r2.0_l1a0.00086", "pca", "ica"])
    out = str(tmp_path / "o.png")
    m.main(["--base-path", str(tmp_path), "--layers", "0,1",
            "--baselines", "pca,ica", "--out", out])
    assert os.path.exists(out)


def test_plot_autointerp_vs_topk_baselines(tmp_path):
    import plot_autointerp_vs_topk_baselines as m

    for layer in (0, 1):
        _protocol_scores(tmp_path / f"l{layer}_residual",
                         ["tied_r2.0_l1a0.00086", "pca_topk"])
    out = str(tmp_path / "o.png")
    m.main(["--base-path", str(tmp_path), "--layers", "0,1",
            "--baselines", "pca_topk", "--out", out])
    assert os.path.exists(out)


def test_plot_autointerp_across_size(tmp_path):
    import plot_autointerp_across_size as m

    _protocol_scores(tmp_path / "l2_residual",
                     ["tied_r1.0_l1a0.00086", "tied_r2.0_l1a0.00086"])
    out = str(tmp_path / "o.png")
    m.main(["--base-path", str(tmp_path), "--layers", "2",
            "--ratios", "1.0,2.0", "--out", out])
    assert os.path.exists(out)


def test_plot_autointerp_across_chunks(tmp_path):
    import plot_autointerp_across_chunks as m

    _protocol_scores(tmp_path / "l2_residual",
                     ["tied_r2.0_nc1_l1a0.00072", "tied_r2.0_nc4_l1a0.00072"])
    out = str(tmp_path / "o.png")
    m.main(["--base-path", str(tmp_path), "--layers", "2",
            "--chunks", "1,4", "--out", out])
    assert os.path.exists(out)


@pytest.mark.timeout(300)
def test_erasure_study_end_to_end(tmp_path):
    """compute -> every reader plot, with a learned dict (all 5 methods)."""
    import erasure_plot as m

    from sparse_coding_amd.models.learned_dict import TiedSAE

    ld_path = str(tmp_path / "ld.pt")
    torch.save(TiedSAE(torch.randn(N, 64), torch.zeros(N)), ld_path)  # d matches tiny-gptneox hidden
    out_dir = str(tmp_path / "erasure")
    m.main(["compute", "--model-name", "tiny-gptneox", "--layers", "0,1",
            "--layer-loc", "residual", "--learned-dict", ld_path,
            "--n-prompts", "64", "--seq-len", "8", "--ks", "1,2",
            "--device", "cpu", "--out-dir", out_dir])
    assert os.path.exists(os.path.join(out_dir, "eval_layer_0_gender.pt"))
    m.main(["scores-across-depth", "--out-dir", out_dir, "--layers", "0,1"])
    assert os.path.exists(os.path.join(out_dir, "erasure_across_depth_gender.png"))
    m.main(["leace-across-depth", "--out-dir", out_dir, "--layers", "0,1"])
    m.main(["kl-across-depth", "--out-dir", out_dir, "--layers", "0,1"])
    m.main(["erasure-scores", "--out-dir", out_dir, "--layers", "0,1"])
    assert os.path.exists(os.path.join(out_dir, "erasure_by_kl_div.png"))
    # bottleneck-scores reader on its schema
    torch.save({"dicts": [(0.1, list(range(4)), 0.8, 0.1), (0.2, list(range(8)), 0.7, 0.2)]},
               tmp_path / "bn.pt")
    m.main(["bottleneck-scores", "--scores", str(tmp_path / "bn.pt"),
            "--out-dir", str(tmp_path / "graphs")])
    assert os.path.exists(tmp_path / "graphs" / "bottleneck_scores.png")


def test_leace_eraser_fit_apply_split():
    """LEACE estimator: fit on train, apply on held-out; concept AUROC drops
    to ~chance while overall geometry is preserved."""
    from sparse_coding_amd.metrics.standard_metrics import logistic_regression_auroc
    from sparse_coding_amd.sweep.erasure import LeaceEraser

    torch.manual_seed(0)
    n, d = 1024, 16
    labels = torch.randint(0, 2, (n,))
    acts = torch.randn(n, d)
    acts[:, 3] += labels.float() * 2.0  # planted concept direction
    base = logistic_regression_auroc(acts[512:], labels[512:], max_iter=200)
    er = LeaceEraser.fit(acts[:512], labels[:512])
    # on the fit split, linear guarding is (near-)exact
    a_fit = logistic_regression_auroc(er(acts[:512]), labels[:512], max_iter=200)
    assert a_fit < 0.6, a_fit
    # held-out: sampling noise in the eraser leaves some leakage, but the
    # probe must lose most of its signal
    erased = er(acts[512:])
    a = logistic_regression_auroc(erased, labels[512:], max_iter=200)
    assert base > 0.9 and a < base - 0.2, (base, a)
    assert (erased - acts[512:]).norm() < acts[512:].norm()  # small edit


def test_leace_multiclass():
    from sparse_coding_amd.sweep.erasure import LeaceEraser

    torch.manual_seed(1)
    n, d = 900, 12
    labels = torch.randint(0, 3, (n,))
    acts = torch.randn(n, d)
    for c in range(3):
        acts[labels == c, c] += 3.0
    er = LeaceEraser.fit(acts, labels)
    erased = er(acts)
    # class means collapse together after erasure
    means = torch.stack([erased[labels == c].mean(dim=0) for c in range(3)])
    spread = (means - means.mean(dim=0)).norm(dim=1).max()
    before = torch.stack([acts[labels == c].mean(dim=0) for c in range(3)])
    spread_before = (before - before.mean(dim=0)).norm(dim=1).max()
    assert spread < 0.25 * spread_before


def test_fvu_sparsity_named_variants(tmp_path, chunk_file):
    import fvu_sparsity_plot_gpt2sm as g
    import fvu_sparsity_plot_mlp_center as c

    _sweep_layout(tmp_path, ["sweep"], epochs=(9,))
    ld_path = f"{tmp_path}/sweep/_9/learned_dicts.pt"
    out1, out2 = str(tmp_path / "g.png"), str(tmp_path / "c.png")
    g.main(["--learned-dicts", ld_path, "--chunk", chunk_file,
            "--device", "cpu", "--out", out1])
    c.main(["--learned-dicts", ld_path, "--chunk", chunk_file,
            "--device", "cpu", "--out", out2])
    assert os.path.exists(out1) and os.path.exists(out2)
