"""Metrics suite correctness (FVU, moments, MMCS, capacity, probes)."""

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm
from sparse_coding_amd.models.learned_dict import Identity, TiedSAE, UntiedSAE


def _stub_dict(encode_fn, n_feats):
    return type("Stub", (), {"n_feats": n_feats, "encode": staticmethod(encode_fn),
                             "center": staticmethod(lambda b: b)})()


def test_fvu_identity_zero():
    x = torch.randn(200, 8)
    ident = Identity(8)
    fvu = sm.fraction_variance_unexplained(ident, x)
    assert fvu.abs() < 1e-6
    assert abs(sm.r_squared(ident, x) - 1.0) < 1e-6


def test_fvu_zero_model_one():
    """A dict reconstructing ~0 has FVU ≈ total/variance ratio > ~1."""
    x = torch.randn(500, 8) + 3.0  # nonzero mean
    zero_sae = UntiedSAE(torch.zeros(4, 8), torch.randn(4, 8), torch.full((4,), -1e9))
    fvu = sm.fraction_variance_unexplained(zero_sae, x)
    # residual = x itself; total = centered variance → fvu > 1 with mean 3
    assert fvu > 1.0


def test_streaming_moments_match_exact():
    torch.manual_seed(0)
    acts = torch.randn(10000, 1)
    stub = _stub_dict(lambda b: b, 1)
    times_active, mean, var, skew, kurt, m4 = sm.calc_moments_streaming(stub, acts, batch_size=512)
    x = acts[:, 0]
    assert torch.allclose(mean, x.mean(), atol=1e-4)
    assert torch.allclose(var, x.var(unbiased=False), atol=1e-3)
    exact_skew = (x**3).mean() / x.var(unbiased=False) ** 1.5
    exact_kurt = (x**4).mean() / x.var(unbiased=False) ** 2
    assert torch.allclose(skew, exact_skew, atol=1e-2)
    assert torch.allclose(kurt, exact_kurt, atol=1e-2)


def test_mmcs_self_is_one():
    d = torch.randn(32, 16)
    sae = TiedSAE(d, torch.zeros(32))
    assert abs(sm.mmcs(sae, sae).item() - 1.0) < 1e-5
    assert abs(sm.mmcs_to_fixed(sae, sae.get_learned_dict()).item() - 1.0) < 1e-5


def test_mmcs_orthogonal_low():
    eye = torch.eye(8)
    a = TiedSAE(eye[:4], torch.zeros(4))
    b = TiedSAE(eye[4:], torch.zeros(4))
    assert sm.mmcs(a, b).item() < 1e-5


def test_hungarian_mmcs():
    g = torch.eye(6)
    perm = g[torch.randperm(6)]
    assert abs(sm.hungarian_mmcs(g, perm).item() - 1.0) < 1e-6


def test_mean_nonzero_and_l0():
    x = torch.randn(100, 8)
    ident = Identity(8)
    assert sm.mean_l0(ident, x) <= 8.0
    props = sm.mean_nonzero_activations(ident, x)
    assert props.shape == (8,)


def test_dead_feature_fraction():
    enc = torch.zeros(10, 4)
    enc[:5] = torch.randn(5, 4)
    sae = UntiedSAE(enc, torch.randn(10, 4), torch.zeros(10))
    x = torch.randn(300, 4)
    frac = sm.dead_feature_fraction(sae, x)
    assert 0.3 <= frac <= 0.8  # ~half the features can never fire


def test_capacity_and_neurons_per_feature():
    eye = TiedSAE(torch.eye(8), torch.zeros(8))
    caps = sm.capacity_per_feature(eye)
    assert torch.allclose(caps, torch.ones(8), atol=1e-5)
    assert abs(sm.neurons_per_feature(eye).item() - 1.0) < 1e-5


def test_ever_active_counts():
    x = torch.randn(500, 8)
    ident = Identity(8)
    n = sm.batched_calc_feature_n_ever_active(ident, x, batch_size=100, threshold=1)
    assert n == 8


def test_probes():
    torch.manual_seed(0)
    x = torch.randn(200, 4)
    labels = (x[:, 0] > 0).long()
    auroc = sm.logistic_regression_auroc(x, labels, max_iter=200)
    assert auroc > 0.9
    assert sm.ridge_regression_auroc(x, labels) > 0.9


def test_expected_interference():
    d = torch.eye(6)
    batch = (torch.rand(50, 6) > 0.5).float()
    cap = sm.calc_expected_interference(d, batch)
    assert cap.shape == (6,)
    assert (cap <= 1.0 + 1e-5).all()


def test_clustering():
    d = TiedSAE(torch.randn(30, 8), torch.zeros(30))
    labels, centers = sm.cluster_directions_kmeans(d, n_clusters=4)
    assert len(labels) == 30 and centers.shape == (4, 8)
    labels_h = sm.cluster_directions_hierarchical(d, n_clusters=4)
    assert len(set(labels_h)) == 4


def test_ica_identifiability():
    """ICA wrapper invariants (role of reference test/test_ica.py:13-69):
    deterministic re-encode; near-identity unmixing on independent Laplace
    axes; identifiable on non-Gaussian data across seeds."""
    import numpy as np

    from sparse_coding_amd.models.ica import ICAEncoder

    np.random.seed(0)
    X = torch.tensor(np.random.laplace(0, 1, (1000, 2)))
    ica = ICAEncoder(2)
    out = ica.train(X)
    again = ica.encode(X)
    assert np.allclose(np.asarray(out), np.asarray(again), atol=1e-5)

    comps = ica.ica.components_ / np.linalg.norm(ica.ica.components_, axis=1)[:, None]
    comps = comps[np.argsort(comps[:, 0])]
    assert np.allclose(abs(comps), np.eye(2), atol=1e-1)

    # identifiable on non-Gaussian data: two runs agree up to sign/order
    np.random.seed(42)
    X = torch.tensor(np.random.laplace(0, 1, (1000, 4)))
    ica1, ica2 = ICAEncoder(4), ICAEncoder(4)
    ica1.train(X)
    ica2.train(X)
    o1 = np.argsort(abs(ica1.ica.components_[:, 0]))
    o2 = np.argsort(abs(ica2.ica.components_[:, 0]))
    assert np.allclose(abs(ica1.ica.components_[o1]), abs(ica2.ica.components_[o2]), atol=1e-3)


def test_run_mmcs_with_larger():
    """Hungarian-matched MMCS grid vs next-larger dict (reference :811-842):
    a larger dict that CONTAINS the smaller one matches at ~1.0."""
    from sparse_coding_amd.metrics.standard_metrics import run_mmcs_with_larger

    torch.manual_seed(0)
    d = 16
    small = torch.randn(8, d)
    large = torch.cat([small * 2.0, torch.randn(8, d)])  # scaled copies inside
    grid = [[small, large]]
    av, above, hists = run_mmcs_with_larger(grid, threshold=0.9)
    assert av.shape == (1, 2)
    assert av[0, 0] > 0.99
    assert above[0, 0] == 100.0
    assert hists[0][0].shape == (8,)
    # unrelated dicts match poorly
    grid2 = [[torch.randn(8, d), torch.randn(64, d)]]
    av2, above2, _ = run_mmcs_with_larger(grid2)
    assert av2[0, 0] < 0.9


def test_plot_capacity_scatter(tmp_path):
    from sparse_coding_amd.metrics.standard_metrics import plot_capacity_scatter
    from sparse_coding_amd.models.learned_dict import TiedSAE

    torch.manual_seed(1)
    dicts = [(TiedSAE(torch.randn(16, 8), torch.zeros(16)), {"l1_alpha": 1e-3})
             for _ in range(2)]
    base = str(tmp_path / "cap")
    plot_capacity_scatter(dicts, save_name=base)
    import os

    assert os.path.exists(base + "_0.png")
    assert os.path.exists(base + "_1.png")
    assert os.path.exists(base + "_hist.png")


def test_cluster_vectors_export(tmp_path):
    from sparse_coding_amd.metrics.standard_metrics import cluster_vectors
    from sparse_coding_amd.models.learned_dict import TiedSAE

    torch.manual_seed(2)
    ld = TiedSAE(torch.randn(40, 8), torch.zeros(40))
    loc = str(tmp_path / "top_clusters.txt")
    top = cluster_vectors(ld, n_clusters=5, top_clusters=3, save_loc=loc, perplexity=5.0)
    assert len(top) == 3
    lines = open(loc).read().strip().split("\n")
    assert len(lines) == 3
    # clusters are disjoint id lists covering <= n_feats
    ids = [int(x) for line in lines for x in line.strip("[]").split(",") if x.strip()]
    assert len(ids) == len(set(ids))


def test_make_one_chunk_per_layer(tmp_path):
    """Layer-chunk helper writes the l{N}_{loc} layout (reference :582-601)."""
    from sparse_coding_amd.metrics.standard_metrics import make_one_chunk_per_layer

    make_one_chunk_per_layer(model_name="tiny-gptneox", out_root=str(tmp_path),
                             layer_locs=("residual",), n_layers=2, device="cpu",
                             chunk_size_gb=0.0002, max_length=16, model_batch_size=2)
    import os

    for layer in range(2):
        assert os.path.exists(tmp_path / f"l{layer}_residual" / "0.pt")
