"""Property-based tests (hypothesis) for numerical invariants that
example-based tests can miss."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=25, deadline=None)
@given(st.text(max_size=200), st.integers(min_value=1, max_value=16))
def test_parse_simulation_total(text, n_tokens):
    """parse_simulation never raises and always returns one clamped level
    per token, for ARBITRARY model output."""
    from sparse_coding_amd.interpret.protocol import MAX_ACT_LEVEL, parse_simulation

    tokens = [f"t{i}" for i in range(n_tokens)]
    levels = parse_simulation(text, tokens)
    assert len(levels) == n_tokens
    assert all(0.0 <= v <= MAX_ACT_LEVEL for v in levels)


@settings(max_examples=10, deadline=None)
@given(st.integers(min_value=2, max_value=5), st.integers(min_value=8, max_value=24),
       st.integers(min_value=0, max_value=1000))
def test_leace_zeroes_class_mean_gap(n_classes, d, seed):
    """LEACE theorem: on the fit data, erased activations have (near-)equal
    class means — the cross-covariance with the concept is annihilated."""
    from sparse_coding_amd.sweep.erasure import LeaceEraser

    rng = torch.Generator().manual_seed(seed)
    n = 400
    labels = torch.randint(0, n_classes, (n,), generator=rng)
    acts = torch.randn(n, d, generator=rng)
    for c in range(n_classes):
        acts[labels == c] += torch.randn(d, generator=rng) * 2.0
    er = LeaceEraser.fit(acts, labels)
    erased = er(acts)
    means = torch.stack([erased[labels == c].mean(dim=0) for c in range(n_classes)
                         if (labels == c).any()])
    gap = (means - means.mean(dim=0)).norm(dim=1).max()
    before = torch.stack([acts[labels == c].mean(dim=0) for c in range(n_classes)
                          if (labels == c).any()])
    gap_before = (before - before.mean(dim=0)).norm(dim=1).max()
    assert gap < 0.1 * gap_before + 1e-3, (gap, gap_before)


@settings(max_examples=10, deadline=None)
@given(st.integers(min_value=2, max_value=6), st.integers(min_value=4, max_value=12),
       st.integers(min_value=0, max_value=1000))
def test_hungarian_mmcs_bounds_and_identity(n_small, d, seed):
    """Hungarian-matched MMCS is in [0, 1]; matching a dict against a
    superset of itself gives ~1."""
    from sparse_coding_amd.metrics.standard_metrics import run_mmcs_with_larger

    rng = np.random.default_rng(seed)
    small = torch.from_numpy(rng.standard_normal((n_small, d))).float()
    large = torch.cat([small * 3.0, torch.from_numpy(rng.standard_normal((4, d))).float()])
    av, above, _ = run_mmcs_with_larger([[small, large]])
    assert 0.0 <= av[0, 0] <= 1.0 + 1e-6
    assert av[0, 0] > 0.999
    assert above[0, 0] == 100.0


@settings(max_examples=15, deadline=None)
@given(st.integers(min_value=1, max_value=4), st.integers(min_value=1, max_value=64),
       st.integers(min_value=1, max_value=40), st.integers(min_value=0, max_value=100))
def test_weighted_reservoir_is_subset_of_stream(m, b, n_track, seed):
    """The resampler pool only ever contains examples that were observed."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.engine.resample import EnsembleResampler
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(seed)
    d = 8
    models = [FunctionalTiedSAE.init(d, 16, 1e-3) for _ in range(m)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, backend="torch")
    rs = EnsembleResampler(ens, n_track=n_track, protocol="anthropic")
    seen = []
    for _ in range(3):
        x = torch.randn(b, d)
        seen.append(x)
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    seen_t = torch.cat(seen)
    pool = rs.pool_examples.reshape(-1, d)
    filled = pool[pool.abs().sum(dim=1) > 0]
    for row in filled:
        assert (seen_t - row).abs().sum(dim=1).min() < 1e-6
