"""FunctionalEnsemble engine: grads vs autograd, training progress,
serialization, no_stacking path, every trainable signature under vmap."""

import numpy as np
import pytest
import torch

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble, stack_dict, unstack_dict
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.models.lista import FunctionalLISTADenoisingSAE, FunctionalResidualDenoisingSAE
from sparse_coding_amd.models.positive import FunctionalPositiveTiedSAE
from sparse_coding_amd.models.sae_signatures import (
    FunctionalMaskedSAE,
    FunctionalMaskedTiedSAE,
    FunctionalReverseSAE,
    FunctionalSAE,
    FunctionalThresholdingSAE,
    FunctionalTiedCenteredSAE,
    FunctionalTiedSAE,
)
from sparse_coding_amd.models.semilinear import SemiLinearSAE
from sparse_coding_amd.models.topk import TopKEncoder

D, N, B = 16, 32, 64


def _make(sig, n_models=3, **kw):
    models = [sig.init(D, N, 10 ** (-4 + i), **kw) for i in range(n_models)]
    return FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, backend="torch")


def test_stack_unstack_roundtrip():
    models = [FunctionalSAE.init(D, N, 1e-3) for _ in range(3)]
    params = [m[0] for m in models]
    stacked = stack_dict(params)
    assert stacked["encoder"].shape == (3, N, D)
    unstacked = unstack_dict(stacked, 3)
    for i in range(3):
        assert torch.equal(unstacked[i]["encoder"], params[i]["encoder"])


@pytest.mark.parametrize(
    "sig",
    [FunctionalSAE, FunctionalTiedSAE, FunctionalTiedCenteredSAE, FunctionalReverseSAE,
     FunctionalThresholdingSAE, FunctionalPositiveTiedSAE],
)
def test_grads_match_autograd(sig):
    torch.manual_seed(0)
    ens = _make(sig)
    x = torch.randn(B, D)
    grads, (loss_data, aux) = ens.compute_grads(x)
    assert aux["c"].shape == (3, B, N)
    for m in range(3):
        p = {k: v[m].clone().requires_grad_() for k, v in ens.params.items()}
        b = {k: v[m] for k, v in ens.buffers.items()}
        loss, _ = sig.loss(p, b, x)
        loss.backward()
        assert torch.allclose(loss, loss_data["loss"][m], atol=1e-5)
        for k in p:
            if p[k].grad is None:
                continue
            assert torch.allclose(grads[k][m], p[k].grad, atol=1e-5), f"{sig.__name__}.{k} model {m}"


@pytest.mark.parametrize("sig,kw", [
    (FunctionalMaskedTiedSAE, {}),
    (FunctionalMaskedSAE, {}),
])
def test_masked_signatures(sig, kw):
    torch.manual_seed(0)
    # stack width 2N, real sizes N and N/2
    models = [sig.init(D, N, 2 * N, 1e-3), sig.init(D, N // 2, 2 * N, 1e-3)]
    ens = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, backend="torch")
    x = torch.randn(B, D)
    losses, aux = ens.step_batch(x)
    # masked coefficients stay zero
    assert (aux["c"][1][:, N // 2:] == 0).all()
    lds = ens.to_learned_dicts()
    assert lds[0].n_feats == N and lds[1].n_feats == N // 2


def test_lista_and_semilinear_and_residual():
    torch.manual_seed(0)
    x = torch.randn(B, D)
    for sig, init_args in [
        (FunctionalLISTADenoisingSAE, (D, N, 3, 1e-3)),
        (FunctionalResidualDenoisingSAE, (D, N, 2, 1e-3)),
        (SemiLinearSAE, (D, N, 1e-3)),
    ]:
        models = [sig.init(*init_args) for _ in range(2)]
        ens = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, backend="torch")
        l0, _ = ens.step_batch(x)
        for _ in range(20):
            losses, _ = ens.step_batch(x)
        assert (losses["loss"] < l0["loss"]).all(), sig.__name__


def test_topk_no_stacking():
    torch.manual_seed(0)
    models = [TopKEncoder.init(D, N, k) for k in (2, 4, 8)]
    ens = FunctionalEnsemble(models, TopKEncoder, adam, {"lr": 1e-3}, no_stacking=True, backend="torch")
    x = torch.randn(B, D)
    l0, aux = ens.step_batch(x)
    # code sparsity equals k
    for i, k in enumerate((2, 4, 8)):
        nz = (aux["c"][i] != 0).sum(dim=-1).float()
        assert nz.max() <= k
    for _ in range(30):
        losses, _ = ens.step_batch(x)
    assert (losses["loss"] <= l0["loss"] + 1e-6).all()


def test_training_reduces_loss_and_state_roundtrip():
    torch.manual_seed(0)
    ens = _make(FunctionalTiedSAE)
    x = torch.randn(B, D)
    l0, _ = ens.step_batch(x)
    for _ in range(40):
        pass_losses, _ = ens.step_batch(x)
    assert (pass_losses["loss"] < l0["loss"]).all()

    from sparse_coding_amd.utils.tree import tree_map

    # deep-copied state → an independent ensemble with an identical trajectory
    state = ens.state_dict()
    state_copy = dict(state)
    for key in ("params", "buffers", "optim_states"):
        state_copy[key] = tree_map(lambda t: t.clone(), state[key])
    ens2 = FunctionalEnsemble.from_state(state_copy)
    l_a, _ = ens.step_batch(x)
    l_b, _ = ens2.step_batch(x)
    assert torch.allclose(l_a["loss"], l_b["loss"], atol=1e-7)


def test_optimizer_state_writeback():
    """optim_states tensors must be updated in place (shared-memory contract)."""
    ens = _make(FunctionalSAE)
    mu_before = ens.optim_states["mu"]["encoder"]
    ptr = mu_before.data_ptr()
    x = torch.randn(B, D)
    ens.step_batch(x)
    assert ens.optim_states["mu"]["encoder"].data_ptr() == ptr
    assert ens.optim_states["mu"]["encoder"].abs().sum() > 0
    assert (ens.optim_states["step"] == 1).all()


def test_adam_trajectory_matches_torch_adam_single_model():
    """One ensemble model must follow exactly torch.optim.Adam on the same loss."""
    torch.manual_seed(0)
    sig = FunctionalSAE
    model = sig.init(D, N, 1e-3)
    ens = FunctionalEnsemble([model], sig, adam, {"lr": 1e-2}, backend="torch")

    p_ref = {k: v[0].clone().requires_grad_() for k, v in ens.params.items()}
    buf = {k: v[0].clone() for k, v in ens.buffers.items()}
    opt = torch.optim.Adam(p_ref.values(), lr=1e-2)

    x = torch.randn(B, D)
    for _ in range(5):
        ens.step_batch(x)
        opt.zero_grad()
        loss, _ = sig.loss(p_ref, buf, x)
        loss.backward()
        opt.step()

    for k in p_ref:
        assert torch.allclose(ens.params[k][0], p_ref[k].detach(), atol=1e-5), k
