"""Dead-neuron resampling protocols (engine/resample.py): the reference's
worst-example rule (huge_batch_size.py:224-254) and the Anthropic-style
loss^2-weighted protocol with post-resample lr warmup.  CPU (torch backend);
the fused k_resample + lr_mult path is covered in test_full_stack_gpu.py."""

import pytest
import torch

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.engine.resample import EnsembleResampler
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.models.sae_signatures import FunctionalSAE, FunctionalTiedSAE

D, N, B = 32, 128, 64


def _ens(sig=FunctionalTiedSAE, n_models=2):
    models = [sig.init(D, N, 1e-3) for _ in range(n_models)]
    return FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, backend="torch")


def _kill_half(ens):
    with torch.no_grad():
        ens.params["encoder_bias"][:, : N // 2] = -1e6


@pytest.mark.parametrize("protocol", ["worst", "anthropic"])
def test_resample_replaces_dead(protocol):
    torch.manual_seed(0)
    ens = _ens()
    _kill_half(ens)
    rs = EnsembleResampler(ens, n_track=32, protocol=protocol)
    x = torch.randn(B, D)
    for _ in range(3):
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    assert (rs.fired[:, : N // 2] == 0).all()
    before = ens.params["encoder"][:, :32].clone()
    counts = rs.resample()
    assert (counts == 32).all()
    after = ens.params["encoder"][:, :32]
    assert not torch.allclose(before, after)
    assert (ens.optim_states["mu"]["encoder"][:, :32] == 0).all()
    assert (ens.params["encoder_bias"][:, :32] == 0).all()


def test_anthropic_encoder_scale_uses_alive_rows():
    """Replacement rows land at ratio x mean ALIVE row norm."""
    torch.manual_seed(1)
    ens = _ens()
    _kill_half(ens)
    # make alive rows big and dead rows tiny so the two means differ a lot
    with torch.no_grad():
        ens.params["encoder"][:, : N // 2] *= 0.01
        ens.params["encoder"][:, N // 2 :] *= 10.0
    rs = EnsembleResampler(ens, n_track=16, protocol="anthropic")
    x = torch.randn(B, D)
    for _ in range(2):
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    alive_mean = torch.norm(ens.params["encoder"][:, N // 2 :], dim=-1).mean(dim=1)
    rs.resample()
    new_norms = torch.norm(ens.params["encoder"][:, :16], dim=-1)
    expect = 0.2 * alive_mean
    assert torch.allclose(new_norms, expect[:, None].expand_as(new_norms), rtol=0.05)


def test_anthropic_pool_prefers_high_loss_examples():
    """The weighted reservoir should be dominated by examples whose
    reconstruction loss is orders of magnitude larger."""
    torch.manual_seed(2)
    ens = _ens(n_models=1)
    rs = EnsembleResampler(ens, n_track=64, protocol="anthropic")
    big = torch.randn(D) * 30.0  # huge residual -> huge loss^2 weight
    small = torch.randn(D) * 0.01
    for _ in range(20):
        batch = torch.cat(
            [small.expand(B - 4, D), big.expand(4, D)], dim=0
        ).contiguous()
        _, aux = ens.step_batch(batch)
        rs.observe(batch, aux)
    pool = rs.pool_examples[0]
    frac_big = (pool - big).norm(dim=-1).lt(1e-3).float().mean().item()
    assert frac_big > 0.9, frac_big


def test_untied_resample_sets_decoder_unit_rows():
    torch.manual_seed(3)
    ens = _ens(FunctionalSAE)
    _kill_half(ens)
    rs = EnsembleResampler(ens, n_track=16, protocol="anthropic")
    x = torch.randn(B, D)
    for _ in range(2):
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    rs.resample()
    dec_norms = torch.norm(ens.params["decoder"][:, :16], dim=-1)
    assert torch.allclose(dec_norms, torch.ones_like(dec_norms), atol=1e-4)
    assert (ens.optim_states["mu"]["decoder"][:, :16] == 0).all()


def test_replaced_mask_matches_rule():
    torch.manual_seed(4)
    ens = _ens(n_models=1)
    rs = EnsembleResampler(ens, n_track=4, protocol="anthropic")
    rs.fired = torch.tensor([[1.0, 0, 0, 1, 0, 0, 0, 0] + [1.0] * (N - 8)])
    mask = rs._replaced_mask()
    # first 4 dead in index order: 1,2,4,5
    assert mask[0, [1, 2, 4, 5]].all()
    assert not mask[0, [0, 3, 6, 7]].any() or not mask[0, [6, 7]].any()
    assert mask.sum() == 4


def test_resampler_rejects_multilayer_encoders():
    from sparse_coding_amd.models.lista import FunctionalLISTADenoisingSAE

    models = [FunctionalLISTADenoisingSAE.init(16, 32, 2, 1e-3)]
    ens = FunctionalEnsemble(models, FunctionalLISTADenoisingSAE, adam, {"lr": 1e-3},
                             backend="torch")
    with pytest.raises(ValueError, match="no 'encoder'/'dict'"):
        EnsembleResampler(ens)
