"""Core unit tests: tree utils, functional optimizers, config."""

import torch

from sparse_coding_amd.config import EnsembleArgs, TrainArgs
from sparse_coding_amd.functional.optim import adam, apply_updates, sgd
from sparse_coding_amd.utils.tree import tree_flatten, tree_map, tree_unflatten


def test_tree_roundtrip():
    tree = {"b": torch.ones(2), "a": [torch.zeros(3), {"x": torch.full((1,), 2.0)}]}
    leaves, spec = tree_flatten(tree)
    assert len(leaves) == 3
    rebuilt = tree_unflatten(spec, leaves)
    assert torch.equal(rebuilt["b"], tree["b"])
    assert torch.equal(rebuilt["a"][1]["x"], tree["a"][1]["x"])


def test_tree_map_multi():
    a = {"x": torch.ones(3)}
    b = {"x": torch.full((3,), 2.0)}
    out = tree_map(lambda u, v: u + v, a, b)
    assert torch.equal(out["x"], torch.full((3,), 3.0))


def test_adam_matches_torch_optim():
    torch.manual_seed(0)
    p_ref = torch.randn(10, 4).requires_grad_()
    p_fx = {"w": p_ref.detach().clone()}

    opt_ref = torch.optim.Adam([p_ref], lr=1e-2)
    tx = adam(lr=1e-2)
    state = tx.init(p_fx)

    for _ in range(5):
        g = torch.randn(10, 4)
        p_ref.grad = g.clone()
        opt_ref.step()
        updates, state = tx.update({"w": g}, state)
        apply_updates(p_fx, updates)

    assert torch.allclose(p_ref.detach(), p_fx["w"], atol=1e-6)


def test_adam_vmap_per_model_steps():
    """Bias correction must use each model's own step count under vmap."""
    tx = adam(lr=1e-3)
    params = {"w": torch.randn(3, 5)}
    state = {"mu": {"w": torch.zeros(3, 5)}, "nu": {"w": torch.zeros(3, 5)}, "step": torch.zeros(3)}
    grads = {"w": torch.randn(3, 5)}
    updates, new_state = torch.vmap(tx.update)(grads, state)
    assert new_state["step"].tolist() == [1.0, 1.0, 1.0]
    # first step of adam: update = -lr * sign-ish(g) (m_hat/sqrt(v_hat) = g/|g|)
    expected = -1e-3 * grads["w"] / (grads["w"].abs() + 1e-8)
    assert torch.allclose(updates["w"], expected, atol=1e-6)


def test_sgd():
    tx = sgd(lr=0.1)
    params = {"w": torch.ones(4)}
    state = tx.init(params)
    updates, state = tx.update({"w": torch.ones(4)}, state)
    assert torch.allclose(updates["w"], torch.full((4,), -0.1))


def test_config_cli_parsing():
    cfg = TrainArgs.from_cli(["--layer", "5", "--use_wandb", "false", "--lr", "0.01"])
    assert cfg.layer == 5
    assert cfg.use_wandb is False
    assert abs(cfg.lr - 0.01) < 1e-12


def test_config_no_argv_touch():
    cfg = EnsembleArgs()
    assert cfg.activation_width == 512


def test_config_dict_protocol():
    cfg = TrainArgs()
    d = dict(cfg)
    assert d["layer"] == 2
