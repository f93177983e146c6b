"""Functional, vmap-compatible optimizers.

Drop-in replacement for the subset of ``torchopt`` the reference uses
(reference ``autoencoders/ensemble.py:25-31,94-95,123,182-191``): a
GradientTransformation object with ``init(params) -> state`` and
``update(grads, state) -> (updates, new_state)``, where ``updates`` are
*additive* deltas applied by :func:`apply_updates` (params += updates).

Everything is written so that ``torch.vmap`` over the leading (ensemble-model)
dimension works: state leaves are tensors (including the step count), and all
math is expressed with tensor ops.

The HIP engine (``sparse_coding_amd.engine.hip_step``) implements the same
Adam recurrence fused into the gradient kernels; this module is the
correctness oracle for it.
"""

from __future__ import annotations

from typing import Any, Callable, NamedTuple, Tuple

import torch

from sparse_coding_amd.utils.tree import tree_map


class GradientTransformation(NamedTuple):
    init: Callable[[Any], Any]
    update: Callable[[Any, Any], Tuple[Any, Any]]


def adam(lr: float = 1e-3, betas: Tuple[float, float] = (0.9, 0.999), eps: float = 1e-8) -> GradientTransformation:
    """Adam with bias correction, matching torch.optim.Adam / torchopt.adam.

    update_t = -lr * m_hat / (sqrt(v_hat) + eps)
    """
    b1, b2 = betas

    def init(params):
        def zero_like(p):
            return torch.zeros_like(p)

        return {
            "mu": tree_map(zero_like, params),
            "nu": tree_map(zero_like, params),
            "step": torch.zeros((), dtype=torch.float32),
        }

    def update(grads, state):
        step = state["step"] + 1.0
        bc1 = 1.0 - b1**step
        bc2 = 1.0 - b2**step

        new_mu = tree_map(lambda m, g: b1 * m + (1.0 - b1) * g, state["mu"], grads)
        new_nu = tree_map(lambda v, g: b2 * v + (1.0 - b2) * g * g, state["nu"], grads)

        def delta(m, v):
            m_hat = m / bc1
            v_hat = v / bc2
            return -lr * m_hat / (torch.sqrt(v_hat) + eps)

        updates = tree_map(delta, new_mu, new_nu)
        return updates, {"mu": new_mu, "nu": new_nu, "step": step}

    return GradientTransformation(init, update)


def sgd(lr: float = 1e-3, momentum: float = 0.0) -> GradientTransformation:
    def init(params):
        if momentum == 0.0:
            return {"step": torch.zeros((), dtype=torch.float32)}
        return {
            "mom": tree_map(torch.zeros_like, params),
            "step": torch.zeros((), dtype=torch.float32),
        }

    def update(grads, state):
        step = state["step"] + 1.0
        if momentum == 0.0:
            updates = tree_map(lambda g: -lr * g, grads)
            return updates, {"step": step}
        new_mom = tree_map(lambda m, g: momentum * m + g, state["mom"], grads)
        updates = tree_map(lambda m: -lr * m, new_mom)
        return updates, {"mom": new_mom, "step": step}

    return GradientTransformation(init, update)


def apply_updates(params, updates) -> None:
    """In-place params += updates (same contract as torchopt.apply_updates)."""
    from sparse_coding_amd.utils.tree import tree_flatten

    p_leaves, _ = tree_flatten(params)
    u_leaves, _ = tree_flatten(updates)
    with torch.no_grad():
        for p, u in zip(p_leaves, u_leaves):
            p.add_(u)


def optim_str_to_func(optim_str: str):
    """Reference parity: autoencoders/ensemble.py:25-31."""
    if optim_str == "adam":
        return adam
    if optim_str == "sgd":
        return sgd
    raise ValueError(f"Unknown optimizer string: {optim_str}")
