"""FunctionalEnsemble: train M dictionary models as one batched program.

API parity with reference ``autoencoders/ensemble.py:68-193`` (construction
from a list of (params, buffers) models, ``step_batch``, ``unstack``,
``state_dict``/``from_state``, ``to_shared_memory``, ``no_stacking``
fallback).

Two execution backends:

* ``torch`` — ``torch.vmap(torch.func.grad(sig.loss))`` + a vmapped
  functional Adam/SGD (``sparse_coding_amd.functional.optim``).  Runs on CPU
  and ROCm; this is the semantics oracle.
* ``hip``  — the fused CDNA4 kernel pipeline
  (``sparse_coding_amd.engine.hip_step``): batched encoder-GEMM → ReLU →
  decoder-GEMM → MSE+L1 forward, analytic backward through the in-forward
  decoder row renormalization, fused Adam — one hipGraph per step.  Used
  automatically on gfx950 for supported signatures; raises loudly if the
  extension is missing on a GPU device.

The writeback in ``step_batch`` copies new optimizer state into the existing
(sometimes shared-memory) tensors; the reference's clone-then-copy no-op
(ensemble.py:185-189) is deliberately NOT reproduced — see SURVEY.md §5.
"""

from __future__ import annotations

from typing import List, Type

import torch

from sparse_coding_amd.functional import optim as fx_optim
from sparse_coding_amd.models.sae_signatures import DictSignature
from sparse_coding_amd.utils.tree import (
    tree_flatten,
    tree_map,
    tree_map_,
    tree_unflatten,
)

# re-exports for API parity
optim_str_to_func = fx_optim.optim_str_to_func


def stack_dict(models: list, device=None):
    """Stack a list of congruent trees into one tree of [M, ...] leaves
    (reference ensemble.py:50-56)."""
    flats = [tree_flatten(m) for m in models]
    spec = flats[0][1]
    stacked = []
    for leaves in zip(*[f[0] for f in flats]):
        t = torch.stack(list(leaves)).to(device=device)
        stacked.append(t)
    return tree_unflatten(spec, stacked)


def unstack_dict(tree, n_models: int, device=None):
    leaves, spec = tree_flatten(tree)
    per_model: List[list] = [[] for _ in range(n_models)]
    for leaf in leaves:
        for i in range(n_models):
            # copy=True: unstacking is a SNAPSHOT even when device is
            # unchanged (a plain .to() would alias live training memory on
            # same-device unstacks and every "checkpoint" would silently
            # track the ensemble)
            per_model[i].append(leaf[i].to(device=device, copy=True))
    return [tree_unflatten(spec, ls) for ls in per_model]


class FunctionalEnsemble:
    def __init__(
        self,
        models,
        sig: Type[DictSignature],
        optimizer_func,
        optimizer_kwargs,
        device=None,
        no_stacking: bool = False,
        backend: str = "auto",
    ):
        if device is None:
            # first tensor of the first model's params decides
            first_leaf = tree_flatten(models[0][0])[0][0]
            device = first_leaf.device
        self.device = device

        self.n_models = len(models)
        params, buffers = tuple(zip(*models))
        self.params = stack_dict(list(params), device=self.device)
        self.buffers = stack_dict(list(buffers), device=self.device)

        self.sig = sig
        self.no_stacking = no_stacking
        self.backend = backend

        self.optimizer_func = optimizer_func
        self.optimizer_kwargs = optimizer_kwargs
        self.optimizer = optimizer_func(**optimizer_kwargs)
        self.optim_states = self._init_optim_states()

        self.init_functions()

    # -- optimizer state ----------------------------------------------------
    def _init_optim_states(self):
        """State over stacked params: leaves are [M, ...]; the step counter is
        one float per model so vmap sees a per-model scalar."""
        state = self.optimizer.init(self.params)
        if "step" in state:
            state["step"] = torch.zeros(self.n_models, device=self.device)
        return state

    # -- functional machinery ----------------------------------------------
    def init_functions(self):
        sig = self.sig

        def calc_grads_single(params, buffers, batch):
            return torch.func.grad(sig.loss, has_aux=True)(params, buffers, batch)

        if self.no_stacking:

            def calc_grads(params, buffers, batch):
                grads, auxs = [], []
                for i in range(self.n_models):
                    p_i = tree_map(lambda t: t[i], params)
                    b_i = tree_map(lambda t: t[i], buffers)
                    g, a = calc_grads_single(p_i, b_i, batch[i])
                    grads.append(g)
                    auxs.append(a)
                return stack_dict(grads), stack_dict(auxs)

            self.calc_grads = calc_grads
        else:
            self.calc_grads = torch.vmap(calc_grads_single)
        self.update = torch.vmap(self.optimizer.update)

        self._hip_step = None
        if self.backend in ("auto", "hip"):
            from sparse_coding_amd.engine import hip_step

            self._hip_step = hip_step.maybe_make_step(self, required=(self.backend == "hip"))

    # -- (de)serialization ---------------------------------------------------
    @staticmethod
    def from_state(state_dict) -> "FunctionalEnsemble":
        self = FunctionalEnsemble.__new__(FunctionalEnsemble)
        self.device = state_dict["device"]
        self.n_models = state_dict["n_models"]
        self.params = state_dict["params"]
        self.buffers = state_dict["buffers"]
        self.sig = state_dict["sig"]
        self.no_stacking = state_dict["no_stacking"]
        self.optimizer_func = state_dict["optimizer_func"]
        self.optimizer_kwargs = state_dict["optimizer_kwargs"]
        self.optim_states = state_dict["optim_states"]
        self.backend = state_dict.get("backend", "auto")
        self.optimizer = self.optimizer_func(**self.optimizer_kwargs)
        self.init_functions()
        return self

    def state_dict(self):
        return {
            "device": self.device,
            "n_models": self.n_models,
            "params": self.params,
            "buffers": self.buffers,
            "sig": self.sig,
            "no_stacking": self.no_stacking,
            "optimizer_func": self.optimizer_func,
            "optimizer_kwargs": self.optimizer_kwargs,
            "optim_states": self.optim_states,
            "backend": self.backend,
        }

    def unstack(self, device=None):
        params = unstack_dict(self.params, self.n_models, device=device)
        buffers = unstack_dict(self.buffers, self.n_models, device=device)
        return list(zip(params, buffers))

    def to_learned_dicts(self) -> list:
        return [self.sig.to_learned_dict(p, b) for p, b in self.unstack(device="cpu")]

    def to_device(self, device):
        self.device = device
        self.params = tree_map(lambda t: t.to(device), self.params)
        self.buffers = tree_map(lambda t: t.to(device), self.buffers)
        self.optim_states = tree_map(lambda t: t.to(device), self.optim_states)
        self._hip_step = None
        self.init_functions()

    def to_shared_memory(self):
        tree_map_(lambda t: t.share_memory_(), self.params)
        tree_map_(lambda t: t.share_memory_(), self.buffers)
        tree_map_(lambda t: t.share_memory_(), self.optim_states)

    # -- training ------------------------------------------------------------
    def step_batch(self, minibatches: torch.Tensor, expand_dims: bool = True):
        with torch.no_grad():
            if self._hip_step is not None:
                return self._hip_step.step(minibatches, expand_dims=expand_dims)
            return self._step_batch_torch(minibatches, expand_dims)

    def _step_batch_torch(self, minibatches, expand_dims):
        if expand_dims:
            minibatches = minibatches.expand(self.n_models, *minibatches.shape)

        grads, (loss_data, aux_data) = self.calc_grads(self.params, self.buffers, minibatches)
        updates, new_states = self.update(grads, self.optim_states)

        # write the new optimizer state back into the existing tensors so
        # shared-memory views (cluster dispatch) observe the update
        new_leaves, _ = tree_flatten(new_states)
        leaves, _ = tree_flatten(self.optim_states)
        for leaf, new_leaf in zip(leaves, new_leaves):
            leaf.copy_(new_leaf)

        fx_optim.apply_updates(self.params, updates)
        return loss_data, aux_data

    # convenience: gradient-only pass, used by the DP trainer to insert an
    # all-reduce between grad computation and the optimizer update
    def compute_grads(self, minibatches, expand_dims: bool = True):
        with torch.no_grad():
            if expand_dims:
                minibatches = minibatches.expand(self.n_models, *minibatches.shape)
            return self.calc_grads(self.params, self.buffers, minibatches)

    def apply_grads(self, grads):
        with torch.no_grad():
            updates, new_states = self.update(grads, self.optim_states)
            new_leaves, _ = tree_flatten(new_states)
            leaves, _ = tree_flatten(self.optim_states)
            for leaf, new_leaf in zip(leaves, new_leaves):
                leaf.copy_(new_leaf)
            fx_optim.apply_updates(self.params, updates)


FunctionalEnsemble.__module__ = "autoencoders.ensemble"
