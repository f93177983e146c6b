"""Fused CDNA4 training step for SAE ensembles.

Implements the reference's vmapped grad+Adam `step_batch`
(autoencoders/ensemble.py:119-123,175-193 + the signature losses in
sae_ensemble.py / topk_encoder.py / mlp_tests.py) as a pipeline of
hand-written gfx950 HIP kernels (``sparse_coding_amd/ops/hip/sae_kernels.hip``):

  k_row_norms    : per-row ||w|| and 1/max(||w||,1e-8) of the dictionary
  k_enc_fwd      : encoder GEMM -> epilogue by mode: bias+ReLU (0), raw
                   scores for TopK (1), threshold gate (2), reverse
                   bias-subtract (3); + L1 partial, fired counts, optional
                   per-model coef mask (K9) and per-model x stride
  k_dec_fwd      : decoder GEMM (rows scaled by inv-norm) -> residual r
                   (+MSE partial)
  k_gc           : code-grad GEMM -> relu/sign mask + l1 term -> g_pre
                   (+bias-grad column sums); k_gc_thresh adds the gate
                   derivative and gain/scale column-sum grads
  k_grad_w       : batch-contraction GEMMs -> dL/d(W_hat) (and encoder grad)
  k_project_adam : analytic gradient of w/max(||w||,eps) (the in-forward
                   decoder renormalization) + fused Adam; optional separate
                   "used" weights + clamp-derivative mask (positive SAE)
  k_bias_adam    : Adam on any [M,k] vector param (+ L2-decay gradient)

Fused-step coverage: tied, untied, masked tied/untied, thresholding,
reverse, tied-centered, positive-tied, TopK, and LISTA (HipLISTAStep:
hand-derived backward through the unrolled learned-ISTA layers, every GEMM
on the MFMA kernels).  The semilinear (MLP-encoder) and residual-denoising
SAEs stay on the vmap backend: plain batched GEMMs, rocBLAS-served.

All GEMMs run on the exact-f32 MFMA path (v_mfma_f32_32x32x2_f32): fp32 end
to end, the reference's training dtype.  Validated against the
torch/vmap oracle in tests/test_hip_numerics.py.

For multi-GPU DP the step splits at the gradient boundary
(`grads_phase` / `update_phase`) so the RCCL all-reduce of [M,n,d] grads
overlaps with nothing locally but lands between grad GEMMs and Adam.
"""

from __future__ import annotations

from typing import Optional

import torch

from sparse_coding_amd import ops as _ops
from sparse_coding_amd.models import sae_signatures as sigs

EPS_NORM = 1e-8


def _identity_centering(buffers) -> bool:
    rot = buffers.get("center_rot")
    scale = buffers.get("center_scale")
    trans = buffers.get("center_trans")
    if rot is None and scale is None and trans is None:
        return True
    d = rot.shape[-1]
    eye = torch.eye(d, device=rot.device, dtype=rot.dtype).expand_as(rot)
    return torch.equal(rot, eye) and bool((scale == 1).all()) and bool((trans == 0).all())


def maybe_make_step(ensemble, required: bool = False) -> Optional["HipSAEStep"]:
    """Return a HipSAEStep if (sig, device, buffers) are supported.

    On a CUDA device with a supported signature, a missing extension is a
    hard error: GPU runs must not silently fall back to eager."""
    import os

    if os.environ.get("SPARSE_CODING_AMD_FORCE_TORCH") == "1":
        return None

    dev = torch.device(ensemble.device) if not isinstance(ensemble.device, torch.device) else ensemble.device
    if dev.type != "cuda":
        if required:
            raise RuntimeError(f"backend='hip' requested but ensemble device is {dev}")
        return None

    sig = ensemble.sig
    from sparse_coding_amd.models.topk import TopKEncoder

    if sig is TopKEncoder:
        ext = _ops.get_extension(required=True)
        return HipTopKStep(ensemble, ext)

    if sig is sigs.FunctionalThresholdingSAE:
        ext = _ops.get_extension(required=True)
        return HipThresholdStep(ensemble, ext)

    if sig is sigs.FunctionalReverseSAE:
        ext = _ops.get_extension(required=True)
        return HipSAEStep(ensemble, ext, tied=True, reverse=True)

    if sig is sigs.FunctionalTiedCenteredSAE:
        ext = _ops.get_extension(required=True)
        return HipCenteredStep(ensemble, ext)

    from sparse_coding_amd.models.positive import FunctionalPositiveTiedSAE

    if sig is FunctionalPositiveTiedSAE:
        ext = _ops.get_extension(required=True)
        return HipPositiveStep(ensemble, ext)

    from sparse_coding_amd.models.lista import FunctionalLISTADenoisingSAE

    if sig is FunctionalLISTADenoisingSAE:
        ext = _ops.get_extension(required=True)
        return HipLISTAStep(ensemble, ext)

    if sig is sigs.FunctionalMaskedTiedSAE:
        ext = _ops.get_extension(required=True)
        return HipSAEStep(ensemble, ext, tied=True, masked=True)
    if sig is sigs.FunctionalMaskedSAE:
        ext = _ops.get_extension(required=True)
        return HipSAEStep(ensemble, ext, tied=False, masked=True)

    tied = sig is sigs.FunctionalTiedSAE
    untied = sig is sigs.FunctionalSAE
    if not (tied or untied):
        if required:
            raise RuntimeError(f"backend='hip' requested but signature {sig.__name__} has no fused step yet")
        return None
    if tied and not _identity_centering(ensemble.buffers):
        if required:
            raise RuntimeError("fused tied step requires identity centering buffers")
        return None

    ext = _ops.get_extension(required=True)  # loud on GPU
    return HipSAEStep(ensemble, ext, tied=tied)


class HipSAEStep:
    """Workspaces + kernel-sequence launcher for one ensemble."""

    def __init__(self, ensemble, ext, tied: bool, masked: bool = False, reverse: bool = False):
        self.ens = ensemble
        self.ext = ext
        self.tied = tied
        self.masked = masked
        # reverse SAE (sae_ensemble.py:447-503): code = pre * [pre+b > 0],
        # |code| in the L1, sign-aware backward, NO bias gradient from the
        # code path (only the L2-decay term moves the bias)
        self.reverse = reverse
        self._ws = {}

        p = ensemble.params
        self.n_models, self.n_dict, self.d_act = p["encoder"].shape
        if not tied:
            assert p["decoder"].shape == p["encoder"].shape

        dev = p["encoder"].device
        b = ensemble.buffers
        self.l1_alpha = b["l1_alpha"].detach().to(dev, torch.float32).reshape(self.n_models).contiguous()
        bd = b.get("bias_decay")
        # the masked signatures carry a bias_decay buffer but never apply it
        # (reference sae_ensemble.py:347-373,425-444) — force zero
        if bd is None or masked:
            bd = torch.zeros(self.n_models, device=dev)
        self.bias_decay = bd.detach().to(dev, torch.float32).reshape(self.n_models).contiguous()
        self.dict_sizes = None
        if masked:
            # K9: coefficient columns >= dict_size[m] are masked to zero in
            # the encoder epilogue; every downstream grad is then zero via
            # the c>0 gating, so the rest of the pipeline is unchanged
            self.dict_sizes = b["dict_size"].detach().reshape(self.n_models).to(dev, torch.int32).contiguous()

        opt = ensemble.optimizer_kwargs
        self.lr = float(opt.get("lr", 1e-3))
        betas = opt.get("betas", (0.9, 0.999))
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(opt.get("eps", 1e-8))
        name = getattr(ensemble.optimizer_func, "__name__", "adam")
        if name != "adam":
            raise RuntimeError(f"fused HIP step supports adam only, got {name}")

        # persistent workspaces, sized lazily on first batch
        self._B = None
        # hipGraph capture of the whole step (single-GPU path): replayed
        # after 2 eager warmup steps; disabled via SPARSE_CODING_AMD_NO_GRAPH=1
        import os as _os

        self.use_graph = _os.environ.get("SPARSE_CODING_AMD_NO_GRAPH") != "1"
        self._graph = None
        self._eager_steps = 0

    # -- workspace management -------------------------------------------------
    def _alloc(self, B: int):
        from sparse_coding_amd.ops.kconfig import kernel_config

        self.kc = kernel_config()
        M, n, d = self.n_models, self.n_dict, self.d_act
        dev = self.ens.params["encoder"].device
        f = lambda *shape: torch.empty(shape, device=dev, dtype=torch.float32)
        self.c = f(M, B, n)
        self.gpre = f(M, B, n)
        self.r = f(M, B, d)
        self.norms = f(M, n)
        self.inv_norms = f(M, n)
        self.loss_parts = f(M, 2)
        self.fired = torch.zeros(M, n, device=dev)
        self.g_bias = f(M, n)
        self.gw = f(M, n, d)
        if self.reverse:
            self.kc["staging"] = "t"
        if self.kc["staging"] == "pre":
            # pre-transposed operands: every GEMM's staging direct/b128
            self.xT = f(d, B)
            self.rT = f(M, d, B)
            self.WT = f(M, d, n)
            if not self.tied:
                self.WencT = f(M, d, n)
        if not self.tied:
            self.gw_enc = f(M, n, d)
        self._B = B
        self._graph = None
        self._eager_steps = 0

    # -- phases ---------------------------------------------------------------
    def grads_phase(self, x: torch.Tensor, on_grads=None):
        """Forward + backward: fills self.gw (+gw_enc), g_bias, loss parts.

        on_grads: optional callback(list_of_tensors) fired as gradient
        tensors become final — used by the DP trainer to launch RCCL
        all-reduces that overlap the remaining grad GEMMs (the [M,n,d]
        weight grads are chunked over model halves for this).
        """
        ens, ext = self.ens, self.ext
        M, n, d = self.n_models, self.n_dict, self.d_act
        B = x.shape[0]
        if self._B != B:
            self._alloc(B)

        x = x.contiguous()
        p = ens.params
        enc = p["encoder"]
        bias = p["encoder_bias"]
        dict_w = enc if self.tied else p["decoder"]

        self.loss_parts.zero_()
        self.g_bias.zero_()
        self._fired_step = self.fired  # accumulated across steps for resampling

        kc = self.kc
        bk, prio = kc["bk"], kc["prio"]
        bk_dec = kc["bk_dec"] or bk
        bk_gw = kc["bk_grad_w"] or bk

        ext.row_norms(dict_w, self.norms, self.inv_norms, EPS_NORM)
        if kc["staging"] == "pre":
            # pre-transpose once per step so the big GEMMs stage all operands
            # directly (b128 LDS writes): x^T, What^T (inv-norm pre-applied)
            ext.transpose_scale(x, self.xT, None)
            ext.transpose_scale(dict_w, self.WT, self.inv_norms)
            if self.tied:
                ext.enc_fwd2(self.xT, self.WT, bias, self.c, self.loss_parts, self.fired, 0, bk, prio,
                             dict_sizes=self.dict_sizes)
            else:
                ext.transpose_scale(enc, self.WencT, None)
                ext.enc_fwd2(self.xT, self.WencT, bias, self.c, self.loss_parts, self.fired, 0, bk, prio,
                             dict_sizes=self.dict_sizes)
            ext.dec_fwd(self.c, dict_w, self.inv_norms, x, self.r, self.loss_parts, bk_dec, prio)
            ext.transpose_scale(self.r, self.rT, None)
            ext.gc2(self.rT, self.WT, self.c, self.l1_alpha, self.gpre, self.g_bias, bk, prio)
        else:
            # transpose-in-staging GEMMs (no separate transpose kernels)
            enc_inv = self.inv_norms if self.tied else None
            mode = 3 if self.reverse else 0
            ext.enc_fwd(x, enc, bias, enc_inv, self.c, self.loss_parts, self.fired, mode, bk, prio,
                        dict_sizes=self.dict_sizes)
            ext.dec_fwd(self.c, dict_w, self.inv_norms, x, self.r, self.loss_parts, bk_dec, prio)
            ext.gc(self.r, dict_w, self.inv_norms, self.c, self.l1_alpha, self.gpre, self.g_bias, bk, prio,
                   gc_mode=1 if self.reverse else 0)

        if on_grads is not None:
            on_grads([self.g_bias])  # final after k_gc

        gscale = 2.0 / (B * d)
        if self.tied and on_grads is not None and M > 1:
            # chunk the weight-grad GEMMs over model halves so the first
            # half's all-reduce rides under the second half's compute
            half = M // 2
            for sl in (slice(0, half), slice(half, M)):
                ext.grad_w(self.c[sl], self.r[sl], self.gw[sl], gscale, 0.0, bk_gw, prio)
                ext.grad_w(self.gpre[sl], x, self.gw[sl], 1.0, 1.0, bk_gw, prio)
                on_grads([self.gw[sl]])
        elif self.tied:
            ext.grad_w(self.c, self.r, self.gw, gscale, 0.0, bk_gw, prio)
            ext.grad_w(self.gpre, x, self.gw, 1.0, 1.0, bk_gw, prio)
        else:
            ext.grad_w(self.c, self.r, self.gw, gscale, 0.0, bk_gw, prio)
            if on_grads is not None:
                on_grads([self.gw])
            ext.grad_w(self.gpre, x, self.gw_enc, 1.0, 0.0, bk_gw, prio)
            if on_grads is not None:
                on_grads([self.gw_enc])
        return B

    def update_phase(self, B: int):
        """Adam updates (projection through the renorm for the dictionary)."""
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        step_no = st["step"]
        p = ens.params

        if self.tied:
            ext.project_adam(p["encoder"], self.gw, self.norms,
                             st["mu"]["encoder"], st["nu"]["encoder"], step_no,
                             self.n_dict, self.lr, self.beta1, self.beta2,
                             self.eps, EPS_NORM, True)
        else:
            ext.project_adam(p["decoder"], self.gw, self.norms,
                             st["mu"]["decoder"], st["nu"]["decoder"], step_no,
                             self.n_dict, self.lr, self.beta1, self.beta2,
                             self.eps, EPS_NORM, True)
            ext.project_adam(p["encoder"], self.gw_enc, self.norms,
                             st["mu"]["encoder"], st["nu"]["encoder"], step_no,
                             self.n_dict, self.lr, self.beta1, self.beta2,
                             self.eps, EPS_NORM, False)
        ext.bias_adam(p["encoder_bias"], self.g_bias, self.bias_decay,
                      st["mu"]["encoder_bias"], st["nu"]["encoder_bias"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)

    def _loss_data(self, B: int):
        d = self.d_act
        mse = self.loss_parts[:, 0] / (B * d)
        l1 = self.l1_alpha * self.loss_parts[:, 1] / B
        if self.masked:  # masked losses carry no bias-decay term
            return {"loss": mse + l1, "l_reconstruction": mse, "l_l1": l1}
        bias = self.ens.params["encoder_bias"]
        l_bd = self.bias_decay * torch.norm(bias, 2, dim=-1)
        total = mse + l1 + l_bd
        return {"loss": total, "l_reconstruction": mse, "l_l1": l1, "l_bias_decay": l_bd}

    # -- public ---------------------------------------------------------------
    def _capture(self, x: torch.Tensor) -> None:
        try:
            self.x_static = x.clone()
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                B = self.grads_phase(self.x_static)
                self.update_phase(B)
            self._graph = g
        except Exception as e:  # noqa: BLE001 - graphs are an optimization only
            print(f"[hip_step] hipGraph capture failed ({e}); staying eager")
            self.use_graph = False
            self._graph = None

    def step(self, minibatches: torch.Tensor, expand_dims: bool = True):
        if not expand_dims:
            raise NotImplementedError("per-model batches not supported by the HIP step")
        B = minibatches.shape[0]
        if self.use_graph:
            if self._B == B and self._graph is None and self._eager_steps >= 2:
                self._capture(minibatches.contiguous())
            if self._graph is not None:
                self.x_static.copy_(minibatches)
                self._graph.replay()
                return self._loss_data(B), {"c": self.c}
        B = self.grads_phase(minibatches)
        self.update_phase(B)
        self._eager_steps += 1
        return self._loss_data(B), {"c": self.c}

    def dp_grad_tensors(self):
        """Tensors to all-reduce for data parallelism (average across ranks)."""
        ts = [self.gw, self.g_bias]
        if not self.tied:
            ts.append(self.gw_enc)
        return ts


class HipCenteredStep(HipSAEStep):
    """Fused step for FunctionalTiedCenteredSAE (sae_ensemble.py:164-230):
    the tied pipeline runs on the per-model centered input x' = x - t[m]
    (enc/dec kernels take an [M,B,d] x with a model stride), and the
    learnable center's gradient falls out of workspaces that already exist:
        dL/dt = gscale * sum_b r_b  -  g_bias @ What
    (g_bias is the column sum of gpre from k_gc)."""

    def __init__(self, ensemble, ext):
        super().__init__(ensemble, ext, tied=True)

    def _alloc(self, B: int):
        super()._alloc(B)
        M, n, d = self.n_models, self.n_dict, self.d_act
        dev = self.ens.params["encoder"].device
        self.xc = torch.empty(M, B, d, device=dev)
        self.g_center = torch.empty(M, d, device=dev)
        self.zero_decay = torch.zeros_like(self.bias_decay)
        self.kc["staging"] = "t"

    def grads_phase(self, x: torch.Tensor, on_grads=None):
        ens, ext = self.ens, self.ext
        B = x.shape[0]
        if self._B != B:
            self._alloc(B)
        x = x.contiguous()
        p = ens.params
        enc = p["encoder"]
        bias = p["encoder_bias"]
        center = p["center"]
        bk, prio = self.kc["bk"], self.kc["prio"]
        bk_gw = self.kc["bk_grad_w"] or bk

        self.loss_parts.zero_()
        self.g_bias.zero_()

        torch.sub(x.unsqueeze(0), center.unsqueeze(1), out=self.xc)
        ext.row_norms(enc, self.norms, self.inv_norms, EPS_NORM)
        ext.enc_fwd(self.xc, enc, bias, self.inv_norms, self.c,
                    self.loss_parts, self.fired, 0, bk, prio)
        ext.dec_fwd(self.c, enc, self.inv_norms, self.xc, self.r,
                    self.loss_parts, self.kc["bk_dec"] or bk, prio)
        ext.gc(self.r, enc, self.inv_norms, self.c, self.l1_alpha,
               self.gpre, self.g_bias, bk, prio)

        gscale = 2.0 / (B * self.d_act)
        ext.grad_w(self.c, self.r, self.gw, gscale, 0.0, bk_gw, prio)
        ext.grad_w(self.gpre, self.xc, self.gw, 1.0, 1.0, bk_gw, prio)
        # center gradient from existing reductions (docstring derivation)
        torch.sum(self.r, dim=1, out=self.g_center)
        self.g_center.mul_(gscale)
        self.g_center.sub_(torch.einsum("mn,mnd->md", self.g_bias * self.inv_norms, enc))
        if on_grads is not None:
            on_grads([self.gw, self.g_bias, self.g_center])
        return B

    def update_phase(self, B: int):
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        step_no = st["step"]
        p = ens.params
        ext.project_adam(p["encoder"], self.gw, self.norms,
                         st["mu"]["encoder"], st["nu"]["encoder"], step_no,
                         self.n_dict, self.lr, self.beta1, self.beta2,
                         self.eps, EPS_NORM, True)
        ext.bias_adam(p["encoder_bias"], self.g_bias, self.zero_decay,
                      st["mu"]["encoder_bias"], st["nu"]["encoder_bias"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)
        ext.bias_adam(p["center"], self.g_center, self.zero_decay,
                      st["mu"]["center"], st["nu"]["center"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)

    def _loss_data(self, B: int):
        mse = self.loss_parts[:, 0] / (B * self.d_act)
        l1 = self.l1_alpha * self.loss_parts[:, 1] / B
        return {"loss": mse + l1, "l_reconstruction": mse, "l_l1": l1}

    def dp_grad_tensors(self):
        return [self.gw, self.g_bias, self.g_center]


class HipPositiveStep(HipSAEStep):
    """Fused step for FunctionalPositiveTiedSAE (mlp_tests.py:68-125):
    the tied pipeline runs on Wc = clamp(W, 0) and the shifted input
    x' = x + 0.18; the projected gradient is masked by the clamp derivative
    before Adam updates the raw (signed) encoder."""

    def __init__(self, ensemble, ext):
        super().__init__(ensemble, ext, tied=True)

    def _alloc(self, B: int):
        super()._alloc(B)
        M, n, d = self.n_models, self.n_dict, self.d_act
        dev = self.ens.params["encoder"].device
        self.Wc = torch.empty(M, n, d, device=dev)
        self.x_shift = torch.empty(B, d, device=dev)
        self.kc["staging"] = "t"

    def grads_phase(self, x: torch.Tensor, on_grads=None):
        from sparse_coding_amd.models.positive import INPUT_SHIFT

        ens, ext = self.ens, self.ext
        B = x.shape[0]
        if self._B != B:
            self._alloc(B)
        x = x.contiguous()
        p = ens.params
        enc = p["encoder"]
        bias = p["encoder_bias"]
        bk, prio = self.kc["bk"], self.kc["prio"]
        bk_gw = self.kc["bk_grad_w"] or bk

        self.loss_parts.zero_()
        self.g_bias.zero_()

        torch.clamp(enc, min=0.0, out=self.Wc)
        torch.add(x, INPUT_SHIFT, out=self.x_shift)
        ext.row_norms(self.Wc, self.norms, self.inv_norms, EPS_NORM)
        ext.enc_fwd(self.x_shift, self.Wc, bias, self.inv_norms, self.c,
                    self.loss_parts, self.fired, 0, bk, prio)
        ext.dec_fwd(self.c, self.Wc, self.inv_norms, self.x_shift, self.r,
                    self.loss_parts, self.kc["bk_dec"] or bk, prio)
        ext.gc(self.r, self.Wc, self.inv_norms, self.c, self.l1_alpha,
               self.gpre, self.g_bias, bk, prio)

        gscale = 2.0 / (B * self.d_act)
        ext.grad_w(self.c, self.r, self.gw, gscale, 0.0, bk_gw, prio)
        ext.grad_w(self.gpre, self.x_shift, self.gw, 1.0, 1.0, bk_gw, prio)
        if on_grads is not None:
            on_grads([self.gw, self.g_bias])
        return B

    def update_phase(self, B: int):
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        step_no = st["step"]
        p = ens.params
        ext.project_adam(p["encoder"], self.gw, self.norms,
                         st["mu"]["encoder"], st["nu"]["encoder"], step_no,
                         self.n_dict, self.lr, self.beta1, self.beta2,
                         self.eps, EPS_NORM, True,
                         w_used=self.Wc, clamp_mask=True)
        ext.bias_adam(p["encoder_bias"], self.g_bias, self.bias_decay,
                      st["mu"]["encoder_bias"], st["nu"]["encoder_bias"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)


class HipThresholdStep(HipSAEStep):
    """Fused step for FunctionalThresholdingSAE.

    Tied dictionary with a learned soft-threshold gate instead of bias+ReLU
    (reference sae_ensemble.py:232-289): the encoder GEMM runs with a gate
    epilogue (k_enc_fwd_t mode 2) that keeps u = (c+gain)/max(a^2,eps) for
    the backward, and k_gc_thresh_t pushes dL/dcode through g'(u) while
    accumulating the per-feature gain/scale gradients as column sums.
    Weight grads + renorm-projected Adam are the tied-SAE kernels.
    """

    def __init__(self, ensemble, ext):
        super().__init__(ensemble, ext, tied=True)

    def _alloc(self, B: int):
        super()._alloc(B)
        M, n, d = self.n_models, self.n_dict, self.d_act
        dev = self.ens.params["encoder"].device
        self.u = torch.empty(M, B, n, device=dev)
        self.g_gain = torch.zeros(M, n, device=dev)
        self.g_scale = torch.zeros(M, n, device=dev)
        self.dummy_bias = torch.zeros(M, n, device=dev)
        self.zero_decay = torch.zeros_like(self.bias_decay)
        # the gate epilogue lives only in the transpose-in-staging kernels
        self.kc["staging"] = "t"

    def grads_phase(self, x: torch.Tensor, on_grads=None):
        ens, ext = self.ens, self.ext
        B = x.shape[0]
        if self._B != B:
            self._alloc(B)
        x = x.contiguous()
        p = ens.params
        enc = p["encoder"]
        a = p["activation_scale"]
        gain = p["activation_gain"]
        bk, prio = self.kc["bk"], self.kc["prio"]
        bk_gw = self.kc["bk_grad_w"] or bk

        self.loss_parts.zero_()
        self.g_gain.zero_()
        self.g_scale.zero_()

        ext.row_norms(enc, self.norms, self.inv_norms, EPS_NORM)
        ext.enc_fwd(x, enc, self.dummy_bias, self.inv_norms, self.c,
                    self.loss_parts, self.fired, 2, bk, prio, a, gain, self.u)
        ext.dec_fwd(self.c, enc, self.inv_norms, x, self.r, self.loss_parts,
                    self.kc["bk_dec"] or bk, prio)
        ext.gc_thresh(self.r, enc, self.inv_norms, self.c, self.u, a,
                      self.l1_alpha, self.gpre, self.g_gain, self.g_scale, bk, prio)

        gscale = 2.0 / (B * self.d_act)
        ext.grad_w(self.c, self.r, self.gw, gscale, 0.0, bk_gw, prio)
        ext.grad_w(self.gpre, x, self.gw, 1.0, 1.0, bk_gw, prio)
        if on_grads is not None:
            on_grads([self.gw, self.g_gain, self.g_scale])
        return B

    def update_phase(self, B: int):
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        step_no = st["step"]
        p = ens.params
        zero_decay = self.zero_decay
        ext.project_adam(p["encoder"], self.gw, self.norms,
                         st["mu"]["encoder"], st["nu"]["encoder"], step_no,
                         self.n_dict, self.lr, self.beta1, self.beta2,
                         self.eps, EPS_NORM, True)
        ext.bias_adam(p["activation_scale"], self.g_scale, zero_decay,
                      st["mu"]["activation_scale"], st["nu"]["activation_scale"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)
        ext.bias_adam(p["activation_gain"], self.g_gain, zero_decay,
                      st["mu"]["activation_gain"], st["nu"]["activation_gain"],
                      step_no, self.lr, self.beta1, self.beta2, self.eps)

    def _loss_data(self, B: int):
        mse = self.loss_parts[:, 0] / (B * self.d_act)
        l1 = self.l1_alpha * self.loss_parts[:, 1] / B
        return {"loss": mse + l1, "l_reconstruction": mse, "l_l1": l1}

    def dp_grad_tensors(self):
        return [self.gw, self.g_gain, self.g_scale]


class HipLISTAStep:
    """Fused training step for FunctionalLISTADenoisingSAE (SURVEY.md K11;
    reference residual_denoising_autoencoder.py:15-122).

    The unrolled learned-ISTA encoder is GEMM-dominated; every GEMM here
    runs on the hand-written MFMA kernels (enc/dec-shaped products reuse
    k_enc_fwd mode 1 with the per-model x stride, the batch contractions
    reuse k_grad_w), with the shrinkage/momentum elementwise glue in torch
    ops on persistent workspaces.  The backward is derived analytically:

      fwd  per layer: e = b - y A_hat;  r = y + e W_l^T;
                      x = sign(r) relu(|r|-theta_l);  y' = x + m_l (x - x_prev)
      bwd  per layer: g_x  = (1+m) g_y ;  g_x_prev -= m g_y
                      g_r  = g_x * [|r| > theta];  g_theta = -sum_b g_r sign(r)
                      g_m  = sum(g_y (x - x_prev)) * [0 < rho < 1]
                      g_W  = g_r^T e;   g_e = g_r W_l
                      g_y_prev = g_r - g_e A_hat^T   (+ momentum carry)
                      g_Ahat  -= y_prev^T g_e
      plus the decode/encode-init contributions g_Ahat += gscale c^T rr
      + g_y0^T b, all pushed through the row-renorm projection by
      k_project_adam.  Validated against the torch.func.grad oracle in
      tests/test_hip_numerics.py::test_lista_step_matches_torch.
    """

    def __init__(self, ensemble, ext):
        self.ens = ensemble
        self.ext = ext
        p = ensemble.params
        self.n_models, self.n_dict, self.d_act = p["decoder"].shape
        self.n_layers = len(p["encoder_layers"])

        opt = ensemble.optimizer_kwargs
        self.lr = float(opt.get("lr", 1e-3))
        betas = opt.get("betas", (0.9, 0.999))
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(opt.get("eps", 1e-8))
        name = getattr(ensemble.optimizer_func, "__name__", "adam")
        if name != "adam":
            raise RuntimeError(f"fused HIP step supports adam only, got {name}")
        self.l1_alpha = ensemble.buffers["l1_alpha"].detach().reshape(self.n_models).contiguous()
        self._B = None

    def _alloc(self, B: int):
        from sparse_coding_amd.ops.kconfig import kernel_config

        self.kc = kernel_config()
        M, n, d, L = self.n_models, self.n_dict, self.d_act, self.n_layers
        dev = self.ens.params["decoder"].device
        f = lambda *shape: torch.empty(shape, device=dev, dtype=torch.float32)
        self.norms = f(M, n)
        self.inv_norms = f(M, n)
        self.ones_mn = torch.ones(M, n, device=dev)
        self.zeros_bd = torch.zeros(B, d, device=dev)
        self.scratch_lp = f(M, 2)
        self.loss_parts = f(M, 2)
        self.fired = torch.zeros(M, n, device=dev)
        # forward saves
        self.y = [f(M, B, n) for _ in range(L + 1)]
        self.x = [f(M, B, n) for _ in range(L + 1)]
        self.r = [f(M, B, n) for _ in range(L)]
        self.neg_e = [f(M, B, d) for _ in range(L)]
        self.s = f(M, B, n)       # e @ W_l^T scratch
        self.rr = f(M, B, d)      # decode residual
        # backward workspaces
        self.g_y = f(M, B, n)
        self.g_r = f(M, B, n)
        self.g_xc = torch.zeros(M, B, n, device=dev)  # momentum carry: g wrt x_{l-1}
        self.g_e = f(M, B, d)
        self.t_n = f(M, B, n)
        self.gA = f(M, n, d)
        self.gW = [f(M, n, d) for _ in range(L)]
        self.g_theta = [f(M, n) for _ in range(L)]
        self.g_rho = [f(M) for _ in range(L)]
        self.zero_decay = torch.zeros(M, device=dev)
        self._B = B

    def grads_phase(self, x_in: torch.Tensor, on_grads=None):
        ens, ext = self.ens, self.ext
        M, n, d, L = self.n_models, self.n_dict, self.d_act, self.n_layers
        B = x_in.shape[0]
        if self._B != B:
            self._alloc(B)
        b = x_in.contiguous()
        p = ens.params
        A = p["decoder"]
        layers = p["encoder_layers"]
        bk, prio = self.kc["bk"], self.kc["prio"]
        bk_gw = self.kc["bk_grad_w"] or bk

        self.loss_parts.zero_()
        ext.row_norms(A, self.norms, self.inv_norms, EPS_NORM)

        # ---- forward ----
        # y0 = b @ A_hat^T
        ext.enc_fwd(b, A, self.ones_mn, self.inv_norms, self.y[0],
                    self.scratch_lp, self.fired, 1, bk, prio)
        self.x[0].copy_(self.y[0])
        self.ms = [torch.clamp(layers[l]["rho"], 0.0, 1.0) for l in range(L)]
        for l in range(L):
            W_l, theta = layers[l]["W"], layers[l]["theta"]
            # neg_e = y A_hat - b
            ext.dec_fwd(self.y[l], A, self.inv_norms, b, self.neg_e[l],
                        self.scratch_lp, self.kc["bk_dec"] or bk, prio)
            # s = neg_e @ W_l^T ;  r = y - s
            ext.enc_fwd(self.neg_e[l], W_l, self.ones_mn, None, self.s,
                        self.scratch_lp, self.fired, 1, bk, prio)
            torch.sub(self.y[l], self.s, out=self.r[l])
            # x = shrink(r, theta);  y' = x + m (x - x_prev)
            torch.sub(torch.abs(self.r[l]), theta.unsqueeze(1), out=self.x[l + 1])
            self.x[l + 1].clamp_(min=0.0).mul_(torch.sign(self.r[l]))
            m = self.ms[l].reshape(M, 1, 1)
            torch.add(self.x[l + 1], (self.x[l + 1] - self.x[l]) * m, out=self.y[l + 1])

        c = self.y[L]
        # decode: rr = c A_hat - b (+ MSE partial)
        ext.dec_fwd(c, A, self.inv_norms, b, self.rr, self.loss_parts,
                    self.kc["bk_dec"] or bk, prio)
        self._l1_sum = c.abs().sum(dim=(1, 2))

        # ---- backward ----
        gscale = 2.0 / (B * d)
        # g_c = gscale * rr @ A_hat^T + l1/B * sign(c)
        ext.enc_fwd(self.rr, A, self.ones_mn, self.inv_norms, self.g_y,
                    self.scratch_lp, self.fired, 1, bk, prio)
        self.g_y.mul_(gscale)
        self.g_y.add_(torch.sign(c) * (self.l1_alpha / B).reshape(M, 1, 1))
        # g_Ahat from decode path
        ext.grad_w(c, self.rr, self.gA, gscale, 0.0, bk_gw, prio)

        # two grad streams walk the layers together: g_y (through r/e) and
        # g_xc, the momentum carry -m_{l+1} g_y_{l+1}, which reaches x_l and
        # must flow through LAYER l's shrink, not through y_l
        self.g_xc.zero_()
        for l in range(L - 1, -1, -1):
            W_l, theta = layers[l]["W"], layers[l]["theta"]
            m = self.ms[l].reshape(M, 1, 1)
            dx = self.x[l + 1] - self.x[l]
            # rho grad (clamp passes only inside (0,1))
            rho = layers[l]["rho"].reshape(M)
            gate = ((rho > 0.0) & (rho < 1.0)).float()
            torch.sum(self.g_y * dx, dim=(1, 2), out=self.g_rho[l])
            self.g_rho[l].mul_(gate)
            # g_x_l = (1+m) g_y_l + carry from layer l+1
            torch.mul(self.g_y, 1.0 + m, out=self.g_r)
            self.g_r.add_(self.g_xc)
            torch.mul(self.g_y, -m, out=self.g_xc)  # carry for x_{l-1}
            # shrink backward
            mask = (self.r[l].abs() > theta.unsqueeze(1)).float()
            self.g_r.mul_(mask)
            torch.sum(self.g_r * torch.sign(self.r[l]), dim=1, out=self.g_theta[l])
            self.g_theta[l].neg_()
            # g_W_l = g_r^T e = -(g_r^T neg_e)
            ext.grad_w(self.g_r, self.neg_e[l], self.gW[l], -1.0, 0.0, bk_gw, prio)
            if on_grads is not None:
                on_grads([self.gW[l], self.g_theta[l], self.g_rho[l]])
            # g_e = g_r @ W_l
            ext.dec_fwd(self.g_r, W_l, self.ones_mn, self.zeros_bd, self.g_e,
                        self.scratch_lp, self.kc["bk_dec"] or bk, prio)
            # g_Ahat -= y_prev^T g_e
            ext.grad_w(self.y[l], self.g_e, self.gA, -1.0, 1.0, bk_gw, prio)
            # g_y_prev = g_r - g_e @ A_hat^T
            ext.enc_fwd(self.g_e, A, self.ones_mn, self.inv_norms, self.t_n,
                        self.scratch_lp, self.fired, 1, bk, prio)
            torch.sub(self.g_r, self.t_n, out=self.g_y)

        # x0 = y0: the remaining carry lands on y0 directly
        self.g_y.add_(self.g_xc)
        # y0 = b @ A_hat^T: g_Ahat += g_y0^T b
        ext.grad_w(self.g_y, b, self.gA, 1.0, 1.0, bk_gw, prio)
        if on_grads is not None:
            on_grads([self.gA])
        return B

    def update_phase(self, B: int):
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        step_no = st["step"]
        p = ens.params
        ext.project_adam(p["decoder"], self.gA, self.norms,
                         st["mu"]["decoder"], st["nu"]["decoder"], step_no,
                         self.n_dict, self.lr, self.beta1, self.beta2,
                         self.eps, EPS_NORM, True)
        for l in range(self.n_layers):
            mu = st["mu"]["encoder_layers"][l]["W"]
            nu = st["nu"]["encoder_layers"][l]["W"]
            ext.project_adam(p["encoder_layers"][l]["W"], self.gW[l], self.norms,
                             mu, nu, step_no, self.n_dict,
                             self.lr, self.beta1, self.beta2,
                             self.eps, EPS_NORM, False)
            ext.bias_adam(p["encoder_layers"][l]["theta"], self.g_theta[l],
                          self.zero_decay,
                          st["mu"]["encoder_layers"][l]["theta"],
                          st["nu"]["encoder_layers"][l]["theta"],
                          step_no, self.lr, self.beta1, self.beta2, self.eps)
            self._apply_rho_adam(l, step_no)

    def _apply_rho_adam(self, l, step_no):
        st = self.ens.optim_states
        rho = self.ens.params["encoder_layers"][l]["rho"]
        mu = st["mu"]["encoder_layers"][l]["rho"]
        nu = st["nu"]["encoder_layers"][l]["rho"]
        g = self.g_rho[l].reshape(rho.shape)
        t = step_no.reshape(rho.shape)
        mu.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
        nu.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
        bc1 = 1 - torch.pow(self.beta1, t)
        bc2 = 1 - torch.pow(self.beta2, t)
        rho.sub_(self.lr * (mu / bc1) / ((nu / bc2).sqrt() + self.eps))

    def _loss_data(self, B: int):
        mse = self.loss_parts[:, 0] / (B * self.d_act)
        l1 = self.l1_alpha * self._l1_sum / B
        return {"loss": mse + l1, "l_reconstruction": mse, "l_l1": l1}

    def step(self, minibatches: torch.Tensor, expand_dims: bool = True):
        if not expand_dims:
            raise NotImplementedError("per-model batches not supported by the HIP step")
        B = self.grads_phase(minibatches)
        self.update_phase(B)
        return self._loss_data(B), {"c": self.y[self.n_layers]}

    def dp_grad_tensors(self):
        return [self.gA] + self.gW + self.g_theta + self.g_rho


class HipTopKStep:
    """Fused TopK-encoder training step.

    Same GEMM pipeline as the tied SAE but: raw-score encoder (no bias/relu),
    per-row top-k selection + scatter + relu between encode and decode, no L1
    term, and the dictionary normalization has NO eps clamp
    (reference topk_encoder.py:31).  The per-model k lives in
    buffers["sparsity"]; selection runs as one torch.topk per model.
    """

    EPS = 1e-30  # un-clamped normalization (reference divides by the raw norm)

    def __init__(self, ensemble, ext):
        self.ens = ensemble
        self.ext = ext
        p = ensemble.params
        self.n_models, self.n_dict, self.d_act = p["dict"].shape
        dev = p["dict"].device
        self.ks = [int(k) for k in ensemble.buffers["sparsity"].reshape(-1).tolist()]
        self.zero_l1 = torch.zeros(self.n_models, device=dev)
        self.dummy_bias = torch.zeros(self.n_models, self.n_dict, device=dev)

        opt = ensemble.optimizer_kwargs
        self.lr = float(opt.get("lr", 1e-3))
        betas = opt.get("betas", (0.9, 0.999))
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(opt.get("eps", 1e-8))
        name = getattr(ensemble.optimizer_func, "__name__", "adam")
        if name != "adam":
            raise RuntimeError(f"fused HIP step supports adam only, got {name}")
        self._B = None

    def _alloc(self, B):
        M, n, d = self.n_models, self.n_dict, self.d_act
        dev = self.ens.params["dict"].device
        f = lambda *shape: torch.empty(shape, device=dev, dtype=torch.float32)
        self.scores = f(M, B, n)
        self.c = f(M, B, n)
        self.gpre = f(M, B, n)
        self.r = f(M, B, d)
        self.norms = f(M, n)
        self.inv_norms = f(M, n)
        self.loss_parts = f(M, 2)
        self.fired = torch.zeros(M, n, device=dev)
        self.g_bias_scratch = f(M, n)
        self.gw = f(M, n, d)
        self._B = B

    def grads_phase(self, x):
        ens, ext = self.ens, self.ext
        B = x.shape[0]
        if self._B != B:
            self._alloc(B)
        x = x.contiguous()
        W = ens.params["dict"]

        self.loss_parts.zero_()
        self.g_bias_scratch.zero_()

        ext.row_norms(W, self.norms, self.inv_norms, self.EPS)
        ext.enc_fwd(x, W, self.dummy_bias, self.inv_norms,
                    self.scores, self.loss_parts, self.fired, 1)
        # top-k selection per model (k varies across the ensemble)
        self.c.zero_()
        for m, k in enumerate(self.ks):
            sc = self.scores[m]
            top = torch.topk(sc, k, dim=-1)
            vals = torch.clamp(top.values, min=0.0)
            self.c[m].scatter_(-1, top.indices, vals)
        self.fired += (self.c > 0).float().sum(dim=1)

        ext.dec_fwd(self.c, W, self.inv_norms, x, self.r, self.loss_parts)
        ext.gc(self.r, W, self.inv_norms, self.c, self.zero_l1,
               self.gpre, self.g_bias_scratch)
        gscale = 2.0 / (B * self.d_act)
        ext.grad_w(self.c, self.r, self.gw, gscale, 0.0)
        ext.grad_w(self.gpre, x, self.gw, 1.0, 1.0)
        return B

    def update_phase(self, B):
        ens, ext = self.ens, self.ext
        st = ens.optim_states
        st["step"] += 1.0
        ext.project_adam(ens.params["dict"], self.gw, self.norms,
                         st["mu"]["dict"], st["nu"]["dict"], st["step"],
                         self.n_dict, self.lr, self.beta1, self.beta2,
                         self.eps, self.EPS, True)

    def _loss_data(self, B):
        mse = self.loss_parts[:, 0] / (B * self.d_act)
        return {"loss": mse}

    def step(self, minibatches, expand_dims=True):
        if not expand_dims:
            # per-model batches not needed: TopK stacks fine on this path
            raise NotImplementedError
        B = self.grads_phase(minibatches)
        self.update_phase(B)
        return self._loss_data(B), {"c": self.c}

    def dp_grad_tensors(self):
        return [self.gw]
