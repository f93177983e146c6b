"""Dead-neuron resampling for ensembles.

The reference implements resampling only in its DDP experiment
(huge_batch_size.py:224-254): dead = never fired over a window; dead encoder
rows re-initialized from the worst-reconstructed examples; Adam state zeroed
for the resampled slices.  Here the same rule is applied to the stacked
ensemble, entirely on-device, with two selectable protocols:

``protocol="worst"`` (the reference's rule, round-1 default): replacement
directions come from the top-``n_track`` worst-reconstructed examples.

``protocol="anthropic"`` (the retention protocol of the resampling
literature; default for long runs): replacement directions are sampled
WITHOUT replacement with probability proportional to the SQUARED
per-example reconstruction loss, via an Efraimidis–Spirakis weighted
reservoir maintained on device (key = log(u)/w, keep the ``n_track``
largest keys — an exact loss^2-weighted sample of the whole observation
window, not just its worst tail).  On top of the replacement writes:

* the encoder row is scaled to ``encoder_norm_ratio`` x the mean norm of
  the ALIVE encoder rows (not all rows — dead rows shrink under decay and
  would drag the target norm down);
* the per-feature learning rate of every replaced feature is ramped from
  ``warmup_start`` x lr back to lr over ``warmup_steps`` steps, so Adam
  cannot blast the fresh direction away on its first batches.  The ramp is
  written in place into ``HipSAEStep.lr_mult``, which the fused Adam
  kernels (k_project_adam / k_bias_adam) read every step — hipGraph-safe
  because the tensor pointer never changes.  The torch/vmap oracle backend
  ignores the warmup (it is a per-step numerical reference, not the
  production trainer).

Mechanics shared by both protocols:

* fired counts come for free from the fused forward kernel (k_enc_fwd
  accumulates them into HipSAEStep.fired) or are recomputed from aux["c"]
  on the torch backend;
* the example pool is merged per step with a batched top-K (no host sync);
* the replacement writes + Adam-state zeroing run in one k_resample launch
  on GPU (batched index ops on the stacked [M, n, d] tensors on CPU).
"""

from __future__ import annotations

from typing import Optional

import torch


class EnsembleResampler:
    """Tracks fired counts + a replacement-example pool for a
    FunctionalEnsemble and resamples dead features on demand."""

    def __init__(self, ensemble, n_track: int = 512, encoder_norm_ratio: float = 0.2,
                 protocol: str = "worst", warmup_steps: int = 1000,
                 warmup_start: float = 0.1, dead_threshold: Optional[float] = None):
        if protocol not in ("worst", "anthropic"):
            raise ValueError(f"unknown resample protocol {protocol!r}")
        self.ens = ensemble
        self.n_track = n_track
        self.encoder_norm_ratio = encoder_norm_ratio
        self.protocol = protocol
        self.warmup_steps = warmup_steps
        self.warmup_start = warmup_start
        # dead = fires on fewer than this fraction of observed examples.
        # 0 = the reference's literal never-fired rule (huge_batch_size.py:230);
        # the anthropic protocol defaults to 1e-5 because a feature firing
        # once per ~1M examples is dead for every practical purpose yet
        # never trips the ==0 rule over a long window.
        if dead_threshold is None:
            dead_threshold = 1e-5 if protocol == "anthropic" else 0.0
        self.dead_threshold = dead_threshold
        self.examples_seen = 0
        # TopK ensembles name their weight "dict"; SAEs use "encoder"
        if "encoder" in ensemble.params:
            self.w_key = "encoder"
        elif "dict" in ensemble.params:
            self.w_key = "dict"
        else:
            raise ValueError(
                "EnsembleResampler supports single-matrix encoders (tied/untied "
                "SAEs, TopK); this ensemble's params "
                f"({sorted(ensemble.params)}) have no 'encoder'/'dict' weight "
                "(multi-layer encoders have no per-feature row to resample)")
        M, n, d = ensemble.params[self.w_key].shape
        dev = ensemble.params[self.w_key].device
        self.fired = torch.zeros(M, n, device=dev)
        # pool scores: per-example loss ("worst") or reservoir key ("anthropic")
        self.pool_scores = torch.full((M, n_track), -float("inf"), device=dev)
        self.pool_examples = torch.zeros(M, n_track, d, device=dev)
        self._warmup_mask: Optional[torch.Tensor] = None  # [M, n] bool
        self._warmup_t = 0

    # round-1 attribute names kept as aliases (scripts/tests referenced them)
    @property
    def worst_losses(self):
        return self.pool_scores

    @property
    def worst_examples(self):
        return self.pool_examples

    @torch.no_grad()
    def observe(self, batch: torch.Tensor, aux: Optional[dict] = None) -> None:
        """Update fired counts + the example pool after a step, and advance
        any active post-resample lr warmup.

        On the HIP backend, per-example losses come from the residual
        workspace; on the torch backend from aux["c"] and a re-decode.
        """
        self.examples_seen += batch.shape[0]
        hs = getattr(self.ens, "_hip_step", None)
        if hs is not None:
            self.fired += hs.fired
            hs.fired.zero_()
            per_ex = hs.r.pow(2).mean(dim=-1)  # [M, B]
        else:
            c = aux["c"]  # [M, B, n]
            self.fired += (c > 0).float().sum(dim=1)
            # reconstruct residuals through the learned dict per model
            from sparse_coding_amd.models.learned_dict import normalize_rows

            p = self.ens.params
            w = p.get("decoder", p[self.w_key])
            what = normalize_rows(w)
            x_hat = torch.einsum("mbn,mnd->mbd", c, what)
            per_ex = (x_hat - batch.unsqueeze(0)).pow(2).mean(dim=-1)

        if self.protocol == "anthropic":
            # Efraimidis–Spirakis A-Res: key = log(u)/w with w = loss^2;
            # keeping the n_track largest keys over the stream is an exact
            # weighted sample without replacement of the window
            w = per_ex.pow(2).clamp_min(1e-30)
            u = torch.rand_like(per_ex).clamp_min(1e-12)
            scores = torch.log(u) / w
        else:
            scores = per_ex

        # batched top-K merge of the pool
        M = per_ex.shape[0]
        cat_scores = torch.cat([self.pool_scores, scores], dim=1)
        top = torch.topk(cat_scores, self.n_track, dim=1)
        self.pool_scores = top.values
        examples = torch.cat(
            [self.pool_examples, batch.unsqueeze(0).expand(M, *batch.shape)], dim=1
        )
        self.pool_examples = torch.gather(
            examples, 1, top.indices[:, :, None].expand(-1, -1, examples.shape[-1])
        )
        self._tick_warmup()

    @torch.no_grad()
    def _tick_warmup(self) -> None:
        if self._warmup_mask is None:
            return
        hs = getattr(self.ens, "_hip_step", None)
        if hs is None or not hasattr(hs, "lr_mult"):
            self._warmup_mask = None
            return
        self._warmup_t += 1
        if self._warmup_t >= self.warmup_steps:
            hs.lr_mult.fill_(1.0)
            self._warmup_mask = None
            return
        frac = self._warmup_t / self.warmup_steps
        factor = self.warmup_start + (1.0 - self.warmup_start) * frac
        hs.lr_mult.fill_(1.0)
        hs.lr_mult.masked_fill_(self._warmup_mask, factor)

    def _effective_fired(self) -> torch.Tensor:
        """fired counts with sub-threshold (rare-firing) features zeroed, so
        the ==0 dead rule sees them as dead."""
        if self.dead_threshold <= 0.0 or self.examples_seen == 0:
            return self.fired
        min_count = self.dead_threshold * self.examples_seen
        return torch.where(self.fired < min_count, torch.zeros_like(self.fired), self.fired)

    def _replaced_mask(self) -> torch.Tensor:
        """[M, n] bool: the dead features the replacement pass will rewrite
        (first n_track dead per model, in index order — k_resample's rule)."""
        dead = self._effective_fired() == 0
        rank = torch.cumsum(dead.to(torch.int32), dim=1) - 1
        return dead & (rank < self.n_track)

    def _enc_scale(self) -> torch.Tensor:
        """Per-model replacement norm: ratio x mean ALIVE row norm under the
        anthropic protocol, ratio x mean row norm under "worst" (round-1 /
        reference behavior)."""
        p = self.ens.params
        norms = torch.norm(p[self.w_key], dim=-1)  # [M, n]
        if self.protocol == "anthropic":
            alive = (self._effective_fired() > 0).float()
            denom = alive.sum(dim=1).clamp_min(1.0)
            mean_norm = (norms * alive).sum(dim=1) / denom
            # all-dead model: fall back to the all-rows mean
            mean_norm = torch.where(denom > 1.0, mean_norm, norms.mean(dim=1))
        else:
            mean_norm = norms.mean(dim=1)
        return self.encoder_norm_ratio * mean_norm  # [M]

    @torch.no_grad()
    def resample(self) -> torch.Tensor:
        """Replace never-fired features; returns per-model replacement counts."""
        ens = self.ens
        p = ens.params
        st = ens.optim_states
        wk = self.w_key
        M, n, d = p[wk].shape

        if self.protocol == "anthropic":
            replaced = self._replaced_mask()

        if p[wk].is_cuda:
            from sparse_coding_amd import ops as _ops

            ext = _ops.get_extension(required=True)
            counts = self._resample_fused(ext)
        else:
            counts = self._resample_cpu()

        if self.protocol == "anthropic":
            hs = getattr(ens, "_hip_step", None)
            if hs is not None and hasattr(hs, "lr_mult") and bool(replaced.any()):
                if self._warmup_mask is None:
                    self._warmup_mask = replaced
                else:
                    self._warmup_mask |= replaced
                self._warmup_t = 0
                hs.lr_mult.fill_(1.0)
                hs.lr_mult.masked_fill_(self._warmup_mask, self.warmup_start)
        return counts

    @torch.no_grad()
    def _resample_cpu(self) -> torch.Tensor:
        ens = self.ens
        p = ens.params
        st = ens.optim_states
        wk = self.w_key
        M, n, d = p[wk].shape
        counts = torch.zeros(M, dtype=torch.long)
        enc_scale = self._enc_scale()

        eff = self._effective_fired()
        for m in range(M):
            dead = torch.where(eff[m] == 0)[0]
            k = min(int(dead.numel()), self.n_track)
            if k == 0:
                continue
            dead = dead[:k]
            pool = self.pool_examples[m, :k]
            pool_unit = pool / torch.clamp(torch.norm(pool, dim=-1, keepdim=True), 1e-8)

            p[wk][m, dead] = pool_unit * enc_scale[m]
            st["mu"][wk][m, dead] = 0
            st["nu"][wk][m, dead] = 0
            if "decoder" in p:
                p["decoder"][m, dead] = pool_unit
                st["mu"]["decoder"][m, dead] = 0
                st["nu"]["decoder"][m, dead] = 0
            if "encoder_bias" in p:
                p["encoder_bias"][m, dead] = 0
                st["mu"]["encoder_bias"][m, dead] = 0
                st["nu"]["encoder_bias"][m, dead] = 0
            counts[m] = k

        self.fired.zero_()
        self.examples_seen = 0
        self.pool_scores.fill_(-float("inf"))
        return counts

    @torch.no_grad()
    def _resample_fused(self, ext) -> torch.Tensor:
        """K14 fully on device (ops k_resample): one launch ranks the dead
        features per model in index order, rewrites rows from the example
        pool, and zeroes the Adam slices — no host loop or sync."""
        ens = self.ens
        p = ens.params
        st = ens.optim_states
        wk = self.w_key
        M, n, d = p[wk].shape
        dev = p[wk].device

        pool_unit = self.pool_examples / torch.clamp(
            torch.norm(self.pool_examples, dim=-1, keepdim=True), 1e-8)
        enc_scale = self._enc_scale().contiguous()
        counts = torch.zeros(M, device=dev, dtype=torch.int32)

        kwargs = {}
        if "decoder" in p:
            kwargs = dict(dec=p["decoder"], mu_d=st["mu"]["decoder"], nu_d=st["nu"]["decoder"])
        bkw = {}
        if "encoder_bias" in p:
            bkw = dict(bias=p["encoder_bias"], mu_b=st["mu"]["encoder_bias"],
                       nu_b=st["nu"]["encoder_bias"])
        ext.resample(self._effective_fired().contiguous(), pool_unit.contiguous(), enc_scale,
                     p[wk], st["mu"][wk], st["nu"][wk],
                     counts_out=counts, **kwargs, **bkw)

        self.fired.zero_()
        self.examples_seen = 0
        self.pool_scores.fill_(-float("inf"))
        return counts.long().cpu()
