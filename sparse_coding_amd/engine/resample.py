"""Dead-neuron resampling for ensembles.

The reference implements resampling only in its DDP experiment
(huge_batch_size.py:224-254): dead = never fired over a window; dead encoder
rows re-initialized from the worst-reconstructed examples; Adam state zeroed
for the resampled slices.  Here the same rule is applied to the stacked
ensemble, entirely on-device:

* fired counts come for free from the fused forward kernel (k_enc_fwd
  accumulates them into HipSAEStep.fired) or are recomputed from aux["c"]
  on the torch backend;
* worst examples are tracked per model with a batched top-K merge
  (no host sync per step);
* the replacement writes + Adam-state zeroing are batched index ops on the
  stacked [M, n, d] tensors.

Replacement scale: unit worst-example direction × 0.2 × mean encoder row
norm (the convention of the resampling literature; the reference's literal
expression at :243 divides by the mean norm instead — with its transposed
weight layout — so the intent, re-scaling new rows well below typical rows,
is preserved rather than the expression).
"""

from __future__ import annotations

from typing import Optional

import torch


class EnsembleResampler:
    """Tracks fired counts + worst examples for a FunctionalEnsemble and
    resamples dead features on demand."""

    def __init__(self, ensemble, n_track: int = 512, encoder_norm_ratio: float = 0.2):
        self.ens = ensemble
        self.n_track = n_track
        self.encoder_norm_ratio = encoder_norm_ratio
        # TopK ensembles name their weight "dict"; SAEs use "encoder"
        self.w_key = "encoder" if "encoder" in ensemble.params else "dict"
        M, n, d = ensemble.params[self.w_key].shape
        dev = ensemble.params[self.w_key].device
        self.fired = torch.zeros(M, n, device=dev)
        self.worst_losses = torch.full((M, n_track), -float("inf"), device=dev)
        self.worst_examples = torch.zeros(M, n_track, d, device=dev)

    @torch.no_grad()
    def observe(self, batch: torch.Tensor, aux: Optional[dict] = None) -> None:
        """Update fired counts + worst-example pool after a step.

        On the HIP backend, per-example losses come from the residual
        workspace; on the torch backend from aux["c"] and a re-decode.
        """
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

        # batched top-K merge of the worst examples
        M = per_ex.shape[0]
        losses = torch.cat([self.worst_losses, per_ex], dim=1)
        top = torch.topk(losses, self.n_track, dim=1)
        self.worst_losses = top.values
        examples = torch.cat(
            [self.worst_examples, batch.unsqueeze(0).expand(M, *batch.shape)], dim=1
        )
        self.worst_examples = torch.gather(
            examples, 1, top.indices[:, :, None].expand(-1, -1, examples.shape[-1])
        )

    @torch.no_grad()
    def resample(self) -> torch.Tensor:
        """Replace never-fired features; returns per-model replacement counts."""
        ens = self.ens
        p = ens.params
        st = ens.optim_states
        wk = self.w_key
        M, n, d = p[wk].shape

        if p[wk].is_cuda:
            from sparse_coding_amd import ops as _ops

            ext = _ops.get_extension(required=True)
            return self._resample_fused(ext)
        counts = torch.zeros(M, dtype=torch.long)

        for m in range(M):
            dead = torch.where(self.fired[m] == 0)[0]
            k = min(int(dead.numel()), self.n_track)
            if k == 0:
                continue
            dead = dead[:k]
            worst = self.worst_examples[m, :k]
            worst_unit = worst / torch.clamp(torch.norm(worst, dim=-1, keepdim=True), 1e-8)
            avg_norm = torch.norm(p[wk][m], dim=-1).mean()

            p[wk][m, dead] = worst_unit * self.encoder_norm_ratio * avg_norm
            st["mu"][wk][m, dead] = 0
            st["nu"][wk][m, dead] = 0
            if "decoder" in p:
                p["decoder"][m, dead] = worst_unit
                st["mu"]["decoder"][m, dead] = 0
                st["nu"]["decoder"][m, dead] = 0
            if "encoder_bias" in p:
                p["encoder_bias"][m, dead] = 0
                st["mu"]["encoder_bias"][m, dead] = 0
                st["nu"]["encoder_bias"][m, dead] = 0
            counts[m] = k

        self.fired.zero_()
        self.worst_losses.fill_(-float("inf"))
        return counts

    @torch.no_grad()
    def _resample_fused(self, ext) -> torch.Tensor:
        """K14 fully on device (ops k_resample): one launch ranks the dead
        features per model in index order, rewrites rows from the worst-
        example pool, and zeroes the Adam slices — no host loop or sync."""
        ens = self.ens
        p = ens.params
        st = ens.optim_states
        wk = self.w_key
        M, n, d = p[wk].shape
        dev = p[wk].device

        worst_unit = self.worst_examples / torch.clamp(
            torch.norm(self.worst_examples, dim=-1, keepdim=True), 1e-8)
        avg_norm = torch.norm(p[wk], dim=-1).mean(dim=1)  # [M]
        enc_scale = (self.encoder_norm_ratio * avg_norm).contiguous()
        counts = torch.zeros(M, device=dev, dtype=torch.int32)

        kwargs = {}
        if "decoder" in p:
            kwargs = dict(dec=p["decoder"], mu_d=st["mu"]["decoder"], nu_d=st["nu"]["decoder"])
        bkw = {}
        if "encoder_bias" in p:
            bkw = dict(bias=p["encoder_bias"], mu_b=st["mu"]["encoder_bias"],
                       nu_b=st["nu"]["encoder_bias"])
        ext.resample(self.fired, worst_unit.contiguous(), enc_scale,
                     p[wk], st["mu"][wk], st["nu"][wk],
                     counts_out=counts, **kwargs, **bkw)

        self.fired.zero_()
        self.worst_losses.fill_(-float("inf"))
        return counts.long().cpu()
