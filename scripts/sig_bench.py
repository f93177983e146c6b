"""Per-signature step-time comparison: fused HIP step vs the vmap oracle.

Runs every fused-step signature at a common shape and prints one JSON line
per (signature, backend) with ms/step — the coverage table for
profiles/README.md.  GPU box:  python scripts/sig_bench.py
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def build(sig_name, d, n, M, device):
    from sparse_coding_amd.models import positive, sae_signatures as sigs
    from sparse_coding_amd.models.lista import FunctionalLISTADenoisingSAE
    from sparse_coding_amd.models.topk import TopKEncoder

    l1s = np.logspace(-4, -3, M)
    no_stacking = False
    if sig_name == "tied":
        sig = sigs.FunctionalTiedSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "untied":
        sig = sigs.FunctionalSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "masked_tied":
        sig = sigs.FunctionalMaskedTiedSAE
        models = [sig.init(d, n // (2 ** (i % 2)), n, float(l1), device=device)
                  for i, l1 in enumerate(l1s)]
    elif sig_name == "thresholding":
        sig = sigs.FunctionalThresholdingSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "reverse":
        sig = sigs.FunctionalReverseSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "centered":
        sig = sigs.FunctionalTiedCenteredSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "whitened":
        sig = sigs.FunctionalTiedSAE
        torch.manual_seed(7)
        models = []
        for l1 in l1s:
            q, _ = torch.linalg.qr(torch.randn(d, d, device=device))
            models.append(sig.init(d, n, float(l1), device=device, rotation=q,
                                   translation=torch.randn(d, device=device) * 0.1,
                                   scaling=torch.rand(d, device=device) + 0.5))
    elif sig_name == "positive":
        sig = positive.FunctionalPositiveTiedSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    elif sig_name == "topk":
        sig = TopKEncoder
        models = [TopKEncoder.init(d, n, 32) for _ in range(M)]
        no_stacking = True  # vmap can't stack data-dependent topk
    elif sig_name == "lista":
        sig = FunctionalLISTADenoisingSAE
        models = [sig.init(d, n, 3, float(l1)) for l1 in l1s]
    elif sig_name == "residual":
        from sparse_coding_amd.models.lista import FunctionalResidualDenoisingSAE

        sig = FunctionalResidualDenoisingSAE
        models = [sig.init(d, n, 2, float(l1)) for l1 in l1s]
    elif sig_name == "semilinear":
        from sparse_coding_amd.models.semilinear import SemiLinearSAE

        sig = SemiLinearSAE
        models = [sig.init(d, n, float(l1), device=device) for l1 in l1s]
    else:
        raise ValueError(sig_name)
    return sig, models, no_stacking


def time_backend(sig_name, backend, args, device):
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam

    torch.manual_seed(0)
    sig, models, no_stacking = build(sig_name, args.d_model, args.dict_size,
                                     args.n_models, device)
    ens = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, device=device,
                             backend=backend,
                             no_stacking=no_stacking and backend == "torch")
    x = torch.randn(args.batch, args.d_model, device=device)
    for _ in range(args.warmup):
        ens.step_batch(x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        ens.step_batch(x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    return dt * 1e3, type(ens._hip_step).__name__ if ens._hip_step else "vmap"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=6)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--dict-size", type=int, default=4096)
    p.add_argument("--n-models", type=int, default=8)
    p.add_argument("--only", default="", help="run just one signature")
    p.add_argument("--backend", default="", help="run just one backend (hip|torch)")
    args = p.parse_args()
    device = "cuda:0"

    sig_names = ("tied", "untied", "masked_tied", "thresholding", "reverse",
                 "centered", "whitened", "positive", "topk", "lista",
                 "residual", "semilinear")
    if args.only:
        sig_names = tuple(s for s in sig_names if s == args.only)
    backends = (args.backend,) if args.backend else ("hip", "torch")
    for sig_name in sig_names:
        row = {"sig": sig_name}
        for backend in backends:
            try:
                ms, impl = time_backend(sig_name, backend, args, device)
                row[backend + "_ms"] = round(ms, 3)
                if backend == "hip":
                    row["impl"] = impl
            except Exception as e:  # noqa: BLE001
                row[backend + "_ms"] = f"error: {type(e).__name__}: {e}"[:120]
        if isinstance(row.get("hip_ms"), float) and isinstance(row.get("torch_ms"), float):
            row["speedup"] = round(row["torch_ms"] / row["hip_ms"], 2)
        print(json.dumps(row), flush=True)


if __name__ == "__main__":
    main()
