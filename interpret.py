"""Autointerp CLI (reference interpret.py:764-815 subcommands).

Subcommands:
  make_fragments  build + cache the per-feature activation table
  interpret       select records (+ optional LLM explain/score via a plugin)
  read_results    summarize + plot saved scores

The reference hard-requires an OpenAI secrets.json at import; here LLM
calls are a plugin: pass --explainer module.path:function to use one.
"""

from __future__ import annotations

import argparse
import importlib
import os
import sys

import torch

from sparse_coding_amd.config import InterpArgs  # noqa: F401 (API parity)
from sparse_coding_amd.data.activation_dataset import load_model, synthetic_token_batches
from sparse_coding_amd.interpret.interpret import (
    get_df,
    interpret_features,
    make_feature_activation_dataset,
    plot_scores,
    read_results,
)


def _load_plugin(spec: str):
    mod, fn = spec.split(":")
    return getattr(importlib.import_module(mod), fn)


def main(argv=None):
    p = argparse.ArgumentParser()
    sub = p.add_subparsers(dest="cmd", required=True)

    f = sub.add_parser("make_fragments")
    f.add_argument("--learned-dict", required=True)
    f.add_argument("--model-name", default="pythia-70m-deduped")
    f.add_argument("--layer", type=int, default=2)
    f.add_argument("--layer-loc", default="residual")
    f.add_argument("--n-fragments", type=int, default=256)
    f.add_argument("--df-n-feats", type=int, default=200)
    f.add_argument("--cache", default="interp_cache.pt")
    f.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")

    i = sub.add_parser("interpret")
    i.add_argument("--cache", default="interp_cache.pt")
    i.add_argument("--output-folder", default="interp_results")
    i.add_argument("--n-feats-explain", type=int, default=10)
    i.add_argument("--sort-mode", default="max")
    i.add_argument("--explainer", default="", help="module.path:function -> explain_fn(records)->str")
    i.add_argument("--simulator", default="", help="module.path:function -> simulate_fn(expl, records)->float")

    r = sub.add_parser("read_results")
    r.add_argument("--output-folder", default="interp_results")
    r.add_argument("--plot", default="scores.png")

    args = p.parse_args(argv)

    if args.cmd == "make_fragments":
        ld = torch.load(args.learned_dict, map_location="cpu", weights_only=False)
        if isinstance(ld, list):  # learned_dicts.pt: take the first entry
            ld = ld[0][0]
        ld.to_device(args.device)
        model = load_model(args.model_name, device=args.device)
        batches = list(synthetic_token_batches(model.config.vocab_size, 4, 64, args.n_fragments // 4))

        def build():
            return make_feature_activation_dataset(
                model, ld, args.layer, args.layer_loc, batches,
                device=args.device, max_features=args.df_n_feats,
            )

        get_df(args.cache, build, rebuild=True)
        print(f"wrote {args.cache}")

    elif args.cmd == "interpret":
        acts, tokens, feats = get_df(args.cache, lambda: (_ for _ in ()).throw(FileNotFoundError(args.cache)))
        explain_fn = _load_plugin(args.explainer) if args.explainer else None
        simulate_fn = _load_plugin(args.simulator) if args.simulator else None
        results = interpret_features(
            acts, tokens, feats[: args.n_feats_explain],
            explain_fn=explain_fn, simulate_fn=simulate_fn,
            output_folder=args.output_folder, sort_mode=args.sort_mode,
        )
        print(f"wrote {len(results)} feature records to {args.output_folder}")

    elif args.cmd == "read_results":
        recs = read_results(args.output_folder)
        scores = [r["score"] for r in recs.values() if r.get("score") == r.get("score")]
        print(f"{len(recs)} features, {len(scores)} scored")
        if scores:
            plot_scores({"results": scores}, save_path=args.plot)


if __name__ == "__main__":
    main()
