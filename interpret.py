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
from sparse_coding_amd.interpret import drivers as interp_drivers  # noqa: F401 (API parity)


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
    r.add_argument("--score-mode", default="top", choices=["top", "random", "top_random", "all"])
    r.add_argument("--protocol-layout", action="store_true",
                   help="read the reference feature_N/explanation.txt layout (drivers.read_scores)")

    # reference CLI surface (interpret.py:764-815): run / run_group / chunks
    pr = sub.add_parser("run", help="full explain+simulate protocol on one learned dict")
    pr.add_argument("--learned-dict", required=True)
    pr.add_argument("--model-name", default="pythia-70m-deduped")
    pr.add_argument("--layer", type=int, default=2)
    pr.add_argument("--layer-loc", default="residual")
    pr.add_argument("--n-feats-explain", type=int, default=10)
    pr.add_argument("--df-n-feats", type=int, default=200)
    pr.add_argument("--n-fragments", type=int, default=256)
    pr.add_argument("--save-loc", default="auto_interp_results/run")
    pr.add_argument("--client", default="mock", help="mock | hf:<local model path>")
    pr.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")

    g = sub.add_parser("run_group", help="split a learned_dicts.pt and run the protocol per dict")
    g.add_argument("--results-loc", required=True)
    g.add_argument("--model-name", default="pythia-70m-deduped")
    g.add_argument("--layer", type=int, default=2)
    g.add_argument("--layer-loc", default="residual")
    g.add_argument("--n-feats-explain", type=int, default=10)
    g.add_argument("--out-base", default="auto_interp_results")
    g.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")

    c = sub.add_parser("chunks", help="protocol across chunk-count checkpoints of a sweep dir")
    c.add_argument("--base-dir", required=True)
    c.add_argument("--save-dir", default="auto_interp_results_overtime")
    c.add_argument("--l1-val", type=float, default=8.577e-4)
    c.add_argument("--layer", type=int, default=2)
    c.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")

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
        if args.protocol_layout:
            modes = ["top", "random", "top_random"] if args.score_mode == "all" else [args.score_mode]
            for mode in modes:
                by_name = interp_drivers.read_scores(args.output_folder, mode)
                flat = {name: sc for name, (ndxs, sc) in by_name.items()}
                print(f"[{mode}] " + ", ".join(f"{k}: n={len(v)}" for k, v in flat.items()))
                if flat:
                    plot_scores(flat, save_path=args.plot.replace(".png", f"_{mode}.png"))
        else:
            recs = read_results(args.output_folder)
            scores = [r["score"] for r in recs.values() if r.get("score") == r.get("score")]
            print(f"{len(recs)} features, {len(scores)} scored")
            if scores:
                plot_scores({"results": scores}, save_path=args.plot)

    elif args.cmd in ("run", "run_group", "chunks"):
        from sparse_coding_amd.config import InterpArgs

        cfg = InterpArgs()
        cfg.model_name = args.model_name if hasattr(args, "model_name") else cfg.model_name
        cfg.layer = args.layer
        cfg.device = args.device
        if hasattr(args, "layer_loc"):
            cfg.layer_loc = args.layer_loc
        if hasattr(args, "n_feats_explain"):
            cfg.n_feats_explain = args.n_feats_explain
            cfg.df_n_feats = max(getattr(args, "df_n_feats", 200), args.n_feats_explain)
        if args.cmd == "run":
            client = None
            if args.client.startswith("hf:"):
                from transformers import AutoModelForCausalLM, AutoTokenizer

                from sparse_coding_amd.interpret.protocol import HFLocalClient

                path = args.client[3:]
                m = AutoModelForCausalLM.from_pretrained(path).to(args.device)
                t = AutoTokenizer.from_pretrained(path)
                client = HFLocalClient(m, t, device=args.device)
            cfg.save_loc = args.save_loc
            ld = torch.load(args.learned_dict, map_location="cpu", weights_only=False)
            if isinstance(ld, list):
                ld = ld[0][0]
            scores = interp_drivers.run(ld, cfg, client=client, n_fragments=args.n_fragments)
            print(f"scored {len(scores)} features -> {cfg.save_loc}")
        elif args.cmd == "run_group":
            out = interp_drivers.run_from_grouped(cfg, args.results_loc, out_base=args.out_base)
            print(f"ran {len(out)} dicts -> {args.out_base}")
        else:
            out = interp_drivers.interpret_across_chunks(args.base_dir, args.save_dir, cfg, args.l1_val)
            print(f"ran {len(out)} (dict, chunk) combos -> {args.save_dir}")


if __name__ == "__main__":
    main()
