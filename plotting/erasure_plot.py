"""The erasure study driver + plots (reference plotting/erasure_plot.py,
C27): compute per-layer concept-erasure scores on a labeled prompt set and
render the reference's five figures.

Subcommands:
  compute  — for each layer: capture prompt activations from the host LM,
             fit LEACE / mean / mean-affine / dict-feature / random-feature
             erasers, score probe prediction ability + mean edit magnitude +
             distributional KL, and save per-layer .pt files in the
             reference's schemas (eval_layer_{L}_{task}.pt,
             kl_div_scores_layer_{L}.pt, general_{L}_{task}.pt).
  scores-across-depth   — reference plot_scores_across_depth (:199-215)
  leace-across-depth    — reference plot_leace_scores_across_depth (:129-197)
  kl-across-depth       — reference plot_kl_div_across_depth (:283-337)
  erasure-scores        — reference plot_erasure_scores (:59-127)
  bottleneck-scores     — reference plot_bottleneck_scores (:12-57)
"""

from __future__ import annotations

import argparse
import itertools
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import os


import torch


def _plt():
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    return plt


# ---------------------------------------------------------------------------
# compute
# ---------------------------------------------------------------------------

def compute_layer(acts: torch.Tensor, labels: torch.Tensor, learned_dict=None,
                  ks=(1, 4, 16)) -> dict:
    """Score every erasure method on one layer's labeled activations.

    Returns the reference eval-file schema: {"base": score,
    "leace"/"means"/"mean_affine": (score, mean_edit),
    "dict"/"random": [(k, score, mean_edit), ...]} plus a "kl" dict.
    Score = probe prediction ability (logistic AUROC); KL = symmetric KL of a
    Gaussian fit of erased vs original activations (distribution shift)."""
    from sparse_coding_amd.metrics.standard_metrics import logistic_regression_auroc
    from sparse_coding_amd.sweep.erasure import LeaceEraser, ablate_top_features, mean_erase

    def score(a):
        return logistic_regression_auroc(a, labels, max_iter=200)

    def gauss_kl(a):
        mu0, s0 = acts.mean(dim=0), acts.var(dim=0) + 1e-6
        mu1, s1 = a.mean(dim=0), a.var(dim=0) + 1e-6
        kl01 = 0.5 * ((s0 / s1) + (mu1 - mu0) ** 2 / s1 - 1 + torch.log(s1 / s0)).sum()
        kl10 = 0.5 * ((s1 / s0) + (mu0 - mu1) ** 2 / s0 - 1 + torch.log(s0 / s1)).sum()
        return 0.5 * (kl01 + kl10).item()

    out: dict = {"base": score(acts)}
    kl: dict = {}

    leace = LeaceEraser.fit(acts, labels)
    a = leace(acts)
    out["leace"] = (score(a), leace.mean_edit(acts))
    kl["LEACE"] = gauss_kl(a)

    a = mean_erase(acts, labels)
    out["means"] = (score(a), (a - acts).norm(dim=-1).mean().item())
    kl["means"] = gauss_kl(a)

    a = mean_erase(acts, labels, affine=True)
    out["mean_affine"] = (score(a), (a - acts).norm(dim=-1).mean().item())
    kl["mean_affine"] = gauss_kl(a)

    if learned_dict is not None:
        drows, rrows, dkl, rkl = [], [], [], []
        n = learned_dict.n_feats
        g = torch.Generator().manual_seed(0)
        for k in ks:
            a = ablate_top_features(learned_dict, acts, labels, k)
            drows.append((k, score(a), (a - acts).norm(dim=-1).mean().item()))
            dkl.append(gauss_kl(a))
            # random-feature control: ablate k random features
            code = learned_dict.encode(learned_dict.center(acts))
            ridx = torch.randperm(n, generator=g)[:k]
            removed = code[:, ridx] @ learned_dict.get_learned_dict()[ridx]
            a = acts - learned_dict.uncenter(removed) + learned_dict.uncenter(torch.zeros_like(removed))
            rrows.append((k, score(a), (a - acts).norm(dim=-1).mean().item()))
            rkl.append(gauss_kl(a))
        out["dict"] = drows
        out["random"] = rrows
        kl["dict"] = dkl
        kl["random"] = rkl
    out["kl"] = kl
    return out


def cmd_compute(args):
    from sparse_coding_amd.data.activation_dataset import capture_activation_hook, load_model
    from sparse_coding_amd.data.eval_prompts import gender_prompt_batch

    device = args.device
    model = load_model(args.model_name, device=device)
    ld = None
    if args.learned_dict:
        ld = torch.load(args.learned_dict, map_location="cpu", weights_only=False)
        if isinstance(ld, list):
            ld = ld[0][0]
    tokens, labels = gender_prompt_batch(n=args.n_prompts, vocab_size=model.config.vocab_size,
                                         seq_len=args.seq_len)
    tokens = tokens.to(device)
    os.makedirs(args.out_dir, exist_ok=True)
    layers = [int(x) for x in args.layers.split(",")]
    for layer in layers:
        store = []
        with torch.no_grad(), capture_activation_hook(model, layer, args.layer_loc, store):
            model(input_ids=tokens)
        acts = store[0].float().reshape(tokens.shape[0], tokens.shape[1], -1)[:, -1].cpu()
        res = compute_layer(acts, labels, learned_dict=ld,
                            ks=[int(k) for k in args.ks.split(",")])
        kl = res.pop("kl")
        torch.save(res, os.path.join(args.out_dir, f"eval_layer_{layer}_{args.task}.pt"))
        torch.save({"leace": res["leace"], "mean": res["means"],
                    "mean_affine": res["mean_affine"], "base": res["base"]},
                   os.path.join(args.out_dir, f"general_{layer}_{args.task}.pt"))
        torch.save(kl, os.path.join(args.out_dir, f"kl_div_scores_layer_{layer}.pt"))
        print(f"layer {layer}: base={res['base']:.3f} leace={res['leace'][0]:.3f} "
              f"means={res['means'][0]:.3f}")
    print(f"wrote {args.out_dir}")


# ---------------------------------------------------------------------------
# plots (readers of the computed .pt files, reference figure layouts)
# ---------------------------------------------------------------------------

def do_dataset_plot(files, name, layers, title, out_dir):
    """Reference do_dataset_plot (:216-280): two stacked panels, prediction
    ability + mean edit magnitude, per method across layers."""
    plt = _plt()
    base_score = files[0]["base"]
    fig, (ax2, ax1) = plt.subplots(2, 1, sharex=True)
    for ax in (ax1, ax2):
        ax.grid(True, alpha=0.5, linestyle="dashed")
        ax.set_axisbelow(True)
        ax.set_xticks(range(len(layers)))
        ax.set_xticklabels(layers)
    series = [("Mean", "means", "x", "orange"), ("Dict. Feature", "dict", ".", "green"),
              ("Rand. Feature", "random", ".", "red")]
    for label, key, marker, color in series:
        if any(key not in f for f in files):
            continue  # e.g. no learned dict supplied at compute time
        vals = [f[key] for f in files]
        sc = [v[0][1] if isinstance(v, list) else v[0] for v in vals]
        ed = [v[0][2] if isinstance(v, list) else v[1] for v in vals]
        ax1.plot(sc, label=label, marker=marker, color=color)
        ax2.plot(ed, label=label, marker=marker, color=color)
    ax1.axhline(y=base_score, color="red", linestyle="dashed", label="Base Perf.")
    ax1.axhline(y=0.5, color="grey", linestyle="dashed", label="Majority")
    ax1.set_ylabel("Model Prediction Ability")
    ax2.set_xlabel("Layer")
    ax2.set_ylabel("Mean Edit Magnitude")
    ax2.set_ylim(bottom=0)
    handles, labels_ = ax1.get_legend_handles_labels()
    ax2.legend(handles, labels_, loc="upper center", facecolor="white", framealpha=1, ncol=2)
    fig.suptitle(title)
    path = os.path.join(out_dir, f"erasure_across_depth_{name}.png")
    fig.savefig(path)
    print(f"saved {path}")


def cmd_scores_across_depth(args):
    layers = [int(x) for x in args.layers.split(",")]
    files = [torch.load(os.path.join(args.out_dir, f"eval_layer_{l}_{args.task}.pt"),
                        weights_only=False) for l in layers]
    do_dataset_plot(files, args.task, layers, "Concept Erasure on the Primary Task", args.out_dir)
    if args.transfer_task:
        tfiles = [torch.load(os.path.join(args.out_dir, f"eval_layer_{l}_{args.transfer_task}.pt"),
                             weights_only=False) for l in layers]
        do_dataset_plot(tfiles, args.transfer_task, layers,
                        "Transferred Concept Erasure on the Secondary Task", args.out_dir)


def cmd_leace_across_depth(args):
    plt = _plt()
    layers = [int(x) for x in args.layers.split(",")]
    files = [torch.load(os.path.join(args.out_dir, f"general_{l}_{args.task}.pt"),
                        weights_only=False) for l in layers]
    base_score = files[0]["base"]
    fig, (ax2, ax1) = plt.subplots(2, 1, sharex=True)
    for ax in (ax1, ax2):
        ax.grid(True, alpha=0.5, linestyle="dashed")
        ax.set_axisbelow(True)
        ax.set_xticks(range(len(layers)))
        ax.set_xticklabels(layers)
    for label, key, marker in (("LEACE", "leace", "+"), ("Mean", "mean", "x"),
                               ("Mean, Affine", "mean_affine", ".")):
        ax1.plot([f[key][0] for f in files], label=label, marker=marker)
        ax2.plot([f[key][1] for f in files], label=label, marker=marker)
    ax1.axhline(y=base_score, color="red", linestyle="dashed", label="Base Perf.")
    ax1.axhline(y=0.5, color="grey", linestyle="dashed", label="Majority")
    ax1.set_ylabel("Model Prediction Ability")
    ax2.set_xlabel("Layer")
    ax2.set_ylabel("Mean Edit Magnitude")
    ax2.set_ylim(bottom=0)
    handles, labels_ = ax1.get_legend_handles_labels()
    ax2.legend(handles, labels_, loc="upper center", facecolor="white", framealpha=1, ncol=2)
    fig.suptitle(args.title)
    path = os.path.join(args.out_dir, f"erasure_across_depth_general_{args.task}.png")
    fig.savefig(path)
    print(f"saved {path}")


def cmd_kl_across_depth(args):
    plt = _plt()
    layers = [int(x) for x in args.layers.split(",")]
    files = [torch.load(os.path.join(args.out_dir, f"kl_div_scores_layer_{l}.pt"),
                        weights_only=False) for l in layers]
    fig, ax1 = plt.subplots(1, 1, figsize=(6, 3))
    ax1.grid(True, alpha=0.5, linestyle="dashed")
    ax1.set_axisbelow(True)
    for label, key, marker in (("LEACE", "LEACE", "+"), ("Mean", "means", "x"),
                               ("Dict. Feature", "dict", "."), ("Rand. Feature", "random", ".")):
        vals = [f[key] for f in files if key in f]
        if not vals:
            continue
        pts = [v[0] if isinstance(v, list) else v for v in vals]
        ax1.plot(pts, label=label, marker=marker)
    ax1.set_xticks(range(len(layers)))
    ax1.set_xticklabels(layers)
    ax1.set_yscale("log")
    ax1.set_xlabel("Layer")
    ax1.set_ylabel("KL-Divergence")
    fig.suptitle("KL-Divergence From Base Model Under Erasure")
    ax1.legend(facecolor="white", framealpha=1, loc="upper left")
    fig.tight_layout()
    path = os.path.join(args.out_dir, "kl_across_depth.png")
    fig.savefig(path)
    print(f"saved {path}")


def cmd_erasure_scores(args):
    """Reference plot_erasure_scores (:59-127): per-method scatter of
    prediction ability vs edit magnitude and vs KL."""
    plt = _plt()
    layer = args.layers.split(",")[0]
    res = torch.load(os.path.join(args.out_dir, f"eval_layer_{layer}_{args.task}.pt"),
                     weights_only=False)
    kl = torch.load(os.path.join(args.out_dir, f"kl_div_scores_layer_{layer}.pt"),
                    weights_only=False)
    colors = ["red", "blue", "green", "orange", "purple"]
    markers = ["x", "+", "*", "o", "v"]
    for xaxis, xlab, fname in ((1, "Mean Edit", "erasure_by_edit_magnitude.png"),
                               (2, "KL Divergence", "erasure_by_kl_div.png")):
        fig, ax = plt.subplots()
        for i, key in enumerate(("leace", "means", "mean_affine", "dict", "random")):
            if key not in res:
                continue
            v = res[key]
            if isinstance(v, list):
                ys = [r[1] for r in v]
                xs = [r[2] for r in v] if xaxis == 1 else kl.get(key, [0] * len(v))
            else:
                ys = [v[0]]
                xs = [v[1]] if xaxis == 1 else [kl.get(key.upper() if key == "leace" else key, 0)]
            ax.scatter(xs, ys, c=colors[i], marker=markers[i], label=key, alpha=0.5)
        ax.axhline(y=res["base"], color="red", linestyle="dashed", label="Base")
        ax.set_xlabel(xlab)
        ax.set_ylabel("Prediction Ability")
        ax.legend()
        path = os.path.join(args.out_dir, fname)
        fig.savefig(path)
        print(f"saved {path}")


def cmd_bottleneck_scores(args):
    """Reference plot_bottleneck_scores (:12-57): per-key task metric vs
    bottleneck size from a dict_scores .pt of {key: [(tau, graph, metric,
    corruption), ...]}."""
    plt = _plt()
    scores = torch.load(args.scores, weights_only=False)
    colors = ["red", "blue", "green", "orange", "purple", "brown", "pink", "gray", "olive", "cyan"]
    styles = ["solid", "dashed", "dashdot", "dotted"]
    fig, ax = plt.subplots()
    for (style, color), (key, score) in zip(itertools.product(styles, colors), scores.items()):
        tau, graph, task_metric, corruption = zip(*score)
        ax.plot([len(g) for g in graph], task_metric, c=color, linestyle=style,
                label=key, alpha=0.5)
    ax.set_xlabel("Bottleneck Size")
    ax.set_ylabel("Per-Task Metric")
    ax.legend(fontsize=6)
    path = os.path.join(args.out_dir, "bottleneck_scores.png")
    os.makedirs(args.out_dir, exist_ok=True)
    fig.savefig(path)
    print(f"saved {path}")


def main(argv=None):
    p = argparse.ArgumentParser()
    sub = p.add_subparsers(dest="cmd", required=True)

    c = sub.add_parser("compute")
    c.add_argument("--model-name", default="pythia-410m")
    c.add_argument("--layers", default="0,2,4,6,8,10,12,14,16,18,20,22")
    c.add_argument("--layer-loc", default="residual")
    c.add_argument("--learned-dict", default="")
    c.add_argument("--task", default="gender")
    c.add_argument("--n-prompts", type=int, default=256)
    c.add_argument("--seq-len", type=int, default=16)
    c.add_argument("--ks", default="1,4,16")
    c.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    c.add_argument("--out-dir", default="output_erasure")

    for name in ("scores-across-depth", "leace-across-depth", "kl-across-depth", "erasure-scores"):
        s = sub.add_parser(name)
        s.add_argument("--out-dir", default="output_erasure")
        s.add_argument("--layers", default="0,2,4,6,8,10,12,14,16,18,20,22")
        s.add_argument("--task", default="gender")
        s.add_argument("--transfer-task", default="")
        s.add_argument("--title", default="various settings")

    b = sub.add_parser("bottleneck-scores")
    b.add_argument("--scores", required=True)
    b.add_argument("--out-dir", default="graphs")

    args = p.parse_args(argv)
    {"compute": cmd_compute, "scores-across-depth": cmd_scores_across_depth,
     "leace-across-depth": cmd_leace_across_depth, "kl-across-depth": cmd_kl_across_depth,
     "erasure-scores": cmd_erasure_scores, "bottleneck-scores": cmd_bottleneck_scores}[args.cmd](args)


if __name__ == "__main__":
    main()
