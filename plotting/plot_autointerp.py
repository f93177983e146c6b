"""Autointerp score comparisons across dictionary families.

Covers reference ``plotting/plot_autointerp_*.py``: violin/means of
autointerp scores per dict family (trained SAEs at several L1s vs PCA / ICA /
neuron baselines), on the fixed -0.2..0.6 axis used throughout the reference.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse
import os
from typing import Dict, List

import numpy as np

from sparse_coding_amd.interpret.interpret import plot_scores, read_results


def collect_scores(results_root: str) -> Dict[str, List[float]]:
    """Each subfolder of results_root is one dict family's interpretation
    output (feature_*.json files)."""
    out: Dict[str, List[float]] = {}
    for entry in sorted(os.listdir(results_root)):
        folder = os.path.join(results_root, entry)
        if not os.path.isdir(folder):
            continue
        recs = read_results(folder)
        scores = [r["score"] for r in recs.values() if np.isfinite(r.get("score", float("nan")))]
        if scores:
            out[entry] = scores
    return out


def summarize(scores: Dict[str, List[float]]) -> str:
    lines = [f"{'family':30s} {'n':>5} {'mean':>7} {'median':>7}"]
    for name, vals in sorted(scores.items()):
        lines.append(f"{name:30s} {len(vals):>5} {np.mean(vals):7.3f} {np.median(vals):7.3f}")
    return "\n".join(lines)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--results-root", required=True)
    p.add_argument("--out", default="autointerp_scores.png")
    args = p.parse_args()
    scores = collect_scores(args.results_root)
    print(summarize(scores))
    plot_scores(scores, save_path=args.out)


if __name__ == "__main__":
    main()
