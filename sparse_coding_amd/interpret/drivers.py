"""Autointerp drivers: the reference's folder/grouped/baseline/chunk runners
and score readers (interpret.py:388-761), with paths parameterized instead of
the reference's hard-coded ``/mnt/ssd-cluster`` mounts, and the LLM protocol
served by an injectable client (protocol.py) instead of the OpenAI API.

Driver surface (reference symbol -> here):
  run                        :388  -> run
  run_folder                 :414  -> run_folder
  make_tag_name              :426  -> make_tag_name
  run_from_grouped           :439  -> run_from_grouped
  read_transform_scores      :456  -> read_transform_scores
  read_scores                :488  -> read_scores
  parse_folder_name          :505  -> parse_folder_name
  run_list_of_learned_dicts  :522  -> run_list_of_learned_dicts
  interpret_across_baselines :539  -> interpret_across_baselines
  interpret_across_big_sweep :583  -> interpret_across_big_sweep
  interpret_across_chunks    :648  -> interpret_across_chunks
  get_score                  :404  -> get_score
"""

from __future__ import annotations

import copy
import os
from datetime import datetime
from typing import Dict, List, Tuple

import torch


# ---------------------------------------------------------------------------
# score readers (reference :404-503) — format-compatible with the reference's
# explanation.txt layout
# ---------------------------------------------------------------------------

def get_score(lines: List[str], mode: str) -> float:
    """Reference get_score (:404-413): pull one of the three aggregate scores
    out of an explanation.txt's lines."""
    if mode == "top":
        return float(lines[-3].split(" ")[-1])
    if mode == "random":
        return float(lines[-2].split(" ")[-1])
    if mode == "top_random":
        score_line = [line for line in lines if "Score: " in line][0]
        return float(score_line.split(" ")[1])
    raise ValueError(f"Unknown mode: {mode}")


def read_transform_scores(transform_loc: str, score_mode: str,
                          verbose: bool = False) -> Tuple[List[int], List[float]]:
    """Reference :456-486: scan feature_N folders under one transform."""
    ndxs: List[int] = []
    scores: List[float] = []
    if not os.path.isdir(transform_loc):
        return ndxs, scores
    feat_folders = [x for x in os.listdir(transform_loc) if x.startswith("feature_")]
    for feature_folder in feat_folders:
        feature_ndx = int(feature_folder.split("_")[1])
        expl = os.path.join(transform_loc, feature_folder, "explanation.txt")
        if not os.path.exists(expl):
            continue
        # unfiltered split: get_score's -3/-2 indexing counts the trailing
        # empty line of the file, exactly as the reference does (:478-481)
        lines = open(expl).read().split("\n")
        score = get_score(lines, score_mode)
        if verbose:
            print(f"{feature_ndx=}, {transform_loc=}, {score=}")
        ndxs.append(feature_ndx)
        scores.append(score)
    return ndxs, scores


def read_scores(results_folder: str, score_mode: str = "top") -> Dict[str, Tuple[List[int], List[float]]]:
    """Reference :488-503: all transforms under a results folder, the
    'sparse_coding' transform listed first."""
    assert score_mode in ("top", "random", "top_random")
    scores: Dict[str, Tuple[List[int], List[float]]] = {}
    transforms = [t for t in os.listdir(results_folder)
                  if os.path.isdir(os.path.join(results_folder, t))]
    if "sparse_coding" in transforms:
        transforms.remove("sparse_coding")
        transforms = ["sparse_coding"] + transforms
    for transform in transforms:
        ndxs, sc = read_transform_scores(os.path.join(results_folder, transform), score_mode)
        if ndxs:
            scores[transform] = (ndxs, sc)
    return scores


def make_tag_name(hparams: Dict) -> str:
    """Reference :426-436."""
    tag = ""
    if "tied" in hparams:
        tag += f"tied_{hparams['tied']}"
    if "dict_size" in hparams:
        tag += f"dict_size_{hparams['dict_size']}"
    if "l1_alpha" in hparams:
        tag += f"l1_alpha_{hparams['l1_alpha']:.2}"
    if "bias_decay" in hparams:
        tag += "0.0" if hparams["bias_decay"] == 0 else f"{hparams['bias_decay']:.1}"
    return tag


def parse_folder_name(folder_name: str) -> Tuple[str, str, int, float, str]:
    """Reference :505-519: e.g. tied_residual_l2_r4 -> components."""
    tied, layer_loc, layer_str, ratio_str, *extras = folder_name.split("_")
    extra_str = "_".join(extras) if extras else ""
    layer = int(layer_str[1:])
    ratio = float(ratio_str[1:])
    if ratio == 0:
        ratio = 0.5
    return tied, layer_loc, layer, ratio, extra_str


# ---------------------------------------------------------------------------
# the single-dict runner (reference run :388-401)
# ---------------------------------------------------------------------------

def run(learned_dict, cfg, client=None, model=None, token_batches=None,
        n_fragments: int = 256) -> Dict[int, float]:
    """Build (or load) the fragment activation table for one learned dict and
    run the explain/simulate protocol over cfg.n_feats_explain features.

    client: any object with batch_complete(prompts)->responses; defaults to
    the deterministic offline MockLLMClient.  model: the host LM (loaded
    from cfg.model_name when omitted).  token_batches: fragment token
    sources; synthetic when omitted (no-network environment).
    """
    from sparse_coding_amd.data.activation_dataset import load_model, synthetic_token_batches
    from sparse_coding_amd.interpret.interpret import get_df, make_feature_activation_dataset
    from sparse_coding_amd.interpret.protocol import MockLLMClient, interpret_protocol

    assert cfg.df_n_feats >= cfg.n_feats_explain
    if client is None:
        client = MockLLMClient()
    device = cfg.device
    tokenizer = None
    if model is None:
        model = load_model(cfg.model_name, device=device)
    tokenizer = getattr(model, "_sc_tokenizer", None)
    if token_batches is None:
        token_batches = list(synthetic_token_batches(
            model.config.vocab_size, 4, 64, max(1, n_fragments // 4)))

    os.makedirs(cfg.save_loc or ".", exist_ok=True)
    cache = os.path.join(cfg.save_loc or ".", "fragment_table.pt")

    def build():
        learned_dict.to_device(device)
        return make_feature_activation_dataset(
            model, learned_dict, cfg.layer, cfg.layer_loc, token_batches,
            tokenizer=tokenizer, device=device, max_features=cfg.df_n_feats)

    acts, tokens, feats = get_df(cache, build)
    return interpret_protocol(
        acts, tokens, feats[: cfg.n_feats_explain], client,
        cfg.save_loc or ".", tokenizer=tokenizer)


def run_list_of_learned_dicts(dicts: List[Tuple[str, object]], cfg, **kw):
    """Reference :522-529."""
    out = {}
    for name, ld in dicts:
        print(f"Running {name}")
        sub = copy.deepcopy(cfg)
        sub.save_loc = os.path.join(cfg.save_loc or ".", name)
        out[name] = run(ld, sub, **kw)
    return out


def run_folder(cfg, **kw):
    """Reference run_folder (:414-423): every .pt/.pkl dict in a folder."""
    base_folder = cfg.load_interpret_autoencoder
    encoders = [x for x in os.listdir(base_folder) if x.endswith((".pt", ".pkl"))]
    print(f"Found {len(encoders)} encoders in {base_folder}")
    out = {}
    for i, encoder in enumerate(encoders):
        print(f"Running encoder {i} of {len(encoders)}: {encoder}")
        ld = torch.load(os.path.join(base_folder, encoder),
                        map_location="cpu", weights_only=False)
        sub = copy.deepcopy(cfg)
        sub.save_loc = os.path.join(cfg.save_loc or "auto_interp_results", encoder)
        out[encoder] = run(ld, sub, **kw)
    return out


def run_from_grouped(cfg, results_loc: str, out_base: str = "auto_interp_results", **kw):
    """Reference :439-454: split a learned_dicts.pt into per-dict files named
    by hyperparameters, then run_folder over them."""
    results = torch.load(results_loc, map_location="cpu", weights_only=False)
    time_str = datetime.now().strftime("%Y-%m-%d_%H-%M-%S")
    group_dir = os.path.join(out_base, time_str)
    os.makedirs(group_dir, exist_ok=True)
    for learned_dict, hparams_dict in results:
        filename = make_tag_name(hparams_dict) + ".pt"
        torch.save(learned_dict, os.path.join(group_dir, filename))
    cfg = copy.deepcopy(cfg)
    cfg.load_interpret_autoencoder = group_dir
    return run_folder(cfg, **kw)


# ---------------------------------------------------------------------------
# sweep-scale drivers (reference :539-688), path-parameterized
# ---------------------------------------------------------------------------

def interpret_across_baselines(baselines_dir: str, save_dir: str, cfg,
                               layer_loc: str = "residual", **kw):
    """Reference :539-580: every baseline dict per layer folder (l{N}_{loc})."""
    os.makedirs(save_dir, exist_ok=True)
    out = {}
    for folder in sorted(os.listdir(baselines_dir)):
        layer_str, floc = folder.split("_", 1)
        if floc != layer_loc:
            continue
        for baseline_file in sorted(os.listdir(os.path.join(baselines_dir, folder))):
            if "nmf" in baseline_file or not baseline_file.endswith(".pt"):
                continue
            sub = copy.deepcopy(cfg)
            sub.layer = int(layer_str[1:])
            sub.layer_loc = floc
            sub.save_loc = os.path.join(save_dir, folder, baseline_file[:-3])
            ld = torch.load(os.path.join(baselines_dir, folder, baseline_file),
                            map_location="cpu", weights_only=False)
            out[f"{folder}/{baseline_file}"] = run(ld, sub, **kw)
    return out


def _matching_encoder(autoencoders, l1_val: float):
    matching = [ae for ae in autoencoders if abs(ae[1]["l1_alpha"] - l1_val) < 1e-4]
    if len(matching) != 1:
        print(f"Found {len(matching)} matching encoders")
    return matching[0][0]


def interpret_across_big_sweep(base_dir: str, save_dir: str, cfg, l1_val: float,
                               n_chunks_training: int = 10, ratio: float = 2,
                               layer_loc: str = "residual", tied: str = "tied", **kw):
    """Reference :583-645: the l1-matched dict from every sweep folder."""
    os.makedirs(save_dir, exist_ok=True)
    out = {}
    for folder in sorted(os.listdir(base_dir)):
        try:
            f_tied, f_loc, layer, f_ratio, extra = parse_folder_name(folder)
        except Exception:  # noqa: BLE001 - reference's bare except (:597)
            continue
        if f_loc != layer_loc or f_tied != tied or f_ratio != ratio or extra:
            continue
        ckpt = os.path.join(base_dir, folder, f"_{n_chunks_training - 1}", "learned_dicts.pt")
        autoencoders = torch.load(ckpt, map_location="cpu", weights_only=False)
        enc = _matching_encoder(autoencoders, l1_val)
        sub = copy.deepcopy(cfg)
        sub.layer = layer
        sub.layer_loc = f_loc
        sub.save_loc = os.path.join(save_dir, f"l{layer}_{f_loc}", f"{f_tied}_r{f_ratio}_l1a{l1_val:.2}")
        out[folder] = run(enc, sub, **kw)
    return out


def interpret_across_chunks(base_dir: str, save_dir: str, cfg, l1_val: float,
                            chunks: Tuple[int, ...] = (1, 4, 16, 32), **kw):
    """Reference :648-688: the same dict at several training-chunk counts."""
    os.makedirs(save_dir, exist_ok=True)
    out = {}
    for folder in sorted(os.listdir(base_dir)):
        for n_chunks in chunks:
            tied, layer_loc, layer, ratio, _ = parse_folder_name(folder)
            if layer != cfg.layer:
                continue
            ckpt = os.path.join(base_dir, folder, f"_{n_chunks - 1}", "learned_dicts.pt")
            if not os.path.exists(ckpt):
                continue
            autoencoders = torch.load(ckpt, map_location="cpu", weights_only=False)
            enc = _matching_encoder(autoencoders, l1_val)
            sub = copy.deepcopy(cfg)
            sub.layer = layer
            sub.layer_loc = layer_loc
            sub.save_loc = os.path.join(
                save_dir, f"l{layer}_{layer_loc}", f"{tied}_r{ratio}_nc{n_chunks}_l1a{l1_val:.2}")
            out[f"{folder}@{n_chunks}"] = run(enc, sub, **kw)
    return out
