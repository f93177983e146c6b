"""Sweep driver: chunk-epoch training loop over ensembles of SAEs.

Parity with reference ``big_sweep.py`` (sweep :298-386, ensemble_train_loop
:159-199, unstacked_to_learned_dicts :202-225, init_model_dataset :240-266,
init_synthetic_dataset :269-295, checkpoint layout ``_{i}/learned_dicts.pt``
+ ``config.yaml`` at chunk counts 2^j and final :378-384).

Differences (MI355X build): wandb is optional (RunLogger JSONL always on);
the host LM is an HF transformers model (activation_dataset); dispatch
across GPUs goes through sweep/cluster_runs (process-per-ensemble) or the
RCCL DP trainer for a single big ensemble; and training is RESUMABLE — the
reference never saves optimizer state, here every checkpoint
also writes ``resume_state.pt`` (params + Adam moments + chunk cursor,
atomic rename) and ``cfg.resume=True`` continues from it.
"""

from __future__ import annotations

import datetime
import os
from itertools import product
from typing import Any, Dict, List, Tuple

import numpy as np
import torch
import torch.multiprocessing as mp
import yaml

from sparse_coding_amd.data.activation_dataset import get_activation_size, load_model, setup_data
from sparse_coding_amd.data.random_dataset import SparseMixDataset
from sparse_coding_amd.metrics import standard_metrics
from sparse_coding_amd.sweep.cluster_runs import dispatch_job_on_chunk
from sparse_coding_amd.utils.logging import RunLogger


def get_model(cfg):
    """Host LM + tokenizer.  Random-init from the local config table (no
    network); tokenizer is only needed for real text corpora."""
    model = load_model(cfg.model_name, device=cfg.device)
    tokenizer = None
    try:
        from transformers import AutoTokenizer

        tokenizer = AutoTokenizer.from_pretrained(cfg.model_name)
    except Exception:  # noqa: BLE001 - offline
        pass
    return model, tokenizer


def filter_learned_dicts(learned_dicts, hyperparam_filters):
    from math import isclose

    out = []
    for ld, hp in learned_dicts:
        ok = True
        for k, val in hyperparam_filters.items():
            if isinstance(val, float):
                ok &= isclose(hp[k], val, rel_tol=1e-3)
            else:
                ok &= hp[k] == val
        if ok:
            out.append((ld, hp))
    return out


def format_hyperparam_val(val):
    if isinstance(val, float):
        return f"{val:.2E}".replace("+", "")
    return str(val)


def make_hyperparam_name(setting):
    return "_".join(f"{k}_{format_hyperparam_val(v)}" for k, v in setting.items())


def ensemble_train_loop(ensemble, cfg, args, ensemble_name, sampler, dataset, progress_counter):
    """Inner loop run by each dispatched worker (reference :159-199).

    Adds a throughput counter the reference lacks: activations/sec per
    ensemble, logged at chunk end (SURVEY.md §5 "rebuild adds its own
    throughput counters").
    """
    import time as _time

    torch.set_grad_enabled(False)
    torch.manual_seed(0)
    np.random.seed(0)

    logger = getattr(cfg, "logger", None)

    # opt-in dead-feature resampling inside the sweep (the reference wires
    # resampling only into its DDP experiment; here any sweep can use the
    # anthropic/worst protocols — engine/resample.py).  The dead window is
    # one chunk (the reference's rule, huge_batch_size.py:230); the rewrite
    # lands in the shared-memory params, so the parent sees it like any
    # training update.
    resample_every = int(getattr(cfg, "resample_every_chunks", 0) or 0)
    resampler = None
    if resample_every:
        from sparse_coding_amd.engine.resample import EnsembleResampler

        resampler = EnsembleResampler(
            ensemble,
            n_track=int(getattr(cfg, "resample_n_track", 512)),
            protocol=getattr(cfg, "resample_protocol", "anthropic"),
            warmup_steps=int(getattr(cfg, "resample_warmup_steps", 1000)))

    # stage the whole chunk into this GPU's HBM once: a 2 GB chunk costs one
    # H2D copy amortized over ~1000 batches instead of a per-batch unpinned
    # H2D stall (288 GB per GPU — whole-chunk residency is the design)
    device = torch.device(args["device"])
    if device.type == "cuda" and not dataset.is_cuda:
        try:
            dataset = dataset.to(device, non_blocking=False)
        except torch.cuda.OutOfMemoryError:
            pass  # fall back to per-batch H2D below

    t0 = _time.perf_counter()
    n_acts = 0

    for i, batch_idxs in enumerate(sampler):
        batch = dataset[batch_idxs].to(args["device"])
        losses, aux = ensemble.step_batch(batch)
        n_acts += batch.shape[0]
        if resampler is not None:
            resampler.observe(batch, aux)

        if logger is not None and i % getattr(cfg, "log_every", 10) == 0:
            num_nonzero = aux["c"].count_nonzero(dim=-1).float().mean(dim=-1)
            log = {}
            for m in range(ensemble.n_models):
                hp = {}
                for ep in cfg.ensemble_hyperparams:
                    hp[ep] = args[ep]
                for bp in cfg.buffer_hyperparams:
                    hp[bp] = ensemble.buffers[bp][m].item()
                name = make_hyperparam_name(hp)
                for k in losses.keys():
                    log[f"{ensemble_name}_{name}_{k}"] = losses[k][m].item()
                log[f"{ensemble_name}_{name}_num_nonzero"] = num_nonzero[m].item()
            logger.log(log)

        progress_counter.value = i

    if resampler is not None and (getattr(cfg, "_chunk_i", 0) + 1) % resample_every == 0:
        counts = resampler.resample()
        if logger is not None:
            logger.log({f"{ensemble_name}_resampled": int(sum(counts))})

    if device.type == "cuda":
        torch.cuda.synchronize(device)
    dt = _time.perf_counter() - t0
    if logger is not None and dt > 0:
        logger.log({f"{ensemble_name}_acts_per_sec": n_acts / dt,
                    f"{ensemble_name}_chunk_wall_s": dt})


def unstacked_to_learned_dicts(ensemble, args, ensemble_hyperparams, buffer_hyperparams):
    """CPU LearnedDicts tagged with their hyperparams (reference :202-225)."""
    learned_dicts = []
    for params, buffers in ensemble.unstack(device="cpu"):
        hp_values: Dict[str, Any] = {}
        for ep in ensemble_hyperparams:
            if ep not in args:
                raise ValueError(f"Hyperparameter {ep} not found in args")
            hp_values[ep] = args[ep]
        for bp in buffer_hyperparams:
            if bp not in buffers:
                raise ValueError(f"Hyperparameter {bp} not found in buffers")
            hp_values[bp] = buffers[bp].item()
        learned_dicts.append((ensemble.sig.to_learned_dict(params, buffers), hp_values))
    return learned_dicts


def log_standard_metrics(learned_dicts, chunk, chunk_num, hyperparam_ranges, cfg):
    """Periodic image metrics: MMCS grids across dict sizes + sparsity
    histograms (reference :86-156)."""
    logger = getattr(cfg, "logger", None)
    if logger is None:
        return
    n_samples = min(2000, len(chunk))
    sample = chunk[np.random.choice(len(chunk), size=n_samples, replace=False)]

    for ld, setting in learned_dicts:
        name = make_hyperparam_name(setting)
        n_ever = standard_metrics.batched_calc_feature_n_ever_active(ld, sample, threshold=1)
        logger.log({f"{name}_n_active": n_ever, f"{name}_prop_active": n_ever / ld.n_feats}, commit=False)

    dict_sizes = hyperparam_ranges.get("dict_size", [])
    l1_values = hyperparam_ranges.get("l1_alpha", [])
    if len(dict_sizes) > 1 and len(l1_values) > 0:
        grid_hps = [k for k in hyperparam_ranges if k not in ("l1_alpha", "dict_size")]
        for setting_vals in product(*[hyperparam_ranges[k] for k in grid_hps]):
            setting = dict(zip(grid_hps, setting_vals))
            scores = np.zeros((len(l1_values), len(dict_sizes) - 1))
            ok = True
            for i, l1 in enumerate(l1_values):
                small = filter_learned_dicts(learned_dicts, {**setting, "l1_alpha": l1, "dict_size": dict_sizes[0]})
                if not small:
                    ok = False
                    break
                for j, ds in enumerate(dict_sizes[1:]):
                    larger = filter_learned_dicts(learned_dicts, {**setting, "l1_alpha": l1, "dict_size": ds})
                    if not larger:
                        ok = False
                        break
                    scores[i, j] = standard_metrics.mcs_duplicates(small[0][0], larger[0][0]).mean().item()
            if ok:
                fig = standard_metrics.plot_grid(scores, l1_values, dict_sizes[1:], "l1_alpha", "dict_size", cmap="viridis")
                logger.log_image(f"mmcs_grid_{chunk_num}/{make_hyperparam_name(setting)}", fig)

    for ld, setting in learned_dicts:
        fig = standard_metrics.plot_hist(
            standard_metrics.mean_nonzero_activations(ld, sample), "Mean nonzero activations", "Frequency", bins=20
        )
        logger.log_image(f"sparsity_hist_{chunk_num}/{make_hyperparam_name(setting)}", fig)


def generate_synthetic_dataset(cfg, generator, chunk_size, n_chunks):
    batch_size = generator.batch_size
    n_samples = chunk_size // batch_size
    for i in range(n_chunks):
        print(f"Generating chunk {i + 1}/{n_chunks}")
        chunk = torch.zeros((n_samples * batch_size, cfg.activation_width), dtype=torch.float32, device="cpu")
        for j in range(n_samples):
            chunk[j * batch_size : (j + 1) * batch_size] = generator.send(None).cpu()
        torch.save(chunk, os.path.join(cfg.dataset_folder, f"{i}.pt"))


def init_model_dataset(cfg) -> int:
    cfg.activation_width = get_activation_size(cfg.model_name, cfg.layer_loc)
    if len(os.listdir(cfg.dataset_folder)) == 0:
        print(f"Activations in {cfg.dataset_folder} do not exist, creating them")
        transformer, tokenizer = get_model(cfg)
        n = setup_data(
            tokenizer,
            transformer,
            dataset_name=cfg.dataset_name,
            dataset_folder=cfg.dataset_folder,
            layer=cfg.layer,
            layer_loc=cfg.layer_loc,
            n_chunks=cfg.n_chunks,
            device=cfg.device,
            chunk_size_gb=cfg.chunk_size_gb,
            center_dataset=cfg.center_dataset,
            model_name=cfg.model_name,
        )
        del transformer, tokenizer
        return n
    n = 0
    for f in os.listdir(cfg.dataset_folder):
        if f.endswith(".pt"):
            n += torch.load(os.path.join(cfg.dataset_folder, f), map_location="cpu").shape[0]
    return n


def init_synthetic_dataset(cfg) -> None:
    if len(os.listdir(cfg.dataset_folder)) != 0:
        print(f"Activations in {cfg.dataset_folder} already exist, loading them")
        return
    device = cfg.device if torch.cuda.is_available() else "cpu"
    generator = SparseMixDataset(
        cfg.activation_width,
        cfg.n_ground_truth_components,
        cfg.gen_batch_size,
        cfg.feature_num_nonzero,
        cfg.feature_prob_decay,
        cfg.noise_magnitude_scale,
        device,
        sparse_component_covariance=None
        if cfg.correlated_components
        else torch.eye(cfg.n_ground_truth_components, device=device),
        t_type=torch.float16,
    )
    chunk_size = int(cfg.chunk_size_gb * 1024**3)
    chunk_activations = chunk_size // (cfg.activation_width * 2)
    generate_synthetic_dataset(cfg, generator, chunk_activations, cfg.n_chunks)
    torch.save(generator, os.path.join(cfg.output_folder, "generator.pt"))


def sweep(ensemble_init_func, cfg):
    """The trainer entry point (reference :298-386)."""
    # the reference flips grad mode globally and leaves it off
    # (big_sweep.py:299); restore it on exit so library callers are safe
    _prev_grad_mode = torch.is_grad_enabled()
    torch.set_grad_enabled(False)
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
        mp.set_start_method("spawn", force=True)
    torch.manual_seed(0)
    np.random.seed(0)

    start_time = datetime.datetime.now().strftime("%Y%m%d-%H%M%S")
    os.makedirs(cfg.dataset_folder, exist_ok=True)
    os.makedirs(cfg.output_folder, exist_ok=True)

    cfg.logger = RunLogger(
        cfg.output_folder,
        name=f"ensemble_{cfg.model_name}_{start_time[4:]}",
        use_wandb=getattr(cfg, "use_wandb", False),
        config=cfg.as_dict() if hasattr(cfg, "as_dict") else None,
    )

    if cfg.use_synthetic_dataset:
        init_synthetic_dataset(cfg)
    else:
        init_model_dataset(cfg)

    print("Initialising ensembles...", end=" ")
    ensembles, ensemble_hyperparams, buffer_hyperparams, hyperparam_ranges = ensemble_init_func(cfg)
    cfg.ensemble_hyperparams = ensemble_hyperparams
    cfg.buffer_hyperparams = buffer_hyperparams
    print("Ensembles initialised.")

    chunk_files = [f for f in os.listdir(cfg.dataset_folder) if f.endswith(".pt")]
    n_chunks = len(chunk_files)
    chunk_order = np.random.permutation(n_chunks)
    if getattr(cfg, "n_repetitions", None) is not None:
        chunk_order = np.tile(chunk_order, cfg.n_repetitions)

    # -- resume (this framework; the reference has no resume path) ----------
    resume_path = os.path.join(cfg.output_folder, "resume_state.pt")
    start_chunk = 0
    if getattr(cfg, "resume", False) and os.path.exists(resume_path):
        saved = torch.load(resume_path, map_location="cpu", weights_only=False)
        if len(saved["chunk_order"]) == len(chunk_order):
            chunk_order = np.asarray(saved["chunk_order"])
        else:
            # run extended/shortened (e.g. n_repetitions raised to continue
            # training): keep the fresh order, whose prefix matches because
            # sweep() reseeds np before drawing it
            print("[resume] chunk_order length changed; keeping the new schedule")
        start_chunk = min(saved["chunk_pos"], len(chunk_order))
        for (ensemble, _, name), st in zip(ensembles, saved["ensemble_states"]):
            dev = ensemble.device
            ensemble.params = _to_device_tree(st["params"], dev)
            ensemble.optim_states = _to_device_tree(st["optim_states"], dev)
            ensemble.buffers = _to_device_tree(st["buffers"], dev)
            ensemble._hip_step = None
            ensemble.init_functions()
        print(f"Resumed from {resume_path} at chunk {start_chunk}/{len(chunk_order)}")

    means = None
    if getattr(cfg, "center_activations", False) and start_chunk > 0:
        means = torch.load(os.path.join(cfg.output_folder, "means.pt"), map_location="cpu")

    # persistent workers: one spawn per ensemble for the WHOLE sweep instead
    # of one per (ensemble, chunk) — kills the per-chunk CUDA-context spawn
    # cost (sweep/worker_pool.py)
    pool = None
    if getattr(cfg, "persistent_workers", False):
        from sparse_coding_amd.sweep.worker_pool import PersistentWorkerPool

        pool = PersistentWorkerPool(ensembles, cfg, ensemble_train_loop)

    learned_dicts = []
    if start_chunk > 0:
        # a fully-resumed run (nothing left to train) still returns dicts
        for ensemble, arg, _ in ensembles:
            learned_dicts.extend(unstacked_to_learned_dicts(ensemble, arg, ensemble_hyperparams, buffer_hyperparams))
    for i, chunk_idx in enumerate(chunk_order):
        cfg._chunk_i = i  # read by ensemble_train_loop's resample schedule
        if i < start_chunk:
            continue
        print(f"Chunk {i + 1}/{len(chunk_order)}")
        chunk = torch.load(os.path.join(cfg.dataset_folder, f"{chunk_idx}.pt")).to(device="cpu", dtype=torch.float32)
        if getattr(cfg, "center_activations", False):
            if means is None:
                means = chunk.mean(dim=0)
                torch.save(means, os.path.join(cfg.output_folder, "means.pt"))
            chunk -= means

        if pool is not None:
            pool.run_chunk(chunk)
        else:
            dispatch_job_on_chunk(ensembles, cfg, chunk, ensemble_train_loop)

        learned_dicts = []
        for ensemble, arg, _ in ensembles:
            learned_dicts.extend(unstacked_to_learned_dicts(ensemble, arg, ensemble_hyperparams, buffer_hyperparams))

        if getattr(cfg, "wandb_images", False) and i % 10 == 0:
            log_standard_metrics(learned_dicts, chunk, i, hyperparam_ranges, cfg)

        del chunk
        if i == len(chunk_order) - 1 or (i + 1) in [2**j for j in range(3, 10)]:
            iter_folder = os.path.join(cfg.output_folder, f"_{i}")
            os.makedirs(iter_folder, exist_ok=True)
            torch.save(learned_dicts, os.path.join(iter_folder, "learned_dicts.pt"))
            with open(os.path.join(iter_folder, "config.yaml"), "w") as f:
                cfg_dict = cfg.as_dict() if hasattr(cfg, "as_dict") else dict(cfg)
                cfg_dict.pop("logger", None)
                yaml.dump({k: v for k, v in cfg_dict.items() if isinstance(v, (int, float, str, bool, list, type(None)))}, f)
            _save_resume_state(resume_path, ensembles, chunk_order, i + 1)

    if pool is not None:
        pool.close()
    cfg.logger.close()
    torch.set_grad_enabled(_prev_grad_mode)
    return learned_dicts


def _to_device_tree(tree, device):
    from sparse_coding_amd.utils.tree import tree_map

    return tree_map(lambda t: t.to(device) if hasattr(t, "to") else t, tree)


def _save_resume_state(path, ensembles, chunk_order, chunk_pos):
    """Full training state (params + optimizer moments + chunk cursor),
    written atomically so a crash mid-save never corrupts the last good
    state.  The reference never persists optimizer state."""
    states = []
    for ensemble, _, name in ensembles:
        st = ensemble.state_dict()
        states.append({
            "name": name,
            "params": _to_device_tree(st["params"], "cpu"),
            "buffers": _to_device_tree(st["buffers"], "cpu"),
            "optim_states": _to_device_tree(st["optim_states"], "cpu"),
        })
    tmp = path + ".tmp"
    torch.save({"chunk_order": np.asarray(chunk_order).tolist(),
                "chunk_pos": int(chunk_pos),
                "ensemble_states": states}, tmp)
    os.replace(tmp, path)
