"""Distributed-path tests on CPU: gloo world_size=2 DP equivalence, cluster
dispatch, big-SAE resampling."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.functional.optim import adam
from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

D, N, B = 16, 32, 128


def _make_ensemble(seed=0, device="cpu"):
    torch.manual_seed(seed)
    models = [FunctionalTiedSAE.init(D, N, 1e-3) for _ in range(2)]
    return FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=device, backend="torch")


def _dp_worker(rank, world_size, port, batch, out_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)

        from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer, shard_batch

        ens = _make_ensemble(seed=0)  # same seed → identical replicas
        trainer = DataParallelEnsembleTrainer(ens, bucket_bytes=1 << 16)
        for _ in range(3):
            local = shard_batch(batch, rank, world_size)
            trainer.step(local)
        if rank == 0:
            # numpy → pickled by value: the child may exit before the parent
            # drains the queue (torch tensors would ship an fd to a dead sharer)
            out_q.put({k: v.detach().cpu().numpy() for k, v in ens.params.items()})
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put({"_error": f"rank {rank}: {traceback.format_exc()}"})
        raise


@pytest.mark.timeout(120)
def test_dp_matches_single_process():
    """2-rank sharded training == single-process full-batch training."""
    torch.manual_seed(42)
    batch = torch.randn(B, D)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29531
    procs = [ctx.Process(target=_dp_worker, args=(r, 2, port, batch, q)) for r in range(2)]
    for p in procs:
        p.start()
    dp_params = q.get(timeout=100)
    assert "_error" not in dp_params, dp_params.get("_error")
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    ens = _make_ensemble(seed=0)
    for _ in range(3):
        # full-batch grad == mean of shard grads (all losses are means)
        ens.step_batch(batch)

    for k in dp_params:
        assert torch.allclose(torch.from_numpy(dp_params[k]), ens.params[k], atol=1e-5), k


def _cluster_job(ensemble, cfg, args, name, sampler, dataset, progress_counter):
    for i, idxs in enumerate(sampler):
        batch = dataset[idxs].to(args["device"])
        ensemble.step_batch(batch)
        progress_counter.value = i


@pytest.mark.timeout(120)
def test_cluster_dispatch_two_ensembles():
    from types import SimpleNamespace

    from sparse_coding_amd.sweep.cluster_runs import dispatch_job_on_chunk

    mp.set_start_method("spawn", force=True)
    chunk = torch.randn(512, D)
    e1, e2 = _make_ensemble(seed=1), _make_ensemble(seed=2)
    before = e1.params["encoder"].clone()
    cfg = SimpleNamespace(batch_size=128, show_progress=False)
    ensembles = [
        (e1, {"batch_size": 128, "device": "cpu"}, "a"),
        (e2, {"batch_size": 128, "device": "cpu"}, "b"),
    ]
    dispatch_job_on_chunk(ensembles, cfg, chunk, _cluster_job)
    # children trained through shared memory → parent sees updated params
    assert not torch.allclose(before, e1.params["encoder"])


def test_big_sae_resampling(tmp_path):
    from sparse_coding_amd.parallel.big_sae import (
        BigSAE,
        WorstExampleTracker,
        resample_dead_features,
        train_big_sae,
    )

    torch.manual_seed(0)
    model = BigSAE(D, N, 1e-3)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    x = torch.randn(64, D)
    for _ in range(3):
        opt.zero_grad()
        loss, mse, l1, c, per_ex = model(x)
        loss.backward()
        opt.step()

    tracker = WorstExampleTracker(N, D, "cpu")
    tracker.update(x, per_ex.detach())
    c_totals = (c > 0).float().sum(0).detach()
    c_totals[:5] = 0  # force 5 dead features
    n = resample_dead_features(model, opt, c_totals, tracker)
    assert n == 5
    # adam state zeroed on those rows
    st = opt.state[model.encoder]
    assert (st["exp_avg"][:5] == 0).all()
    assert (st["exp_avg"][5:] != 0).any()
    # replaced encoder rows have the scaled norm
    enc_norms = torch.norm(model.encoder[:5], dim=-1)
    assert (enc_norms > 0).all()

    # single-process training loop over chunk files
    for i in range(2):
        torch.save(torch.randn(256, D, dtype=torch.float16), tmp_path / f"{i}.pt")
    m = train_big_sae(
        [str(tmp_path / f"{i}.pt") for i in range(2)],
        activation_size=D, n_features=N, batch_size=64,
        reinit_every_chunks=1, device="cpu", log_fn=lambda *a: None,
    )
    assert isinstance(m, BigSAE)


def test_grad_bucket_allreducer_layout():
    from sparse_coding_amd.parallel.dp import GradBucketAllReducer

    grads = {"a": torch.randn(100), "b": torch.randn(300), "c": torch.randn(50)}
    red = GradBucketAllReducer(grads, bucket_bytes=1024)  # 256 floats per bucket
    total = sum(b.numel() for b in red.buckets)
    assert total == 450
    # without dist init, all_reduce_ is a no-op
    before = {k: v.clone() for k, v in grads.items()}
    red.all_reduce_(grads)
    for k in grads:
        assert torch.equal(before[k], grads[k])


@pytest.mark.timeout(300)
def test_persistent_pool_matches_dispatch(tmp_path):
    """cfg.persistent_workers=True (one spawn per ensemble for the whole
    sweep) must produce bit-identical dicts to the per-chunk dispatcher."""
    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    def make_cfg(sub, persistent):
        cfg = SyntheticEnsembleArgs()
        cfg.use_synthetic_dataset = True
        cfg.activation_width = 16
        cfg.n_ground_truth_components = 24
        cfg.gen_batch_size = 256
        cfg.feature_num_nonzero = 3
        cfg.noise_magnitude_scale = 0.0
        cfg.chunk_size_gb = 16 * 256 * 4 * 2 / 1024**3
        cfg.n_chunks = 2
        cfg.n_repetitions = 2
        cfg.batch_size = 128
        cfg.device = "cpu"
        cfg.dataset_folder = str(tmp_path / sub / "data")
        cfg.output_folder = str(tmp_path / sub / "out")
        cfg.use_wandb = False
        cfg.wandb_images = False
        cfg.persistent_workers = persistent
        return cfg

    def init_func(c):
        return make_grid_ensembles(c, FunctionalTiedSAE, [1e-4, 1e-3], [1.0], devices=["cpu"])

    dicts_a = big_sweep.sweep(init_func, make_cfg("a", False))
    dicts_b = big_sweep.sweep(init_func, make_cfg("b", True))
    for (ld_a, hp_a), (ld_b, hp_b) in zip(dicts_a, dicts_b):
        assert hp_a == hp_b
        assert torch.equal(ld_a.get_learned_dict(), ld_b.get_learned_dict())
        assert torch.equal(ld_a.encoder_bias, ld_b.encoder_bias)


def _shard_worker(rank, world_size, port, chunk, out_q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          WORLD_SIZE=str(world_size), RANK=str(rank), LOCAL_RANK=str(rank))
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        from sparse_coding_amd.parallel.chunk_feed import ShardedEnsembleRunner
        from sparse_coding_amd.sweep.big_sweep import ensemble_train_loop

        torch.manual_seed(100 + rank)  # DIFFERENT ensemble per rank
        l1 = [1e-4, 1e-3][rank]
        models = [FunctionalTiedSAE.init(D, N, l1) for _ in range(2)]
        ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                                 device="cpu", backend="torch")

        class Cfg:
            batch_size = 64
            logger = None
            log_every = 1000

        runner = ShardedEnsembleRunner(ens, Cfg(), {"batch_size": 64, "device": "cpu",
                                                    "l1_alpha": l1, "dict_size": N},
                                       f"shard{rank}", ensemble_train_loop, "cpu")
        # only rank 0 holds the chunk; others receive it by broadcast
        runner.run_chunk(chunk if rank == 0 else None)
        dicts = runner.gather_learned_dicts([], ["l1_alpha"])
        if rank == 0:
            out_q.put({
                "n_dicts": len(dicts),
                "l1s": sorted(round(hp["l1_alpha"], 6) for _, hp in dicts),
                "first_dict": dicts[0][0].get_learned_dict().numpy(),
            })
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put({"_error": f"rank {rank}: {traceback.format_exc()}"})
        raise


@pytest.mark.timeout(120)
def test_sharded_ensemble_chunk_broadcast():
    """P1 over collectives: rank 0 broadcasts the chunk, each rank trains a
    DIFFERENT ensemble, dicts gather back to rank 0."""
    torch.manual_seed(7)
    chunk = torch.randn(512, D)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_shard_worker, args=(r, 2, 29553, chunk, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = q.get(timeout=110)
    for p in procs:
        p.join(timeout=30)
    assert "_error" not in res, res.get("_error")
    assert res["n_dicts"] == 4  # 2 ranks x 2 models
    assert res["l1s"] == [0.0001, 0.0001, 0.001, 0.001]
    import numpy as np

    assert np.isfinite(res["first_dict"]).all()


def _sharded_sweep_worker(rank, world_size, port, tmpdir, out_q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          WORLD_SIZE=str(world_size), RANK=str(rank), LOCAL_RANK=str(rank))
        from sparse_coding_amd.config import SyntheticEnsembleArgs
        from sparse_coding_amd.sweep.sharded_sweep import sharded_sweep

        cfg = SyntheticEnsembleArgs()
        cfg.use_synthetic_dataset = True
        cfg.activation_width = 16
        cfg.n_ground_truth_components = 24
        cfg.gen_batch_size = 256
        cfg.feature_num_nonzero = 3
        cfg.noise_magnitude_scale = 0.0
        cfg.chunk_size_gb = 16 * 256 * 4 * 2 / 1024**3
        cfg.n_chunks = 2
        cfg.batch_size = 128
        cfg.dataset_folder = os.path.join(tmpdir, "data")
        cfg.output_folder = os.path.join(tmpdir, "out")
        cfg.use_wandb = False
        cfg.ensemble_hyperparams = ["dict_size"]
        cfg.buffer_hyperparams = ["l1_alpha"]

        def init_for_rank(c, r, w):
            torch.manual_seed(200 + r)
            l1 = [1e-4, 1e-3][r]
            models = [FunctionalTiedSAE.init(16, 32, l1) for _ in range(2)]
            ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                                     device="cpu", backend="torch")
            return ens, {"batch_size": 128, "device": "cpu", "dict_size": 32, "l1_alpha": l1}, f"r{r}"

        dicts = sharded_sweep(init_for_rank, cfg)
        if rank == 0:
            out_q.put({"n": len(dicts),
                       "ckpt": os.path.exists(os.path.join(cfg.output_folder, "_1", "learned_dicts.pt"))})
        import torch.distributed as dist_

        dist_.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put({"_error": f"rank {rank}: {traceback.format_exc()}"})
        raise


@pytest.mark.timeout(180)
def test_sharded_sweep_end_to_end(tmp_path):
    """The torchrun-able sharded sweep on gloo world 2: rank 0 generates +
    broadcasts chunks, each rank trains its grid slice, checkpoints gather
    in the reference layout."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_sharded_sweep_worker, args=(r, 2, 29563, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = q.get(timeout=170)
    for p in procs:
        p.join(timeout=30)
    assert "_error" not in res, res.get("_error")
    assert res["n"] == 4 and res["ckpt"]


@pytest.mark.timeout(120)
def test_dispatch_lite_collect():
    """dispatch_lite/collect_lite (reference cluster_runs.py:50-97): async
    single-ensemble dispatch, join via collect."""
    from types import SimpleNamespace

    from sparse_coding_amd.sweep.big_sweep import ensemble_train_loop
    from sparse_coding_amd.sweep.cluster_runs import collect_lite, dispatch_lite

    ens = _make_ensemble(seed=3)
    before = ens.params["encoder"].clone()
    chunk = torch.randn(256, D)
    cfg = SimpleNamespace(batch_size=64, show_progress=False, logger=None,
                          ensemble_hyperparams=[], buffer_hyperparams=["l1_alpha"],
                          log_every=1000)
    handle = dispatch_lite(cfg, chunk, ens, "lite", ensemble_train_loop)
    collect_lite([handle])
    assert not torch.allclose(ens.params["encoder"], before)
    assert torch.isfinite(ens.params["encoder"]).all()


# ---------------------------------------------------------------------------
# rs_ag (reduce-scatter + sharded Adam + all-gather) DP mode
# ---------------------------------------------------------------------------

def _rs_ag_worker(rank, world_size, port, batch, dp_mode, out_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)

        from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer, shard_batch

        ens = _make_ensemble(seed=0)
        trainer = DataParallelEnsembleTrainer(ens, dp_mode=dp_mode)
        for _ in range(3):
            local = shard_batch(batch, rank, world_size)
            trainer.step(local)
        trainer.consolidate_optim_state()
        if rank == 0:
            out = {k: v.detach().cpu().numpy() for k, v in ens.params.items()}
            out["_mu_encoder"] = ens.optim_states["mu"]["encoder"].cpu().numpy()
            out["_nu_encoder"] = ens.optim_states["nu"]["encoder"].cpu().numpy()
            out_q.put(out)
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put({"_error": f"rank {rank}: {traceback.format_exc()}"})
        raise


@pytest.mark.timeout(180)
def test_rs_ag_matches_allreduce_and_single():
    """dp_mode=rs_ag == dp_mode=allreduce == single process, params AND
    (consolidated) Adam moments, on gloo world 2."""
    torch.manual_seed(7)
    batch = torch.randn(B, D)

    results = {}
    for i, mode in enumerate(("rs_ag", "allreduce")):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_rs_ag_worker, args=(r, 2, 29561 + i, batch, mode, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        res = q.get(timeout=150)
        assert "_error" not in res, res.get("_error")
        for p in procs:
            p.join(timeout=60)
            assert p.exitcode == 0
        results[mode] = res

    for k in results["rs_ag"]:
        a, b = torch.from_numpy(results["rs_ag"][k]), torch.from_numpy(results["allreduce"][k])
        assert torch.allclose(a, b, atol=1e-7), (k, (a - b).abs().max())

    ens = _make_ensemble(seed=0)
    for _ in range(3):
        ens.step_batch(batch)
    for k, v in ens.params.items():
        assert torch.allclose(torch.from_numpy(results["rs_ag"][k]), v, atol=1e-5), k
    assert torch.allclose(torch.from_numpy(results["rs_ag"]["_mu_encoder"]),
                          ens.optim_states["mu"]["encoder"], atol=1e-5)


def _dp_resample_worker(rank, world_size, port, batch, out_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)

        from sparse_coding_amd.engine.resample import EnsembleResampler
        from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer, shard_batch

        ens = _make_ensemble(seed=0)
        with torch.no_grad():
            ens.params["encoder_bias"][:, : N // 2] = -1e6
        trainer = DataParallelEnsembleTrainer(ens)
        rs = EnsembleResampler(ens, n_track=8, protocol="anthropic")
        for _ in range(3):
            local = shard_batch(batch, rank, world_size)
            _, aux = trainer.step(local)
            rs.observe(local, aux)
        counts = trainer.resample(rs)
        out_q.put((rank, counts.numpy(),
                   {k: v.detach().cpu().numpy() for k, v in ens.params.items()}))
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put((rank, None, {"_error": f"rank {rank}: {traceback.format_exc()}"}))
        raise


@pytest.mark.timeout(180)
def test_dp_resample_keeps_replicas_identical():
    """trainer.resample(): fired summed across ranks, pool broadcast from
    rank 0 — replicas must stay bit-identical after the rewrite."""
    torch.manual_seed(9)
    batch = torch.randn(B, D)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_resample_worker, args=(r, 2, 29566, batch, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        rank, counts, params = q.get(timeout=150)
        assert "_error" not in params, params.get("_error")
        res[rank] = (counts, params)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    c0, p0 = res[0]
    c1, p1 = res[1]
    assert (c0 == c1).all() and c0.sum() > 0
    for k in p0:
        assert (p0[k] == p1[k]).all(), k


# ---------------------------------------------------------------------------
# generator-rank -> trainer-rank chunk streaming
# ---------------------------------------------------------------------------

def _gen_trainer_worker(rank, world_size, port, tmpdir, out_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)

        from sparse_coding_amd.config import EnsembleArgs
        from sparse_coding_amd.sweep.sharded_sweep import generator_trainer_sweep

        cfg = EnsembleArgs()
        cfg.model_name = "tiny-gptneox"
        cfg.layer = 1
        cfg.layer_loc = "residual"
        cfg.model_batch_size = 2
        cfg.max_length = 16
        cfg.chunk_activations = 256
        cfg.n_chunks = 2
        cfg.n_repetitions = 1
        cfg.batch_size = 64
        cfg.lr = 1e-3
        cfg.output_folder = os.path.join(tmpdir, "out")
        cfg.use_wandb = False
        cfg.ensemble_hyperparams = ["dict_size"]
        cfg.buffer_hyperparams = ["l1_alpha"]

        def init_for_rank(c, r, w):
            from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
            from sparse_coding_amd.functional.optim import adam
            from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

            torch.manual_seed(100 + r)
            d = 64  # tiny-gptneox hidden
            models = [FunctionalTiedSAE.init(d, 2 * d, l1) for l1 in (1e-4, 1e-3)]
            ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3},
                                     backend="torch")
            return ens, {"batch_size": c.batch_size, "device": "cpu", "dict_size": 2 * d}, f"r{r}"

        dicts = generator_trainer_sweep(init_for_rank, cfg)
        if rank == 0:
            out_q.put(("ok", len(dicts), cfg.output_folder))
        dist.destroy_process_group()
    except Exception:  # noqa: BLE001
        import traceback

        out_q.put(("error", f"rank {rank}: {traceback.format_exc()}", None))
        raise


@pytest.mark.timeout(300)
def test_generator_trainer_sweep_gloo(tmp_path):
    """Rank 0 generates host-LM activation chunks, rank 1 trains; checkpoints
    gather to rank 0 in the reference layout."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_gen_trainer_worker, args=(r, 2, 29581, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    status, n_dicts, out_folder = q.get(timeout=250)
    assert status == "ok", n_dicts
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert n_dicts == 2  # rank 1's two-l1 ensemble
    final = torch.load(os.path.join(out_folder, "_1", "learned_dicts.pt"), weights_only=False)
    assert len(final) == 2
    assert type(final[0][0]).__module__ == "autoencoders.learned_dict"
