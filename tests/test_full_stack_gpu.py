"""Full-stack slice on GPU (BASELINE.json config 2): random-init Pythia-70m
layer-2 residual activations -> chunk store -> fused HIP ensemble training ->
learned_dicts.pt -> FVU/L0 report.  All @gpu."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_pythia70m_resid_slice(tmp_path):
    from sparse_coding_amd.data.activation_dataset import (
        load_model,
        make_activation_dataset_hf,
        synthetic_token_batches,
    )
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.metrics import standard_metrics as sm
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep.big_sweep import unstacked_to_learned_dicts

    torch.manual_seed(0)
    np.random.seed(0)

    # 1. activation dataset from the hooked host LM
    model = load_model("pythia-70m-deduped", device=DEV)
    chunk_size = 16384
    total = make_activation_dataset_hf(
        synthetic_token_batches(model.config.vocab_size, 8, 256, 10),
        model, [2], "residual",
        chunk_size=chunk_size, n_chunks=1,
        output_folder=str(tmp_path), device=DEV, model_name="pythia-70m",
    )
    del model
    torch.cuda.empty_cache()
    chunk = torch.load(tmp_path / "0.pt").float()
    assert chunk.shape == (chunk_size, 512)

    # 2. 4-way L1 ensemble, 8x dict, fused HIP backend
    d, n_dict, M, B = 512, 4096, 4, 1024
    l1s = np.logspace(-4, -2, M)
    models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=DEV) for l1 in l1s]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    assert ens._hip_step is not None

    data = chunk.to(DEV)
    for epoch in range(6):
        perm = torch.randperm(data.shape[0], device=DEV)
        for s in range(0, data.shape[0] - B + 1, B):
            losses, aux = ens.step_batch(data[perm[s : s + B]])
    assert torch.isfinite(losses["loss"]).all()

    # 3. checkpoint in reference format + quality metrics
    args = {"dict_size": n_dict, "batch_size": B, "device": DEV}
    lds = unstacked_to_learned_dicts(ens, args, ["dict_size"], ["l1_alpha"])
    torch.save(lds, tmp_path / "learned_dicts.pt")
    loaded = torch.load(tmp_path / "learned_dicts.pt", weights_only=False)
    assert type(loaded[0][0]).__module__ == "autoencoders.learned_dict"

    sample = chunk[:4096]
    fvus, l0s = [], []
    for ld, hp in loaded:
        fvus.append(sm.fraction_variance_unexplained(ld, sample).item())
        l0s.append(sm.mean_l0(ld, sample).item())
    print("fvu:", fvus, "l0:", l0s)
    # low-l1 models must reconstruct reasonably after 6 epochs on one chunk
    assert fvus[0] < 0.5
    # stronger l1 -> sparser codes (monotone trend at the extremes)
    assert l0s[-1] < l0s[0]


def test_dp_hip_grads_phase_split():
    """grads_phase/update_phase split == step_batch on one GPU (the path the
    multi-GPU DP trainer exercises between all-reduces)."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(1)
    M, B, d, n = 2, 256, 64, 128
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens_a = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
               for p, b in ens_a.unstack()]
    ens_b = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    ens_b._hip_step.use_graph = False

    x = torch.randn(B, d, device=DEV)
    for _ in range(3):
        ens_a.step_batch(x)
        hb = ens_b._hip_step
        Bn = hb.grads_phase(x)
        hb.update_phase(Bn)
    for k in ens_a.params:
        assert torch.allclose(ens_a.params[k], ens_b.params[k], atol=1e-6), k


def test_gpt2small_mlpout_grid():
    """BASELINE config 3 shape: GPT-2-small MLP-out (d=768), 32-SAE
    (8 l1 x 4 ratios as 4 ensembles) — fused steps run and converge."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(2)
    d, B = 768, 512
    x = torch.randn(B, d, device=DEV)
    l1s = np.logspace(-4, -2, 8)
    for ratio in (1, 2, 4, 8):
        n_dict = d * ratio
        models = [FunctionalTiedSAE.init(d, n_dict, float(l1), device=DEV) for l1 in l1s]
        ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
        l0, _ = ens.step_batch(x)
        for _ in range(3):
            losses, _ = ens.step_batch(x)
        assert torch.isfinite(losses["loss"]).all(), ratio
        assert (losses["loss"] <= l0["loss"] + 1e-5).all(), ratio
        del ens
        torch.cuda.empty_cache()


def test_pythia14b_topk_resample():
    """BASELINE config 5 shape: TopK (k=32) + resampling at d=2048."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.engine.resample import EnsembleResampler
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.topk import TopKEncoder

    torch.manual_seed(3)
    d, n, B, M = 2048, 8192, 512, 2
    models = [TopKEncoder.init(d, n, 32) for _ in range(M)]
    ens = FunctionalEnsemble(models, TopKEncoder, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    rs = EnsembleResampler(ens, n_track=128)
    x = torch.randn(B, d, device=DEV)
    l0, aux = ens.step_batch(x)
    for _ in range(3):
        losses, aux = ens.step_batch(x)
        rs.observe(x, aux)
    assert torch.isfinite(losses["loss"]).all()
    assert (losses["loss"] <= l0["loss"]).all()
    # topk leaves most features unfired on a fixed batch -> resampler acts
    counts = rs.resample()
    assert (counts > 0).all()


def test_chunked_grads_match_unchunked():
    """grads_phase(on_grads=...) with model-group chunked grad_w GEMMs must
    produce the same gradients as the monolithic path (the multi-GPU
    all-reduce-overlap path)."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(4)
    M, B, d, n = 4, 256, 64, 128
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    hs = ens._hip_step
    hs.use_graph = False
    x = torch.randn(B, d, device=DEV)

    hs.grads_phase(x)
    gw_mono = hs.gw.clone()
    gb_mono = hs.g_bias.clone()

    seen = []
    hs.grads_phase(x, on_grads=lambda ts: seen.extend(t.shape for t in ts))
    assert torch.allclose(hs.gw, gw_mono, atol=1e-6)
    assert torch.allclose(hs.g_bias, gb_mono, atol=1e-6)
    assert len(seen) == 1 + min(M, 4)  # g_bias + model-group gw slices


def test_dp_phase_split_matches_step():
    """The DP trainer's grads_phase(on_grads)/update_phase split — including
    the model-half chunked grad_w path taken only when a callback is
    installed — must produce the same update as the fused step()."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(21)
    M, B, d, n = 4, 512, 128, 512
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens_a = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
               for p, b in ens_a.unstack()]
    ens_b = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")

    x = torch.randn(B, d, device=DEV)
    seen = []
    for _ in range(3):
        ens_a.step_batch(x)
        hs = ens_b._hip_step
        nb = hs.grads_phase(x, on_grads=lambda ts: seen.extend(t.shape for t in ts))
        hs.update_phase(nb)
    torch.cuda.synchronize()
    # callback fired for g_bias + model-group gw chunks per step
    assert (M, n) in seen and (M // min(M, 4), n, d) in seen
    for k in ens_a.params:
        err = (ens_a.params[k] - ens_b.params[k]).abs().max().item()
        assert err < 1e-6, (k, err)


def test_cluster_dispatch_gpu(tmp_path):
    """P1 (process-per-ensemble, shared chunk) with CUDA-resident ensembles:
    two children attach to shared GPU tensors and train on one chunk."""
    import numpy as np

    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep.cluster_runs import dispatch_job_on_chunk
    from sparse_coding_amd.sweep.big_sweep import ensemble_train_loop

    torch.manual_seed(22)
    d, n = 32, 64
    ensembles = []
    for gi, l1s in enumerate([(1e-4, 1e-3), (3e-3,)]):
        models = [FunctionalTiedSAE.init(d, n, l1, device=DEV) for l1 in l1s]
        ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
        ensembles.append((ens, {"batch_size": 128, "device": DEV, "dict_size": n, "l1_alpha": list(l1s)}, f"ens{gi}"))

    before = [e[0].params["encoder"].clone() for e in ensembles]
    chunk = torch.randn(1024, d)

    # spawn-context children pickle cfg: must not be a test-local class
    from types import SimpleNamespace

    cfg = SimpleNamespace(batch_size=128, show_progress=False, logger=None,
                          ensemble_hyperparams=[], buffer_hyperparams=["l1_alpha"],
                          log_every=100)
    dispatch_job_on_chunk(ensembles, cfg, chunk, ensemble_train_loop)
    for (ens, _, _), enc0 in zip(ensembles, before):
        assert torch.isfinite(ens.params["encoder"]).all()
        assert not torch.allclose(ens.params["encoder"], enc0)


def test_dp_rccl_one_rank_overlap_path(tmp_path):
    """The 8-GPU scaling bench's exact code path — fused step split +
    all-reduces launched from a side stream via RCCL — on a 1-rank nccl
    process group (all_reduce is then an on-device no-op, but the full
    RCCL/stream call sequence runs)."""
    import os

    import torch.distributed as dist

    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        torch.manual_seed(31)
        M, B, d, n = 4, 1024, 128, 512
        models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
        ens_dp = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
        models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
                   for p, b in ens_dp.unstack()]
        ens_ref = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")

        trainer = DataParallelEnsembleTrainer(ens_dp, force_dp_path=True)
        x = torch.randn(B, d, device=DEV)
        for _ in range(3):
            losses, _ = trainer.step(x)
            ens_ref.step_batch(x)
        torch.cuda.synchronize()
        assert torch.isfinite(losses["loss"]).all()
        for k in ens_ref.params:
            err = (ens_dp.params[k] - ens_ref.params[k]).abs().max().item()
            assert err < 1e-6, (k, err)
    finally:
        dist.destroy_process_group()


def test_flagship_soak_1000_steps():
    """1000 fused steps at the flagship shape: finite losses throughout,
    loss improves, fired counters accumulate monotonically."""
    import numpy as np

    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(41)
    M, B, d, n = 4, 2048, 512, 4096
    models = [FunctionalTiedSAE.init(d, n, float(l1), device=DEV)
              for l1 in np.logspace(-4, -3, M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    pool = [torch.randn(B, d, device=DEV) for _ in range(4)]
    l0 = None
    for i in range(1000):
        losses, _ = ens.step_batch(pool[i % 4])
        if i == 0:
            l0 = losses["loss"].clone()
        if i % 250 == 0:
            assert torch.isfinite(losses["loss"]).all(), i
    torch.cuda.synchronize()
    assert torch.isfinite(losses["loss"]).all()
    assert (losses["loss"] < l0).all()
    fired = ens._hip_step.fired
    assert (fired.sum(dim=1) > 0).all()


def test_sweep_end_to_end_gpu(tmp_path):
    """The full sweep() driver on GPU: synthetic chunks -> spawn-dispatched
    ensemble children on cuda:0 (fused step) -> reference checkpoint layout."""
    import os

    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = 128
    cfg.n_ground_truth_components = 256
    cfg.gen_batch_size = 512
    cfg.feature_num_nonzero = 8
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = 128 * 512 * 8 * 2 / 1024**3  # 8 batches/chunk
    cfg.n_chunks = 2
    cfg.batch_size = 512
    cfg.device = DEV
    cfg.dataset_folder = str(tmp_path / "data")
    cfg.output_folder = str(tmp_path / "out")
    cfg.use_wandb = False
    cfg.wandb_images = False

    def init_func(c):
        return make_grid_ensembles(c, FunctionalTiedSAE, [1e-4, 1e-3], [4.0], devices=[DEV])

    dicts = big_sweep.sweep(init_func, cfg)
    assert len(dicts) == 2
    final = os.path.join(cfg.output_folder, "_1")
    assert os.path.exists(os.path.join(final, "learned_dicts.pt"))
    ld, hp = dicts[0]
    assert torch.isfinite(ld.get_learned_dict()).all()


def test_perplexity_under_reconstruction_gpu():
    """C20's flagship quality metric on GPU: identity dict must reproduce
    the clean perplexity; a random dict must not beat it."""
    from sparse_coding_amd.data.activation_dataset import load_model, synthetic_token_batches
    from sparse_coding_amd.metrics import standard_metrics as sm
    from sparse_coding_amd.models.learned_dict import Identity, RandomDict

    torch.manual_seed(43)
    model = load_model("pythia-70m", device=DEV)
    d = model.config.hidden_size
    tokens = torch.cat(list(synthetic_token_batches(model.config.vocab_size, 4, 64, 2)))

    clean = sm.calculate_perplexity(model, None, None, 2, "residual", tokens, device=DEV)
    ident = Identity(d)
    ident.to_device(DEV)
    p_ident = sm.calculate_perplexity(model, None, ident, 2, "residual", tokens, device=DEV)
    rand = RandomDict(d, 2 * d)
    rand.to_device(DEV)
    p_rand = sm.calculate_perplexity(model, None, rand, 2, "residual", tokens, device=DEV)

    assert abs(p_ident - clean) / clean < 1e-3
    # the LM here is random-init (no network), so corrupting the stream
    # cannot meaningfully "hurt" — but it must CHANGE the output
    assert abs(p_rand - clean) / clean > 1e-3


def test_activation_dataset_throughput_gpu(tmp_path):
    """The data plane's hot loop: hooked LM forward ->
    fp16 chunks.  Sanity + a throughput print for the record."""
    import time

    from sparse_coding_amd.data.activation_dataset import (
        load_model,
        make_activation_dataset_hf,
        synthetic_token_batches,
    )

    n_batches, bsz, seq = 24, 16, 256
    for dtype, tag in ((None, "fp32"), (torch.bfloat16, "bf16")):
        model = load_model("pythia-70m", device=DEV, dtype=dtype)
        out = tmp_path / tag
        t0 = time.perf_counter()
        total = make_activation_dataset_hf(
            synthetic_token_batches(model.config.vocab_size, bsz, seq, n_batches),
            model, [2], "residual",
            chunk_size=65536, n_chunks=2,
            output_folder=str(out), device=DEV, model_name="pythia-70m",
        )
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        chunk = torch.load(out / "0.pt")
        assert chunk.dtype == torch.float16 and chunk.shape[1] == model.config.hidden_size
        assert torch.isfinite(chunk.float()).all()
        print(f"[data-plane {tag}] {total} activations in {dt:.2f}s = {total/dt:,.0f} acts/s")
        del model
        torch.cuda.empty_cache()


def test_baselines_runner_gpu(tmp_path):
    """C21: per-layer baseline suite (streaming PCA on device + topk-PCA /
    random / identity-relu exports) against a real fp16 chunk file."""
    from sparse_coding_amd.sweep.baselines import run_layer_baselines

    torch.manual_seed(44)
    d = 64
    chunk = (torch.randn(4096, d) @ torch.randn(d, d)).to(torch.float16)
    chunk_path = str(tmp_path / "0.pt")
    torch.save(chunk, chunk_path)

    out = run_layer_baselines(2, chunk_path, str(tmp_path / "base"),
                              device=DEV, sparsity=8, do_ica=False)
    import os

    assert set(out) == {"pca", "pca_topk", "pca_rot", "random", "identity_relu"}
    ld = torch.load(os.path.join(tmp_path / "base", "pca_topk_l2.pt"), weights_only=False)
    # saved baselines load as reference-module-path LearnedDicts
    assert type(ld).__module__.startswith("autoencoders.")
    ld.to_device(DEV)  # baselines are saved as CPU copies
    c = ld.encode(torch.randn(16, d, device=DEV))
    assert (c != 0).sum(dim=-1).max() <= 8


def test_chunk_broadcast_rccl_one_rank():
    """BroadcastChunkFeeder's RCCL call sequence on a 1-rank nccl group
    (shape broadcast + payload broadcast land on-device)."""
    import os

    import torch.distributed as dist

    from sparse_coding_amd.parallel.chunk_feed import BroadcastChunkFeeder

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29559")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        feeder = BroadcastChunkFeeder(DEV, src=0)
        chunk = torch.randn(4096, 64)
        out = feeder.feed(chunk)
        assert out.is_cuda and out.shape == chunk.shape
        assert torch.allclose(out.cpu(), chunk)
    finally:
        dist.destroy_process_group()


def test_dp_rccl_graph_capture_one_rank():
    """Opt-in hipGraph capture of the whole DP step INCLUDING the RCCL
    all-reduces (ROADMAP item 2), validated at world size 1: captured
    replays must keep matching the plain fused step."""
    import os

    import torch.distributed as dist

    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        torch.manual_seed(33)
        M, B, d, n = 4, 1024, 128, 512
        models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
        ens_dp = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
        models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
                   for p, b in ens_dp.unstack()]
        ens_ref = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")

        trainer = DataParallelEnsembleTrainer(ens_dp, force_dp_path=True, graph_capture=True)
        x = torch.randn(B, d, device=DEV)
        for i in range(6):  # 2 eager + capture + replays
            losses, _ = trainer.step(x)
            ens_ref.step_batch(x)
        torch.cuda.synchronize()
        assert trainer._graph is not None, "graph capture did not engage"
        assert torch.isfinite(losses["loss"]).all()
        for k in ens_ref.params:
            err = (ens_dp.params[k] - ens_ref.params[k]).abs().max().item()
            assert err < 1e-6, (k, err)
    finally:
        dist.destroy_process_group()


def test_lr_mult_freezes_rows():
    """k_project_adam/k_bias_adam per-row lr multiplier: rows at 0 must not
    move; rows at 1 must match a run without lr_mult."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(41)
    M, B, d, n = 2, 256, 64, 256
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
               for p, b in ens.unstack()]
    ens_ref = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")

    hs = ens._hip_step
    hs.lr_mult[:, : n // 2] = 0.0
    x = torch.randn(B, d, device=DEV)
    before = ens.params["encoder"][:, : n // 2].clone()
    before_bias = ens.params["encoder_bias"][:, : n // 2].clone()
    # one step from identical states: unfrozen rows must match the
    # all-ones-lr_mult run exactly (after this step the frozen rows make
    # the forwards diverge, so only the first step is comparable)
    ens.step_batch(x)
    ens_ref.step_batch(x)
    torch.cuda.synchronize()
    err = (ens.params["encoder"][:, n // 2 :] - ens_ref.params["encoder"][:, n // 2 :]).abs().max()
    assert err.item() == 0.0
    for _ in range(2):
        ens.step_batch(x)
    torch.cuda.synchronize()
    assert torch.equal(ens.params["encoder"][:, : n // 2], before)
    assert torch.equal(ens.params["encoder_bias"][:, : n // 2], before_bias)


def test_anthropic_resample_warmup_gpu():
    """Fused anthropic protocol: resample rewrites dead rows, sets lr_mult to
    warmup_start on replaced rows, and ramps back to 1 over warmup_steps."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.engine.resample import EnsembleResampler
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(42)
    M, B, d, n = 2, 256, 64, 256
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    rs = EnsembleResampler(ens, n_track=64, protocol="anthropic",
                           warmup_steps=4, warmup_start=0.25)
    with torch.no_grad():
        ens.params["encoder_bias"][:, : n // 2] = -1e6
    x = torch.randn(B, d, device=DEV)
    for _ in range(3):
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    counts = rs.resample()
    assert (counts == 64).all()
    hs = ens._hip_step
    torch.cuda.synchronize()
    assert (hs.lr_mult[:, :64] == 0.25).all()
    assert (hs.lr_mult[:, 64:] == 1.0).all()
    # replaced encoder rows: unit direction x 0.2 x alive mean norm
    alive_norm = torch.norm(ens.params["encoder"][:, n // 2 :], dim=-1).mean(dim=1)
    new_norms = torch.norm(ens.params["encoder"][:, :64], dim=-1)
    assert torch.allclose(new_norms, (0.2 * alive_norm)[:, None].expand_as(new_norms), rtol=0.08)
    # ramp: after warmup_steps observes lr_mult returns to 1 everywhere
    for _ in range(4):
        _, aux = ens.step_batch(x)
        rs.observe(x, aux)
    torch.cuda.synchronize()
    assert (hs.lr_mult == 1.0).all()


@pytest.mark.parametrize("sig_name", ["tied", "untied"])
def test_rs_ag_one_rank_matches_step(sig_name):
    """dp_mode=rs_ag at world 1 (force path: same kernel/collective code
    shape, local shard copies) must match the plain fused step bit-exactly."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalSAE, FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    sig = FunctionalTiedSAE if sig_name == "tied" else FunctionalSAE
    torch.manual_seed(51)
    M, B, d, n = 4, 512, 128, 512
    models = [sig.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens = FunctionalEnsemble(models, sig, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
               for p, b in ens.unstack()]
    ens_ref = FunctionalEnsemble(models2, sig, adam, {"lr": 1e-3}, device=DEV, backend="hip")

    trainer = DataParallelEnsembleTrainer(ens, force_dp_path=True, dp_mode="rs_ag")
    x = torch.randn(B, d, device=DEV)
    for _ in range(3):
        losses, _ = trainer.step(x)
        ens_ref.step_batch(x)
    torch.cuda.synchronize()
    assert torch.isfinite(losses["loss"]).all()
    # atomic-order noise in the column-sum reductions (g_bias) makes repeat
    # runs differ in the last ulp; 1e-6 matches the other equivalence tests
    for k in ens_ref.params:
        err = (ens.params[k] - ens_ref.params[k]).abs().max().item()
        assert err < 1e-6, (k, err)
    for k in ("mu", "nu"):
        for pk in ens_ref.optim_states[k]:
            err = (ens.optim_states[k][pk] - ens_ref.optim_states[k][pk]).abs().max().item()
            assert err < 1e-5, (k, pk, err)


def test_whitened_tied_step_matches_oracle():
    """HipWhitenedStep (general affine centering, K7) vs the vmap oracle:
    same losses and same trained weights over several steps."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.engine.hip_step import HipWhitenedStep
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE

    torch.manual_seed(61)
    M, B, d, n = 3, 256, 64, 192
    models = []
    for i in range(M):
        q, _ = torch.linalg.qr(torch.randn(d, d))
        models.append(FunctionalTiedSAE.init(
            d, n, 10 ** (-4 + 0.3 * i), device=DEV,
            rotation=q.to(DEV), translation=torch.randn(d, device=DEV) * 0.3,
            scaling=(torch.rand(d, device=DEV) + 0.5)))
    ens_hip = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    assert isinstance(ens_hip._hip_step, HipWhitenedStep)
    models2 = [({k: v.clone() for k, v in p.items()}, {k: v.clone() for k, v in b.items()})
               for p, b in ens_hip.unstack()]
    ens_ref = FunctionalEnsemble(models2, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="torch")

    x = torch.randn(B, d, device=DEV)
    for i in range(4):
        l_hip, aux_hip = ens_hip.step_batch(x)
        l_ref, aux_ref = ens_ref.step_batch(x)
        for k in ("loss", "l_reconstruction", "l_l1"):
            err = (l_hip[k] - l_ref[k]).abs().max() / l_ref[k].abs().max().clamp_min(1e-9)
            assert err < 1e-3, (i, k, err)
    torch.cuda.synchronize()
    err = (ens_hip.params["encoder"] - ens_ref.params["encoder"]).abs().max()
    assert err < 2e-3, err
    err_b = (ens_hip.params["encoder_bias"] - ens_ref.params["encoder_bias"]).abs().max()
    assert err_b < 2e-3, err_b


def test_sweep_resume_equivalence_gpu(tmp_path):
    """resume_state.pt round-trips the FUSED step's Adam state (VERDICT weak
    #7: the CPU test covers the torch backend only).  Long-horizon bitwise
    replay is impossible on GPU (atomic-order noise in the column-sum
    reductions), so the test checks what resume actually guarantees:
    (a) a sweep-written resume_state restores params AND moments exactly,
    (b) the restored ensemble's next fused steps track the original's on the
    same batches at single-step tolerance."""
    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.big_sweep import _save_resume_state, _to_device_tree
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    # (a) sweep-level: run 2 chunks via the real driver, then restore from
    # its resume_state and check exact state equality with the checkpointed
    # learned dicts
    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = 128
    cfg.n_ground_truth_components = 256
    cfg.gen_batch_size = 512
    cfg.feature_num_nonzero = 8
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = 128 * 512 * 4 * 2 / 1024**3
    cfg.n_chunks = 2
    cfg.batch_size = 512
    cfg.device = DEV
    cfg.dataset_folder = str(tmp_path / "data")
    cfg.output_folder = str(tmp_path / "out")
    cfg.use_wandb = False
    cfg.wandb_images = False

    def init_func(c):
        return make_grid_ensembles(c, FunctionalTiedSAE, [1e-3], [2.0], devices=[DEV])

    dicts = big_sweep.sweep(init_func, cfg)
    resume_path = os.path.join(cfg.output_folder, "resume_state.pt")
    assert os.path.exists(resume_path)
    saved = torch.load(resume_path, map_location="cpu", weights_only=False)
    st = saved["ensemble_states"][0]
    (ld, _), = dicts
    # restored params == final checkpointed dict, bit-exact
    assert torch.equal(st["params"]["encoder"][0].cpu(), ld.encoder.cpu())
    assert (st["optim_states"]["step"] > 0).all()
    for mom in ("mu", "nu"):
        assert st["optim_states"][mom]["encoder"].abs().sum() > 0  # moments persisted

    # (b) engine-level: restore into a fresh ensemble and verify the next
    # fused steps track a never-serialized twin on identical batches
    torch.manual_seed(77)
    models = [FunctionalTiedSAE.init(128, 256, 1e-3, device=DEV) for _ in range(2)]
    ens_a = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    warm = torch.randn(512, 128, device=DEV)
    for _ in range(50):
        ens_a.step_batch(warm)
    _save_resume_state(str(tmp_path / "rs.pt"), [(ens_a, {}, "a")], np.arange(2), 1)
    saved2 = torch.load(tmp_path / "rs.pt", map_location="cpu", weights_only=False)
    models_b = [FunctionalTiedSAE.init(128, 256, 1e-3, device=DEV) for _ in range(2)]
    ens_b = FunctionalEnsemble(models_b, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    st2 = saved2["ensemble_states"][0]
    ens_b.params = _to_device_tree(st2["params"], DEV)
    ens_b.optim_states = _to_device_tree(st2["optim_states"], DEV)
    ens_b.buffers = _to_device_tree(st2["buffers"], DEV)
    ens_b._hip_step = None
    ens_b.init_functions()
    assert ens_b._hip_step is not None
    for i in range(5):
        x = torch.randn(512, 128, device=DEV)
        la, _ = ens_a.step_batch(x)
        lb, _ = ens_b.step_batch(x)
        assert torch.allclose(la["loss"], lb["loss"], rtol=1e-5), i
    torch.cuda.synchronize()
    err = (ens_a.params["encoder"] - ens_b.params["encoder"]).abs().max().item()
    assert err < 1e-5, err
    err_m = (ens_a.optim_states["nu"]["encoder"] - ens_b.optim_states["nu"]["encoder"]).abs().max().item()
    assert err_m < 1e-6, err_m


def test_colsum_matches_torch():
    """k_colsum vs torch.sum(dim=1), plain and absval, edge n not a
    multiple of the 256-column block."""
    from sparse_coding_amd import ops

    ext = ops.get_extension(required=True)
    torch.manual_seed(91)
    for M, B, n in ((3, 517, 384), (8, 2048, 4096), (2, 64, 100)):
        x = torch.randn(M, B, n, device=DEV)
        out = torch.empty(M, n, device=DEV)
        ext.colsum(x, out)
        ref = x.sum(dim=1)
        assert torch.allclose(out, ref, atol=1e-3 * B ** 0.5), (M, B, n)
        ext.colsum(x, out, 2.5, True)
        ref = 2.5 * x.abs().sum(dim=1)
        assert torch.allclose(out, ref, rtol=1e-5, atol=1e-3 * B ** 0.5), (M, B, n)


def test_rs_ag_respects_lr_mult():
    """The sharded Adam must read the same per-row lr_mult slices the full
    update does: frozen rows stay frozen under dp_mode=rs_ag."""
    from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
    from sparse_coding_amd.functional.optim import adam
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.parallel.dp import DataParallelEnsembleTrainer

    torch.manual_seed(92)
    M, B, d, n = 2, 256, 64, 256
    models = [FunctionalTiedSAE.init(d, n, 1e-3, device=DEV) for _ in range(M)]
    ens = FunctionalEnsemble(models, FunctionalTiedSAE, adam, {"lr": 1e-3}, device=DEV, backend="hip")
    trainer = DataParallelEnsembleTrainer(ens, force_dp_path=True, dp_mode="rs_ag")
    hs = ens._hip_step
    hs.lr_mult[:, : n // 2] = 0.0
    before = ens.params["encoder"][:, : n // 2].clone()
    before_live = ens.params["encoder"][:, n // 2 :].clone()
    x = torch.randn(B, d, device=DEV)
    for _ in range(3):
        trainer.step(x)
    torch.cuda.synchronize()
    assert torch.equal(ens.params["encoder"][:, : n // 2], before)
    assert not torch.equal(ens.params["encoder"][:, n // 2 :], before_live)  # others moved


def test_sweep_with_resampling_gpu(tmp_path):
    """In-sweep anthropic resampling through the dispatched FUSED path:
    k_resample + lr_mult run inside the spawned child against shared CUDA
    tensors; the rewrite must be visible to the parent."""
    from sparse_coding_amd.config import SyntheticEnsembleArgs
    from sparse_coding_amd.models.sae_signatures import FunctionalTiedSAE
    from sparse_coding_amd.sweep import big_sweep
    from sparse_coding_amd.sweep.experiments import make_grid_ensembles

    cfg = SyntheticEnsembleArgs()
    cfg.use_synthetic_dataset = True
    cfg.activation_width = 128
    cfg.n_ground_truth_components = 256
    cfg.gen_batch_size = 512
    cfg.feature_num_nonzero = 8
    cfg.noise_magnitude_scale = 0.0
    cfg.chunk_size_gb = 128 * 512 * 4 * 2 / 1024**3
    cfg.n_chunks = 2
    cfg.batch_size = 512
    cfg.device = DEV
    cfg.dataset_folder = str(tmp_path / "data")
    cfg.output_folder = str(tmp_path / "out")
    cfg.use_wandb = False
    cfg.wandb_images = False
    cfg.resample_every_chunks = 1
    cfg.resample_n_track = 8
    cfg.resample_warmup_steps = 4

    def init_func(c):
        out = make_grid_ensembles(c, FunctionalTiedSAE, [1e-3], [1.0], devices=[DEV])
        for ens, _, _ in out[0]:
            with torch.no_grad():
                ens.params["encoder_bias"][:, :8] = -1e6  # guaranteed dead
        return out

    dicts = big_sweep.sweep(init_func, cfg)
    (ld, _), = dicts
    assert (ld.encoder_bias[:8] > -1e5).all()  # rewritten by k_resample
    assert torch.isfinite(ld.get_learned_dict()).all()
