"""Activation-dataset generation: host LM → hooked layer → fp16 chunks on disk.

Covers reference ``activation_dataset.py`` (C14): the location maps
(``make_tensor_name`` :69-106, ``get_activation_size`` :39-58), the HF
forward-hook capture path (``make_activation_dataset_hf`` :393-496), the
pack-and-tokenize pipeline (:136-235), and ``setup_data``/``setup_data_new``
(:505-611).  Chunk layout on disk is the reference's: ``{folder}/{i}.pt``
containing a ``[N, d]`` fp16 tensor (:499-503).

MI355X-native differences:
 * TransformerLens/baukit are replaced by plain HF ``transformers`` modules +
   forward hooks (module paths resolved per architecture below); TL-style
   tensor names are still produced for checkpoint/folder parity.
 * Capture accumulates on-GPU and drains to a pinned host buffer with
   ``non_blocking`` copies so the LM forward and the D2H copy overlap.
 * No network in this environment: models are built RANDOM-INIT from a local
   config table (``MODEL_TABLE``) unless a local checkpoint path is given,
   and the token stream falls back to a synthetic sampler.
"""

from __future__ import annotations

import contextlib
import os
from typing import Any, Dict, Iterable, List, Literal, Optional, Tuple

import torch

MODEL_BATCH_SIZE = 4
CHUNK_SIZE_GB = 2.0
MAX_SENTENCE_LEN = 256

LAYER_LOCS = ["residual", "mlp", "attn", "attn_concat", "mlpout"]

# (family, hidden, n_layer, n_head, intermediate, vocab)
MODEL_TABLE: Dict[str, Tuple[str, int, int, int, int, int]] = {
    "tiny-gptneox": ("gptneox", 64, 2, 4, 256, 512),  # test-scale host LM
    "pythia-70m": ("gptneox", 512, 6, 8, 2048, 50304),
    "pythia-70m-deduped": ("gptneox", 512, 6, 8, 2048, 50304),
    "EleutherAI/pythia-70m-deduped": ("gptneox", 512, 6, 8, 2048, 50304),
    "pythia-160m": ("gptneox", 768, 12, 12, 3072, 50304),
    "pythia-160m-deduped": ("gptneox", 768, 12, 12, 3072, 50304),
    "pythia-410m": ("gptneox", 1024, 24, 16, 4096, 50304),
    "pythia-410m-deduped": ("gptneox", 1024, 24, 16, 4096, 50304),
    "pythia-1.4b": ("gptneox", 2048, 24, 16, 8192, 50304),
    "pythia-1.4b-deduped": ("gptneox", 2048, 24, 16, 8192, 50304),
    "pythia-2.8b": ("gptneox", 2560, 32, 32, 10240, 50304),
    "gpt2": ("gpt2", 768, 12, 12, 3072, 50257),
    "gpt2-small": ("gpt2", 768, 12, 12, 3072, 50257),
    "gpt2-medium": ("gpt2", 1024, 24, 16, 4096, 50257),
    "gpt2-large": ("gpt2", 1280, 36, 20, 5120, 50257),
    "gpt2-xl": ("gpt2", 1600, 48, 25, 6400, 50257),
}


def _lookup(model_name: str):
    key = model_name
    if key not in MODEL_TABLE and "/" in key:
        key = key.split("/")[-1]
    if key not in MODEL_TABLE:
        raise NotImplementedError(f"Model {model_name} not in MODEL_TABLE")
    return MODEL_TABLE[key]


def check_transformerlens_model(model_name: str) -> bool:
    """Name-recognition parity shim: True if we know the architecture."""
    try:
        _lookup(model_name)
        return True
    except NotImplementedError:
        return False


def get_activation_size(model_name: str, layer_loc: str) -> int:
    """Width of the hooked activation (reference :39-58)."""
    assert layer_loc in LAYER_LOCS, f"Layer location {layer_loc} not supported"
    family, hidden, n_layer, n_head, inter, vocab = _lookup(model_name)
    if layer_loc in ("residual", "mlpout", "attn", "attn_concat"):
        return hidden
    if layer_loc == "mlp":
        return inter
    raise AssertionError


def make_tensor_name(layer: int, layer_loc: str, model_name: str) -> str:
    """TL-style tensor name used for folder layout parity (reference :69-106)."""
    assert layer_loc in LAYER_LOCS, f"Layer location {layer_loc} not supported"
    if layer_loc in ("residual", "attn"):
        return f"blocks.{layer}.hook_resid_post"
    if layer_loc == "attn_concat":
        return f"blocks.{layer}.attn.hook_z"
    if layer_loc == "mlp":
        return f"blocks.{layer}.mlp.hook_post"
    if layer_loc == "mlpout":
        return f"blocks.{layer}.hook_mlp_out"
    raise AssertionError


def load_model(model_name: str, device="cpu", random_init: bool = True, local_path: Optional[str] = None,
               dtype: Optional[torch.dtype] = None):
    """Build the host LM.  With no network access, models come up random-init
    from the config table (BASELINE.json configs specify random-init weights);
    pass ``local_path`` to load real weights from disk.

    ``dtype``: run the host LM in bf16/fp16 to roughly double data-plane
    throughput — the captured activations are stored fp16 regardless
    (reference activation_dataset.py:364-388), so the chunk format and the
    fp32 TRAINING dtype are unaffected."""
    from transformers import AutoModelForCausalLM, GPT2Config, GPT2LMHeadModel, GPTNeoXConfig, GPTNeoXForCausalLM

    if local_path is not None:
        model = AutoModelForCausalLM.from_pretrained(local_path)
        if dtype is not None:
            model = model.to(dtype)
        return model.to(device)

    family, hidden, n_layer, n_head, inter, vocab = _lookup(model_name)
    if family == "gptneox":
        cfg = GPTNeoXConfig(
            hidden_size=hidden,
            num_hidden_layers=n_layer,
            num_attention_heads=n_head,
            intermediate_size=inter,
            vocab_size=vocab,
            max_position_embeddings=2048,
        )
        model = GPTNeoXForCausalLM(cfg)
    else:
        cfg = GPT2Config(
            n_embd=hidden,
            n_layer=n_layer,
            n_head=n_head,
            n_inner=inter,
            vocab_size=vocab,
            n_positions=1024,
        )
        model = GPT2LMHeadModel(cfg)
    model.eval()
    if dtype is not None:
        model = model.to(dtype)
    return model.to(device)


def _family_of(model) -> str:
    name = type(model).__name__.lower()
    if "neox" in name:
        return "gptneox"
    if "gpt2" in name:
        return "gpt2"
    raise NotImplementedError(f"Unsupported host model {type(model)}")


def resolve_hook_point(model, layer: int, layer_loc: str):
    """Map (layer, layer_loc) to (module, kind) where kind tells how to read
    the activation: 'output0' = module output (tuple→[0]), 'input0' = first
    forward arg."""
    fam = _family_of(model)
    if fam == "gptneox":
        block = model.gpt_neox.layers[layer]
        if layer_loc in ("residual", "attn"):
            return block, "output0"
        if layer_loc == "mlp":
            return block.mlp.act, "output0"
        if layer_loc == "mlpout":
            return block.mlp, "output0"
        if layer_loc == "attn_concat":
            return block.attention.dense, "input0"
    else:
        block = model.transformer.h[layer]
        if layer_loc in ("residual", "attn"):
            return block, "output0"
        if layer_loc == "mlp":
            return block.mlp.act, "output0"
        if layer_loc == "mlpout":
            return block.mlp, "output0"
        if layer_loc == "attn_concat":
            return block.attn.c_proj, "input0"
    raise NotImplementedError(layer_loc)


@contextlib.contextmanager
def capture_activation_hook(model, layer: int, layer_loc: str, store: List[torch.Tensor]):
    """Forward hook appending the flattened [b*l, d] activation to `store`."""
    module, kind = resolve_hook_point(model, layer, layer_loc)

    if kind == "output0":
        def hook(mod, inputs, output):
            out = output[0] if isinstance(output, tuple) else output
            store.append(out.reshape(-1, out.shape[-1]))
            return output

        handle = module.register_forward_hook(hook)
    else:
        def pre_hook(mod, inputs):
            x = inputs[0]
            store.append(x.reshape(-1, x.shape[-1]))

        handle = module.register_forward_pre_hook(pre_hook)
    try:
        yield
    finally:
        handle.remove()


@contextlib.contextmanager
def replace_activation_hook(model, layer: int, layer_loc: str, learned_dict):
    """Replace the hooked activation with learned_dict.predict(activation)
    (for perplexity-under-reconstruction, reference standard_metrics.py:693-699).
    learned_dict=None → identity (clean baseline)."""
    if learned_dict is None:
        yield
        return

    module, kind = resolve_hook_point(model, layer, layer_loc)
    if kind != "output0":
        raise NotImplementedError("replace_activation_hook supports output locations only")

    def hook(mod, inputs, output):
        is_tuple = isinstance(output, tuple)
        out = output[0] if is_tuple else output
        shape = out.shape
        flat = out.reshape(-1, shape[-1])
        replaced = learned_dict.predict(flat.to(torch.float32)).to(out.dtype).reshape(shape)
        if is_tuple:
            return (replaced,) + tuple(output[1:])
        return replaced

    handle = module.register_forward_hook(hook)
    try:
        yield
    finally:
        handle.remove()


# ---------------------------------------------------------------------------
# token streams
# ---------------------------------------------------------------------------

def synthetic_token_batches(vocab_size: int, batch_size: int, seq_len: int, n_batches: int, seed: int = 0):
    """Deterministic random token stream (no-network stand-in for a corpus)."""
    g = torch.Generator().manual_seed(seed)
    for _ in range(n_batches):
        yield torch.randint(0, vocab_size, (batch_size, seq_len), generator=g)


def chunk_and_tokenize(dataset, tokenizer, text_key: str = "text", max_length: int = MAX_SENTENCE_LEN, num_proc: Optional[int] = None):
    """Pack a text dataset into fixed-length token rows (reference :136-235).

    Concatenates tokenized documents and splits into max_length windows.
    Returns a datasets.Dataset with an 'input_ids' column.
    """
    if num_proc is None:
        num_proc = min((os.cpu_count() or 2) // 2, 8) or 1

    def tok_fn(examples):
        ids = tokenizer(examples[text_key], add_special_tokens=False)["input_ids"]
        flat: List[int] = []
        for row in ids:
            flat.extend(row)
        n = (len(flat) // max_length) * max_length
        rows = [flat[i : i + max_length] for i in range(0, n, max_length)]
        return {"input_ids": rows}

    return dataset.map(
        tok_fn,
        batched=True,
        num_proc=num_proc,
        remove_columns=dataset.column_names,
    )


# ---------------------------------------------------------------------------
# capture loop
# ---------------------------------------------------------------------------


class _PinnedStager:
    """FIFO ring of reusable pinned D2H staging buffers.

    ``stage(t, out)`` async-copies a CUDA tensor into a pinned slot; when the
    ring is full the oldest slot is retired first (event sync + copy to a
    pageable tensor appended to ``out``), so ``out`` receives tensors in
    staging order and pinned memory stays bounded at ~``depth`` batches.
    """

    def __init__(self, depth: int = 4):
        import collections

        self.depth = depth
        self.inflight = collections.deque()  # (flat pinned buf, event, shape)
        self.free: List[torch.Tensor] = []

    def stage(self, t: torch.Tensor, out: List[torch.Tensor]) -> None:
        if len(self.inflight) >= self.depth:
            out.append(self._retire())
        buf = None
        for i, f in enumerate(self.free):
            if f.numel() >= t.numel() and f.dtype == t.dtype:
                buf = self.free.pop(i)
                break
        if buf is None:
            buf = torch.empty(t.numel(), dtype=t.dtype, pin_memory=True)
        view = buf[: t.numel()].view(t.shape)
        view.copy_(t, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record()
        self.inflight.append((buf, ev, t.shape))

    def _retire(self) -> torch.Tensor:
        import math

        buf, ev, shape = self.inflight.popleft()
        ev.synchronize()
        out = buf[: math.prod(shape)].view(shape).clone()  # pageable copy
        self.free.append(buf)
        return out

    def retire_all(self, out: List[torch.Tensor]) -> None:
        while self.inflight:
            out.append(self._retire())


def make_activation_dataset_hf(
    token_batches: Iterable[torch.Tensor],
    model,
    layers: List[int],
    layer_loc: str,
    chunk_size: int,
    n_chunks: int,
    output_folder: str = "activation_data",
    device: str = "cuda:0",
    precision: Literal["float16", "float32"] = "float16",
    model_name: str = "",
    flush_every: int = 8,
) -> int:
    """Run the LM over token batches, capture one or many layers, write fp16
    ``{folder}/{tensor_name}/{i}.pt`` chunks.  Returns total activations.

    The GPU-side store holds up to ``flush_every`` batches of [b*l, d] fp16
    tensors, then drains them to host with non_blocking copies on the default
    stream (overlapping the next forward's compute).
    """
    dtype = torch.float16 if precision == "float16" else torch.float32
    model.eval()

    stores: Dict[int, List[torch.Tensor]] = {l: [] for l in layers}
    host_parts: Dict[int, List[torch.Tensor]] = {l: [] for l in layers}
    host_counts: Dict[int, int] = {l: 0 for l in layers}
    chunk_idx: Dict[int, int] = {l: 0 for l in layers}
    stagers: Dict[int, _PinnedStager] = {l: _PinnedStager() for l in layers}
    total = 0

    folders = {}
    for l in layers:
        tname = make_tensor_name(l, layer_loc, model_name)
        folder = os.path.join(output_folder, tname) if len(layers) > 1 else output_folder
        os.makedirs(folder, exist_ok=True)
        folders[l] = folder

    def drain(l: int):
        # NOTE: the capture hook holds a reference to stores[l]; mutate it in
        # place, never rebind it.  D2H goes through a bounded ring of REUSED
        # pinned staging buffers (pinned destination keeps the non_blocking
        # copy async; the ring keeps the pinned footprint at ~depth batches
        # instead of a whole chunk — ADVICE.md round-1).
        while stores[l]:
            t = stores[l].pop(0).to(dtype)
            if t.is_cuda:
                stagers[l].stage(t, host_parts[l])
                host_counts[l] += t.shape[0]
            else:
                host_parts[l].append(t)
                host_counts[l] += t.shape[0]

    def flush_chunk(l: int, final: bool = False):
        if host_counts[l] == 0:
            return
        stagers[l].retire_all(host_parts[l])
        data = torch.cat(host_parts[l], dim=0)
        while data.shape[0] >= chunk_size or (final and data.shape[0] > 0):
            part = data[:chunk_size]
            save_activation_chunk(part, chunk_idx[l], folders[l])
            chunk_idx[l] += 1
            data = data[chunk_size:]
            if chunk_idx[l] >= n_chunks:
                data = data[:0]
                break
        host_parts[l] = [data] if data.shape[0] else []
        host_counts[l] = data.shape[0]

    with torch.no_grad():
        hooks = contextlib.ExitStack()
        with hooks:
            for l in layers:
                hooks.enter_context(capture_activation_hook(model, l, layer_loc, stores[l]))

            for bi, batch in enumerate(token_batches):
                batch = batch.to(device)
                model(input_ids=batch)
                total += batch.numel()
                if (bi + 1) % flush_every == 0:
                    for l in layers:
                        drain(l)
                        if host_counts[l] >= chunk_size:
                            flush_chunk(l)
                if all(chunk_idx[l] >= n_chunks for l in layers):
                    break

            for l in layers:
                drain(l)
                flush_chunk(l, final=True)

    return total


def save_activation_chunk(data: torch.Tensor, n_saved_chunks: int, dataset_folder: str) -> None:
    """Reference chunk file layout (:499-503): {folder}/{i}.pt fp16 [N, d]."""
    os.makedirs(dataset_folder, exist_ok=True)
    torch.save(data.contiguous(), os.path.join(dataset_folder, f"{n_saved_chunks}.pt"))


# ---------------------------------------------------------------------------
# top-level entry points (reference setup_data :544, setup_data_new :505)
# ---------------------------------------------------------------------------

def _token_stream_for(cfg_like: Dict[str, Any], tokenizer, model, dataset_name, n_batches, max_length, batch_size, device):
    vocab = model.config.vocab_size
    if dataset_name in (None, "", "synthetic", "random"):
        return synthetic_token_batches(vocab, batch_size, max_length, n_batches)
    # local datasets path (network-less environment): load_from_disk or text file
    try:
        import datasets as hf_datasets

        if os.path.isdir(dataset_name):
            ds = hf_datasets.load_from_disk(dataset_name)
        else:
            ds = hf_datasets.load_dataset(dataset_name, split="train")
        tokenized = chunk_and_tokenize(ds, tokenizer, max_length=max_length)

        def gen():
            buf = []
            for row in tokenized:
                buf.append(torch.tensor(row["input_ids"], dtype=torch.long))
                if len(buf) == batch_size:
                    yield torch.stack(buf)
                    buf = []

        return gen()
    except Exception as e:  # noqa: BLE001
        print(f"[activation_dataset] could not load dataset {dataset_name!r} ({e}); using synthetic tokens")
        return synthetic_token_batches(vocab, batch_size, max_length, n_batches)


def setup_data(
    tokenizer,
    transformer,
    dataset_name: str,
    dataset_folder: str,
    layer,
    layer_loc: str = "residual",
    n_chunks: int = 1,
    chunk_size_gb: float = CHUNK_SIZE_GB,
    device: str = "cuda:0",
    center_dataset: bool = False,
    model_name: str = "",
    max_length: int = MAX_SENTENCE_LEN,
    model_batch_size: int = MODEL_BATCH_SIZE,
) -> int:
    """Generate activation chunks for one or many layers (reference :544-611)."""
    layers = layer if isinstance(layer, (list, tuple)) else [layer]
    d = None
    if model_name:
        d = get_activation_size(model_name, layer_loc)
    else:
        d = transformer.config.hidden_size
    chunk_activations = int(chunk_size_gb * 1024**3) // (d * 2)  # fp16
    toks_per_batch = model_batch_size * max_length
    n_batches = (chunk_activations * n_chunks) // toks_per_batch + 1

    stream = _token_stream_for({}, tokenizer, transformer, dataset_name, n_batches, max_length, model_batch_size, device)
    total = make_activation_dataset_hf(
        stream,
        transformer,
        list(layers),
        layer_loc,
        chunk_size=chunk_activations,
        n_chunks=n_chunks,
        output_folder=dataset_folder,
        device=device,
        model_name=model_name,
    )

    if center_dataset:
        # subtract the first chunk's mean from every chunk (reference big_sweep.py:359-364)
        first = torch.load(os.path.join(dataset_folder, "0.pt")).float()
        mean = first.mean(dim=0)
        for i in range(n_chunks):
            p = os.path.join(dataset_folder, f"{i}.pt")
            if os.path.exists(p):
                t = torch.load(p).float() - mean
                torch.save(t.to(torch.float16), p)
        torch.save(mean, os.path.join(dataset_folder, "mean.pt"))
    return total


def setup_data_new(*args, **kwargs) -> int:
    """Alias of setup_data — the HF-hook path IS the native path here
    (reference setup_data_new :505 wraps make_activation_dataset_hf)."""
    return setup_data(*args, **kwargs)
