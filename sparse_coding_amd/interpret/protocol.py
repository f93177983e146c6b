"""The neuron-explainer explain/simulate protocol, implemented natively.

Parity with reference ``interpret.py:265-386`` which drives OpenAI's
``neuron_explainer`` package (TokenActivationPairExplainer over GPT-4,
ExplanationNeuronSimulator over davinci, correlation scoring).  This module
re-implements the protocol itself — prompt construction, record splitting,
simulation parsing, correlation scoring with top/random/combined aggregates,
and the reference's on-disk result format (``feature_N/explanation.txt`` with
``Score:``/``Top only score:``/``Random only score:`` lines, readable by
``drivers.read_scores``) — against an injectable LLM client, because this
environment has no network:

* :class:`MockLLMClient` — deterministic offline client (tests, CI);
* :class:`HFLocalClient` — any local HuggingFace causal LM;
* any object with ``batch_complete(prompts) -> List[str]`` works, so an
  OpenAI-backed client can be dropped in outside this sandbox.

Record-split constants follow the reference (interpret.py:53-57): 64-token
fragments, 20 top + 20 random records, splits of 5; the explainer trains on
the non-validation top splits and the simulator is scored on 5 top + 5
random held-out records (the reference's ``assert len(...) == 10``).
"""

from __future__ import annotations

import math
import os
import pickle
import re
from dataclasses import dataclass
from typing import Dict, List, Sequence, Tuple

import numpy as np

from sparse_coding_amd.interpret.interpret import ActivationRecord

EXAMPLES_PER_SPLIT = 5   # reference OPENAI_EXAMPLES_PER_SPLIT
N_SPLITS = 4             # reference N_SPLITS
TOTAL_EXAMPLES = EXAMPLES_PER_SPLIT * N_SPLITS
MAX_ACT_LEVEL = 10       # the protocol's 0-10 discretized activation scale


# ---------------------------------------------------------------------------
# record splitting (neuron_explainer NeuronRecord semantics)
# ---------------------------------------------------------------------------

@dataclass
class NeuronRecordSplit:
    train: List[ActivationRecord]       # explainer input
    valid_top: List[ActivationRecord]   # held-out top split
    valid_random: List[ActivationRecord]

    @property
    def valid(self) -> List[ActivationRecord]:
        return self.valid_top + self.valid_random


def split_records(top: Sequence[ActivationRecord],
                  random: Sequence[ActivationRecord]) -> NeuronRecordSplit:
    """First top split is validation, the rest train (neuron_explainer's
    ActivationRecordSliceParams convention); 5 random records complete the
    10 scored sequences."""
    top = list(top)[:TOTAL_EXAMPLES]
    return NeuronRecordSplit(
        train=top[EXAMPLES_PER_SPLIT:],
        valid_top=top[:EXAMPLES_PER_SPLIT],
        valid_random=list(random)[:EXAMPLES_PER_SPLIT],
    )


def calculate_max_activation(records: Sequence[ActivationRecord]) -> float:
    return max((max(r.activations) for r in records if r.activations), default=0.0)


def _discretize(a: float, max_act: float) -> int:
    if max_act <= 0:
        return 0
    return min(MAX_ACT_LEVEL, max(0, int(round(a / max_act * MAX_ACT_LEVEL))))


# ---------------------------------------------------------------------------
# prompt construction
# ---------------------------------------------------------------------------

def build_explainer_prompt(train_records: Sequence[ActivationRecord],
                           max_activation: float) -> str:
    """Token-activation-pair explainer prompt (the TokenActivationPairExplainer
    recipe): each example lists token<TAB>level pairs on the 0-10 scale;
    the model is asked for one short behaviour summary."""
    parts = [
        "We are studying a neuron in a neural network. Each token has an "
        f"activation level from 0 to {MAX_ACT_LEVEL} (0 = not active, "
        f"{MAX_ACT_LEVEL} = maximally active).",
        "Look at the token-activation pairs and summarize in a short phrase "
        "what pattern of tokens or context this neuron responds to.",
        "",
    ]
    for i, rec in enumerate(train_records):
        parts.append(f"Example {i + 1}:")
        parts.append("<start>")
        for tok, act in zip(rec.tokens, rec.activations):
            parts.append(f"{tok}\t{_discretize(act, max_activation)}")
        parts.append("<end>")
        parts.append("")
    parts.append("Explanation of neuron behaviour: this neuron activates on")
    return "\n".join(parts)


def build_simulator_prompt(explanation: str, tokens: Sequence[str]) -> str:
    """Per-token activation-prediction prompt (ExplanationNeuronSimulator
    recipe): given the explanation, the model fills in one 0-10 level per
    token, format ``token<TAB>level``."""
    parts = [
        "We are studying a neuron in a neural network. Its behaviour: this "
        f"neuron activates on{explanation}",
        f"For each token below, predict the neuron's activation level from 0 "
        f"to {MAX_ACT_LEVEL}. Answer with one line per token in the format "
        "token<TAB>level.",
        "<start>",
    ]
    for tok in tokens:
        parts.append(f"{tok}\t<level>")
    parts.append("<end>")
    return "\n".join(parts)


def parse_simulation(response: str, tokens: Sequence[str]) -> List[float]:
    """Parse ``token<TAB>level`` lines into per-token levels; malformed or
    missing lines default to 0 (the uncalibrated-simulator convention of
    treating unparseable output as no-activation)."""
    levels: List[float] = []
    lines = [ln for ln in response.splitlines() if "\t" in ln]
    for i, tok in enumerate(tokens):
        val = 0.0
        if i < len(lines):
            tail = lines[i].rsplit("\t", 1)[-1].strip()
            m = re.match(r"-?\d+(\.\d+)?", tail)
            if m:
                val = float(m.group(0))
        levels.append(min(float(MAX_ACT_LEVEL), max(0.0, val)))
    return levels


# ---------------------------------------------------------------------------
# LLM clients
# ---------------------------------------------------------------------------

class MockLLMClient:
    """Deterministic offline client: 'explains' with a token list and
    'simulates' by firing on tokens that appeared highly-activated in the
    explanation.  Gives a positive autointerp score for genuinely
    token-selective features — enough to test the whole pipeline offline."""

    def __init__(self, top_tokens_per_expl: int = 8):
        self.k = top_tokens_per_expl

    def batch_complete(self, prompts: List[str]) -> List[str]:
        return [self._complete(p) for p in prompts]

    def _complete(self, prompt: str) -> str:
        if "Explanation of neuron behaviour" in prompt:
            scores: Dict[str, float] = {}
            for line in prompt.splitlines():
                if "\t" in line:
                    tok, _, lvl = line.rpartition("\t")
                    try:
                        scores[tok] = max(scores.get(tok, 0.0), float(lvl))
                    except ValueError:
                        continue
            top = sorted(scores, key=scores.get, reverse=True)[: self.k]
            return " tokens: " + ", ".join(repr(t) for t in top if scores[t] > 0)
        # simulation: level 10 for tokens named in the explanation, else 0
        expl_line = prompt.splitlines()[0]
        expl_tail = expl_line.split("activates on", 1)[-1]
        named = set(re.findall(r"'([^']*)'", expl_tail))
        out = []
        in_block = False
        for line in prompt.splitlines():
            if line == "<start>":
                in_block = True
                continue
            if line == "<end>":
                break
            if in_block and "\t" in line:
                tok = line.rsplit("\t", 1)[0]
                out.append(f"{tok}\t{MAX_ACT_LEVEL if tok in named else 0}")
        return "\n".join(out)


class HFLocalClient:
    """Local HuggingFace causal-LM client.  Batches prompts through the model
    (greedy, short generations) — the MI355X replacement for the reference's
    asyncio-batched API calls (interpret.py:338-343 MAX_CONCURRENT)."""

    def __init__(self, model, tokenizer, device: str = "cuda:0",
                 max_new_tokens: int = 256, batch_size: int = 8):
        self.model = model
        self.tokenizer = tokenizer
        self.device = device
        self.max_new_tokens = max_new_tokens
        self.batch_size = batch_size
        if tokenizer.pad_token is None:
            tokenizer.pad_token = tokenizer.eos_token

    def batch_complete(self, prompts: List[str]) -> List[str]:
        import torch

        outs: List[str] = []
        for s in range(0, len(prompts), self.batch_size):
            chunk = prompts[s : s + self.batch_size]
            enc = self.tokenizer(chunk, return_tensors="pt", padding=True,
                                 truncation=True, max_length=2048).to(self.device)
            with torch.no_grad():
                gen = self.model.generate(
                    **enc, max_new_tokens=self.max_new_tokens, do_sample=False,
                    pad_token_id=self.tokenizer.pad_token_id,
                )
            for i in range(len(chunk)):
                new = gen[i, enc["input_ids"].shape[1] :]
                outs.append(self.tokenizer.decode(new, skip_special_tokens=True))
        return outs


# ---------------------------------------------------------------------------
# scoring
# ---------------------------------------------------------------------------

def _corr(real: np.ndarray, pred: np.ndarray) -> float:
    if real.std() < 1e-9 or pred.std() < 1e-9:
        return 0.0
    c = float(np.corrcoef(real, pred)[0, 1])
    return 0.0 if math.isnan(c) else c


@dataclass
class ScoredSimulation:
    """Correlation scores over the validation records (the uncalibrated
    simulator's preferred score), with the reference's three aggregates."""

    explanation: str
    real: List[List[float]]      # per valid record
    predicted: List[List[float]]
    n_top: int

    def score(self) -> float:
        return _corr(np.concatenate([np.asarray(r) for r in self.real]),
                     np.concatenate([np.asarray(p) for p in self.predicted]))

    def top_only_score(self) -> float:
        return _corr(np.concatenate([np.asarray(r) for r in self.real[: self.n_top]]),
                     np.concatenate([np.asarray(p) for p in self.predicted[: self.n_top]]))

    def random_only_score(self) -> float:
        if len(self.real) <= self.n_top:
            return 0.0
        return _corr(np.concatenate([np.asarray(r) for r in self.real[self.n_top :]]),
                     np.concatenate([np.asarray(p) for p in self.predicted[self.n_top :]]))


def explain_and_score(
    split: NeuronRecordSplit,
    client,
) -> Tuple[str, ScoredSimulation]:
    """One feature through the full protocol: explain on the train records,
    simulate every validation record, score by correlation."""
    max_act = calculate_max_activation(split.train)
    [explanation] = client.batch_complete(
        [build_explainer_prompt(split.train, max_act)])
    sim_prompts = [build_simulator_prompt(explanation, r.tokens) for r in split.valid]
    responses = client.batch_complete(sim_prompts)
    predicted = [parse_simulation(resp, r.tokens)
                 for resp, r in zip(responses, split.valid)]
    real = [list(r.activations) for r in split.valid]
    return explanation, ScoredSimulation(
        explanation=explanation, real=real, predicted=predicted,
        n_top=len(split.valid_top))


def interpret_protocol(
    acts,
    tokens,
    feature_indices: Sequence[int],
    client,
    save_folder: str,
    tokenizer=None,
    explainer_model_name: str = "local",
    simulator_model_name: str = "local",
) -> Dict[int, float]:
    """The reference ``interpret()`` driver loop (:265-386): per feature,
    select records, explain, simulate, score, and write the reference's
    exact result layout:

        {save_folder}/feature_{n}/explanation.txt   (Score / Top only /
                                                     Random only lines)
        {save_folder}/feature_{n}/scored_simulation.pkl
        {save_folder}/feature_{n}/neuron_record.pkl

    Features already on disk are skipped (resumable, like the reference).
    Returns {feature: combined score}.
    """
    from sparse_coding_amd.interpret.interpret import select_activation_records

    scores: Dict[int, float] = {}
    for fi, feature in enumerate(feature_indices):
        feature_folder = os.path.join(save_folder, f"feature_{feature}")
        if os.path.exists(feature_folder):
            continue
        top, rand = select_activation_records(
            acts, tokens, fi, tokenizer,
            top_k=TOTAL_EXAMPLES, n_random=TOTAL_EXAMPLES)
        # reference :300-316: random records must come from fragments where
        # the feature fires at least once; give up if there are none
        rand = [r for r in rand if max(r.activations, default=0.0) > 0]
        if len(rand) < EXAMPLES_PER_SPLIT:
            os.makedirs(feature_folder, exist_ok=True)  # placeholder: skip marker
            continue
        split = split_records(top, rand)
        explanation, sim = explain_and_score(split, client)
        score = sim.score()
        scores[int(feature)] = score

        os.makedirs(feature_folder, exist_ok=True)
        with open(os.path.join(feature_folder, "explanation.txt"), "w") as f:
            f.write(f"{explanation}\nScore: {score:.2f}\n"
                    f"Explainer model: {explainer_model_name}\n"
                    f"Simulator model: {simulator_model_name}\n")
            f.write(f"Top only score: {sim.top_only_score():.2f}\n")
            f.write(f"Random only score: {sim.random_only_score():.2f}\n")
        with open(os.path.join(feature_folder, "scored_simulation.pkl"), "wb") as f:
            pickle.dump(sim, f)
        with open(os.path.join(feature_folder, "neuron_record.pkl"), "wb") as f:
            pickle.dump(split, f)
    return scores
