"""Evaluation prompt datasets: IOI clean/corrupted pairs, gender-name pairs.

Covers reference ``test_datasets/`` (C26): IOI template prompts with
single-token name filtering (``test_datasets/ioi.py:11-67``), a templated
counterfact-style builder standing in for the Redwood IOIDataset port
(``ioi_counterfact.py``), and gender-name preprocessing
(``preprocess_gender_dataset.py``).  Own name/template pools (no copied
lists): the contract is the structure — ABB/ABA templated pairs tokenized to
equal length — not the particular words.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import torch

ABB_TEMPLATE = "Then, {a} and {b} went to the {place}. {b} gave a {thing} to {a}"
ABA_TEMPLATE = "Then, {a} and {b} went to the {place}. {a} gave a {thing} to {b}"

NAME_POOL = [
    "Alice", "Tom", "Sarah", "Jack", "Emma", "Ryan", "Grace", "Adam", "Lucy",
    "Mark", "Anna", "Paul", "Clara", "Henry", "Rose", "Peter", "Julia", "Sam",
    "Laura", "David", "Nina", "Eric", "Diana", "Luke", "Helen", "Simon",
    "Kate", "Victor", "Amy", "Oscar",
]
PLACE_POOL = ["market", "library", "park", "office", "harbor"]
THING_POOL = ["book", "drink", "basket", "ticket", "snack", "kite"]


def filter_single_token(tokenizer, words: List[str], prefix: str = " ") -> List[str]:
    """Keep words that tokenize (with a leading space) to exactly one token
    (reference ioi.py:22-29)."""
    kept = []
    for w in words:
        if len(tokenizer(prefix + w, add_special_tokens=False)["input_ids"]) == 1:
            kept.append(w)
    return kept


def generate_ioi_dataset(
    tokenizer,
    n_abb_a: int,
    n_abb_b: int,
    seed: int = 42,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Clean/corrupted IOI prompt pairs as equal-shape token tensors
    (reference ioi.py:11-67)."""
    rng = np.random.default_rng(seed)
    names = filter_single_token(tokenizer, NAME_POOL)
    places = filter_single_token(tokenizer, PLACE_POOL)
    things = filter_single_token(tokenizer, THING_POOL)
    assert len(names) >= 2 and places and things, "pools emptied by token filter"

    clean, corrupted = [], []
    for count, (c_tpl, x_tpl) in ((n_abb_a, (ABB_TEMPLATE, ABA_TEMPLATE)),
                                  (n_abb_b, (ABA_TEMPLATE, ABB_TEMPLATE))):
        for _ in range(count):
            a, b = rng.choice(names, size=2, replace=False)
            kw = dict(a=a, b=b, place=rng.choice(places), thing=rng.choice(things))
            clean.append(c_tpl.format(**kw))
            corrupted.append(x_tpl.format(**kw))

    clean_ids = tokenizer(clean, add_special_tokens=False)["input_ids"]
    corr_ids = tokenizer(corrupted, add_special_tokens=False)["input_ids"]
    return torch.tensor(clean_ids), torch.tensor(corr_ids)


def generate_gender_dataset(
    tokenizer,
    n_pairs: int = 100,
    seed: int = 0,
    female_names: Optional[List[str]] = None,
    male_names: Optional[List[str]] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Name-templated prompt pairs labeled by pool membership, for linear
    probe / erasure evals (reference preprocess_gender_dataset.py)."""
    rng = np.random.default_rng(seed)
    female = filter_single_token(tokenizer, female_names or
                                 ["Alice", "Sarah", "Emma", "Grace", "Lucy", "Anna", "Clara", "Rose", "Julia", "Laura"])
    male = filter_single_token(tokenizer, male_names or
                               ["Tom", "Jack", "Ryan", "Adam", "Mark", "Paul", "Henry", "Peter", "Sam", "David"])
    template = "My friend {name} said that"
    prompts, labels = [], []
    for _ in range(n_pairs):
        f = rng.choice(female)
        m = rng.choice(male)
        prompts.append(template.format(name=f))
        labels.append(1)
        prompts.append(template.format(name=m))
        labels.append(0)
    ids = tokenizer(prompts, add_special_tokens=False, padding=True)["input_ids"]
    return torch.tensor(ids), torch.tensor(labels)


def generate_induction_dataset(tokenizer, n_prompts: int = 64, seq_len: int = 32, seed: int = 0) -> torch.Tensor:
    """Repeated-random-token sequences for induction-head evals (the
    reference's test_datasets/induction.py is empty; this fills the slot)."""
    g = torch.Generator().manual_seed(seed)
    vocab = tokenizer.vocab_size if tokenizer is not None else 50257
    half = torch.randint(0, vocab, (n_prompts, seq_len // 2), generator=g)
    return torch.cat([half, half], dim=1)


def gender_prompt_batch(n: int = 256, vocab_size: int = 50304, seq_len: int = 16,
                        pool_size: int = 10, seed: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    """Tokenizer-free synthetic version of the gender prompt set for the
    erasure study (reference preprocess_gender_dataset.py feeds real names;
    this no-network environment substitutes two disjoint token-id pools as
    the "names", which makes the concept linearly decodable from the host
    LM's embeddings — exactly what the erasure pipeline needs).

    Returns (tokens [n, seq_len], labels [n]) where the name token sits at a
    fixed template position and the label is the pool it came from."""
    rng = np.random.default_rng(seed)
    lo = 100  # avoid special-token ids
    female_pool = rng.choice(np.arange(lo, vocab_size // 2), size=pool_size, replace=False)
    male_pool = rng.choice(np.arange(vocab_size // 2, vocab_size - 1), size=pool_size, replace=False)
    tokens = rng.integers(lo, vocab_size - 1, size=(n, seq_len))
    labels = rng.integers(0, 2, size=n)
    name_pos = seq_len // 2
    for i in range(n):
        pool = female_pool if labels[i] == 1 else male_pool
        tokens[i, name_pos] = rng.choice(pool)
    return torch.from_numpy(tokens).long(), torch.from_numpy(labels).long()
