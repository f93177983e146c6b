"""IOI (indirect-object identification) counterfactual dataset — the Redwood
IOIDataset semantics (reference test_datasets/ioi_counterfact.py:1-372, which
ports redwoodresearch/Easy-Transformer ioi_dataset.py).

What the semantics require (and the round-1 prompt-pair generator lacked):

* the template families: BABA ("[B] and [A] ... [B] gave a [OBJECT] to [A]"),
  ABBA (first-clause name order swapped, derived programmatically exactly as
  the reference does at :205-213), the LONG / LATE_IOS / EARLY_IOS variants,
  and the three-name ABC/BAC distractor templates;
* per-prompt metadata: text, IO (indirect object), S (subject), TEMPLATE_IDX
  and the sampled [PLACE]/[OBJECT] nouns (reference :304-336);
* word-index maps: token positions of IO / S1 / S2 / end per prompt (the
  Redwood ``word_idx`` table that the IOI case studies index activations
  with);
* flipped-set generation: replace the IO / S / S2 role with a fresh name
  (Redwood ``gen_flipped_prompts``; the reference's counterfact generator
  is the IO->RAND flip, :321-334);
* the padded counterfact tensor pair with per-prompt sequence lengths
  (reference gen_ioi_dataset :338-372).

The name/noun/template banks are the dataset DEFINITION (originally
Redwood's, MIT) — they are reproduced as data; all machinery around them is
re-designed.  Everything runs without a network: pass a real tokenizer when
available, else the built-in :class:`WordTokenizer` (one token per
whitespace word, hashed ids) keeps the whole pipeline testable offline.
"""

from __future__ import annotations

import random as rd
import re
from typing import Dict, List, Optional, Sequence, Tuple

import torch

NAMES = [
    "Michael", "Christopher", "Jessica", "Matthew", "Ashley", "Jennifer",
    "Joshua", "Amanda", "Daniel", "David", "James", "Robert", "John",
    "Joseph", "Andrew", "Ryan", "Brandon", "Jason", "Justin", "Sarah",
    "William", "Jonathan", "Stephanie", "Brian", "Nicole", "Nicholas",
    "Anthony", "Heather", "Eric", "Elizabeth", "Adam", "Megan", "Melissa",
    "Kevin", "Steven", "Thomas", "Timothy", "Christina", "Kyle", "Rachel",
    "Laura", "Lauren", "Amber", "Brittany", "Danielle", "Richard",
    "Kimberly", "Jeffrey", "Amy", "Crystal", "Michelle", "Tiffany",
    "Jeremy", "Benjamin", "Mark", "Emily", "Aaron", "Charles", "Rebecca",
    "Jacob", "Stephen", "Patrick", "Sean", "Erin", "Jamie", "Kelly",
    "Samantha", "Nathan", "Sara", "Dustin", "Paul", "Angela", "Tyler",
    "Scott", "Katherine", "Andrea", "Gregory", "Erica", "Mary", "Travis",
    "Lisa", "Kenneth", "Bryan", "Lindsey", "Kristen", "Jose", "Alexander",
    "Jesse", "Katie", "Lindsay", "Shannon", "Vanessa", "Courtney",
    "Christine", "Alicia", "Cody", "Allison", "Bradley", "Samuel",
]

BABA_TEMPLATES = [
    "Then, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] had a lot of fun at the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] were working at the [PLACE]. [B] decided to give a [OBJECT] to [A]",
    "Then, [B] and [A] were thinking about going to the [PLACE]. [B] wanted to give a [OBJECT] to [A]",
    "Then, [B] and [A] had a long argument, and afterwards [B] said to [A]",
    "After [B] and [A] went to the [PLACE], [B] gave a [OBJECT] to [A]",
    "When [B] and [A] got a [OBJECT] at the [PLACE], [B] decided to give it to [A]",
    "When [B] and [A] got a [OBJECT] at the [PLACE], [B] decided to give the [OBJECT] to [A]",
    "While [B] and [A] were working at the [PLACE], [B] gave a [OBJECT] to [A]",
    "While [B] and [A] were commuting to the [PLACE], [B] gave a [OBJECT] to [A]",
    "After the lunch, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Afterwards, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] had a long argument. Afterwards [B] said to [A]",
    "The [PLACE] [B] and [A] went to had a [OBJECT]. [B] gave it to [A]",
    "Friends [B] and [A] found a [OBJECT] at the [PLACE]. [B] gave it to [A]",
]

BABA_LONG_TEMPLATES = [
    "Then in the morning, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then in the morning, [B] and [A] had a lot of fun at the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then in the morning, [B] and [A] were working at the [PLACE]. [B] decided to give a [OBJECT] to [A]",
    "Then in the morning, [B] and [A] were thinking about going to the [PLACE]. [B] wanted to give a [OBJECT] to [A]",
    "Then in the morning, [B] and [A] had a long argument, and afterwards [B] said to [A]",
    "After taking a long break [B] and [A] went to the [PLACE], [B] gave a [OBJECT] to [A]",
    "When soon afterwards [B] and [A] got a [OBJECT] at the [PLACE], [B] decided to give it to [A]",
    "When soon afterwards [B] and [A] got a [OBJECT] at the [PLACE], [B] decided to give the [OBJECT] to [A]",
    "While spending time together [B] and [A] were working at the [PLACE], [B] gave a [OBJECT] to [A]",
    "While spending time together [B] and [A] were commuting to the [PLACE], [B] gave a [OBJECT] to [A]",
    "After the lunch in the afternoon, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Afterwards, while spending time together [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then in the morning afterwards, [B] and [A] had a long argument. Afterwards [B] said to [A]",
    "The local big [PLACE] [B] and [A] went to had a [OBJECT]. [B] gave it to [A]",
    "Friends separated at birth [B] and [A] found a [OBJECT] at the [PLACE]. [B] gave it to [A]",
]

BABA_LATE_IOS = [
    "Then, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] had a lot of fun at the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] were working at the [PLACE]. [B] decided to give a [OBJECT] to [A]",
    "Then, [B] and [A] were thinking about going to the [PLACE]. [B] wanted to give a [OBJECT] to [A]",
    "Then, [B] and [A] had a long argument and after that [B] said to [A]",
    "After the lunch, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Afterwards, [B] and [A] went to the [PLACE]. [B] gave a [OBJECT] to [A]",
    "Then, [B] and [A] had a long argument. Afterwards [B] said to [A]",
]

BABA_EARLY_IOS = [
    "Then [B] and [A] went to the [PLACE], and [B] gave a [OBJECT] to [A]",
    "Then [B] and [A] had a lot of fun at the [PLACE], and [B] gave a [OBJECT] to [A]",
    "Then [B] and [A] were working at the [PLACE], and [B] decided to give a [OBJECT] to [A]",
    "Then [B] and [A] were thinking about going to the [PLACE], and [B] wanted to give a [OBJECT] to [A]",
    "Then [B] and [A] had a long argument, and after that [B] said to [A]",
    "After the lunch [B] and [A] went to the [PLACE], and [B] gave a [OBJECT] to [A]",
    "Afterwards [B] and [A] went to the [PLACE], and [B] gave a [OBJECT] to [A]",
    "Then [B] and [A] had a long argument, and afterwards [B] said to [A]",
]

ABC_TEMPLATES = [
    "Then, [A], [B] and [C] went to the [PLACE]. [B] and [C] gave a [OBJECT] to [A]",
    "Afterwards [A], [B] and [C] went to the [PLACE]. [B] and [C] gave a [OBJECT] to [A]",
    "When [A], [B] and [C] arrived at the [PLACE], [B] and [C] gave a [OBJECT] to [A]",
    "Friends [A], [B] and [C] went to the [PLACE]. [B] and [C] gave a [OBJECT] to [A]",
]

BAC_TEMPLATES = [t.replace("[B]", "[A]", 1).replace("[A]", "[B]", 1) for t in ABC_TEMPLATES]


def _swap_first_clause(template: str) -> str:
    """Swap the FIRST [B]/[A] pair only (BABA -> ABBA), the reference's
    in-place loop at :205-213 re-expressed."""
    out = template
    b = out.index("[B]")
    out = out[:b] + "[#]" + out[b + 3 :]
    a = out.index("[A]")
    out = out[:a] + "[B]" + out[a + 3 :]
    return out.replace("[#]", "[A]", 1)


ABBA_TEMPLATES = [_swap_first_clause(t) for t in BABA_TEMPLATES]
ABBA_LATE_IOS = [_swap_first_clause(t) for t in BABA_LATE_IOS]
ABBA_EARLY_IOS = [_swap_first_clause(t) for t in BABA_EARLY_IOS]

VERBS = [" tried", " said", " decided", " wanted", " gave"]
PLACES = ["store", "garden", "restaurant", "school", "hospital", "office", "house", "station"]
OBJECTS = ["ring", "kiss", "bone", "basketball", "computer", "necklace", "drink", "snack"]
ANIMALS = ["dog", "cat", "snake", "elephant", "beetle", "hippo", "giraffe", "tiger",
           "husky", "lion", "panther", "whale", "dolphin", "beaver", "rabbit", "fox",
           "lamb", "ferret"]
NOUNS_DICT = {"[PLACE]": PLACES, "[OBJECT]": OBJECTS}

FAMILIES = {
    "BABA": BABA_TEMPLATES, "ABBA": ABBA_TEMPLATES,
    "BABA_LONG": BABA_LONG_TEMPLATES,
    "BABA_LATE_IOS": BABA_LATE_IOS, "ABBA_LATE_IOS": ABBA_LATE_IOS,
    "BABA_EARLY_IOS": BABA_EARLY_IOS, "ABBA_EARLY_IOS": ABBA_EARLY_IOS,
    "ABC": ABC_TEMPLATES, "BAC": BAC_TEMPLATES,
    "mixed": ABBA_TEMPLATES + BABA_TEMPLATES,
}


def multiple_replace(mapping: Dict[str, str], text: str) -> str:
    regex = re.compile("(%s)" % "|".join(map(re.escape, mapping.keys())))
    return regex.sub(lambda mo: mapping[mo.group(0)], text)


class WordTokenizer:
    """Offline stand-in for a real tokenizer: one token per whitespace word
    (punctuation split off), ids hashed into [reserve, vocab).  Every name is
    single-token by construction, so the single-token filter passes and the
    word-index maps are exact."""

    def __init__(self, vocab_size: int = 50304, reserve: int = 10):
        self.vocab_size = vocab_size
        self.reserve = reserve

    def _words(self, text: str) -> List[str]:
        return re.findall(r"[\w']+|[.,!?;]", text)

    def _id(self, w: str) -> int:
        import hashlib

        h = int(hashlib.md5(w.encode()).hexdigest(), 16)
        return self.reserve + h % (self.vocab_size - self.reserve)

    def __call__(self, text, **kw) -> Dict[str, list]:
        if isinstance(text, str):
            return {"input_ids": [self._id(w) for w in self._words(text.strip())]}
        return {"input_ids": [[self._id(w) for w in self._words(t.strip())] for t in text]}


def _single_token(tokenizer, name: str) -> bool:
    ids = tokenizer(" " + name)["input_ids"]
    return len(ids) == 1


def gen_prompt_counterfact(tokenizer, templates: Sequence[str], names: Sequence[str],
                           nouns_dict: Dict[str, Sequence[str]], N: int,
                           seed: Optional[int] = None) -> Tuple[List[Dict], List[Dict]]:
    """Reference gen_prompt_counterfact (:282-336): N (prompt, counterfact)
    metadata dicts; the counterfact swaps the IO name for a third name."""
    rng = rd.Random(seed)
    prompts, prompts_cf = [], []
    for _ in range(N):
        temp = rng.choice(list(templates))
        temp_id = list(templates).index(temp)
        while True:
            picked = rng.sample(list(names), 3)
            if all(_single_token(tokenizer, n) for n in picked):
                break
        name_1, name_2, name_3 = picked
        nouns = {k: rng.choice(list(v)) for k, v in nouns_dict.items()}
        filled = multiple_replace(nouns, temp)

        meta = dict(nouns)
        meta["text"] = filled.replace("[A]", name_1).replace("[B]", name_2)
        meta["IO"], meta["S"], meta["TEMPLATE_IDX"] = name_1, name_2, temp_id
        prompts.append(meta)

        meta_cf = dict(nouns)
        meta_cf["text"] = filled.replace("[A]", name_3).replace("[B]", name_2)
        meta_cf["IO"], meta_cf["S"], meta_cf["TEMPLATE_IDX"] = name_3, name_2, temp_id
        prompts_cf.append(meta_cf)
    return prompts, prompts_cf


def gen_flipped_prompts(prompts: List[Dict], names: Sequence[str], flip: str = "IO",
                        tokenizer=None, seed: Optional[int] = None) -> List[Dict]:
    """Redwood gen_flipped_prompts semantics: rewrite one name ROLE with a
    fresh name.  flip in {"IO", "S", "S2", "IO,S"}:
      IO   — replace the indirect object everywhere it appears;
      S    — replace the subject everywhere (both S1 and S2);
      S2   — replace only the SECOND occurrence of the subject (makes an
             ABC-like prompt);
      IO,S — swap the IO and S roles (ABB -> BAA).
    """
    rng = rd.Random(seed)
    tokenizer = tokenizer or WordTokenizer()
    out = []
    for meta in prompts:
        new = dict(meta)
        text = meta["text"]
        if flip == "IO,S":
            text = multiple_replace({meta["IO"]: meta["S"], meta["S"]: meta["IO"]}, text)
            new["IO"], new["S"] = meta["S"], meta["IO"]
        else:
            while True:
                repl = rng.choice(list(names))
                if repl not in (meta["IO"], meta["S"]) and _single_token(tokenizer, repl):
                    break
            if flip == "IO":
                text = text.replace(meta["IO"], repl)
                new["IO"] = repl
            elif flip == "S":
                text = text.replace(meta["S"], repl)
                new["S"] = repl
            elif flip == "S2":
                first = text.index(meta["S"])
                second = text.index(meta["S"], first + 1)
                text = text[:second] + repl + text[second + len(meta["S"]) :]
                new["S2"] = repl
            else:
                raise ValueError(f"unknown flip {flip!r}")
        new["text"] = text
        out.append(new)
    return out


def _word_idx_for(tokenizer, meta: Dict) -> Dict[str, int]:
    """Token positions of IO, S1, S2 and the final token.  Exact when each
    name is a single token (guaranteed by the filter)."""
    text = meta["text"]
    io_tok = tokenizer(" " + meta["IO"])["input_ids"][0]
    s_tok = tokenizer(" " + meta["S"])["input_ids"][0]
    ids = tokenizer(text)["input_ids"]
    s_positions = [i for i, t in enumerate(ids) if t == s_tok]
    io_positions = [i for i, t in enumerate(ids) if t == io_tok]
    return {
        "IO": io_positions[0] if io_positions else -1,
        "S1": s_positions[0] if s_positions else -1,
        "S2": s_positions[1] if len(s_positions) > 1 else -1,
        "end": len(ids) - 1,
    }


class IOIDataset:
    """The Redwood IOIDataset surface: toks [N, L], per-prompt metadata,
    word_idx maps, and flipped-set generation returning a new IOIDataset."""

    def __init__(self, prompt_family: str = "mixed", N: int = 64, tokenizer=None,
                 seed: int = 0, prompts: Optional[List[Dict]] = None):
        self.tokenizer = tokenizer or WordTokenizer()
        self.prompt_family = prompt_family
        if prompts is None:
            prompts, _ = gen_prompt_counterfact(
                self.tokenizer, FAMILIES[prompt_family], NAMES, NOUNS_DICT, N, seed=seed)
        self.prompts = prompts
        self.N = len(prompts)
        ids = [self.tokenizer(p["text"])["input_ids"] for p in prompts]
        self.seq_lengths = torch.tensor([len(x) for x in ids])
        L = int(self.seq_lengths.max())
        self.toks = torch.zeros(self.N, L, dtype=torch.long)
        for i, x in enumerate(ids):
            self.toks[i, : len(x)] = torch.tensor(x)
        self.word_idx: Dict[str, torch.Tensor] = {}
        per_prompt = [_word_idx_for(self.tokenizer, p) for p in prompts]
        for key in ("IO", "S1", "S2", "end"):
            self.word_idx[key] = torch.tensor([w[key] for w in per_prompt])

    def gen_flipped_prompts(self, flip: str = "IO", seed: int = 1) -> "IOIDataset":
        flipped = gen_flipped_prompts(self.prompts, NAMES, flip=flip,
                                      tokenizer=self.tokenizer, seed=seed)
        return IOIDataset(self.prompt_family, tokenizer=self.tokenizer, prompts=flipped)

    def io_token_ids(self) -> torch.Tensor:
        return torch.tensor([self.tokenizer(" " + p["IO"])["input_ids"][0] for p in self.prompts])

    def s_token_ids(self) -> torch.Tensor:
        return torch.tensor([self.tokenizer(" " + p["S"])["input_ids"][0] for p in self.prompts])


def gen_ioi_dataset(tokenizer, n_prompts: int, seed: Optional[int] = None):
    """Reference gen_ioi_dataset (:338-372): padded (prompt, counterfact)
    token tensors with the final (answer) token dropped, + seq lengths.
    Pairs whose tokenizations differ in length are regenerated."""
    while True:
        prompts, prompts_cf = gen_prompt_counterfact(
            tokenizer, ABBA_TEMPLATES + BABA_TEMPLATES, NAMES, NOUNS_DICT,
            n_prompts, seed=seed)
        ids = tokenizer([p["text"] for p in prompts])["input_ids"]
        ids_cf = tokenizer([p["text"] for p in prompts_cf])["input_ids"]
        if all(len(a) == len(b) for a, b in zip(ids, ids_cf)):
            break
        seed = None  # resample
    seq_lengths = torch.tensor([len(p) - 1 for p in ids])
    L = int(seq_lengths.max())
    pad = lambda xs: torch.stack([
        torch.tensor(x[:-1] + [0] * (L - len(x[:-1]))) for x in xs])
    return pad(ids), pad(ids_cf), seq_lengths
