"""Redwood IOIDataset semantics (data/ioi_counterfact.py) + the IOI
feature-identification study, fully offline via WordTokenizer."""

import torch

from sparse_coding_amd.data.ioi_counterfact import (
    ABBA_TEMPLATES,
    BABA_TEMPLATES,
    FAMILIES,
    IOIDataset,
    NAMES,
    NOUNS_DICT,
    WordTokenizer,
    gen_flipped_prompts,
    gen_ioi_dataset,
    gen_prompt_counterfact,
)


def test_abba_derivation():
    """ABBA templates: first clause swapped, second clause unchanged."""
    for baba, abba in zip(BABA_TEMPLATES, ABBA_TEMPLATES):
        assert baba != abba
        # both still contain exactly two [A] and two [B] placeholders total
        assert abba.count("[A]") + abba.count("[B]") == baba.count("[A]") + baba.count("[B]")
        # the final placeholder (the IO position) is [A] in both
        assert baba.rstrip().endswith("[A]") == abba.rstrip().endswith("[A]")
        # first-name slot differs: BABA starts with [B], ABBA with [A]
        assert baba.index("[B]") < baba.index("[A]")
        assert abba.index("[A]") < abba.index("[B]")


def test_gen_prompt_counterfact_metadata():
    tok = WordTokenizer()
    prompts, cf = gen_prompt_counterfact(tok, BABA_TEMPLATES, NAMES, NOUNS_DICT, 16, seed=0)
    assert len(prompts) == len(cf) == 16
    for p, c in zip(prompts, cf):
        for key in ("text", "IO", "S", "TEMPLATE_IDX", "[PLACE]"):
            assert key in p or key == "[PLACE]" and "[PLACE]" in p
        assert p["S"] == c["S"]          # subject kept
        assert p["IO"] != c["IO"]        # IO swapped for a third name
        assert p["TEMPLATE_IDX"] == c["TEMPLATE_IDX"]
        assert p["IO"] in p["text"] and p["S"] in p["text"]
        assert c["IO"] in c["text"]


def test_word_idx_positions():
    ds = IOIDataset("BABA", N=12, seed=0)
    for key in ("IO", "S1", "S2", "end"):
        assert key in ds.word_idx
    # BABA: S appears before IO in clause 1, S2 between, IO last
    assert (ds.word_idx["S1"] >= 0).all()
    assert (ds.word_idx["S2"] > ds.word_idx["S1"]).all()
    # the final token is the IO (answer position is end)
    rows = torch.arange(ds.N)
    io_ids = ds.io_token_ids()
    assert (ds.toks[rows, ds.word_idx["end"]] == io_ids).all()


def test_flipped_prompts_semantics():
    ds = IOIDataset("mixed", N=16, seed=3)
    flip_io = ds.gen_flipped_prompts("IO", seed=5)
    assert all(a["IO"] != b["IO"] for a, b in zip(ds.prompts, flip_io.prompts))
    assert all(a["S"] == b["S"] for a, b in zip(ds.prompts, flip_io.prompts))

    flip_s = ds.gen_flipped_prompts("S", seed=6)
    assert all(a["S"] != b["S"] for a, b in zip(ds.prompts, flip_s.prompts))
    assert all(a["IO"] == b["IO"] for a, b in zip(ds.prompts, flip_s.prompts))

    swap = ds.gen_flipped_prompts("IO,S", seed=7)
    assert all(a["IO"] == b["S"] and a["S"] == b["IO"]
               for a, b in zip(ds.prompts, swap.prompts))

    s2 = gen_flipped_prompts(ds.prompts, NAMES, flip="S2", seed=8)
    for a, b in zip(ds.prompts, s2):
        assert b["S2"] != a["S"]
        assert b["text"].count(a["S"]) == a["text"].count(a["S"]) - 1


def test_gen_ioi_dataset_padded_pairs():
    tok = WordTokenizer()
    toks, toks_cf, seq_lengths = gen_ioi_dataset(tok, 24, seed=0)
    assert toks.shape == toks_cf.shape
    assert toks.shape[0] == 24
    assert (seq_lengths <= toks.shape[1]).all()
    # the dropped final token means padded tails are zeros
    rows = torch.arange(24)
    assert (toks[rows, seq_lengths - 1] != 0).all()


def test_all_families_build():
    for fam in FAMILIES:
        ds = IOIDataset(fam, N=4, seed=1)
        assert ds.toks.shape[0] == 4


def test_ioi_feature_ident_end_to_end():
    """Feature identification on the tiny host LM with a planted dict: the
    pipeline ranks features and the ablation changes the logit diff."""
    from sparse_coding_amd.analysis.ioi_feature_ident import run_ioi_feature_ident
    from sparse_coding_amd.data.activation_dataset import load_model
    from sparse_coding_amd.models.learned_dict import TiedSAE

    torch.manual_seed(0)
    model = load_model("tiny-gptneox", device="cpu")
    ld = TiedSAE(torch.randn(32, 64), torch.zeros(32))
    tok = WordTokenizer(vocab_size=model.config.vocab_size)
    out = run_ioi_feature_ident(ld, model, layer=1, n_prompts=16, top_k=4,
                                device="cpu", tokenizer=tok)
    assert len(out["features"]) == 4
    assert all(s >= 0 for s in out["diff_scores"])
    assert out["base_logit_diff"] != out["ablated_logit_diff"]
