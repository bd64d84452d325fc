"""Tokenizer tests (C20): unigram Viterbi encode/decode, T5 pair encoding
with max_length padding (reference Anyscale_job/utils.py:23-29), spiece.model
serialization roundtrip, cross-check against the sentencepiece library when
available."""
import numpy as np
import pytest

from mi355.tokenize import T5Tokenizer, UnigramModel, train_unigram
from mi355.tokenize.unigram import parse_sentencepiece_model, serialize_sentencepiece_model

CORPUS = [
    "the quick brown fox jumps over the lazy dog",
    "pack my box with five dozen liquor jugs",
    "the five boxing wizards jump quickly",
    "how vexingly quick daft zebras jump",
] * 4


@pytest.fixture(scope="module")
def tok():
    return T5Tokenizer.train_new(CORPUS, vocab_size=400, model_max_length=32)


def test_encode_decode_roundtrip(tok):
    text = "the quick brown fox"
    ids = tok.sp.encode_text(text)
    assert tok.sp.decode_ids(ids) == text


def test_call_padding_max_length(tok):
    out = tok(["the quick fox", "dog"], padding="max_length", truncation=True,
              max_length=16, return_tensors="np")
    assert out["input_ids"].shape == (2, 16)
    assert out["attention_mask"].shape == (2, 16)
    # ends with eos then pads
    row = out["input_ids"][1]
    mask = out["attention_mask"][1]
    last = int(mask.sum()) - 1
    assert row[last] == tok.eos_token_id
    assert (row[last + 1:] == tok.pad_token_id).all()


def test_pair_encoding(tok):
    single = tok("the quick fox")["input_ids"]
    pair = tok("the quick fox", text_pair="lazy dog")["input_ids"]
    assert pair[:len(single)] == single  # A </s> prefix preserved
    assert pair.count(tok.eos_token_id) == 2


def test_truncation(tok):
    out = tok("the quick brown fox jumps over the lazy dog " * 10,
              truncation=True, max_length=8)
    assert len(out["input_ids"]) == 8
    assert out["input_ids"][-1] == tok.eos_token_id


def test_batch_decode_skip_special(tok):
    enc = tok(["five boxing wizards"], padding="max_length", max_length=12,
              truncation=True, return_tensors="np")
    texts = tok.batch_decode(enc["input_ids"], skip_special_tokens=True)
    assert texts == ["five boxing wizards"]


def test_save_load_roundtrip(tok, tmp_path):
    tok.save_pretrained(str(tmp_path))
    tok2 = T5Tokenizer.from_pretrained(str(tmp_path))
    t = "the lazy dog jumps"
    assert tok(t)["input_ids"] == tok2(t)["input_ids"]
    assert tok2.model_max_length == 32


def test_spiece_proto_roundtrip():
    pieces = [("<pad>", 0.0, 3), ("</s>", 0.0, 3), ("<unk>", 0.0, 2),
              ("▁hello", -1.5, 1), ("▁world", -2.25, 1), ("lo", -3.0, 1)]
    blob = serialize_sentencepiece_model(pieces)
    back = parse_sentencepiece_model(blob)
    assert [(p, t) for p, _, t in back] == [(p, t) for p, _, t in pieces]
    assert back[3][1] == pytest.approx(-1.5)


def test_viterbi_prefers_higher_score_segmentation():
    pieces = [("<pad>", 0.0, 3), ("</s>", 0.0, 3), ("<unk>", 0.0, 2),
              ("▁ab", -1.0, 1), ("▁a", -2.0, 1), ("b", -2.0, 1), ("▁", -3.0, 1),
              ("a", -3.0, 1)]
    m = UnigramModel(pieces, unk_id=2)
    # "▁ab" (score -1) must beat "▁a"+"b" (-4)
    assert m.encode_text("ab") == [3]


def test_against_sentencepiece_library_if_available(tok, tmp_path):
    sp = pytest.importorskip("sentencepiece")
    tok.save_pretrained(str(tmp_path))
    proc = sp.SentencePieceProcessor()
    proc.Load(str(tmp_path / "spiece.model"))
    for text in ["the quick brown fox", "lazy dog"]:
        ours = tok.sp.encode_text(text)
        theirs = proc.EncodeAsIds(text)
        assert ours == theirs, (tok.convert_ids_to_tokens(ours),
                                proc.EncodeAsPieces(text))


def test_tokenizer_properties_hypothesis():
    """Property-based checks over arbitrary text (hypothesis): ids in range,
    eos-terminated, deterministic, pad/truncation shape contract."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from mi355.tokenize import T5Tokenizer

    tok = T5Tokenizer.train_new(
        ["the quick brown fox jumps over the lazy dog"] * 4 +
        ["pack my box with five dozen liquor jugs"] * 4,
        vocab_size=200, model_max_length=16)

    @settings(max_examples=60, deadline=None)
    @given(st.text(alphabet=st.characters(codec="ascii",
                                          exclude_categories=("Cc", "Cs")),
                   max_size=60))
    def check(text):
        enc = tok([text], padding="max_length", truncation=True,
                  max_length=16, return_tensors="np")
        ids = enc["input_ids"][0]
        mask = enc["attention_mask"][0]
        assert ids.shape == (16,) and mask.shape == (16,)
        assert ids.max() < len(tok) and ids.min() >= 0
        nvalid = int(mask.sum())
        assert (ids[nvalid:] == tok.pad_token_id).all()
        assert nvalid >= 1 and ids[nvalid - 1] == tok.eos_token_id
        enc2 = tok([text], padding="max_length", truncation=True,
                   max_length=16, return_tensors="np")
        assert (enc2["input_ids"][0] == ids).all()
        # decode drops specials and never raises
        tok.batch_decode(ids[None], skip_special_tokens=True)

    check()
