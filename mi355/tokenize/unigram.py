"""SentencePiece-compatible unigram tokenizer core.

Loads a binary ``spiece.model`` (sentencepiece ModelProto — parsed with a
minimal protobuf wire reader, no sentencepiece dependency) and encodes with
Viterbi segmentation over piece log-probs, the same algorithm as
sentencepiece's unigram model. Also provides a small corpus-based trainer
(frequency-pruned substring vocabulary) so tests and synthetic pipelines can
build a tokenizer without any downloaded model file.
"""
from __future__ import annotations

import math
import re
import struct
import unicodedata
from dataclasses import dataclass, field

SPACE = "▁"  # the sentencepiece meta-space
_WS_RUN = re.compile(r"\s+")


# --------------------------------------------------------------------------- #
# minimal protobuf wire parsing for sentencepiece ModelProto                  #
# --------------------------------------------------------------------------- #
def _read_varint(buf: bytes, pos: int):
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _iter_fields(buf: bytes):
    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = _read_varint(buf, pos)
        fnum, wtype = key >> 3, key & 7
        if wtype == 0:  # varint
            val, pos = _read_varint(buf, pos)
        elif wtype == 1:  # fixed64
            val = buf[pos:pos + 8]
            pos += 8
        elif wtype == 2:  # length-delimited
            ln, pos = _read_varint(buf, pos)
            val = buf[pos:pos + ln]
            pos += ln
        elif wtype == 5:  # fixed32
            val = buf[pos:pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wtype}")
        yield fnum, wtype, val


def parse_sentencepiece_model(data: bytes):
    """Returns list of (piece, score, type); type: 1 normal, 2 unk, 3 control,
    4 user-defined, 6 byte."""
    pieces = []
    for fnum, wtype, val in _iter_fields(data):
        if fnum == 1 and wtype == 2:  # repeated SentencePiece
            piece, score, ptype = "", 0.0, 1
            for f2, w2, v2 in _iter_fields(val):
                if f2 == 1 and w2 == 2:
                    piece = v2.decode("utf-8", errors="replace")
                elif f2 == 2 and w2 == 5:
                    score = struct.unpack("<f", v2)[0]
                elif f2 == 3 and w2 == 0:
                    ptype = v2
            pieces.append((piece, score, ptype))
    return pieces


def serialize_sentencepiece_model(pieces) -> bytes:
    """Inverse of the parser — lets save_pretrained() emit a loadable
    spiece.model for tokenizers built from a corpus."""
    out = bytearray()

    def varint(v):
        b = bytearray()
        while True:
            if v < 0x80:
                b.append(v)
                return bytes(b)
            b.append((v & 0x7F) | 0x80)
            v >>= 7

    for piece, score, ptype in pieces:
        body = bytearray()
        pb = piece.encode("utf-8")
        body += varint((1 << 3) | 2) + varint(len(pb)) + pb
        body += varint((2 << 3) | 5) + struct.pack("<f", score)
        body += varint((3 << 3) | 0) + varint(ptype)
        out += varint((1 << 3) | 2) + varint(len(body)) + bytes(body)
    return bytes(out)


# --------------------------------------------------------------------------- #
# unigram model                                                               #
# --------------------------------------------------------------------------- #
@dataclass
class UnigramModel:
    pieces: list  # [(piece, score, type)]
    unk_id: int = 2

    def __getstate__(self):
        return {"pieces": self.pieces, "unk_id": self.unk_id}

    def __setstate__(self, state):
        self.pieces = state["pieces"]
        self.unk_id = state["unk_id"]
        self.__post_init__()

    def __post_init__(self):
        self.piece_to_id = {p: i for i, (p, _, _) in enumerate(self.pieces)}
        self.scores = [s for _, s, _ in self.pieces]
        # prefix trie as nested dict for Viterbi candidate lookup
        self._max_len = max((len(p) for p, _, t in self.pieces if t == 1), default=1)
        self._vocab = {p: (i, s) for i, (p, s, t) in enumerate(self.pieces) if t != 3}
        self._unk_score = min(self.scores, default=-10.0) - 10.0
        # native C++ trie+Viterbi core (csrc/tokenize/unigram_core.cpp)
        self._core = None
        try:
            from mi355 import _tokenize

            core = _tokenize.UnigramCore()
            core.unk_id = self.unk_id
            core.unk_score = self._unk_score
            for i, (piece, score, t) in enumerate(self.pieces):
                if t != 3:  # control pieces never match text
                    core.add_piece(piece.encode("utf-8"), i, score)
            self._core = core
        except ImportError:
            pass

    @classmethod
    def load(cls, path: str) -> "UnigramModel":
        with open(path, "rb") as f:
            pieces = parse_sentencepiece_model(f.read())
        unk = next((i for i, (_, _, t) in enumerate(pieces) if t == 2), 2)
        return cls(pieces, unk_id=unk)

    def save(self, path: str):
        with open(path, "wb") as f:
            f.write(serialize_sentencepiece_model(self.pieces))

    def __len__(self):
        return len(self.pieces)

    # -- encoding ------------------------------------------------------- #
    def encode_text(self, text: str) -> list[int]:
        """Normalize then Viterbi.  Normalization follows sentencepiece's
        default nmt_nfkc spec (what google/flan-t5's spiece.model declares):
        NFKC, whitespace runs collapsed and stripped (remove_extra_
        whitespaces=1), dummy-prefix space, spaces -> U+2581 (byte-parity
        checked against the real spm runtime in tests/test_hf_interchange)."""
        text = unicodedata.normalize("NFKC", text)
        text = _WS_RUN.sub(" ", text).strip()
        text = SPACE + text.replace(" ", SPACE)
        if self._core is not None:
            return list(self._core.encode(text.encode("utf-8")))
        n = len(text)
        best = [(-math.inf, -1, -1)] * (n + 1)  # (score, prev, piece_id)
        best[0] = (0.0, -1, -1)
        for i in range(n):
            sc_i = best[i][0]
            if sc_i == -math.inf:
                continue
            hi = min(n, i + self._max_len)
            for j in range(i + 1, hi + 1):
                cand = self._vocab.get(text[i:j])
                if cand is not None:
                    pid, psc = cand
                    sc = sc_i + psc
                    if sc > best[j][0]:
                        best[j] = (sc, i, pid)
            # unk fallback: single char
            sc = sc_i + self._unk_score
            if sc > best[i + 1][0]:
                best[i + 1] = (sc, i, self.unk_id)
        ids = []
        j = n
        while j > 0:
            _, i, pid = best[j]
            ids.append(pid)
            j = i
        ids.reverse()
        return ids

    def decode_ids(self, ids) -> str:
        parts = []
        for i in ids:
            i = int(i)
            if 0 <= i < len(self.pieces):
                piece, _, t = self.pieces[i]
                if t in (1, 4):
                    parts.append(piece)
                elif t == 2:
                    parts.append(" ⁇ ")  # sp's unk surface
        s = "".join(parts).replace(SPACE, " ")
        # spm removes only the dummy-prefix leading space — a trailing space
        # (e.g. from a final unk's " ⁇ " surface) is preserved
        return s[1:] if s.startswith(" ") else s

    def id_to_piece(self, i: int) -> str:
        return self.pieces[i][0]


def train_unigram(texts, vocab_size: int = 1000,
                  special=("<pad>", "</s>", "<unk>")) -> UnigramModel:
    """Build a functional unigram vocabulary from a corpus: whole words +
    frequent prefixes + single characters, scored by log-frequency."""
    from collections import Counter

    counts: Counter = Counter()
    chars: Counter = Counter()
    for t in texts:
        for w in t.strip().split():
            w = SPACE + w
            counts[w] += 1
            for c in w:
                chars[c] += 1
            for ln in (2, 3, 4):
                if len(w) > ln:
                    counts[w[:ln]] += 1
    pieces = [(s, 0.0, 3 if s != "<unk>" else 2) for s in special]
    # all single chars first (coverage), then most frequent substrings
    total = sum(chars.values()) or 1
    for c, n in chars.most_common():
        pieces.append((c, math.log(n / total) - 5.0, 1))
    budget = vocab_size - len(pieces)
    totw = sum(counts.values()) or 1
    for w, n in counts.most_common(max(budget, 0)):
        if w not in (p for p, _, _ in pieces):
            pieces.append((w, math.log(n / totw), 1))
    return UnigramModel(pieces, unk_id=special.index("<unk>") if "<unk>" in special else 2)
