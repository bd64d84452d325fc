"""T5Tokenizer surface over the unigram core (reference call
sites NLP_workloads/Anyscale_job/utils.py:23-29 — text-PAIR encoding with
padding="max_length", truncation, return_tensors="np" — and
predictor.py:64-66/104 batch_decode(skip_special_tokens=True))."""
from __future__ import annotations

import json
import os

import numpy as np

from .unigram import UnigramModel, train_unigram


class T5Tokenizer:
    """T5 conventions: pad=0 '<pad>', eos=1 '</s>', unk=2 '<unk>'; every
    encoded sequence ends with </s>; pairs are concatenated with an eos
    between (T5's pair encoding: A </s> B </s>)."""

    model_input_names = ["input_ids", "attention_mask"]

    def __init__(self, model: UnigramModel, model_max_length: int = 512,
                 extra_ids: int = 100):
        self.sp = model
        self.model_max_length = model_max_length
        self.extra_ids = extra_ids
        self.pad_token, self.eos_token, self.unk_token = "<pad>", "</s>", "<unk>"
        self.pad_token_id, self.eos_token_id, self.unk_token_id = 0, 1, 2

    # -- constructors --------------------------------------------------- #
    @classmethod
    def from_pretrained(cls, path: str, model_max_length: int = 512, **_):
        if os.path.isdir(path):
            spm = os.path.join(path, "spiece.model")
            cfgf = os.path.join(path, "tokenizer_config.json")
            if os.path.exists(cfgf):
                with open(cfgf) as f:
                    model_max_length = json.load(f).get("model_max_length",
                                                        model_max_length)
            return cls(UnigramModel.load(spm), model_max_length=model_max_length)
        if os.path.isfile(path):
            return cls(UnigramModel.load(path), model_max_length=model_max_length)
        raise FileNotFoundError(
            f"T5Tokenizer.from_pretrained: no spiece.model at {path!r} (no "
            "network in this environment — use T5Tokenizer.train_new(corpus) "
            "or point at a local checkpoint directory)")

    @classmethod
    def train_new(cls, texts, vocab_size: int = 1000, model_max_length: int = 512):
        return cls(train_unigram(texts, vocab_size=vocab_size),
                   model_max_length=model_max_length)

    @property
    def vocab_size(self) -> int:
        return len(self.sp)

    def __len__(self):
        return len(self.sp) + self.extra_ids

    def get_vocab(self):
        v = {p: i for i, (p, _, _) in enumerate(self.sp.pieces)}
        for k in range(self.extra_ids):
            v[f"<extra_id_{k}>"] = len(self.sp) + self.extra_ids - 1 - k
        return v

    # -- encoding -------------------------------------------------------- #
    def _encode_one(self, text: str, text_pair: str | None, max_length: int,
                    truncation: bool, add_special_tokens: bool = True):
        eos = [self.eos_token_id] if add_special_tokens else []
        ids = self.sp.encode_text(text) + eos
        if text_pair is not None:
            ids += self.sp.encode_text(text_pair) + eos
        if truncation and len(ids) > max_length:
            ids = ids[:max_length]
            if add_special_tokens:
                ids[-1] = self.eos_token_id
        return ids

    def __call__(self, text, text_pair=None, padding=False, truncation=False,
                 max_length=None, return_tensors=None,
                 add_special_tokens=True, **_):
        max_length = max_length or self.model_max_length
        batched = isinstance(text, (list, tuple))
        texts = list(text) if batched else [text]
        pairs = (list(text_pair) if batched else [text_pair]) if text_pair is not None \
            else [None] * len(texts)
        encoded = [self._encode_one(t, p, max_length, bool(truncation),
                                    add_special_tokens)
                   for t, p in zip(texts, pairs)]
        if padding == "max_length":
            target = max_length
        elif padding in (True, "longest"):
            target = max(len(e) for e in encoded)
        else:
            target = None
        input_ids, attention_mask = [], []
        for e in encoded:
            if target is not None:
                mask = [1] * len(e) + [0] * (target - len(e))
                e = e + [self.pad_token_id] * (target - len(e))
            else:
                mask = [1] * len(e)
            input_ids.append(e)
            attention_mask.append(mask)
        out = {"input_ids": input_ids, "attention_mask": attention_mask}
        if return_tensors == "np":
            out = {k: np.asarray(v, dtype=np.int64) for k, v in out.items()}
        elif return_tensors == "pt":
            import torch

            out = {k: torch.as_tensor(np.asarray(v, dtype=np.int64))
                   for k, v in out.items()}
        if not batched and return_tensors is None:
            out = {k: v[0] for k, v in out.items()}
        return BatchEncoding(out)

    def encode(self, text, **kw):
        return self(text, **kw)["input_ids"]

    # -- decoding -------------------------------------------------------- #
    def decode(self, ids, skip_special_tokens: bool = False, **_):
        import torch

        if isinstance(ids, torch.Tensor):
            ids = ids.tolist()
        ids = [int(i) for i in np.asarray(ids).reshape(-1)]
        if skip_special_tokens:
            ids = [i for i in ids if i not in
                   (self.pad_token_id, self.eos_token_id) and i < len(self.sp)]
        return self.sp.decode_ids(ids)

    def batch_decode(self, batch, skip_special_tokens: bool = False, **kw):
        return [self.decode(row, skip_special_tokens=skip_special_tokens)
                for row in batch]

    def convert_ids_to_tokens(self, ids):
        return [self.sp.id_to_piece(int(i)) for i in np.asarray(ids).reshape(-1)]

    # -- persistence ------------------------------------------------------ #
    def save_pretrained(self, save_dir: str):
        os.makedirs(save_dir, exist_ok=True)
        self.sp.save(os.path.join(save_dir, "spiece.model"))
        with open(os.path.join(save_dir, "tokenizer_config.json"), "w") as f:
            json.dump({"tokenizer_class": "T5Tokenizer",
                       "model_max_length": self.model_max_length,
                       "extra_ids": self.extra_ids,
                       "pad_token": self.pad_token, "eos_token": self.eos_token,
                       "unk_token": self.unk_token}, f, indent=2)
        return (os.path.join(save_dir, "spiece.model"),)


class BatchEncoding(dict):
    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e


AutoTokenizer = T5Tokenizer
