"""Checkpoint: dict <-> directory duality with HF ``save_pretrained`` layout
(reference consumers predictor.py:39-72
get_model/get_tokenizer/get_preprocessor, Scaling_batch_inference.ipynb:
cell-73 from_dict, cell-83 HuggingFaceCheckpoint.from_model)."""
from __future__ import annotations

import json
import os
import shutil
import tempfile

import cloudpickle as pickle


class Checkpoint:
    """A checkpoint is either an in-memory dict or a directory on disk; both
    views are interconvertible (the reference uses both idioms)."""

    def __init__(self, data: dict | None = None, path: str | None = None):
        self._data = data
        self._path = path

    # -- constructors --------------------------------------------------- #
    @classmethod
    def from_dict(cls, data: dict) -> "Checkpoint":
        return cls(data=dict(data))

    @classmethod
    def from_directory(cls, path: str) -> "Checkpoint":
        return cls(path=os.path.abspath(path))

    # -- views ----------------------------------------------------------- #
    def to_dict(self) -> dict:
        if self._data is not None:
            return dict(self._data)
        out = {}
        pkl = os.path.join(self._path, "_checkpoint_dict.pkl")
        if os.path.exists(pkl):
            with open(pkl, "rb") as f:
                out.update(pickle.load(f))
        out["_directory"] = self._path
        return out

    def to_directory(self, path: str | None = None) -> str:
        if path is None:
            path = tempfile.mkdtemp(prefix="mi355_ckpt_")
        os.makedirs(path, exist_ok=True)
        if self._path is not None and os.path.abspath(path) != self._path:
            for name in os.listdir(self._path):
                src = os.path.join(self._path, name)
                dst = os.path.join(path, name)
                if os.path.isdir(src):
                    shutil.copytree(src, dst, dirs_exist_ok=True)
                else:
                    shutil.copy2(src, dst)
        elif self._data is not None:
            with open(os.path.join(path, "_checkpoint_dict.pkl"), "wb") as f:
                pickle.dump(self._data, f)
        return path

    @property
    def path(self) -> str | None:
        return self._path

    def as_directory(self):
        from contextlib import contextmanager

        @contextmanager
        def cm():
            created = self._path is None
            p = self._path or self.to_directory()
            yield p
            if created:
                shutil.rmtree(p, ignore_errors=True)

        return cm()

    # -- model/artifact accessors (reference predictor.py:39-72) -------- #
    def get_model(self, model_cls=None, **kwargs):
        if self._data is not None and "model" in self._data:
            return self._data["model"]
        assert self._path is not None, "empty checkpoint"
        if model_cls is None:
            from ..models.auto import AutoModel

            model_cls = AutoModel
        return model_cls.from_pretrained(self._path, **kwargs)

    def get_tokenizer(self, tokenizer_cls=None):
        if self._data is not None and "tokenizer" in self._data:
            return self._data["tokenizer"]
        assert self._path is not None
        if tokenizer_cls is None:
            from ..tokenize import T5Tokenizer

            tokenizer_cls = T5Tokenizer
        return tokenizer_cls.from_pretrained(self._path)

    def get_preprocessor(self):
        if self._data is not None:
            return self._data.get("preprocessor")
        pp = os.path.join(self._path or "", "preprocessor.pkl")
        if os.path.exists(pp):
            with open(pp, "rb") as f:
                return pickle.load(f)
        return None

    def set_preprocessor(self, preprocessor):
        if self._data is not None:
            self._data["preprocessor"] = preprocessor
        elif preprocessor is not None:
            with open(os.path.join(self._path, "preprocessor.pkl"), "wb") as f:
                pickle.dump(preprocessor, f)

    def get_metadata(self) -> dict:
        if self._path:
            meta = os.path.join(self._path, "air_metadata.json")
            if os.path.exists(meta):
                with open(meta) as f:
                    return json.load(f)
        if self._data:
            return self._data.get("_metadata", {})
        return {}

    def set_metadata(self, meta: dict):
        if self._path:
            with open(os.path.join(self._path, "air_metadata.json"), "w") as f:
                json.dump(meta, f, indent=2)
        elif self._data is not None:
            self._data["_metadata"] = meta

    def __repr__(self):
        where = self._path if self._path else f"dict[{len(self._data or {})}]"
        return f"Checkpoint({where})"


class HuggingFaceCheckpoint(Checkpoint):
    """Checkpoint in HF save_pretrained layout (reference cell-83)."""

    @classmethod
    def from_model(cls, model, path: str, tokenizer=None,
                   preprocessor=None) -> "HuggingFaceCheckpoint":
        os.makedirs(path, exist_ok=True)
        model.save_pretrained(path)
        if tokenizer is not None and hasattr(tokenizer, "save_pretrained"):
            tokenizer.save_pretrained(path)
        ckpt = cls(path=os.path.abspath(path))
        if preprocessor is not None:
            ckpt.set_preprocessor(preprocessor)
        return ckpt


TorchCheckpoint = Checkpoint  # alias parity
