"""AutoModel: resolve a checkpoint directory's config.json to the right
model class in the mi355 zoo."""
from __future__ import annotations

import json
import os


class AutoModel:
    @staticmethod
    def from_pretrained(path: str, **kwargs):
        with open(os.path.join(path, "config.json")) as f:
            cfg = json.load(f)
        mt = cfg.get("model_type", "")
        if mt == "t5":
            from .t5.modeling import T5ForConditionalGeneration

            return T5ForConditionalGeneration.from_pretrained(path, **kwargs)
        if mt == "segformer":
            from .segformer.modeling import SegformerForSemanticSegmentation

            return SegformerForSemanticSegmentation.from_pretrained(path, **kwargs)
        if mt == "mi355_gbt":
            from .gbt import GBTModel

            return GBTModel.from_pretrained(path)
        raise ValueError(f"AutoModel: unknown model_type {mt!r} at {path}")
