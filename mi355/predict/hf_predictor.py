"""Framework predictors: HuggingFacePredictor (pipeline-style
over the mi355 model zoo — reference Scaling_batch_inference.ipynb:cell-83
task="image-segmentation"; Intro nb XGBoostPredictor analog in
mi355/models/gbt.py) and a generic TorchPredictor."""
from __future__ import annotations

import numpy as np
import pandas as pd
import torch

from ..air import Checkpoint
from .predictor import Predictor


def _pick_device():
    return torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")


class TorchPredictor(Predictor):
    def __init__(self, model, preprocessor=None, use_gpu: bool | None = None):
        super().__init__(preprocessor)
        self.device = _pick_device() if (use_gpu or use_gpu is None and
                                         torch.cuda.is_available()) else torch.device("cpu")
        self.model = model.to(self.device).eval()

    @classmethod
    def from_checkpoint(cls, checkpoint: Checkpoint, model=None, model_cls=None,
                        **kwargs) -> "TorchPredictor":
        if model is None:
            model = checkpoint.get_model(model_cls, **kwargs)
        return cls(model, preprocessor=checkpoint.get_preprocessor())

    def _predict_numpy(self, data: dict, **kwargs):
        tensors = {k: torch.as_tensor(np.asarray(v)).to(self.device)
                   for k, v in data.items()}
        with torch.no_grad():
            out = self.model(**tensors)
        logits = out.logits if hasattr(out, "logits") else out
        return pd.DataFrame({"predictions": list(logits.float().cpu().numpy())})


class HuggingFacePredictor(Predictor):
    """Pipeline-style predictor over the mi355 model zoo."""

    def __init__(self, model=None, tokenizer=None, task: str = "text2text-generation",
                 preprocessor=None, device=None, **pipeline_kwargs):
        super().__init__(preprocessor)
        self.task = task
        self.device = device or _pick_device()
        self.model = model.to(self.device).eval() if model is not None else None
        self.tokenizer = tokenizer
        self.pipeline_kwargs = pipeline_kwargs

    @classmethod
    def from_checkpoint(cls, checkpoint: Checkpoint, *, task: str = "text2text-generation",
                        model_cls=None, tokenizer=None, torch_dtype=None,
                        **kwargs) -> "HuggingFacePredictor":
        if task in ("text2text-generation", "text-generation", "summarization"):
            from ..models.t5.modeling import T5ForConditionalGeneration

            model_cls = model_cls or T5ForConditionalGeneration
        elif task == "image-segmentation":
            from ..models.segformer.modeling import SegformerForSemanticSegmentation

            model_cls = model_cls or SegformerForSemanticSegmentation
        model = checkpoint.get_model(model_cls, torch_dtype=torch_dtype)
        if tokenizer is not None and isinstance(tokenizer, type):
            tokenizer = checkpoint.get_tokenizer(tokenizer)
        elif tokenizer is None and task.startswith("text"):
            try:
                tokenizer = checkpoint.get_tokenizer()
            except Exception:
                tokenizer = None
        return cls(model=model, tokenizer=tokenizer, task=task,
                   preprocessor=checkpoint.get_preprocessor(), **kwargs)

    def _predict_pandas(self, df: pd.DataFrame, **kwargs):
        if self.task in ("text2text-generation", "text-generation", "summarization"):
            return self._generate_text(df, **kwargs)
        if self.task == "image-segmentation":
            return self._segment(df, **kwargs)
        raise ValueError(f"unsupported task {self.task!r}")

    def _generate_text(self, df: pd.DataFrame, **gen_kwargs):
        if "input_ids" in df.columns:
            ids = torch.as_tensor(np.stack(df["input_ids"].to_list())).to(self.device)
            mask = None
            if "attention_mask" in df.columns:
                mask = torch.as_tensor(np.stack(df["attention_mask"].to_list())).to(self.device)
        else:
            text_col = df.columns[0]
            enc = self.tokenizer(list(df[text_col]), padding=True, truncation=True,
                                 return_tensors="np")
            ids = torch.as_tensor(enc["input_ids"]).to(self.device)
            mask = torch.as_tensor(enc["attention_mask"]).to(self.device)
        with torch.no_grad():
            out = self.model.generate(input_ids=ids, attention_mask=mask, **gen_kwargs)
        if self.tokenizer is not None:
            text = self.tokenizer.batch_decode(out.cpu().numpy(),
                                               skip_special_tokens=True)
            return pd.DataFrame({"generated_output": text})
        return pd.DataFrame({"generated_tokens": list(out.cpu().numpy())})

    def _segment(self, df: pd.DataFrame, **kwargs):
        col = "pixel_values" if "pixel_values" in df.columns else df.columns[0]
        pix = torch.as_tensor(np.stack([np.asarray(v, dtype=np.float32)
                                        for v in df[col]])).to(self.device)
        dt = next(self.model.parameters()).dtype
        with torch.no_grad():
            logits = self.model(pixel_values=pix.to(dt)).logits
        pred = logits.float().argmax(dim=1).cpu().numpy()
        return pd.DataFrame({"predictions": [p for p in pred]})
