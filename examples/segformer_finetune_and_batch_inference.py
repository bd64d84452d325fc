#!/usr/bin/env python3
"""The reference's CV pipelines (Computer_vision_workloads/
Scaling_model_training.ipynb + Scaling_batch_inference.ipynb) on the mi355
stack: ADE20K-shaped data -> SegformerImageProcessor BatchMapper ->
2-worker HuggingFaceTrainer with the notebook's explicit AdamW + LambdaLR
passed as ``optimizers=(opt, sched)`` -> BatchPredictor semantic
segmentation with the fused upsample+argmax postprocess.

Env knobs: MI355_SMALL=1 (tiny model + CPU, default when no GPU),
MI355_NUM_WORKERS (default 2 like the reference).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import torch

import ray
from ray.air.config import CheckpointConfig, RunConfig, ScalingConfig
from ray.data.preprocessors import BatchMapper
from ray.train.batch_predictor import BatchPredictor
from ray.train.huggingface import HuggingFaceTrainer
from ray.train.predictor import Predictor

from mi355.data.synthetic import synthetic_ade20k
from mi355.models.segformer import (SegformerConfig, SegformerImageProcessor,
                                    SegformerForSemanticSegmentation)
from mi355.train import Trainer, TrainingArguments

SMALL = os.environ.get("MI355_SMALL", "1" if not torch.cuda.is_available() else "0") == "1"
NUM_WORKERS = int(os.environ.get("MI355_NUM_WORKERS", "2"))
USE_GPU = torch.cuda.is_available() and not SMALL
SIZE = 32 if SMALL else 512
OVERRIDES = (dict(depths=[1, 1, 1, 1], hidden_sizes=[8, 16, 24, 32],
                  num_attention_heads=[1, 2, 3, 4], decoder_hidden_size=32)
             if SMALL else {})
NUM_LABELS = 16 if SMALL else 150


def preprocessor():
    proc = SegformerImageProcessor(size=SIZE, do_reduce_labels=True)

    def fn(df: pd.DataFrame) -> pd.DataFrame:
        enc = proc([np.asarray(im) for im in df["image"]],
                   segmentation_maps=[np.asarray(a) for a in df["annotation"]])
        return pd.DataFrame({"pixel_values": list(enc["pixel_values"]),
                             "labels": list(enc["labels"])})

    return BatchMapper(fn, batch_format="pandas", batch_size=64)


def trainer_init_per_worker(train_dataset, eval_dataset=None, **config):
    torch.manual_seed(42)
    cfg = SegformerConfig(num_labels=NUM_LABELS, **OVERRIDES)
    model = SegformerForSemanticSegmentation(cfg)
    # the reference passes an explicit AdamW + LambdaLR (cell-47)
    opt = torch.optim.AdamW(model.parameters(),
                            lr=config.get("learning_rate", 6e-5))
    sched = torch.optim.lr_scheduler.LambdaLR(opt, lambda e: 0.97 ** e)
    args = TrainingArguments(
        output_dir="segformer-finetuned",
        num_train_epochs=config.get("epochs", 1 if SMALL else 3),
        per_device_train_batch_size=2 if SMALL else 8,
        per_device_eval_batch_size=2 if SMALL else 8,
        evaluation_strategy="epoch", save_strategy="epoch",
        logging_strategy="epoch", no_cuda=not USE_GPU,
        bf16=USE_GPU, disable_tqdm=True, remove_unused_columns=True,
    )
    return Trainer(model=model, args=args, train_dataset=train_dataset,
                   eval_dataset=eval_dataset, optimizers=(opt, sched))


class SegformerPredictor(Predictor):
    """Scaling_batch_inference.ipynb cell-42 pattern: model + processor in
    the replica, fused upsample+argmax postprocess."""

    def __init__(self, model, preprocessor=None):
        super().__init__(preprocessor)
        self.model = model
        self.proc = SegformerImageProcessor(size=SIZE)

    @classmethod
    def from_checkpoint(cls, checkpoint, *, torch_dtype=None, **kw):
        model = checkpoint.get_model(SegformerForSemanticSegmentation,
                                     torch_dtype=torch_dtype)
        if torch.cuda.is_available():
            model = model.to("cuda")
        model.eval()
        return cls(model, preprocessor=checkpoint.get_preprocessor())

    def _predict_pandas(self, df: pd.DataFrame, **kw) -> pd.DataFrame:
        pix = torch.as_tensor(np.stack([np.asarray(p) for p in df["pixel_values"]]))
        pix = pix.to(self.model.device, dtype=next(self.model.parameters()).dtype)
        with torch.no_grad():
            logits = self.model(pixel_values=pix).logits
        masks = self.proc.post_process_semantic_segmentation(
            logits, target_sizes=[(SIZE, SIZE)] * len(df))
        return pd.DataFrame({"mask": [m.cpu().numpy() for m in masks]})


def main():
    ray.init()
    n = 32 if SMALL else 160  # reference: train[:160]
    ds = synthetic_ade20k(n, size=SIZE, num_labels=NUM_LABELS, seed=57)
    train_ds, val_ds = ds.train_test_split(test_size=0.25, shuffle=True, seed=57)

    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=trainer_init_per_worker,
        scaling_config=ScalingConfig(num_workers=NUM_WORKERS, use_gpu=USE_GPU),
        datasets={"train": train_ds, "evaluation": val_ds},
        run_config=RunConfig(
            name="segformer-ade20k",
            checkpoint_config=CheckpointConfig(
                num_to_keep=1, checkpoint_score_attribute="loss",
                checkpoint_score_order="min")),
        preprocessor=preprocessor(),
    )
    result = trainer.fit()
    assert result.error is None, result.error
    print("train metrics:", result.metrics)

    predictor = BatchPredictor.from_checkpoint(
        result.checkpoint, SegformerPredictor,
        torch_dtype=torch.bfloat16 if USE_GPU else None)
    pred = predictor.predict(val_ds.limit(8), batch_size=8,
                             num_gpus_per_worker=1 if USE_GPU else 0)
    masks = pred.to_pandas()["mask"]
    m0 = masks.iloc[0]
    if np.asarray(m0).dtype == object:  # arrow round-trips 2D as nested lists
        m0 = np.stack([np.asarray(r) for r in m0])
    else:
        m0 = np.asarray(m0)
    print(f"predicted {len(masks)} masks, shape {m0.shape}, "
          f"labels used: {np.unique(m0)[:8]}")
    ray.shutdown()
    print("JOB OK")


if __name__ == "__main__":
    main()
