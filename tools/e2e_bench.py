#!/usr/bin/env python3
"""End-to-end fit()-level throughput: the WHOLE
HuggingFaceTrainer job — data pipeline (BatchMapper tokenize), per-epoch
eval, checkpoint save/retention — timed wall-clock, reported as samples/s
next to bench.py's step-level number so nobody mistakes one for the other.

Config mirrors the reference fine-tune (flan-t5-base, per-device batch 2,
seq 512, lr 2e-5, wd 0.01, epoch strategies, keep-1-best-by-eval_loss —
Model_finetuning_and_batch_inference.ipynb cell-32/34/40) on synthetic
Alpaca-shaped data.

Usage (GPU box): python tools/e2e_bench.py [--rows 800] [--epochs 1]
                 [--workers 1]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pandas as pd
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=800)
    ap.add_argument("--epochs", type=int, default=1)
    ap.add_argument("--workers", type=int, default=1)
    ap.add_argument("--model", default="flan-t5-base")
    ap.add_argument("--seq", type=int, default=512)
    args = ap.parse_args()

    import ray
    from ray.air.config import CheckpointConfig, RunConfig, ScalingConfig
    from ray.data.preprocessors import BatchMapper
    from ray.train.huggingface import HuggingFaceTrainer

    from mi355.data.synthetic import synthetic_alpaca, synthetic_alpaca_corpus
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.tokenize import T5Tokenizer
    from mi355.train import Trainer, TrainingArguments

    use_gpu = torch.cuda.is_available()
    tok = T5Tokenizer.train_new(synthetic_alpaca_corpus(256), vocab_size=8000,
                                model_max_length=args.seq)

    def preprocess(batch: pd.DataFrame) -> pd.DataFrame:
        enc = tok(list(batch["instruction"]), text_pair=list(batch["input"]),
                  padding="max_length", truncation=True, max_length=args.seq,
                  return_tensors="np")
        return pd.DataFrame({"input_ids": list(enc["input_ids"]),
                             "attention_mask": list(enc["attention_mask"]),
                             "labels": list(enc["input_ids"].copy())})

    def trainer_init_per_worker(train_dataset, eval_dataset=None, **config):
        cfg = t5_config(args.model)
        cfg.vocab_size = 32128
        model = T5ForConditionalGeneration(cfg)
        if use_gpu:
            model = model.to(torch.bfloat16)
        import tempfile
        targs = TrainingArguments(
            os.path.join(tempfile.gettempdir(), "e2e-bench"), learning_rate=config.get("learning_rate", 2e-5),
            num_train_epochs=args.epochs, weight_decay=0.01,
            per_device_train_batch_size=2, per_device_eval_batch_size=2,
            evaluation_strategy="epoch", save_strategy="epoch",
            logging_strategy="epoch", no_cuda=not use_gpu)
        return Trainer(model=model, args=targs, train_dataset=train_dataset,
                       eval_dataset=eval_dataset)

    ray.init()
    ds = synthetic_alpaca(args.rows)  # already a Dataset
    train_ds, eval_ds = ds.train_test_split(test_size=0.2)
    mapper = BatchMapper(preprocess, batch_format="pandas", batch_size=4096)

    t0 = time.perf_counter()
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=trainer_init_per_worker,
        scaling_config=ScalingConfig(num_workers=args.workers, use_gpu=use_gpu),
        datasets={"train": train_ds, "evaluation": eval_ds},
        run_config=RunConfig(checkpoint_config=CheckpointConfig(
            num_to_keep=1, checkpoint_score_attribute="eval_loss",
            checkpoint_score_order="min")),
        preprocessor=mapper)
    result = trainer.fit()
    wall = time.perf_counter() - t0

    n_train = int(args.rows * 0.8)
    total = n_train * args.epochs
    print(json.dumps({
        "metric": "fit_samples_per_s_e2e",
        "value": round(total / wall, 2),
        "wall_s": round(wall, 1),
        "train_rows": n_train, "epochs": args.epochs,
        "workers": args.workers,
        "includes": "tokenize pipeline + train + per-epoch eval + checkpoint",
        "eval_loss": result.metrics.get("eval_loss"),
        "note": "end-to-end fit() wall; bench.py reports the step-level "
                "number (fwd+bwd+opt only)",
    }))


if __name__ == "__main__":
    main()
