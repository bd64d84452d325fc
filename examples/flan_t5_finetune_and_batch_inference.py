#!/usr/bin/env python3
"""The reference workshop's headless job (NLP_workloads/Anyscale_job/
flan-t5-batch-inference.py) on the mi355 stack, using the drop-in `ray`
namespace: load Alpaca-shaped data -> 80/20 split -> tokenizer BatchMapper
(batch 4096, labels = input_ids copy) -> 2-worker HuggingFaceTrainer
(lr 2e-5, wd 0.01, per-device batch 2, epoch strategies, keep-1-best by
eval_loss) -> BatchPredictor fp16 generation -> pandas join.

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

from mi355.data.synthetic import synthetic_alpaca, synthetic_alpaca_corpus
from mi355.models.t5.configuration import t5_config
from mi355.models.t5.modeling import T5ForConditionalGeneration
from mi355.tokenize import T5Tokenizer
from mi355.train import Trainer, TrainingArguments

SMALL = os.environ.get("MI355_SMALL", "1" if not torch.cuda.is_available() else "0") == "1"
NUM_WORKERS = int(os.environ.get("MI355_NUM_WORKERS", "2"))
USE_GPU = torch.cuda.is_available() and not SMALL
SEQ = 32 if SMALL else 512
MODEL_OVERRIDES = (dict(num_layers=1, num_decoder_layers=1, d_model=64, d_ff=128,
                        num_heads=4, d_kv=16, vocab_size=512, dropout_rate=0.0)
                   if SMALL else {})

tokenizer = T5Tokenizer.train_new(synthetic_alpaca_corpus(256),
                                  vocab_size=(500 if SMALL else 8000),
                                  model_max_length=SEQ)


def preprocess_function(batch: pd.DataFrame) -> pd.DataFrame:
    # Anyscale_job/utils.py:23-31 semantics: (instruction, input) text pairs,
    # max-length padding, labels = copy of input_ids
    enc = tokenizer(list(batch["instruction"]), text_pair=list(batch["input"]),
                    padding="max_length", truncation=True, max_length=SEQ,
                    return_tensors="np")
    return pd.DataFrame({
        "input_ids": list(enc["input_ids"]),
        "attention_mask": list(enc["attention_mask"]),
        "labels": list(enc["input_ids"].copy()),
    })


def trainer_init_per_worker(train_dataset, eval_dataset=None, **config):
    torch.manual_seed(42)
    cfg = t5_config("flan-t5-base", **MODEL_OVERRIDES)
    model = T5ForConditionalGeneration(cfg)
    args = TrainingArguments(
        output_dir="flan-t5-base-finetuned-alpaca",
        learning_rate=config.get("learning_rate", 2e-5),
        num_train_epochs=config.get("epochs", 1 if SMALL else 4),
        weight_decay=config.get("weight_decay", 0.01),
        per_device_train_batch_size=2,
        per_device_eval_batch_size=4,
        evaluation_strategy="epoch", save_strategy="epoch",
        logging_strategy="epoch", no_cuda=not USE_GPU,
        bf16=USE_GPU, disable_tqdm=True,
    )
    return Trainer(model=model, args=args, train_dataset=train_dataset,
                   eval_dataset=eval_dataset, tokenizer=tokenizer)


class HuggingFaceModelPredictor(Predictor):
    """Anyscale_job/predictor.py:27-106 pattern."""

    def __init__(self, model, tokenizer=None, preprocessor=None):
        super().__init__(preprocessor)
        self.model = model
        self.tokenizer = tokenizer

    @classmethod
    def from_checkpoint(cls, checkpoint, *, model_cls, tokenizer=None,
                        torch_dtype=None, **kw):
        model = checkpoint.get_model(model_cls, torch_dtype=torch_dtype)
        if torch.cuda.is_available():
            model = model.to("cuda")
        return cls(model, tokenizer, preprocessor=checkpoint.get_preprocessor())

    def _predict_numpy(self, data, **generate_kwargs):
        ids = torch.as_tensor(np.asarray(data["input_ids"])).to(self.model.device)
        mask = torch.as_tensor(np.asarray(data["attention_mask"])).to(self.model.device)
        out = self.model.generate(input_ids=ids, attention_mask=mask,
                                  **generate_kwargs)
        text = self.tokenizer.batch_decode(out.cpu().numpy(), skip_special_tokens=True)
        return pd.DataFrame({"generated_output": text})


def main():
    ray.init()
    n_rows = 100 if SMALL else 2000  # SMALL_DATA limit(100), cell-21
    hf_dataset = synthetic_alpaca(n_rows + n_rows // 4, seed=57)
    train_ds, validation_ds = hf_dataset.train_test_split(test_size=0.2,
                                                          shuffle=True, seed=57)
    train_ds = train_ds.limit(n_rows)

    preprocessor = BatchMapper(preprocess_function, batch_format="pandas",
                               batch_size=4096)

    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=trainer_init_per_worker,
        scaling_config=ScalingConfig(num_workers=NUM_WORKERS, use_gpu=USE_GPU),
        datasets={"train": train_ds, "evaluation": validation_ds},
        run_config=RunConfig(
            name="flan-t5-alpaca",
            checkpoint_config=CheckpointConfig(
                num_to_keep=1, checkpoint_score_attribute="eval_loss",
                checkpoint_score_order="min")),
        preprocessor=preprocessor,
    )
    result = trainer.fit()
    assert result.error is None, result.error
    print("train metrics:", result.metrics)

    predictor = BatchPredictor.from_checkpoint(
        result.checkpoint, HuggingFaceModelPredictor,
        model_cls=T5ForConditionalGeneration, tokenizer=tokenizer,
        torch_dtype=torch.float16 if USE_GPU else None)
    prediction = predictor.predict(
        validation_ds.limit(16 if SMALL else 256),
        batch_size=256, num_gpus_per_worker=1 if USE_GPU else 0,
        max_new_tokens=8 if SMALL else 128)
    joined = validation_ds.limit(len(prediction.to_pandas())).to_pandas().join(
        prediction.to_pandas())
    print(joined[["instruction", "generated_output"]].head(5).to_string())
    ray.shutdown()
    print("JOB OK")


if __name__ == "__main__":
    main()
