"""ASHA tuner smoke on real hardware (falls back to CPU): 2 trials,
tiny T5, epoch-wise eval_loss reports driving the lockstep early stop."""
import sys
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import torch
import ray
from ray import tune
from ray.air.config import RunConfig, ScalingConfig
from ray.tune import TuneConfig, Tuner
from ray.tune.schedulers import ASHAScheduler
from mi355.data.synthetic import synthetic_alpaca
from mi355.models.t5.configuration import t5_config
from mi355.models.t5.modeling import T5ForConditionalGeneration
from mi355.tokenize import T5Tokenizer
from mi355.train import HuggingFaceTrainer, Trainer, TrainingArguments
from mi355.data.synthetic import synthetic_alpaca_corpus
import pandas as pd
from ray.data.preprocessors import BatchMapper

tok = T5Tokenizer.train_new(synthetic_alpaca_corpus(128), vocab_size=500, model_max_length=32)

def prep(df: pd.DataFrame) -> pd.DataFrame:
    enc = tok(list(df["instruction"]), text_pair=list(df["input"]),
              padding="max_length", truncation=True, max_length=32, return_tensors="np")
    return pd.DataFrame({"input_ids": list(enc["input_ids"]),
                         "attention_mask": list(enc["attention_mask"]),
                         "labels": list(enc["input_ids"].copy())})

def init_worker(train_ds, eval_ds=None, **cfg):
    torch.manual_seed(1)
    model = T5ForConditionalGeneration(t5_config(
        "flan-t5-small", num_layers=2, num_decoder_layers=2, d_model=64, d_ff=128,
        num_heads=4, d_kv=16, vocab_size=512, dropout_rate=0.0))
    args = TrainingArguments(output_dir="tune-out",
                             learning_rate=cfg.get("learning_rate", 2e-5),
                             num_train_epochs=cfg.get("epochs", 2),
                             per_device_train_batch_size=4,
                             evaluation_strategy="epoch", save_strategy="epoch",
                             bf16=torch.cuda.is_available(), disable_tqdm=True)
    return Trainer(model=model, args=args, train_dataset=train_ds,
                   eval_dataset=eval_ds, tokenizer=tok)

def main():
    ray.init()
    ds = synthetic_alpaca(96, seed=3)
    tr, ev = ds.train_test_split(test_size=0.25, shuffle=True, seed=3)
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=init_worker,
        scaling_config=ScalingConfig(num_workers=1, use_gpu=torch.cuda.is_available()),
        datasets={"train": tr, "evaluation": ev},
        preprocessor=BatchMapper(prep, batch_format="pandas", batch_size=64),
    )
    tuner = Tuner(trainer,
                  param_space={"trainer_init_config": {
                      "learning_rate": tune.choice([1e-4, 2e-5]), "epochs": 3}},
                  tune_config=TuneConfig(metric="eval_loss", mode="min",
                                         num_samples=2,
                                         scheduler=ASHAScheduler(max_t=3)),
                  run_config=RunConfig(name="asha-gpu-smoke"))
    grid = tuner.fit()
    best = grid.get_best_result()
    print("ASHA GPU smoke OK, best eval_loss:", best.metrics.get("eval_loss"))


if __name__ == "__main__":
    main()
