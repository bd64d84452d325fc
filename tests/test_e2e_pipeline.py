"""End-to-end integration: the reference's headline notebook pipeline
(Model_finetuning_and_batch_inference.ipynb cells 8-69)
on synthetic Alpaca-shaped data with a tiny random-init T5 — tokenizer
BatchMapper -> 2-worker HuggingFaceTrainer -> best-checkpoint by eval_loss
-> generate smoke -> ASHA Tuner -> BatchPredictor."""
import numpy as np
import pandas as pd
import pytest

import mi355.data as mdata
from mi355 import tune
from mi355.air import CheckpointConfig, RunConfig, ScalingConfig
from mi355.data import BatchMapper
from mi355.data.synthetic import synthetic_alpaca, synthetic_alpaca_corpus
from mi355.predict import BatchPredictor, Predictor
from mi355.tokenize import T5Tokenizer
from mi355.train import HuggingFaceTrainer, Trainer, TrainingArguments
from mi355.tune import ASHAScheduler, TuneConfig, Tuner

SEQ = 24
VOCAB = 512


@pytest.fixture(scope="module")
def tokenizer(tmp_path_factory):
    tok = T5Tokenizer.train_new(synthetic_alpaca_corpus(128), vocab_size=VOCAB - 8,
                                model_max_length=SEQ)
    d = tmp_path_factory.mktemp("tok")
    tok.save_pretrained(str(d))
    return tok


def make_preprocessor(tokenizer):
    def preprocess_function(df: pd.DataFrame) -> pd.DataFrame:
        # mirrors Anyscale_job/utils.py:23-31 — text-pair encode, labels =
        # copy of input_ids (the workshop's input-reconstruction quirk)
        enc = tokenizer(list(df["instruction"]), text_pair=list(df["input"]),
                        padding="max_length", truncation=True, max_length=SEQ,
                        return_tensors="np")
        return pd.DataFrame({
            "input_ids": list(enc["input_ids"]),
            "attention_mask": list(enc["attention_mask"]),
            "labels": list(enc["input_ids"].copy()),
        })

    return BatchMapper(preprocess_function, batch_format="pandas", batch_size=4096)


def trainer_init_per_worker(train_ds, eval_ds, **config):
    import torch

    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    torch.manual_seed(0)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1,
                    d_model=32, d_ff=64, num_heads=2, d_kv=16,
                    vocab_size=VOCAB, dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    args = TrainingArguments(
        output_dir="/tmp/mi355_e2e_out",
        learning_rate=config.get("learning_rate", 2e-4),
        num_train_epochs=config.get("epochs", 2),
        weight_decay=config.get("weight_decay", 0.01),
        per_device_train_batch_size=2,
        per_device_eval_batch_size=4,
        evaluation_strategy="epoch", save_strategy="epoch",
        logging_strategy="epoch", no_cuda=True, disable_tqdm=True,
    )
    return Trainer(model=model, args=args, train_dataset=train_ds,
                   eval_dataset=eval_ds)


@pytest.mark.timeout(900)
def test_full_pipeline(tokenizer, tmp_path):
    # 1. data (cells 13-21): synthetic alpaca, 80/20 split, SMALL_DATA limit
    ds = synthetic_alpaca(60)
    train_ds, valid_ds = ds.train_test_split(test_size=0.2, shuffle=True, seed=57)
    train_ds = train_ds.limit(40)
    valid_ds = valid_ds.limit(12)

    preprocessor = make_preprocessor(tokenizer)

    # 2. fine-tune (cells 32-43): 2 DDP CPU workers
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=trainer_init_per_worker,
        trainer_init_config={"epochs": 2},
        scaling_config=ScalingConfig(num_workers=2, use_gpu=False),
        datasets={"train": train_ds, "evaluation": valid_ds},
        run_config=RunConfig(
            name="e2e", local_dir=str(tmp_path),
            checkpoint_config=CheckpointConfig(
                num_to_keep=1, checkpoint_score_attribute="eval_loss",
                checkpoint_score_order="min")),
        preprocessor=preprocessor,
    )
    result = trainer.fit()
    assert result.error is None, result.error
    assert result.checkpoint is not None
    # metrics JSONL observability (C25)
    import os
    assert os.path.exists(os.path.join(result.path, "metrics.jsonl"))

    # 3. generate smoke test (cells 45-49)
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    model = result.checkpoint.get_model(T5ForConditionalGeneration)
    tok2 = result.checkpoint.get_tokenizer(T5Tokenizer) if False else tokenizer
    enc = tokenizer("describe the quick fox", return_tensors="np",
                    padding="max_length", truncation=True, max_length=SEQ)
    import torch

    gen = model.generate(input_ids=torch.as_tensor(enc["input_ids"]).view(1, -1),
                         max_new_tokens=8)
    assert gen.shape[0] == 1 and gen.shape[1] >= 1

    # 4. ASHA tune over lr/epochs/wd (cells 51-59), 2 trials to stay fast
    tuner = Tuner(
        trainer,
        param_space={"trainer_init_config": {
            "learning_rate": tune.grid_search([2e-4, 2e-2]),
        }},
        tune_config=TuneConfig(metric="eval_loss", mode="min", num_samples=1,
                               max_concurrent_trials=1,
                               scheduler=ASHAScheduler(max_t=16)),
        run_config=RunConfig(name="e2e_tune", local_dir=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 2 and not grid.errors
    best = grid.get_best_result()
    assert "eval_loss" in best.metrics

    # 5. batch inference with the custom predictor pattern (cells 62-67 /
    #    Anyscale_job/predictor.py)
    class HuggingFaceModelPredictor(Predictor):
        def __init__(self, model, tokenizer=None, preprocessor=None):
            super().__init__(preprocessor)
            self.model = model
            self.tokenizer = tokenizer

        @classmethod
        def from_checkpoint(cls, checkpoint, *, model_cls, tokenizer=None, **kw):
            model = checkpoint.get_model(model_cls, **kw)
            return cls(model, tokenizer, preprocessor=checkpoint.get_preprocessor())

        def _predict_numpy(self, data, **generate_kwargs):
            import torch

            ids = torch.as_tensor(np.asarray(data["input_ids"]))
            out = self.model.generate(input_ids=ids, **generate_kwargs)
            text = self.tokenizer.batch_decode(out.numpy(), skip_special_tokens=True)
            return pd.DataFrame({"generated_output": text})

    best_ckpt = best.checkpoint or result.checkpoint
    predictor = BatchPredictor.from_checkpoint(
        best_ckpt, HuggingFaceModelPredictor,
        model_cls=T5ForConditionalGeneration, tokenizer=tokenizer)
    out = predictor.predict(valid_ds, batch_size=8, max_new_tokens=6,
                            feature_columns=None)
    df = out.to_pandas()
    assert "generated_output" in df.columns
    assert len(df) == valid_ds.count()


def test_checkpoint_as_directory_context():
    """Checkpoint dict<->dir duality: as_directory() materializes a dict
    checkpoint and cleans up after."""
    import os

    from mi355.air import Checkpoint

    ck = Checkpoint.from_dict({"weights": [1, 2, 3], "epoch": 7})
    with ck.as_directory() as d:
        assert os.path.isdir(d)
        back = Checkpoint.from_directory(d).to_dict()
        assert back["epoch"] == 7 and back["weights"] == [1, 2, 3]
    roundtrip = Checkpoint.from_dict(ck.to_dict())
    assert roundtrip.to_dict()["epoch"] == 7
