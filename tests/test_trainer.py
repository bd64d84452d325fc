"""Training layer tests (C8-C12, C22): TorchTrainer + HuggingFaceTrainer on
CPU with world_size 2 (gloo) — mirrors the reference pipeline
flan-t5-batch-inference.py:44-113 on tiny synthetic Alpaca-shaped data."""
import numpy as np
import pandas as pd
import pytest

import mi355.data as mdata
from mi355.air import CheckpointConfig, RunConfig, ScalingConfig
from mi355.train import (HuggingFaceTrainer, Trainer, TrainingArguments,
                         TorchTrainer, session)


def synthetic_tokenized(n=32, seq=16, vocab=256):
    rng = np.random.default_rng(0)
    ids = rng.integers(2, vocab, (n, seq))
    return mdata.from_pandas(pd.DataFrame({
        "input_ids": list(ids),
        "attention_mask": [np.ones(seq, dtype=np.int64)] * n,
        "labels": list(ids.copy()),
    }))


def tiny_trainer_init(train_ds, eval_ds, **config):
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    import torch
    torch.manual_seed(0)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1,
                    d_model=32, d_ff=64, num_heads=2, d_kv=16, vocab_size=256,
                    dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    args = TrainingArguments(
        output_dir="/tmp/mi355_test_hf",
        learning_rate=config.get("learning_rate", 2e-4),
        num_train_epochs=config.get("epochs", 2),
        weight_decay=config.get("weight_decay", 0.01),
        per_device_train_batch_size=4,
        per_device_eval_batch_size=4,
        evaluation_strategy="epoch",
        save_strategy="epoch",
        logging_strategy="epoch",
        no_cuda=True,
        disable_tqdm=True,
    )
    return Trainer(model=model, args=args, train_dataset=train_ds,
                   eval_dataset=eval_ds)


@pytest.mark.timeout(600)
def test_hf_trainer_two_workers_cpu(tmp_path):
    ds = synthetic_tokenized()
    train, val = ds.train_test_split(test_size=0.25)
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=tiny_trainer_init,
        trainer_init_config={"epochs": 2},
        scaling_config=ScalingConfig(num_workers=2, use_gpu=False),
        datasets={"train": train, "evaluation": val},
        run_config=RunConfig(
            name="hf_test", local_dir=str(tmp_path),
            checkpoint_config=CheckpointConfig(
                num_to_keep=1, checkpoint_score_attribute="eval_loss",
                checkpoint_score_order="min"),
        ),
    )
    result = trainer.fit()
    assert result.error is None, result.error
    assert "eval_loss" in result.metrics and result.metrics["epoch"] == 2
    assert result.checkpoint is not None
    assert len(result.best_checkpoints) == 1  # num_to_keep=1
    # checkpoint loads back as a model (C12 contract, predictor.py:68)
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    model = result.checkpoint.get_model(T5ForConditionalGeneration)
    assert model.config.vocab_size == 256
    # metrics dataframe has one row per epoch
    assert len(result.metrics_dataframe) == 2


@pytest.mark.timeout(300)
def test_torch_trainer_loop_and_report(tmp_path):
    def loop(config):
        import torch

        shard = session.get_dataset_shard("train")
        n = shard.count()
        w = torch.nn.Parameter(torch.zeros(1))
        opt = torch.optim.SGD([w], lr=config["lr"])
        for epoch in range(3):
            loss = (w - 1.0).pow(2).sum()
            opt.zero_grad(); loss.backward(); opt.step()
            session.report({"epoch": epoch, "loss": float(loss), "rows": n})

    ds = mdata.from_items([{"x": i} for i in range(10)])
    t = TorchTrainer(loop, train_loop_config={"lr": 0.3},
                     scaling_config=ScalingConfig(num_workers=1),
                     datasets={"train": ds},
                     run_config=RunConfig(name="torch_test", local_dir=str(tmp_path)))
    res = t.fit()
    assert res.error is None
    assert res.metrics["epoch"] == 2 and res.metrics["rows"] == 10
    assert res.metrics["loss"] < 1.0


@pytest.mark.timeout(300)
def test_trainer_error_surfaces(tmp_path):
    def bad_loop(config):
        raise RuntimeError("exploding gradient of doom")

    t = TorchTrainer(bad_loop, train_loop_config={},
                     scaling_config=ScalingConfig(num_workers=1),
                     run_config=RunConfig(name="err_test", local_dir=str(tmp_path)))
    res = t.fit()
    assert res.error is not None and "doom" in str(res.error)
