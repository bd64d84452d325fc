"""Trainer on a real GPU (single worker; the 8-GPU path is exercised by the
driver's scaling bench)."""
import numpy as np
import pandas as pd
import pytest

import mi355.data as mdata
from mi355.air import CheckpointConfig, RunConfig, ScalingConfig
from mi355.train import HuggingFaceTrainer, Trainer, TrainingArguments


def synthetic(n=16, seq=32, vocab=512):
    rng = np.random.default_rng(0)
    ids = rng.integers(2, vocab, (n, seq))
    return mdata.from_pandas(pd.DataFrame({
        "input_ids": list(ids),
        "attention_mask": [np.ones(seq, dtype=np.int64)] * n,
        "labels": list(ids.copy()),
    }))


def gpu_trainer_init(train_ds, eval_ds, **config):
    import torch

    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    torch.manual_seed(0)
    cfg = t5_config("flan-t5-small", num_layers=2, num_decoder_layers=2,
                    vocab_size=512)
    model = T5ForConditionalGeneration(cfg)
    args = TrainingArguments(
        output_dir="/tmp/mi355_gpu_hf", learning_rate=2e-4,
        num_train_epochs=2, per_device_train_batch_size=4, bf16=True,
        evaluation_strategy="epoch", save_strategy="epoch", disable_tqdm=True)
    return Trainer(model=model, args=args, train_dataset=train_ds,
                   eval_dataset=eval_ds)


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_hf_trainer_single_gpu(tmp_path):
    ds = synthetic()
    train, val = ds.train_test_split(test_size=0.25)
    res = HuggingFaceTrainer(
        trainer_init_per_worker=gpu_trainer_init,
        scaling_config=ScalingConfig(num_workers=1, use_gpu=True),
        datasets={"train": train, "evaluation": val},
        run_config=RunConfig(name="gpu_hf", local_dir=str(tmp_path),
                             checkpoint_config=CheckpointConfig(
                                 num_to_keep=1,
                                 checkpoint_score_attribute="eval_loss",
                                 checkpoint_score_order="min")),
    ).fit()
    assert res.error is None, res.error
    assert res.metrics["epoch"] == 2 and "eval_loss" in res.metrics
    assert res.checkpoint is not None


@pytest.mark.gpu
def test_bench_hybrid_two_graph_path():
    """The multi-GPU step strategy (graphed fwd+bwd / eager all-reduce /
    graphed optimizer) must work at world=1 via the validation knob."""
    import json
    import os
    import subprocess
    import sys

    env = dict(os.environ, MI355_HYBRID_GRAPH="1")
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "2"],
        capture_output=True, text=True, timeout=300, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["graphs"] is True, out.stdout + out.stderr
    assert d["value"] > 0


@pytest.mark.gpu
def test_gpu_workers_clamp_to_device_count(tmp_path):
    """ScalingConfig asking for more GPU workers than visible devices must
    clamp (RCCL enforces one rank per device; pre-fix this hung at
    rendezvous for 301 s) and still complete training."""
    import torch

    n = torch.cuda.device_count()
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=gpu_trainer_init,
        scaling_config=ScalingConfig(num_workers=n + 1, use_gpu=True),
        datasets={"train": synthetic(16), "evaluation": synthetic(8)},
        run_config=RunConfig(name="clamp-test", local_dir=str(tmp_path)),
    )
    result = trainer.fit()
    assert result.error is None, result.error
