"""SegFormer fine-tune through HuggingFaceTrainer on CPU, 2 DDP workers —
the reference's supplementary pipeline (Scaling_model_training.ipynb:
cell-39/42 image BatchMapper, cell-47 explicit AdamW+LambdaLR passed as
optimizers=(opt, sched), no_cuda=True; cell-50-51 use_gpu=False,
keep-1-best-by-loss)."""
import numpy as np
import pandas as pd
import pytest

from mi355.air import CheckpointConfig, RunConfig, ScalingConfig
from mi355.data import BatchMapper
from mi355.data.synthetic import synthetic_ade20k
from mi355.models.segformer import (SegformerConfig, SegformerImageProcessor)
from mi355.train import HuggingFaceTrainer, Trainer, TrainingArguments

TINY = dict(depths=[1, 1, 1, 1], hidden_sizes=[8, 16, 24, 32],
            num_attention_heads=[1, 2, 3, 4], decoder_hidden_size=32,
            num_labels=12)


def images_preprocessor():
    proc = SegformerImageProcessor(size=32, do_reduce_labels=True)

    def fn(df: pd.DataFrame) -> pd.DataFrame:
        enc = proc([np.asarray(im) for im in df["image"]],
                   segmentation_maps=[np.asarray(a) for a in df["annotation"]])
        return pd.DataFrame({"pixel_values": list(enc["pixel_values"]),
                             "labels": list(enc["labels"])})

    return BatchMapper(fn, batch_format="pandas", batch_size=2)


def trainer_init_per_worker(train_ds, eval_ds, **config):
    import torch

    from mi355.models.segformer.modeling import SegformerForSemanticSegmentation

    torch.manual_seed(0)
    model = SegformerForSemanticSegmentation(SegformerConfig(**TINY))
    args = TrainingArguments(
        output_dir="/tmp/mi355_seg_out", learning_rate=config.get("lr", 1e-4),
        num_train_epochs=config.get("epochs", 2), per_device_train_batch_size=4,
        evaluation_strategy="no", save_strategy="epoch", no_cuda=True,
        remove_unused_columns=False, disable_tqdm=True)
    # the reference passes an explicit AdamW + LambdaLR (cell-47)
    opt = torch.optim.AdamW(model.parameters(), lr=args.learning_rate)
    sched = torch.optim.lr_scheduler.LambdaLR(opt, lambda step: max(0.9 ** step, 0.1))
    return Trainer(model=model, args=args, train_dataset=train_ds,
                   optimizers=(opt, sched))


@pytest.mark.timeout(900)
def test_segformer_trainer_two_cpu_workers(tmp_path):
    ds = synthetic_ade20k(n_images=16, size=40, num_labels=12)
    trainer = HuggingFaceTrainer(
        trainer_init_per_worker=trainer_init_per_worker,
        trainer_init_config={"epochs": 2},
        scaling_config=ScalingConfig(num_workers=2, use_gpu=False),
        datasets={"train": ds},
        run_config=RunConfig(
            name="seg", local_dir=str(tmp_path),
            checkpoint_config=CheckpointConfig(num_to_keep=1,
                                               checkpoint_score_attribute="loss",
                                               checkpoint_score_order="min")),
        preprocessor=images_preprocessor(),
    )
    res = trainer.fit()
    assert res.error is None, res.error
    assert res.metrics["epoch"] == 2 and res.metrics["loss"] > 0
    assert res.checkpoint is not None
    from mi355.models.segformer.modeling import SegformerForSemanticSegmentation

    model = res.checkpoint.get_model(SegformerForSemanticSegmentation)
    assert model.config.num_labels == 12


def test_flat_replica_preserves_batchnorm_dtype():
    """bf16-cast SegFormer keeps BatchNorm fp32 (its own flat group) and the
    fused optimizer steps both dtype groups."""
    import torch

    from mi355.models.segformer import SegformerConfig
    from mi355.models.segformer.modeling import SegformerForSemanticSegmentation
    from mi355.parallel import FlatAdamW, FlatReplica

    torch.manual_seed(3)
    model = SegformerForSemanticSegmentation(
        SegformerConfig(**TINY)).to(torch.bfloat16)
    assert model.decode_head.batch_norm.weight.dtype == torch.float32
    replica = FlatReplica(model)
    assert model.decode_head.batch_norm.weight.dtype == torch.float32
    dtypes = {g.flat_param.dtype for _, g in replica.groups}
    assert dtypes == {torch.bfloat16, torch.float32}
    opt = FlatAdamW(replica, lr=1e-3)
    pix = torch.randn(2, 3, 32, 32)
    labels = torch.randint(0, 12, (2, 32, 32))
    before = model.decode_head.batch_norm.weight.detach().clone()
    opt.zero_grad()
    out = model(pixel_values=pix, labels=labels)
    out.loss.backward()
    replica.finish_grad_sync()
    opt.step()
    assert out.loss.isfinite()
    assert not torch.equal(before, model.decode_head.batch_norm.weight)
