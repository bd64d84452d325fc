"""GPU batch-inference paths: actor-pool BatchPredictor with GPU replicas
(C14's one-predictor-per-GPU design, scaled down to the 1-GPU test box)."""
import numpy as np
import pandas as pd
import pytest
import torch

import mi355
import mi355.data as mdata
from mi355.air import Checkpoint
from mi355.predict import BatchPredictor, HuggingFacePredictor


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_batch_predictor_gpu_actor_pool(tmp_path):
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    torch.manual_seed(0)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1,
                    d_model=128, d_ff=256, num_heads=2, d_kv=64, vocab_size=512,
                    dropout_rate=0.0)
    T5ForConditionalGeneration(cfg).save_pretrained(str(tmp_path))

    mi355.init(num_workers=2, num_gpus=1)
    try:
        ds = mdata.from_pandas(pd.DataFrame({
            "input_ids": [np.arange(16) % 500 + 2 for _ in range(12)],
            "attention_mask": [np.ones(16, dtype=np.int64)] * 12,
        }))
        bp = BatchPredictor.from_checkpoint(
            Checkpoint.from_directory(str(tmp_path)), HuggingFacePredictor,
            task="text2text-generation", torch_dtype=torch.bfloat16)
        out = bp.predict(ds, batch_size=4, max_scoring_workers=2,
                         num_gpus_per_worker=1, max_new_tokens=4, min_new_tokens=4)
        df = out.to_pandas()
        assert len(df) == 12
        assert all(len(t) == 4 for t in df["generated_tokens"])
    finally:
        mi355.shutdown()
