import os, sys, time
sys.path.insert(0, ".")
import torch
from mi355.air.config import ScalingConfig
from mi355.train import Trainer, TrainingArguments
from mi355.train import HuggingFaceTrainer
from mi355.data.synthetic import synthetic_alpaca
from mi355.models.t5.configuration import t5_config
from mi355.models.t5.modeling import T5ForConditionalGeneration
import numpy as np, pandas as pd

def trainer_init_per_worker(train_dataset, eval_dataset=None, **config):
    r = os.environ.get("RANK"); print(f"[child rank={r}] init_per_worker reached, HIP_VISIBLE={os.environ.get('HIP_VISIBLE_DEVICES')}, LOCAL_RANK={os.environ.get('LOCAL_RANK')}, ndev={torch.cuda.device_count()}", flush=True)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1, d_model=64,
                    d_ff=128, num_heads=4, d_kv=16, vocab_size=512)
    m = T5ForConditionalGeneration(cfg).to(torch.bfloat16)
    targs = TrainingArguments("/tmp/dbg-ws2", num_train_epochs=1,
                              per_device_train_batch_size=2, evaluation_strategy="no",
                              save_strategy="no", logging_strategy="no")
    return Trainer(model=m, args=targs, train_dataset=train_dataset)

def pre(b: pd.DataFrame) -> pd.DataFrame:
    n = len(b)
    ids = np.random.randint(3, 500, (n, 16))
    return pd.DataFrame({"input_ids": list(ids), "attention_mask": [np.ones(16, np.int64)]*n,
                         "labels": list(ids.copy())})

from ray.data.preprocessors import BatchMapper
import ray

def main():
    ray.init()
    ds = synthetic_alpaca(32)
    t = HuggingFaceTrainer(trainer_init_per_worker=trainer_init_per_worker,
                           scaling_config=ScalingConfig(num_workers=2, use_gpu=True),
                           datasets={"train": ds},
                           preprocessor=BatchMapper(pre, batch_format="pandas"))
    import threading
    def watchdog():
        time.sleep(90)
        print("WATCHDOG: still running after 90s", flush=True)
        os._exit(7)
    threading.Thread(target=watchdog, daemon=True).start()
    res = t.fit()
    print("fit done, error:", res.error)


if __name__ == "__main__":
    main()
