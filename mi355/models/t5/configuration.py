"""T5 / FLAN-T5 configuration.

Mirrors the architecture hyperparameters of the reference workshop's
``google/flan-t5-{small,base,large}`` models (SURVEY.md §2.3 "Shapes":
vocab 32128, d_model 768, 12 enc + 12 dec layers, 12 heads x d_kv 64,
d_ff 2048 gated, relative-position buckets 32; reference call sites
Model_finetuning_and_batch_inference.ipynb:cell-25, cell-32).
"""
from __future__ import annotations

import json
import os
from dataclasses import asdict, dataclass, field


@dataclass
class T5Config:
    vocab_size: int = 32128
    d_model: int = 768
    d_kv: int = 64
    d_ff: int = 2048
    num_layers: int = 12
    num_decoder_layers: int = 12
    num_heads: int = 12
    relative_attention_num_buckets: int = 32
    relative_attention_max_distance: int = 128
    dropout_rate: float = 0.1
    layer_norm_epsilon: float = 1e-6
    initializer_factor: float = 1.0
    feed_forward_proj: str = "gated-gelu"
    tie_word_embeddings: bool = False  # flan-t5 (T5 v1.1) unties the LM head
    pad_token_id: int = 0
    eos_token_id: int = 1
    decoder_start_token_id: int = 0
    is_encoder_decoder: bool = True
    model_type: str = "t5"
    architectures: list = field(default_factory=lambda: ["T5ForConditionalGeneration"])

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "T5Config":
        known = {f for f in cls.__dataclass_fields__}
        return cls(**{k: v for k, v in d.items() if k in known})

    def save_pretrained(self, save_dir: str):
        os.makedirs(save_dir, exist_ok=True)
        with open(os.path.join(save_dir, "config.json"), "w") as f:
            json.dump(self.to_dict(), f, indent=2, sort_keys=True)

    @classmethod
    def from_pretrained(cls, path: str) -> "T5Config":
        cfg = path if path.endswith("config.json") else os.path.join(path, "config.json")
        with open(cfg) as f:
            return cls.from_dict(json.load(f))


# Named presets matching the reference's model choices (BASELINE.json configs).
T5_PRESETS = {
    "flan-t5-small": dict(
        d_model=512, d_kv=64, d_ff=1024, num_layers=8, num_decoder_layers=8, num_heads=6
    ),
    "flan-t5-base": dict(
        d_model=768, d_kv=64, d_ff=2048, num_layers=12, num_decoder_layers=12, num_heads=12
    ),
    "flan-t5-large": dict(
        d_model=1024, d_kv=64, d_ff=2816, num_layers=24, num_decoder_layers=24, num_heads=16
    ),
    # scale-up presets (google/flan-t5-xl, -xxl): same T5-v1.1 topology;
    # xxl is ~11B params = ~44 GB of bf16 master+optimizer state, well
    # inside one MI355X's 288 GB HBM3E for single-GPU fine-tuning
    "flan-t5-xl": dict(
        d_model=2048, d_kv=64, d_ff=5120, num_layers=24, num_decoder_layers=24, num_heads=32
    ),
    "flan-t5-xxl": dict(
        d_model=4096, d_kv=64, d_ff=10240, num_layers=24, num_decoder_layers=24, num_heads=64
    ),
}


def t5_config(name: str, **overrides) -> T5Config:
    key = name.split("/")[-1]
    if key not in T5_PRESETS:
        raise KeyError(f"unknown T5 preset {name!r}; have {sorted(T5_PRESETS)}")
    kw = dict(T5_PRESETS[key])
    kw.update(overrides)
    return T5Config(**kw)
