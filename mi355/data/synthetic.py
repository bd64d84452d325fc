"""Synthetic dataset generators shaped like the reference's data sources
(SURVEY.md C23 — no network in this environment, BASELINE.json: benchmarks
run on synthetic data): Alpaca-shaped instruction data
(tatsu-lab/alpaca columns instruction/input/output/text), ADE20K-shaped
image+annotation pairs (scene_parse_150), NYC-taxi-shaped tabular rows."""
from __future__ import annotations

import numpy as np
import pandas as pd

from .dataset import Dataset

_WORDS = ("the quick brown fox jumps over lazy dog pack my box with five dozen "
          "liquor jugs how vexingly daft zebras judge wizards at a sphinx of "
          "black quartz please answer this question about summary translate "
          "classify explain write list describe generate").split()


def synthetic_alpaca(n_rows: int = 512, seed: int = 0) -> Dataset:
    rng = np.random.default_rng(seed)

    def sent(lo, hi):
        k = int(rng.integers(lo, hi))
        return " ".join(rng.choice(_WORDS, size=k))

    rows = []
    for _ in range(n_rows):
        instruction = sent(4, 12)
        inp = sent(0, 8) if rng.random() < 0.6 else ""
        output = sent(4, 24)
        text = f"Instruction: {instruction}\nInput: {inp}\nOutput: {output}"
        rows.append({"instruction": instruction, "input": inp,
                     "output": output, "text": text})
    return Dataset.from_pandas(pd.DataFrame(rows))


def synthetic_alpaca_corpus(n_rows: int = 256, seed: int = 0) -> list[str]:
    df = synthetic_alpaca(n_rows, seed).to_pandas()
    return (df["instruction"] + " " + df["input"] + " " + df["output"]).tolist()


def synthetic_ade20k(n_images: int = 16, size: int = 64, num_labels: int = 150,
                     seed: int = 0) -> Dataset:
    rng = np.random.default_rng(seed)
    rows = []
    for _ in range(n_images):
        img = (rng.random((size, size, 3)) * 255).astype(np.uint8)
        ann = rng.integers(0, num_labels + 1, (size, size)).astype(np.uint8)
        # nested lists: Arrow stores ndim>1 image data as list-of-list columns
        rows.append({"image": img.tolist(), "annotation": ann.tolist(),
                     "scene_category": int(rng.integers(0, 16))})
    return Dataset.from_pandas(pd.DataFrame(rows))


def synthetic_taxi(n_rows: int = 1000, seed: int = 0) -> Dataset:
    rng = np.random.default_rng(seed)
    dist = rng.gamma(2.0, 2.0, n_rows)
    dur = dist * 4 + rng.normal(0, 2, n_rows).clip(0.1)
    fare = 2.5 + dist * 2.0 + rng.normal(0, 1, n_rows)
    tip = (rng.random(n_rows) < (0.3 + 0.4 * (fare > fare.mean()))).astype(int)
    return Dataset.from_pandas(pd.DataFrame({
        "trip_distance": dist, "trip_duration": dur, "fare_amount": fare,
        "passenger_count": rng.integers(1, 5, n_rows), "is_big_tip": tip,
    }))
