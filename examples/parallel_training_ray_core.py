#!/usr/bin/env python3
"""The reference's ONLY published quantitative benchmark (SURVEY.md §6;
Overview_of_Ray.ipynb cell-24/44/47-51): train + score NUM_MODELS
RandomForest regressors sequentially, then as parallel `@ray.remote` tasks,
and report the speedup (the workshop reports 60 s -> 10 s, "6x performance
gain", on an M1 MacBook).

Synthetic California-Housing-shaped data (no network in this environment).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from sklearn.ensemble import RandomForestRegressor
from sklearn.metrics import mean_squared_error
from sklearn.model_selection import train_test_split

import ray

NUM_MODELS = int(os.environ.get("NUM_MODELS", "20"))


def make_data(seed=201):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(20640, 8))  # California Housing shape
    w = rng.normal(size=8)
    y = X @ w + 0.3 * rng.normal(size=len(X))
    train_x, test_x, train_y, test_y = train_test_split(
        X, y, test_size=0.2, random_state=201)
    return train_x, train_y, test_x, test_y


def train_and_score_model(train_x, train_y, test_x, test_y, n_estimators):
    start = time.time()
    model = RandomForestRegressor(n_estimators=n_estimators, random_state=201)
    model.fit(train_x, train_y)
    mse = mean_squared_error(test_y, model.predict(test_x))
    took = time.time() - start
    print(f"n_estimators={n_estimators:3d}  mse={mse:6.4f}  {took:5.2f}s")
    return n_estimators, mse


def run_sequential(data, n_list):
    return [train_and_score_model(*data, n) for n in n_list]


def run_parallel(data, n_list):
    remote_fn = ray.remote(train_and_score_model)
    data_refs = [ray.put(d) for d in data]
    futures = [remote_fn.remote(*data_refs, n) for n in n_list]
    return ray.get(futures)


def main():
    ray.init()
    data = make_data()
    n_list = [8 * (i + 1) for i in range(NUM_MODELS)]

    t0 = time.time()
    seq = run_sequential(data, n_list)
    t_seq = time.time() - t0
    print(f"sequential: {t_seq:.1f} s")

    t0 = time.time()
    par = run_parallel(data, n_list)
    t_par = time.time() - t0
    print(f"parallel:   {t_par:.1f} s -> {t_seq / t_par:.1f}x performance gain")

    assert sorted(seq) == sorted(par)
    ray.shutdown()


if __name__ == "__main__":
    main()
