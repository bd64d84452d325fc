"""Preprocessors: BatchMapper, MinMaxScaler, PowerTransformer,
StandardScaler, Chain — the family the reference exercises
(Model_finetuning_and_batch_inference.ipynb:cell-27 BatchMapper batch 4096;
Introduction_to_Ray_AI_Runtime.ipynb:cell-21 MinMaxScaler / cell-25
PowerTransformer) with fit/transform/fit_transform/transform_batch and
checkpoint-embedded persistence (predictor.py:70 get_preprocessor)."""
from __future__ import annotations

import numpy as np
import pandas as pd

from .dataset import Dataset, _from_table, _to_table


class Preprocessor:
    _is_fitted = False

    def fit(self, ds: Dataset) -> "Preprocessor":
        self._fit(ds)
        self._is_fitted = True
        return self

    def transform(self, ds: Dataset) -> Dataset:
        return ds.map_batches(self._transform_pandas, batch_format="pandas",
                              batch_size=self._batch_size())

    def fit_transform(self, ds: Dataset) -> Dataset:
        return self.fit(ds).transform(ds)

    def transform_batch(self, batch):
        """Single in-memory batch (DataFrame / dict of numpy) — the predictor
        path (SURVEY.md C14 auto-applied checkpoint preprocessor)."""
        if isinstance(batch, pd.DataFrame):
            return self._transform_pandas(batch)
        tbl = _to_table(batch)
        out = self._transform_pandas(tbl.to_pandas())
        return _from_table(_to_table(out), "numpy")

    # subclass hooks ---------------------------------------------------- #
    def _fit(self, ds: Dataset):
        pass

    def _transform_pandas(self, df: pd.DataFrame) -> pd.DataFrame:
        raise NotImplementedError

    def _batch_size(self):
        return 4096


class BatchMapper(Preprocessor):
    """Stateless row-batch transform (reference cell-27: tokenization with
    pandas batches of 4096)."""

    def __init__(self, fn, batch_format: str = "pandas", batch_size: int | None = 4096):
        self.fn = fn
        self.batch_format = batch_format
        self.batch_size = batch_size
        self._is_fitted = True

    def transform(self, ds: Dataset) -> Dataset:
        return ds.map_batches(self.fn, batch_format=self.batch_format,
                              batch_size=self.batch_size)

    def transform_batch(self, batch):
        if self.batch_format == "pandas" and not isinstance(batch, pd.DataFrame):
            out = self.fn(_to_table(batch).to_pandas())
            return _from_table(_to_table(out), "numpy")
        if self.batch_format == "numpy" and isinstance(batch, pd.DataFrame):
            out = self.fn(_from_table(_to_table(batch), "numpy"))
            return _to_table(out).to_pandas()
        return self.fn(batch)

    def _fit(self, ds):
        pass


class _ColumnScaler(Preprocessor):
    def __init__(self, columns):
        self.columns = list(columns)
        self.stats_ = {}


class MinMaxScaler(_ColumnScaler):
    def _fit(self, ds: Dataset):
        df = ds.to_pandas()
        for c in self.columns:
            self.stats_[c] = (float(df[c].min()), float(df[c].max()))

    def _transform_pandas(self, df):
        df = df.copy()
        for c in self.columns:
            lo, hi = self.stats_[c]
            span = (hi - lo) or 1.0
            df[c] = (df[c] - lo) / span
        return df


class StandardScaler(_ColumnScaler):
    def _fit(self, ds: Dataset):
        df = ds.to_pandas()
        for c in self.columns:
            self.stats_[c] = (float(df[c].mean()), float(df[c].std() or 1.0))

    def _transform_pandas(self, df):
        df = df.copy()
        for c in self.columns:
            mu, sd = self.stats_[c]
            df[c] = (df[c] - mu) / (sd or 1.0)
        return df


class PowerTransformer(_ColumnScaler):
    """Yeo-Johnson with given or fitted lambda (reference cell-25 uses
    power=0.5 semantics: a fixed-lambda transform)."""

    def __init__(self, columns, power: float | None = None):
        super().__init__(columns)
        self.power = power

    def _fit(self, ds: Dataset):
        if self.power is not None:
            for c in self.columns:
                self.stats_[c] = self.power
            return
        from scipy import stats as sstats

        df = ds.to_pandas()
        for c in self.columns:
            _, lam = sstats.yeojohnson(df[c].to_numpy(dtype=float))
            self.stats_[c] = float(lam)

    @staticmethod
    def _yeo_johnson(x: np.ndarray, lam: float) -> np.ndarray:
        out = np.empty_like(x, dtype=float)
        pos = x >= 0
        if abs(lam) > 1e-9:
            out[pos] = ((x[pos] + 1.0) ** lam - 1.0) / lam
        else:
            out[pos] = np.log1p(x[pos])
        if abs(lam - 2.0) > 1e-9:
            out[~pos] = -(((-x[~pos] + 1.0) ** (2.0 - lam)) - 1.0) / (2.0 - lam)
        else:
            out[~pos] = -np.log1p(-x[~pos])
        return out

    def _transform_pandas(self, df):
        df = df.copy()
        for c in self.columns:
            df[c] = self._yeo_johnson(df[c].to_numpy(dtype=float), self.stats_[c])
        return df


class LabelEncoder(Preprocessor):
    def __init__(self, label_column: str):
        self.label_column = label_column
        self.classes_: list = []

    def _fit(self, ds: Dataset):
        self.classes_ = ds.unique(self.label_column)

    def _transform_pandas(self, df):
        df = df.copy()
        lut = {v: i for i, v in enumerate(self.classes_)}
        df[self.label_column] = df[self.label_column].map(lut)
        return df


class Chain(Preprocessor):
    def __init__(self, *preprocessors):
        self.preprocessors = list(preprocessors)

    def fit(self, ds: Dataset) -> "Chain":
        for p in self.preprocessors:
            ds = p.fit_transform(ds) if not isinstance(p, BatchMapper) else p.transform(ds)
        self._is_fitted = True
        return self

    def transform(self, ds: Dataset) -> Dataset:
        for p in self.preprocessors:
            ds = p.transform(ds)
        return ds

    def transform_batch(self, batch):
        for p in self.preprocessors:
            batch = p.transform_batch(batch)
        return batch
