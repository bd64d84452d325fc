"""Predictor base class (reference
NLP_workloads/Anyscale_job/predictor.py:14-106): ctor stores preprocessor,
``from_checkpoint`` classmethod contract, ``_predict_numpy`` /
``_predict_pandas`` overrides with automatic batch-format conversion."""
from __future__ import annotations

import numpy as np
import pandas as pd

from ..air import Checkpoint


class Predictor:
    def __init__(self, preprocessor=None):
        self._preprocessor = preprocessor

    @classmethod
    def from_checkpoint(cls, checkpoint: Checkpoint, **kwargs) -> "Predictor":
        raise NotImplementedError

    def get_preprocessor(self):
        return self._preprocessor

    def set_preprocessor(self, preprocessor):
        self._preprocessor = preprocessor

    # ------------------------------------------------------------------ #
    def predict(self, data, **kwargs):
        """data: DataFrame | dict of numpy | numpy array."""
        if self._preprocessor is not None:
            data = self._preprocessor.transform_batch(data)
        has_pandas = type(self)._predict_pandas is not Predictor._predict_pandas
        has_numpy = type(self)._predict_numpy is not Predictor._predict_numpy
        if isinstance(data, pd.DataFrame):
            if has_pandas:
                return self._predict_pandas(data, **kwargs)
            return self._predict_numpy(_df_to_numpy(data), **kwargs)
        data_np = _as_numpy_batch(data)
        if has_numpy:
            return self._predict_numpy(data_np, **kwargs)
        return self._predict_pandas(_numpy_to_df(data_np), **kwargs)

    # subclass hooks ---------------------------------------------------- #
    def _predict_numpy(self, data: dict, **kwargs):
        raise NotImplementedError

    def _predict_pandas(self, data: pd.DataFrame, **kwargs):
        raise NotImplementedError


def _df_to_numpy(df: pd.DataFrame) -> dict:
    out = {}
    for c in df.columns:
        col = df[c]
        if len(col) and isinstance(col.iloc[0], (list, np.ndarray)):
            out[c] = np.stack([np.asarray(v) for v in col])
        else:
            out[c] = col.to_numpy()
    return out


def _numpy_to_df(d) -> pd.DataFrame:
    if isinstance(d, np.ndarray):
        return pd.DataFrame({"predictions": list(d)})
    return pd.DataFrame({k: list(v) for k, v in d.items()})


def _as_numpy_batch(data):
    if isinstance(data, np.ndarray):
        return {"__value__": data}
    return {k: np.asarray(v) for k, v in data.items()}
