"""XGBoostPredictor analog over the mi355 GBT model (SURVEY.md C16;
reference Introduction_to_Ray_AI_Runtime.ipynb:cell-57/60/71)."""
from __future__ import annotations

import pandas as pd

from ..air import Checkpoint
from .predictor import Predictor


class XGBoostPredictor(Predictor):
    def __init__(self, model, preprocessor=None):
        super().__init__(preprocessor)
        self.model = model

    @classmethod
    def from_checkpoint(cls, checkpoint: Checkpoint, **_) -> "XGBoostPredictor":
        from ..models.gbt import GBTModel

        model = checkpoint.get_model(GBTModel)
        return cls(model, preprocessor=checkpoint.get_preprocessor())

    def _predict_pandas(self, df: pd.DataFrame, **_):
        cols = self.model.feature_names or list(df.columns)
        preds = self.model.predict(df[cols].to_numpy(dtype=float))
        return pd.DataFrame({"predictions": preds})
