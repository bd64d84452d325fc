"""Histogram gradient-boosted trees (SURVEY.md C18 — the XGBoost capability
the reference exercises via XGBoostTrainer/XGBoostPredictor,
Introduction_to_Ray_AI_Runtime.ipynb:cell-31-35/57-61).

From-scratch numpy implementation (xgboost's C++ core is not a dependency
here): quantile-binned features, second-order (grad/hess) boosting,
level-wise growth with per-level histogram aggregation. Data parallelism
follows xgboost's rabit design (M8): each worker builds histograms on its
row shard and an all-reduce sums them — the ``allreduce`` hook is wired to
torch.distributed by the trainer.
"""
from __future__ import annotations

import json
import os

import numpy as np

MAX_BINS = 64


def _sigmoid(x):
    return 1.0 / (1.0 + np.exp(-x))


class _Tree:
    __slots__ = ("feature", "threshold", "left", "right", "value")

    def __init__(self):
        self.feature, self.threshold = [], []
        self.left, self.right, self.value = [], [], []

    def add_node(self):
        self.feature.append(-1)
        self.threshold.append(0.0)
        self.left.append(-1)
        self.right.append(-1)
        self.value.append(0.0)
        return len(self.feature) - 1

    def predict(self, X: np.ndarray) -> np.ndarray:
        out = np.zeros(len(X))
        node = np.zeros(len(X), dtype=np.int64)
        feature = np.asarray(self.feature)
        thresh = np.asarray(self.threshold)
        left = np.asarray(self.left)
        right = np.asarray(self.right)
        value = np.asarray(self.value)
        active = feature[node] >= 0
        while active.any():
            f = feature[node[active]]
            go_left = X[active, f] < thresh[node[active]]
            nxt = np.where(go_left, left[node[active]], right[node[active]])
            node[active] = nxt
            active = feature[node] >= 0
        return value[node]

    def to_dict(self):
        return {k: list(map(float, getattr(self, k))) if k == "threshold"
                else list(map(int, getattr(self, k))) if k != "value"
                else list(map(float, self.value)) for k in
                ("feature", "threshold", "left", "right", "value")}

    @classmethod
    def from_dict(cls, d):
        t = cls()
        t.feature = list(d["feature"])
        t.threshold = list(d["threshold"])
        t.left = list(d["left"])
        t.right = list(d["right"])
        t.value = list(d["value"])
        return t


class GBTModel:
    def __init__(self, params: dict | None = None):
        p = dict(params or {})
        self.objective = p.get("objective", "reg:squarederror")
        self.eta = p.get("eta", p.get("learning_rate", 0.3))
        self.max_depth = int(p.get("max_depth", 6))
        self.reg_lambda = p.get("lambda", p.get("reg_lambda", 1.0))
        self.min_child_weight = p.get("min_child_weight", 1.0)
        self.gamma = p.get("gamma", 0.0)
        self.base_score = p.get("base_score", 0.5)
        self.trees: list[_Tree] = []
        self.bin_edges: list[np.ndarray] | None = None
        self.feature_names: list[str] | None = None

    # -- gradients ------------------------------------------------------ #
    def _base_margin(self):
        if self.objective.startswith("binary"):
            return float(np.log(self.base_score / (1 - self.base_score)))
        return float(self.base_score)

    def _grad_hess(self, y, margin):
        if self.objective.startswith("binary"):
            p = _sigmoid(margin)
            return p - y, np.maximum(p * (1 - p), 1e-12)
        return margin - y, np.ones_like(y)

    def metric(self, y, margin, name=None):
        if self.objective.startswith("binary"):
            p = np.clip(_sigmoid(margin), 1e-12, 1 - 1e-12)
            logloss = float(-(y * np.log(p) + (1 - y) * np.log(1 - p)).mean())
            err = float(((p > 0.5) != (y > 0.5)).mean())
            return {"logloss": logloss, "error": err}
        return {"rmse": float(np.sqrt(((margin - y) ** 2).mean()))}

    # -- binning -------------------------------------------------------- #
    def make_bins(self, X: np.ndarray):
        self.bin_edges = []
        for f in range(X.shape[1]):
            qs = np.quantile(X[:, f], np.linspace(0, 1, MAX_BINS + 1)[1:-1])
            self.bin_edges.append(np.unique(qs))
        return self.binned(X)

    def binned(self, X: np.ndarray) -> np.ndarray:
        out = np.empty(X.shape, dtype=np.int32)
        for f in range(X.shape[1]):
            out[:, f] = np.searchsorted(self.bin_edges[f], X[:, f], side="right")
        return out

    # -- one boosting round --------------------------------------------- #
    def boost_round(self, Xb: np.ndarray, X: np.ndarray, y: np.ndarray,
                    margin: np.ndarray, allreduce=None) -> np.ndarray:
        """Grow one tree on the (local shard of the) data. ``allreduce(arr)``
        sums float64 arrays across workers (identity when single-process).
        Returns the updated margin for the local rows."""
        n, nfeat = Xb.shape
        nbins = MAX_BINS + 1
        g, h = self._grad_hess(y, margin)
        tree = _Tree()
        root = tree.add_node()
        node_of = np.zeros(n, dtype=np.int64)
        frontier = [root]
        for depth in range(self.max_depth):
            if not frontier:
                break
            nid_index = {nid: i for i, nid in enumerate(frontier)}
            k = len(frontier)
            hist = np.zeros((k, nfeat, nbins, 2))
            rows_mask = np.isin(node_of, frontier)
            rows = np.nonzero(rows_mask)[0]
            li = np.searchsorted(np.asarray(frontier), node_of[rows])
            for f in range(nfeat):
                b = Xb[rows, f]
                flat = (li * nbins + b).astype(np.int64)
                hist[:, f, :, 0] += np.bincount(flat, weights=g[rows],
                                                minlength=k * nbins).reshape(k, nbins)
                hist[:, f, :, 1] += np.bincount(flat, weights=h[rows],
                                                minlength=k * nbins).reshape(k, nbins)
            if allreduce is not None:
                hist = allreduce(hist)
            new_frontier = []
            for nid in frontier:
                i = nid_index[nid]
                Gt = hist[i, 0, :, 0].sum()
                Ht = hist[i, 0, :, 1].sum()
                base = Gt * Gt / (Ht + self.reg_lambda)
                best_gain, best_f, best_b = 0.0, -1, -1
                for f in range(nfeat):
                    cg = np.cumsum(hist[i, f, :, 0])
                    ch = np.cumsum(hist[i, f, :, 1])
                    GL, HL = cg[:-1], ch[:-1]
                    GR, HR = Gt - GL, Ht - HL
                    ok = (HL >= self.min_child_weight) & (HR >= self.min_child_weight)
                    gain = np.where(ok,
                                    GL * GL / (HL + self.reg_lambda) +
                                    GR * GR / (HR + self.reg_lambda) - base,
                                    -np.inf)
                    j = int(np.argmax(gain))
                    if gain[j] > best_gain + self.gamma:
                        best_gain, best_f, best_b = float(gain[j]), f, j
                if best_f < 0 or depth == self.max_depth - 1:
                    tree.value[nid] = float(-Gt / (Ht + self.reg_lambda) * self.eta)
                    continue
                lc, rc = tree.add_node(), tree.add_node()
                tree.feature[nid] = best_f
                edges = self.bin_edges[best_f]
                tree.threshold[nid] = float(edges[best_b]) if best_b < len(edges) \
                    else float(np.inf)
                tree.left[nid], tree.right[nid] = lc, rc
                in_node = node_of == nid
                go_left = in_node & (Xb[:, best_f] <= best_b)
                node_of[go_left] = lc
                node_of[in_node & ~go_left] = rc
                new_frontier.extend([lc, rc])
            frontier = sorted(new_frontier)
        self.trees.append(tree)
        return margin + tree.predict(X)

    # -- top-level fit/predict ------------------------------------------ #
    def fit(self, X, y, num_boost_round=10, evals=None, allreduce=None,
            callback=None):
        X = np.asarray(X, dtype=np.float64)
        y = np.asarray(y, dtype=np.float64)
        Xb = self.make_bins(X)
        margin = np.full(len(y), self._base_margin())
        eval_margins = {name: (np.asarray(ex, dtype=np.float64),
                               np.asarray(ey, dtype=np.float64),
                               np.full(len(ey), self._base_margin()))
                        for name, (ex, ey) in (evals or {}).items()}
        history = []
        for r in range(num_boost_round):
            margin = self.boost_round(Xb, X, y, margin, allreduce=allreduce)
            row = {}
            for m, v in self.metric(y, margin).items():
                row[f"train-{m}"] = v
            for name, (ex, ey, em) in eval_margins.items():
                em += self.trees[-1].predict(ex)
                for m, v in self.metric(ey, em).items():
                    row[f"{name}-{m}"] = v
            row["training_iteration"] = r + 1
            history.append(row)
            if callback is not None and not callback(row):
                break
        return history

    def predict_margin(self, X) -> np.ndarray:
        X = np.asarray(X, dtype=np.float64)
        margin = np.full(len(X), self._base_margin())
        for t in self.trees:
            margin += t.predict(X)
        return margin

    def predict(self, X) -> np.ndarray:
        m = self.predict_margin(X)
        return _sigmoid(m) if self.objective.startswith("binary") else m

    # -- persistence ----------------------------------------------------- #
    def save_pretrained(self, path: str):
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "config.json"), "w") as f:
            json.dump({"model_type": "mi355_gbt", "objective": self.objective,
                       "eta": self.eta, "max_depth": self.max_depth,
                       "lambda": self.reg_lambda, "base_score": self.base_score},
                      f, indent=2)
        with open(os.path.join(path, "gbt_model.json"), "w") as f:
            json.dump({
                "trees": [t.to_dict() for t in self.trees],
                "bin_edges": [e.tolist() for e in (self.bin_edges or [])],
                "feature_names": self.feature_names,
            }, f)

    @classmethod
    def from_pretrained(cls, path: str) -> "GBTModel":
        with open(os.path.join(path, "config.json")) as f:
            cfg = json.load(f)
        model = cls(cfg)
        with open(os.path.join(path, "gbt_model.json")) as f:
            blob = json.load(f)
        model.trees = [_Tree.from_dict(d) for d in blob["trees"]]
        model.bin_edges = [np.asarray(e) for e in blob["bin_edges"]]
        model.feature_names = blob.get("feature_names")
        return model
