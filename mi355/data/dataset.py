"""Arrow-block distributed Dataset.

Covers every Dataset operation the reference exercises
(Introduction_to_Ray_AI_Runtime.ipynb:cell-9..18/58, Scaling_model_training.
ipynb:cell-33-36, Model_finetuning_and_batch_inference.ipynb:cell-18/21/69,
flan-t5-batch-inference.py:32-38): from_items / from_pandas / from_numpy /
from_huggingface / read_parquet / read_csv, limit, take, show, count, schema,
repartition, train_test_split, map_batches, groupby().mean()/sum()/count(),
drop_columns, add_column, random_shuffle, split, to_pandas, iter_batches.

Blocks are pyarrow Tables. Transforms run eagerly; with the mi355 runtime
initialised, ``map_batches`` fans blocks out as remote tasks (parallelism =
block count), or over a GPU actor pool when ``compute=ActorPoolStrategy``
(the BatchPredictor path).
"""
from __future__ import annotations

import builtins
from dataclasses import dataclass

import numpy as np
import pandas as pd
import pyarrow as pa
import pyarrow.parquet as pq


@dataclass
class ActorPoolStrategy:
    """compute= strategy for map_batches: a pool of stateful callable-class
    actors, size in [min_size, max_size]."""
    size: int | None = None
    min_size: int = 1
    max_size: int | None = None


def _to_table(batch) -> pa.Table:
    if isinstance(batch, pa.Table):
        return batch
    if isinstance(batch, pd.DataFrame):
        # multi-dimensional ndarray cells (images, pixel_values, …) become
        # nested-list Arrow columns
        fix = {c: [np.asarray(v).tolist() for v in batch[c]]
               for c in batch.columns
               if len(batch) and isinstance(batch[c].iloc[0], np.ndarray)
               and batch[c].iloc[0].ndim >= 2}
        if fix:
            batch = batch.copy()
            for c, v in fix.items():
                batch[c] = v
        return pa.Table.from_pandas(batch, preserve_index=False)
    if isinstance(batch, dict):
        cols = {}
        for k, v in batch.items():
            v = np.asarray(v)
            if v.ndim > 1:
                cols[k] = pa.array(list(v))
            else:
                cols[k] = pa.array(v)
        return pa.table(cols)
    raise TypeError(f"cannot convert {type(batch)} to an Arrow block")


def _from_table(tbl: pa.Table, fmt: str):
    if fmt in ("pandas", "default"):
        return tbl.to_pandas()
    if fmt == "numpy":
        return {name: np.asarray(col.to_pylist())
                if pa.types.is_list(col.type) or pa.types.is_large_list(col.type)
                else col.to_numpy(zero_copy_only=False)
                for name, col in zip(tbl.column_names, tbl.columns)}
    if fmt in ("pyarrow", "arrow"):
        return tbl
    raise ValueError(f"unknown batch_format {fmt!r}")


class Dataset:
    """Arrow-block dataset.  Blocks are either materialized
    ``pa.Table``s or runtime ``ObjectRef``s pointing into the shared-memory
    object store (map_batches through the task runtime
    keeps results as refs, so chained transforms PIPELINE block-by-block
    through shm and the driver's peak memory is one block, not the dataset).
    Row-streaming consumers (count/schema/take/iter_batches/map_batches)
    resolve refs lazily one block at a time; whole-table consumers
    (to_arrow/to_pandas/sort/groupby/split) materialize, which the DDP
    trainer does anyway when sharding."""

    def __init__(self, blocks: list):
        self._blocks = [b for b in blocks]

    # -- ref-block plumbing --------------------------------------------- #
    @staticmethod
    def _is_ref(b) -> bool:
        return not isinstance(b, pa.Table)

    def _resolve(self, b) -> pa.Table:
        if isinstance(b, pa.Table):
            return b
        from .. import runtime

        return runtime.get(b)

    def _tables(self):
        """Yield blocks as tables, resolving object-store refs lazily —
        bounded driver memory (one block at a time)."""
        for b in self._blocks:
            yield self._resolve(b)

    def materialize(self) -> "Dataset":
        return Dataset(list(self._tables()))

    def window(self, *, blocks_per_window: int = 10) -> "DatasetPipeline":
        k = max(1, int(blocks_per_window))
        wins = [Dataset(self._blocks[i:i + k])
                for i in range(0, max(1, len(self._blocks)), k)]
        return DatasetPipeline(wins)

    def is_fully_materialized(self) -> bool:
        return all(isinstance(b, pa.Table) for b in self._blocks)

    # ------------------------------------------------------------------ #
    # constructors (module-level wrappers below)                         #
    # ------------------------------------------------------------------ #
    @staticmethod
    def from_items(items, parallelism: int = 8) -> "Dataset":
        items = list(items)
        if items and isinstance(items[0], dict):
            df = pd.DataFrame(items)
        else:
            df = pd.DataFrame({"item": items})
        return Dataset.from_pandas(df).repartition(min(parallelism, max(1, len(items))))

    @staticmethod
    def from_pandas(df) -> "Dataset":
        if isinstance(df, list):
            return Dataset([pa.Table.from_pandas(d, preserve_index=False) for d in df])
        return Dataset([pa.Table.from_pandas(df, preserve_index=False)])

    @staticmethod
    def from_numpy(arr) -> "Dataset":
        return Dataset([_to_table({"data": arr})])

    @staticmethod
    def from_arrow(tbl) -> "Dataset":
        return Dataset([tbl] if isinstance(tbl, pa.Table) else list(tbl))

    @staticmethod
    def from_huggingface(ds) -> "Dataset":
        """Accepts a `datasets.Dataset` (arrow-backed) or a DatasetDict."""
        try:
            import datasets as hfds

            if isinstance(ds, hfds.DatasetDict):
                return {k: Dataset.from_huggingface(v) for k, v in ds.items()}
            if isinstance(ds, hfds.Dataset):
                return Dataset([ds.data.table.combine_chunks()])
        except ImportError:
            pass
        raise TypeError(f"from_huggingface: unsupported {type(ds)}")

    @staticmethod
    def read_parquet(paths, parallelism: int = 8) -> "Dataset":
        import glob as _glob
        import os

        if isinstance(paths, str):
            paths = [paths]
        files = []
        for p in paths:
            if os.path.isdir(p):
                files.extend(sorted(_glob.glob(os.path.join(p, "*.parquet"))))
            else:
                files.append(p)
        from .. import runtime

        if runtime.is_initialized() and files:
            # STREAMING read (r2 ROADMAP item): one task per file, read on a
            # worker, block stays in the object store as a ref — the driver
            # never holds the file bytes; downstream map_batches chains
            # pipeline the refs straight through.
            task = runtime.remote(_read_parquet_file)
            return Dataset([task.remote(f) for f in files])
        return Dataset([pq.read_table(f) for f in files])

    @staticmethod
    def read_csv(paths) -> "Dataset":
        if isinstance(paths, str):
            paths = [paths]
        from .. import runtime

        if runtime.is_initialized() and paths:
            task = runtime.remote(_read_csv_file)
            return Dataset([task.remote(p) for p in paths])
        return Dataset([_read_csv_file(p) for p in paths])

    # ------------------------------------------------------------------ #
    # inspection                                                          #
    # ------------------------------------------------------------------ #
    def count(self) -> int:
        return sum(b.num_rows for b in self._tables())

    def __len__(self):
        return self.count()

    def schema(self):
        return self._resolve(self._blocks[0]).schema if self._blocks else None

    def columns(self):
        return list(self._resolve(self._blocks[0]).column_names) \
            if self._blocks else []

    def num_blocks(self) -> int:
        return len(self._blocks)

    def take(self, n: int = 20):
        out = []
        for b in self._tables():
            for row in b.slice(0, builtins.min(n - len(out), b.num_rows)).to_pylist():
                out.append(row)
            if len(out) >= n:
                break
        return out

    def take_all(self):
        return self.take(self.count())

    def show(self, n: int = 20):
        for row in self.take(n):
            print(row)

    def to_pandas(self, limit: int | None = None) -> pd.DataFrame:
        if not self._blocks:
            return pd.DataFrame()
        df = pa.concat_tables(list(self._tables()),
                              promote_options="default").to_pandas()
        return df.head(limit) if limit is not None else df

    def to_arrow(self) -> pa.Table:
        return pa.concat_tables(list(self._tables()), promote_options="default")

    def stats(self) -> str:
        return (f"Dataset(num_blocks={self.num_blocks()}, num_rows={self.count()}, "
                f"schema={self.schema()})")

    def __repr__(self):
        return self.stats()

    # ------------------------------------------------------------------ #
    # transforms                                                          #
    # ------------------------------------------------------------------ #
    def limit(self, n: int) -> "Dataset":
        out = []
        left = n
        for b in self._tables():
            if left <= 0:
                break
            take = builtins.min(left, b.num_rows)
            out.append(b.slice(0, take))
            left -= take
        return Dataset(out)

    def repartition(self, num_blocks: int) -> "Dataset":
        tbl = self.to_arrow()
        n = tbl.num_rows
        if n == 0:
            return Dataset([tbl])
        sizes = [n // num_blocks + (1 if i < n % num_blocks else 0)
                 for i in range(num_blocks)]
        out, off = [], 0
        for s in sizes:
            if s > 0:
                out.append(tbl.slice(off, s))
                off += s
        return Dataset(out)

    def split(self, n: int, equal: bool = False) -> list["Dataset"]:
        """Split into n datasets. With equal=True rows are truncated to
        n*floor(rows/n) first so every shard has exactly the same length —
        required for DDP train shards, where a one-row imbalance would make
        one rank run an extra batch and post all-reduces its peers never
        match (ray.data.Dataset.split(equal=True) semantics)."""
        if equal:
            tbl = self.to_arrow()
            per = tbl.num_rows // n
            return [Dataset([tbl.slice(i * per, per)]) for i in range(n)]
        parts = self.repartition(n)
        return [Dataset([b]) for b in parts._blocks]

    def train_test_split(self, test_size: float = 0.25, shuffle: bool = False,
                         seed: int | None = None):
        tbl = self.to_arrow()
        n = tbl.num_rows
        idx = np.arange(n)
        if shuffle:
            rng = np.random.default_rng(seed)
            rng.shuffle(idx)
            tbl = tbl.take(pa.array(idx))
        n_test = int(n * test_size)
        return Dataset([tbl.slice(0, n - n_test)]), Dataset([tbl.slice(n - n_test)])

    def random_shuffle(self, seed: int | None = None) -> "Dataset":
        tbl = self.to_arrow()
        rng = np.random.default_rng(seed)
        idx = rng.permutation(tbl.num_rows)
        return Dataset([tbl.take(pa.array(idx))]).repartition(max(1, self.num_blocks()))

    def sort(self, key: str, descending: bool = False) -> "Dataset":
        tbl = self.to_arrow()
        order = "descending" if descending else "ascending"
        return Dataset([tbl.sort_by([(key, order)])])

    def drop_columns(self, cols) -> "Dataset":
        return Dataset([b.drop_columns(cols) for b in self._tables()])

    def select_columns(self, cols) -> "Dataset":
        return Dataset([b.select(cols) for b in self._tables()])

    def add_column(self, name: str, fn) -> "Dataset":
        out = []
        for b in self._tables():
            df = b.to_pandas()
            df[name] = fn(df)
            out.append(pa.Table.from_pandas(df, preserve_index=False))
        return Dataset(out)

    def rename_columns(self, mapping: dict) -> "Dataset":
        new_names = [mapping.get(c, c) for c in self.columns()]
        return Dataset([b.rename_columns(new_names) for b in self._tables()])

    def filter(self, fn) -> "Dataset":
        out = []
        for b in self._tables():
            df = b.to_pandas()
            out.append(pa.Table.from_pandas(df[df.apply(fn, axis=1)],
                                            preserve_index=False))
        return Dataset(out)

    def map(self, fn) -> "Dataset":
        def batch_fn(df: pd.DataFrame):
            return pd.DataFrame([fn(row) for row in df.to_dict("records")])

        return self.map_batches(batch_fn, batch_format="pandas")

    # -- map_batches: the workhorse (C6/C7/C14) ------------------------- #
    def map_batches(self, fn, *, batch_size: int | None = 4096,
                    batch_format: str = "default", compute=None,
                    fn_constructor_args=(), fn_constructor_kwargs=None,
                    num_gpus: float = 0, zero_copy_batch: bool = False,
                    **_ignored) -> "Dataset":
        from .. import runtime

        fn_kwargs = fn_constructor_kwargs or {}

        if isinstance(compute, ActorPoolStrategy):
            batches = list(self._iter_block_batches(batch_size))
            return self._map_actor_pool(fn, batches, batch_format, compute,
                                        fn_constructor_args, fn_kwargs, num_gpus)

        if runtime.is_initialized() and not isinstance(fn, type) \
                and (self.num_blocks() > 1
                     or any(self._is_ref(b) for b in self._blocks)):
            # STREAMING task path: one task per block.
            # Ref blocks are handed to the worker AS refs (it fetches from
            # shm directly — the driver never touches the bytes) and results
            # stay in the object store as refs, so a chain of map_batches
            # pipelines block-by-block without driver materialization.
            import cloudpickle

            fnb = cloudpickle.dumps(fn)
            task = runtime.remote(_apply_batch_fn)
            refs = []
            for blk in self._blocks:
                if self._is_ref(blk):
                    refs.append(task.remote(fnb, blk, batch_format, batch_size))
                else:
                    for tbl in _slice_batches(blk, batch_size):
                        refs.append(task.remote(fnb, tbl, batch_format, None))
            return Dataset(refs)

        if isinstance(fn, type):
            fn = fn(*fn_constructor_args, **fn_kwargs)
        out = [_to_table(fn(_from_table(tbl, batch_format)))
               for tbl in self._iter_block_batches(batch_size)]
        return Dataset(out)

    def _map_actor_pool(self, fn_cls, batches, batch_format, strategy,
                        ctor_args, ctor_kwargs, num_gpus):
        from .. import runtime
        from ..runtime import ActorPool

        if not runtime.is_initialized():
            runtime.init()
        size = strategy.size or strategy.max_size or strategy.min_size or 1

        class _BatchActor:
            def __init__(self, cls_b, args, kwargs):
                import cloudpickle as cp

                self.fn = cp.loads(cls_b)(*args, **kwargs)

            def apply(self, tbl, fmt):
                return _to_table(self.fn(_from_table(tbl, fmt)))

        import cloudpickle as cp

        actor_cls = runtime.remote(_BatchActor) if num_gpus == 0 else \
            runtime.remote(_BatchActor).options(num_gpus=num_gpus)
        actors = [actor_cls.remote(cp.dumps(fn_cls), list(ctor_args), dict(ctor_kwargs))
                  for _ in range(size)]
        pool = ActorPool(actors)
        out = pool.map(lambda a, tbl: a.apply.remote(tbl, batch_format), batches)
        for a in actors:
            a._kill()
        return Dataset(out)

    def _iter_block_batches(self, batch_size):
        for b in self._tables():
            yield from _slice_batches(b, batch_size)

    def iter_batches(self, *, batch_size: int = 256, batch_format: str = "default",
                     drop_last: bool = False):
        carry = None
        for b in self._tables():
            tbl = b if carry is None else pa.concat_tables([carry, b])
            carry = None
            off = 0
            while tbl.num_rows - off >= batch_size:
                yield _from_table(tbl.slice(off, batch_size), batch_format)
                off += batch_size
            if off < tbl.num_rows:
                carry = tbl.slice(off)
        if carry is not None and carry.num_rows and not drop_last:
            yield _from_table(carry, batch_format)

    def iter_torch_batches(self, *, batch_size: int = 8, dtypes=None, device=None,
                           drop_last: bool = False):
        import torch

        for batch in self.iter_batches(batch_size=batch_size, batch_format="numpy",
                                       drop_last=drop_last):
            out = {}
            for k, v in batch.items():
                t = torch.as_tensor(np.asarray(v))
                if t.dtype == torch.float64:
                    t = t.float()  # arrow nested-list columns decode as f64
                if dtypes and k in dtypes:
                    t = t.to(dtypes[k])
                if device is not None:
                    t = t.to(device)
                out[k] = t
            yield out

    # ------------------------------------------------------------------ #
    # groupby                                                             #
    # ------------------------------------------------------------------ #
    def groupby(self, key: str) -> "GroupedData":
        return GroupedData(self, key)

    def mean(self, cols=None):
        df = self.to_pandas()
        cols = cols or [c for c in df.columns if np.issubdtype(df[c].dtype, np.number)]
        if isinstance(cols, str):
            cols = [cols]
        return {f"mean({c})": float(df[c].mean()) for c in cols}

    def sum(self, cols=None):
        df = self.to_pandas()
        cols = cols or [c for c in df.columns if np.issubdtype(df[c].dtype, np.number)]
        if isinstance(cols, str):
            cols = [cols]
        return {f"sum({c})": float(df[c].sum()) for c in cols}

    def min(self, col):
        return float(self.to_pandas()[col].min())

    def max(self, col):
        return float(self.to_pandas()[col].max())

    def unique(self, col):
        return sorted(self.to_pandas()[col].unique().tolist())

    def zip(self, other: "Dataset") -> "Dataset":
        a, b = self.to_pandas(), other.to_pandas()
        return Dataset.from_pandas(pd.concat([a.reset_index(drop=True),
                                              b.reset_index(drop=True)], axis=1))

    def union(self, *others: "Dataset") -> "Dataset":
        blocks = list(self._blocks)
        for o in others:
            blocks.extend(o._blocks)
        return Dataset(blocks)


class DatasetPipeline:
    """Windowed streaming execution (Ray's ``Dataset.window`` /
    DatasetPipeline): transforms are recorded as stages and applied
    window-by-window at iteration time, so peak memory is one window's
    blocks (which are themselves object-store refs under the runtime).
    The reference notebooks do not use pipelines; this completes the C6
    dataset surface."""

    def __init__(self, windows: list["Dataset"], stages=None, epochs: int = 1):
        self._windows = windows
        self._stages = list(stages or [])
        self._epochs = epochs

    def num_windows(self) -> int:
        return len(self._windows) * self._epochs

    def map_batches(self, fn, **kwargs) -> "DatasetPipeline":
        return DatasetPipeline(self._windows,
                               self._stages + [("map_batches", fn, kwargs)],
                               self._epochs)

    def repeat(self, times: int = 1) -> "DatasetPipeline":
        return DatasetPipeline(self._windows, self._stages,
                               self._epochs * max(1, times))

    def _transformed(self):
        for _ in range(self._epochs):
            for w in self._windows:
                ds = w
                for kind, fn, kwargs in self._stages:
                    ds = getattr(ds, kind)(fn, **kwargs)
                yield ds

    def iter_batches(self, prefetch_windows: int = 1, **kwargs):
        """Iterate batches window-by-window.  With ``prefetch_windows`` > 0
        a background thread runs the NEXT window's stages (remote-task
        submission under the runtime) while the current window is being
        consumed — pipeline-parallel windows."""
        if prefetch_windows <= 0:
            for ds in self._transformed():
                yield from ds.iter_batches(**kwargs)
            return
        import queue as _q
        import threading

        q: "_q.Queue" = _q.Queue(maxsize=max(1, prefetch_windows))
        sentinel = object()

        def producer():
            try:
                for ds in self._transformed():
                    q.put(ds)
            finally:
                q.put(sentinel)

        threading.Thread(target=producer, daemon=True).start()
        while True:
            ds = q.get()
            if ds is sentinel:
                return
            yield from ds.iter_batches(**kwargs)

    def iter_rows(self):
        for ds in self._transformed():
            for tbl in ds._tables():
                yield from tbl.to_pylist()

    def count(self) -> int:
        return sum(ds.count() for ds in self._transformed())

    def take(self, n: int = 20):
        out = []
        for row in self.iter_rows():
            out.append(row)
            if len(out) >= n:
                break
        return out

    def __repr__(self):
        return (f"DatasetPipeline(windows={len(self._windows)}, "
                f"stages={len(self._stages)}, epochs={self._epochs})")


class GroupedData:
    def __init__(self, ds: Dataset, key: str):
        self._ds = ds
        self._key = key

    def _agg(self, how: str, cols=None) -> Dataset:
        df = self._ds.to_pandas()
        num = df.select_dtypes("number").columns.tolist()
        if self._key in num:
            num.remove(self._key)
        cols = [cols] if isinstance(cols, str) else (cols or num)
        g = df.groupby(self._key)[cols].agg(how).reset_index()
        g.columns = [self._key] + [f"{how}({c})" for c in cols]
        return Dataset.from_pandas(g)

    def mean(self, cols=None) -> Dataset:
        return self._agg("mean", cols)

    def sum(self, cols=None) -> Dataset:
        return self._agg("sum", cols)

    def count(self) -> Dataset:
        df = self._ds.to_pandas()
        g = df.groupby(self._key).size().reset_index(name="count()")
        return Dataset.from_pandas(g)

    def max(self, cols=None) -> Dataset:
        return self._agg("max", cols)

    def min(self, cols=None) -> Dataset:
        return self._agg("min", cols)


def _slice_batches(b: pa.Table, batch_size):
    if batch_size is None or b.num_rows <= batch_size:
        if b.num_rows:
            yield b
        return
    for off in range(0, b.num_rows, batch_size):
        yield b.slice(off, builtins.min(batch_size, b.num_rows - off))


def _apply_batch_fn(fn_bytes, tbl, batch_format, batch_size=None):
    """Worker-side map_batches kernel.  When the driver hands over a whole
    (possibly ref-resolved) block with a batch_size, the block is
    re-batched HERE and the outputs concatenated — the sub-batching happens
    next to the data instead of in the driver."""
    import cloudpickle

    fn = cloudpickle.loads(fn_bytes)
    if batch_size is not None and tbl.num_rows > batch_size:
        parts = [_to_table(fn(_from_table(t, batch_format)))
                 for t in _slice_batches(tbl, batch_size)]
        return pa.concat_tables(parts, promote_options="default")
    return _to_table(fn(_from_table(tbl, batch_format)))


def _read_parquet_file(path):
    """Worker-side file read for the streaming read_parquet path."""
    return pq.read_table(path)


def _read_csv_file(path):
    from pyarrow import csv as pacsv

    return pacsv.read_csv(path)
