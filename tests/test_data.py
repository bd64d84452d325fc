"""Dataset + Preprocessor tests (C6/C7) mirroring the reference's data ops
(Introduction_to_Ray_AI_Runtime.ipynb, Scaling_model_training.ipynb)."""
import numpy as np
import pandas as pd
import pytest

import mi355.data as mdata
from mi355.data import BatchMapper, Chain, Dataset, MinMaxScaler, PowerTransformer


@pytest.fixture
def taxi_like():
    rng = np.random.default_rng(0)
    return mdata.from_pandas(pd.DataFrame({
        "trip_distance": rng.uniform(0, 30, 200),
        "fare_amount": rng.uniform(2, 100, 200),
        "passenger_count": rng.integers(1, 5, 200),
    }))


def test_from_items_count_take():
    ds = mdata.from_items([{"a": i, "b": i * 2} for i in range(10)])
    assert ds.count() == 10
    assert ds.take(3) == [{"a": 0, "b": 0}, {"a": 1, "b": 2}, {"a": 2, "b": 4}]


def test_repartition_and_split(taxi_like):
    ds = taxi_like.repartition(5)
    assert ds.num_blocks() == 5
    assert ds.count() == 200
    train, test = ds.train_test_split(test_size=0.3)
    assert train.count() == 140 and test.count() == 60


def test_limit_schema_columns(taxi_like):
    ds = taxi_like.limit(17)
    assert ds.count() == 17
    assert set(ds.columns()) == {"trip_distance", "fare_amount", "passenger_count"}
    assert ds.schema() is not None


def test_map_batches_pandas(taxi_like):
    def add_tip(df):
        df = df.copy()
        df["tip"] = df["fare_amount"] * 0.2
        return df

    out = taxi_like.map_batches(add_tip, batch_format="pandas", batch_size=32)
    assert "tip" in out.columns()
    assert out.count() == 200


def test_map_batches_numpy():
    ds = mdata.from_items([{"x": float(i)} for i in range(100)])
    out = ds.map_batches(lambda b: {"x": b["x"] * 2}, batch_format="numpy",
                         batch_size=30)
    assert out.to_pandas()["x"].tolist() == [2.0 * i for i in range(100)]


def test_groupby_mean(taxi_like):
    g = taxi_like.groupby("passenger_count").mean("fare_amount")
    df = g.to_pandas()
    assert "mean(fare_amount)" in df.columns
    ref = taxi_like.to_pandas().groupby("passenger_count")["fare_amount"].mean()
    got = dict(zip(df["passenger_count"], df["mean(fare_amount)"]))
    for k, v in ref.items():
        assert got[k] == pytest.approx(v)


def test_drop_columns_iter_batches(taxi_like):
    ds = taxi_like.drop_columns(["passenger_count"])
    assert set(ds.columns()) == {"trip_distance", "fare_amount"}
    sizes = [len(b) for b in ds.iter_batches(batch_size=64, batch_format="pandas")]
    assert sum(sizes) == 200 and max(sizes) == 64


def test_min_max_scaler(taxi_like):
    sc = MinMaxScaler(columns=["trip_distance"])
    out = sc.fit_transform(taxi_like).to_pandas()
    assert out["trip_distance"].min() == pytest.approx(0.0)
    assert out["trip_distance"].max() == pytest.approx(1.0)
    # transform_batch on a raw batch (predictor path)
    b = sc.transform_batch(pd.DataFrame({"trip_distance": [0.0], "fare_amount": [1.0],
                                         "passenger_count": [1]}))
    assert b["trip_distance"].iloc[0] <= 0.0


def test_power_transformer(taxi_like):
    pt = PowerTransformer(columns=["fare_amount"], power=0.5)
    out = pt.fit_transform(taxi_like).to_pandas()
    raw = taxi_like.to_pandas()["fare_amount"].to_numpy()
    expect = ((raw + 1) ** 0.5 - 1) / 0.5
    np.testing.assert_allclose(out["fare_amount"].to_numpy(), expect, rtol=1e-6)


def test_batch_mapper_tokenize_style():
    ds = mdata.from_items([{"instruction": f"do {i}", "input": str(i)} for i in range(8)])

    def preprocess(df):
        return pd.DataFrame({
            "input_ids": [[len(a), len(b)] for a, b in zip(df.instruction, df.input)],
        })

    bm = BatchMapper(preprocess, batch_format="pandas", batch_size=4096)
    out = bm.transform(ds)
    assert out.count() == 8 and out.columns() == ["input_ids"]


def test_chain(taxi_like):
    chain = Chain(MinMaxScaler(columns=["trip_distance"]),
                  BatchMapper(lambda df: df.assign(z=df.trip_distance + 1)))
    out = chain.fit(taxi_like).transform(taxi_like).to_pandas()
    assert out["z"].max() <= 2.0


def test_iter_torch_batches():
    import torch
    ds = mdata.from_items([{"x": [1.0, 2.0], "y": 3} for _ in range(10)])
    batches = list(ds.iter_torch_batches(batch_size=4))
    assert batches[0]["x"].shape == (4, 2)
    assert batches[0]["y"].dtype in (torch.int64,)


def test_sort_and_unique(taxi_like):
    s = taxi_like.sort("fare_amount", descending=True).take(2)
    assert s[0]["fare_amount"] >= s[1]["fare_amount"]
    assert taxi_like.unique("passenger_count") == [1, 2, 3, 4]


def test_map_batches_via_runtime_tasks():
    """Block transforms fan out as runtime tasks when mi355.init is active
    (transforms run as C2 tasks)."""
    import mi355

    mi355.init(num_workers=2, num_gpus=0)
    try:
        ds = mdata.from_items([{"x": float(i)} for i in range(64)]).repartition(4)
        out = ds.map_batches(lambda df: df.assign(y=df.x * 3),
                             batch_format="pandas", batch_size=16)
        assert out.to_pandas()["y"].tolist() == [3.0 * i for i in range(64)]
    finally:
        mi355.shutdown()


def test_map_batches_actor_pool_strategy():
    from mi355.data import ActorPoolStrategy

    class AddState:
        def __init__(self, base):
            self.base = base

        def __call__(self, df):
            return df.assign(z=df.x + self.base)

    ds = mdata.from_items([{"x": float(i)} for i in range(20)]).repartition(2)
    out = ds.map_batches(AddState, compute=ActorPoolStrategy(size=2),
                         fn_constructor_args=(100,), batch_format="pandas",
                         batch_size=10)
    assert sorted(out.to_pandas()["z"].tolist()) == [100.0 + i for i in range(20)]


def test_dataset_roundtrip_properties_hypothesis():
    """Property: any mixed-typed list-of-dicts survives
    from_items -> blocks -> to_pandas with values intact and ops composing."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    import mi355.data as mdata

    row = st.fixed_dictionaries({
        "i": st.integers(min_value=-10**6, max_value=10**6),
        "f": st.floats(allow_nan=False, allow_infinity=False, width=32),
        "s": st.text(alphabet=st.characters(codec="ascii",
                                            exclude_categories=("Cc", "Cs")),
                     max_size=12),
    })

    @settings(max_examples=25, deadline=None)
    @given(st.lists(row, min_size=1, max_size=24))
    def check(rows):
        ds = mdata.Dataset.from_items(rows)
        assert ds.count() == len(rows)
        df = ds.to_pandas()
        assert list(df["i"]) == [r["i"] for r in rows]
        assert list(df["s"]) == [r["s"] for r in rows]
        lim = ds.limit(3)
        assert lim.count() == min(3, len(rows))
        doubled = ds.map_batches(lambda d: d.assign(i=d["i"] * 2),
                                 batch_format="pandas", batch_size=8)
        assert list(doubled.to_pandas()["i"]) == [r["i"] * 2 for r in rows]

    check()


def test_filter_add_rename_select(taxi_like):
    ds = taxi_like
    flt = ds.filter(lambda row: row["passenger_count"] > 2)
    assert all(r["passenger_count"] > 2 for r in flt.take_all())
    plus = ds.add_column("double_fare", lambda df: df["fare_amount"] * 2)
    df = plus.to_pandas()
    assert (df["double_fare"] == df["fare_amount"] * 2).all()
    ren = plus.rename_columns({"double_fare": "fare2"})
    assert "fare2" in ren.columns() and "double_fare" not in ren.columns()
    sel = ren.select_columns(["fare2"])
    assert sel.columns() == ["fare2"]


def test_random_shuffle_preserves_rows(taxi_like):
    ds = taxi_like
    sh = ds.random_shuffle(seed=3)
    assert sh.count() == ds.count()
    assert sorted(r["fare_amount"] for r in sh.take_all()) == \
        sorted(r["fare_amount"] for r in ds.take_all())
    sh2 = ds.random_shuffle(seed=3)
    assert [r["fare_amount"] for r in sh.take_all()] == \
        [r["fare_amount"] for r in sh2.take_all()]  # seed-deterministic


def test_split_equal_truncates_to_even_shards():
    """split(n, equal=True) must yield shards of identical length (DDP train
    sharding: a one-row imbalance would desync all-reduce counts  — ADVICE
    r1 medium)."""
    import pyarrow as pa
    from mi355.data import Dataset

    ds = Dataset.from_arrow(pa.table({"x": list(range(10))}))
    shards = ds.split(3, equal=True)
    assert [s.count() for s in shards] == [3, 3, 3]
    # rows are a prefix partition, no duplicates
    got = sorted(v for s in shards for v in s.to_pandas()["x"].tolist())
    assert got == list(range(9))
    # equal=False keeps every row
    shards = ds.split(3)
    assert sum(s.count() for s in shards) == 10


def test_map_batches_streams_blocks_through_object_store():
    """with the runtime up, chained map_batches keeps
    blocks as object-store refs (driver never materializes the whole
    dataset); streaming consumers resolve lazily and whole-table consumers
    still work."""
    import pandas as pd

    import mi355
    from mi355.data import Dataset

    mi355.init(num_workers=2)
    ds = Dataset.from_items([{"x": i} for i in range(100)], parallelism=5)
    assert ds.num_blocks() == 5

    a = ds.map_batches(lambda df: pd.DataFrame({"x": df["x"] * 2}),
                       batch_format="pandas", batch_size=None)
    b = a.map_batches(lambda df: pd.DataFrame({"x": df["x"] + 1}),
                      batch_format="pandas", batch_size=None)
    # both stages returned REF blocks: nothing materialized in the driver
    assert not a.is_fully_materialized()
    assert not b.is_fully_materialized()
    # streaming consumers
    assert b.count() == 100
    assert b.take(3) == [{"x": 1}, {"x": 3}, {"x": 5}]
    got = []
    for batch in b.iter_batches(batch_size=16, batch_format="pandas"):
        assert len(batch) <= 16
        got.extend(batch["x"].tolist())
    assert sorted(got) == sorted(i * 2 + 1 for i in range(100))
    # whole-table consumer materializes correctly
    assert b.to_pandas()["x"].sum() == sum(i * 2 + 1 for i in range(100))


def test_map_batches_ref_blocks_rebatch_in_worker():
    """A ref block larger than batch_size is re-batched next to the data
    (worker side), not in the driver."""
    import pandas as pd

    import mi355
    from mi355.data import Dataset

    mi355.init(num_workers=2)
    ds = Dataset.from_items([{"x": i} for i in range(64)], parallelism=2)
    a = ds.map_batches(lambda df: df, batch_format="pandas", batch_size=None)
    assert not a.is_fully_materialized()
    seen_sizes = []

    def probe(df):
        seen_sizes.append(len(df))
        return df

    b = a.map_batches(probe, batch_format="pandas", batch_size=8)
    assert b.count() == 64  # probe ran worker-side; sizes list stays empty here
    assert seen_sizes == []  # the closure ran in ANOTHER process


def test_read_parquet_streams_through_object_store(tmp_path):
    """r2 ROADMAP item: with the runtime up, read_parquet reads each file
    on a WORKER and keeps the block as an object-store ref; a chained
    map_batches pipelines the refs without driver materialization."""
    import pandas as pd
    import pyarrow as pa
    import pyarrow.parquet as pq

    import mi355
    from mi355.data import Dataset

    for i in range(3):
        pq.write_table(pa.table({"x": list(range(i * 10, i * 10 + 10))}),
                       tmp_path / f"part-{i}.parquet")

    mi355.init(num_workers=2)
    ds = Dataset.read_parquet(str(tmp_path))
    assert ds.num_blocks() == 3
    assert not ds.is_fully_materialized(), "blocks should be refs"
    out = ds.map_batches(lambda df: pd.DataFrame({"x": df["x"] + 1}),
                         batch_format="pandas", batch_size=None)
    assert not out.is_fully_materialized()
    assert out.count() == 30
    assert sorted(out.to_pandas()["x"]) == list(range(1, 31))
    # eager path without the runtime still works
    mi355.shutdown()
    ds2 = Dataset.read_parquet(str(tmp_path))
    assert ds2.is_fully_materialized()
    assert ds2.count() == 30


def test_window_pipeline():
    """Dataset.window() -> DatasetPipeline: stages apply per window at
    iteration time; repeat() multiplies epochs."""
    import pandas as pd

    from mi355.data import Dataset

    ds = Dataset.from_items([{"x": i} for i in range(40)], parallelism=8)
    pipe = ds.window(blocks_per_window=2)
    assert pipe.num_windows() == 4
    out = pipe.map_batches(lambda df: pd.DataFrame({"x": df["x"] * 10}),
                           batch_format="pandas", batch_size=None)
    assert out.count() == 40
    assert sorted(r["x"] for r in out.iter_rows()) == [i * 10 for i in range(40)]
    got = []
    for b in out.iter_batches(batch_size=7, batch_format="pandas"):
        assert len(b) <= 7
        got.extend(b["x"].tolist())
    assert sorted(got) == [i * 10 for i in range(40)]
    assert out.take(3) == [{"x": 0}, {"x": 10}, {"x": 20}]
    rep = out.repeat(3)
    assert rep.num_windows() == 12 and rep.count() == 120


def test_window_pipeline_prefetch_under_runtime():
    """Prefetched windows still yield every batch exactly once and in
    window order, with the stages running as remote tasks."""
    import pandas as pd

    import mi355
    from mi355.data import Dataset

    mi355.init(num_workers=2)
    ds = Dataset.from_items([{"x": i} for i in range(60)], parallelism=6)
    pipe = ds.window(blocks_per_window=2).map_batches(
        lambda df: pd.DataFrame({"x": df["x"] + 100}),
        batch_format="pandas", batch_size=None)
    got = []
    for b in pipe.iter_batches(prefetch_windows=2, batch_size=10,
                               batch_format="pandas"):
        got.extend(b["x"].tolist())
    assert got == [i + 100 for i in range(60)]
