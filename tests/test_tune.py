"""Tuner + ASHA tests (C13) — ASHA determinism via a scripted trainable
(SURVEY.md §4 implication list) and a small end-to-end sweep."""
import pytest

from mi355 import tune
from mi355.air import RunConfig
from mi355.train import session
from mi355.tune import ASHAScheduler, TuneConfig, Tuner


def test_search_space_sampling():
    import random

    rng = random.Random(0)
    space = {"lr": tune.choice([1, 2, 3]), "wd": tune.uniform(0.0, 1.0),
             "n": tune.randint(5, 10)}
    from mi355.tune.search import sample_space

    for _ in range(20):
        cfg = sample_space(space, rng)
        assert cfg["lr"] in (1, 2, 3)
        assert 0.0 <= cfg["wd"] <= 1.0
        assert 5 <= cfg["n"] < 10


def test_asha_stops_bad_trials_deterministically():
    # scripted metrics: 4 trials report loss = base - 0.1*t at rungs
    sched = ASHAScheduler(time_attr="epoch", metric="loss", mode="min",
                          max_t=16, grace_period=1, reduction_factor=4)
    decisions = {}
    # trial scores at epoch 1 (rung): 1.0, 2.0, 3.0, 4.0 -> only top 1/4 survives
    for i, score in enumerate([1.0, 2.0, 3.0, 4.0]):
        decisions[i] = sched.on_result(f"t{i}", {"epoch": 1, "loss": score})
    assert decisions[0] == "continue"        # best survives
    assert decisions[2] == "stop" and decisions[3] == "stop"
    # max_t enforced
    assert sched.on_result("t0", {"epoch": 16, "loss": 0.1}) == "stop"


@pytest.mark.timeout(600)
def test_tuner_end_to_end(tmp_path):
    def trainable(config):
        # quadratic bowl: trials with lr closer to 0.5 get lower final loss
        loss = (config["lr"] - 0.5) ** 2
        for epoch in range(3):
            if not session.report({"epoch": epoch + 1, "loss": loss + 1.0 / (epoch + 1)}):
                return

    tuner = Tuner(
        trainable,
        param_space={"lr": tune.grid_search([0.1, 0.5, 0.9])},
        tune_config=TuneConfig(metric="loss", mode="min", num_samples=1,
                               max_concurrent_trials=2),
        run_config=RunConfig(name="tunetest", local_dir=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 3
    assert not grid.errors
    best = grid.get_best_result()
    assert best.config["lr"] == 0.5
    df = grid.get_dataframe()
    assert "loss" in df.columns and len(df) == 3


@pytest.mark.timeout(600)
def test_tuner_asha_early_stop(tmp_path):
    def trainable(config):
        for epoch in range(8):
            ok = session.report({"epoch": epoch + 1, "loss": config["base"]})
            if not ok:
                session.report({"epoch": -1, "loss": config["base"],
                                "stopped_at": epoch + 1})
                return

    tuner = Tuner(
        trainable,
        param_space={"base": tune.grid_search([0.1, 0.2, 0.3, 0.4])},
        tune_config=TuneConfig(
            metric="loss", mode="min", num_samples=1, max_concurrent_trials=1,
            scheduler=ASHAScheduler(max_t=8, grace_period=1, reduction_factor=4),
        ),
        run_config=RunConfig(name="asha", local_dir=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 4 and not grid.errors
    assert grid.get_best_result().config["base"] == pytest.approx(0.1)
    # at least one of the worse trials was stopped before epoch 8
    stopped = [r for r in grid if r.metrics.get("epoch") != 8]
    assert stopped, "ASHA should early-stop the worst trials"


@pytest.mark.timeout(600)
def test_tuner_trial_error_isolated(tmp_path):
    def trainable(config):
        if config["x"] == 2:
            raise ValueError("bad trial")
        session.report({"epoch": 1, "loss": config["x"]})

    grid = Tuner(
        trainable, param_space={"x": tune.grid_search([1, 2, 3])},
        tune_config=TuneConfig(metric="loss", mode="min", max_concurrent_trials=1),
        run_config=RunConfig(name="errs", local_dir=str(tmp_path)),
    ).fit()
    assert len(grid.errors) == 1
    assert grid.get_best_result().config["x"] == 1
