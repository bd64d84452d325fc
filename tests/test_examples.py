"""The examples/ scripts (the reference's pipelines rebuilt on the drop-in
`ray` namespace) must run end-to-end in SMALL mode on CPU."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def _run(script, extra_env=None, timeout=420):
    env = dict(os.environ, MI355_SMALL="1", MI355_NUM_WORKERS="2",
               **(extra_env or {}))
    out = subprocess.run([sys.executable, str(REPO / "examples" / script)],
                         capture_output=True, text=True, timeout=timeout,
                         cwd=REPO, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    return out.stdout


@pytest.mark.timeout(480)
def test_flan_t5_example_small():
    assert "JOB OK" in _run("flan_t5_finetune_and_batch_inference.py")


@pytest.mark.timeout(480)
def test_segformer_example_small():
    assert "JOB OK" in _run("segformer_finetune_and_batch_inference.py")


@pytest.mark.timeout(480)
def test_parallel_rf_example():
    out = _run("parallel_training_ray_core.py", extra_env={"NUM_MODELS": "3"})
    assert "performance gain" in out
