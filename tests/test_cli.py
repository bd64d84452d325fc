"""Job CLI tests (C24)."""
import json
import os
import subprocess
import sys

import pytest
import yaml


def test_job_submit_runs_entrypoint(tmp_path):
    out = tmp_path / "ok.txt"
    spec = {"name": "testjob",
            "entrypoint": f"{sys.executable} -c \"open('{out}','w').write('ran')\"",
            "compute_config": "local", "cluster_env": "default"}
    specfile = tmp_path / "job.yml"
    specfile.write_text(yaml.safe_dump(spec))
    r = subprocess.run([sys.executable, "-m", "mi355.cli", "job", "submit",
                        str(specfile)], capture_output=True, text=True,
                       cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr
    assert out.read_text() == "ran"
    assert "SUCCEEDED" in r.stdout


def test_job_submit_propagates_failure(tmp_path):
    spec = {"name": "failjob", "entrypoint": f"{sys.executable} -c 'raise SystemExit(3)'"}
    specfile = tmp_path / "job.yml"
    specfile.write_text(yaml.safe_dump(spec))
    r = subprocess.run([sys.executable, "-m", "mi355.cli", "job", "submit",
                        str(specfile)], capture_output=True, text=True,
                       cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 3
    assert "FAILED" in r.stdout


def test_status_lists_runs_and_nested_trials(tmp_path, capsys):
    import json

    from mi355.cli import cmd_status

    run = tmp_path / "myrun"
    (run / "checkpoint_000001").mkdir(parents=True)
    (run / "metrics.jsonl").write_text(json.dumps({"epoch": 1, "loss": 0.5}) + "\n")
    trial = tmp_path / "tune" / "trial_0000"
    (trial / "checkpoint_000001").mkdir(parents=True)
    (trial / "metrics.jsonl").write_text(json.dumps({"eval_loss": 0.1}) + "\n")

    class A:
        dir = str(tmp_path)

    assert cmd_status(A()) == 0
    out = capsys.readouterr().out
    assert "myrun: 1 checkpoint(s)" in out
    assert "trial_0000: 1 checkpoint(s)" in out and "eval_loss" in out


@pytest.mark.timeout(600)
@pytest.mark.parametrize("spec", ["examples/flan-t5-job.yml",
                                  "examples/segformer-job.yml"])
def test_committed_job_specs_run(spec, tmp_path, monkeypatch):
    """The job ymls shipped in examples/ must submit and succeed (SMALL/CPU)."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MI355_SMALL="1", MI355_NUM_WORKERS="2")
    out = subprocess.run([sys.executable, "-m", "mi355.cli", "job", "submit",
                          os.path.join(repo, spec)],
                         capture_output=True, text=True, timeout=540,
                         cwd=repo, env=env)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    assert "SUCCEEDED" in out.stdout
