"""Job CLI tests (C24)."""
import json
import os
import subprocess
import sys

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
