"""The MI355_* env-knob registry (mi355/flags.py) must stay
in sync with what the code actually reads."""
import re
import subprocess
from pathlib import Path

from mi355.flags import FLAGS, snapshot

REPO = Path(__file__).resolve().parent.parent


def test_every_used_env_var_is_registered():
    out = subprocess.run(
        ["grep", "-rhoE", r"MI355_[A-Z0-9_]+", "mi355", "bench.py", "examples",
         "tools", "--include=*.py", "--include=*.hip", "--include=*.cpp"],
        capture_output=True, text=True, cwd=REPO).stdout.split()
    used = set(out)
    registered = {f.name for f in FLAGS}
    missing = used - registered
    assert not missing, f"env vars read by code but not in flags.py: {missing}"


def test_snapshot_shape():
    snap = snapshot()
    assert len(snap) == len(FLAGS)
    assert all(isinstance(v, str) for v in snap.values())
