"""mi355 job CLI (reference anyscale job submit +
flan-t5-batch-inference-job-setup.yml:1-7 — name / compute_config /
cluster_env / entrypoint). `python -m mi355.cli job submit job.yml` runs the
entrypoint under the local runtime with env + logging set up; `status`
tails run metrics."""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import time


def _load_yaml(path: str) -> dict:
    import yaml

    with open(path) as f:
        return yaml.safe_load(f)


def cmd_submit(args):
    spec = _load_yaml(args.spec)
    name = spec.get("name", os.path.basename(args.spec))
    entrypoint = spec["entrypoint"]
    env = dict(os.environ)
    env.update({str(k): str(v) for k, v in (spec.get("env") or {}).items()})
    env.setdefault("MI355_JOB_NAME", name)
    workdir = spec.get("working_dir", os.path.dirname(os.path.abspath(args.spec)) or ".")
    print(f"[mi355 job] name={name!r} entrypoint={entrypoint!r} cwd={workdir}")
    t0 = time.time()
    proc = subprocess.run(entrypoint, shell=True, cwd=workdir, env=env)
    dt = time.time() - t0
    status = "SUCCEEDED" if proc.returncode == 0 else f"FAILED(rc={proc.returncode})"
    print(f"[mi355 job] {name}: {status} in {dt:.1f}s")
    return proc.returncode


def cmd_status(args):
    base = os.path.expanduser(args.dir or "~/mi355_results")
    if not os.path.isdir(base):
        print(f"no results under {base}")
        return 0
    def show(run, run_dir, indent=""):
        metrics = os.path.join(run_dir, "metrics.jsonl")
        line = ""
        if os.path.exists(metrics):
            with open(metrics) as f:
                rows = f.readlines()
            if rows:
                line = rows[-1].strip()
        entries = os.listdir(run_dir)
        ckpts = [d for d in entries if d.startswith("checkpoint")]
        print(f"{indent}{run}: {len(ckpts)} checkpoint(s) {line}")
        for sub in sorted(entries):  # tune trials nest one level down
            sub_dir = os.path.join(run_dir, sub)
            if sub.startswith("trial_") and os.path.isdir(sub_dir):
                show(sub, sub_dir, indent + "  ")

    for run in sorted(os.listdir(base)):
        run_dir = os.path.join(base, run)
        if os.path.isdir(run_dir):
            show(run, run_dir)
    return 0


def cmd_flags(_args):
    from ..flags import FLAGS, snapshot

    snap = snapshot()
    for f in FLAGS:
        mark = "*" if f.name in os.environ else " "
        print(f"{mark} {f.name:<24} {snap[f.name]:<10} {f.help}")
    return 0


def cmd_dashboard(args):
    """Start the web dashboard standalone (C25; init() also starts it)."""
    from ..dashboard import start_dashboard

    host, port = start_dashboard(port=args.port)
    print(f"mi355 dashboard at http://{host}:{port} (ctrl-c to stop)")
    import time

    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        return 0


def cmd_gpus(_args):
    try:
        out = subprocess.run(["rocm-smi", "--showuse", "--showmemuse"],
                             capture_output=True, text=True, timeout=20)
        print(out.stdout or out.stderr)
    except FileNotFoundError:
        import torch

        print(f"torch.cuda.device_count() = {torch.cuda.device_count()}")
    return 0


def main(argv=None):
    ap = argparse.ArgumentParser(prog="mi355")
    sub = ap.add_subparsers(dest="cmd", required=True)
    job = sub.add_parser("job")
    jobsub = job.add_subparsers(dest="jobcmd", required=True)
    s = jobsub.add_parser("submit")
    s.add_argument("spec")
    s.set_defaults(fn=cmd_submit)
    dash = sub.add_parser("dashboard")
    dash.add_argument("--port", type=int, default=None)
    dash.set_defaults(fn=cmd_dashboard)
    st = sub.add_parser("status")
    st.add_argument("--dir", default=None)
    st.set_defaults(fn=cmd_status)
    fl = sub.add_parser("flags", help="list every MI355_* env knob")
    fl.set_defaults(fn=cmd_flags)
    g = sub.add_parser("gpus")
    g.set_defaults(fn=cmd_gpus)
    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
