"""Data-parallel trainer orchestration (call stack §3.1).

``DataParallelTrainer.fit()`` spawns ``scaling_config.num_workers`` worker
processes (one per GPU when ``use_gpu``), sets the torch.distributed
env rendezvous (RANK/WORLD_SIZE/MASTER_ADDR — M1), shards the datasets,
runs the user train loop under the session API, streams per-epoch reports
back, applies CheckpointConfig top-k retention, and returns a Result.
Gradient all-reduce inside the loop runs on RCCL over xGMI ("nccl" backend)
via parallel.FlatReplica; the CPU test path uses gloo exactly like the
reference's ``use_gpu=False`` mode (Scaling_model_training.ipynb:cell-50).
"""
from __future__ import annotations

import os
import shutil
import socket
import time
import traceback
import multiprocessing as mp

import cloudpickle

from ..air import (Checkpoint, CheckpointConfig, FailureConfig, Result,
                   RunConfig, ScalingConfig)
from . import session as _session_mod


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class _PrefixedOut:
    """Worker stdout/stderr tagged with the rank (SURVEY.md C25 — the
    reference surfaces worker prints at the driver via Ray log redirection)."""

    def __init__(self, inner, tag):
        self.inner, self.tag, self._nl = inner, tag, True

    def write(self, text):
        for chunk in text.splitlines(keepends=True):
            if self._nl:
                self.inner.write(self.tag)
            self.inner.write(chunk)
            self._nl = chunk.endswith("\n")

    def flush(self):
        self.inner.flush()


def _worker_entry(rank, world, port, use_gpu, loop_bytes, loop_config,
                  shard_tables, report_q, control_q, resume_dir, trial_name):
    try:
        import sys

        tag = f"({trial_name + ' ' if trial_name else ''}rank={rank}) "
        sys.stdout = _PrefixedOut(sys.stdout, tag)
        sys.stderr = _PrefixedOut(sys.stderr, tag)
        os.environ.update({
            "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        })
        if use_gpu:
            import torch

            n = torch.cuda.device_count()
            os.environ.setdefault("HIP_VISIBLE_DEVICES", str(rank % max(n, 1)))
        from ..data import Dataset
        from ..parallel import init_distributed

        if world > 1:
            print(f"rendezvous: world={world} port={port} "
                  f"visible_gpus={os.environ.get('HIP_VISIBLE_DEVICES', '<all>')}",
                  flush=True)
            init_distributed()
        shards = {name: Dataset(tbls) for name, tbls in (shard_tables or {}).items()}
        ctx = _session_mod.TrainContext(world_rank=rank, world_size=world,
                                        local_rank=rank, trial_name=trial_name)
        sess = _session_mod._Session(
            context=ctx, report_q=report_q, control_q=control_q,
            dataset_shards=shards,
            resume_checkpoint=Checkpoint.from_directory(resume_dir) if resume_dir else None,
        )
        _session_mod._set_session(sess)
        loop = cloudpickle.loads(loop_bytes)
        if loop_config is not None:
            loop(loop_config)
        else:
            loop()
        report_q.put(("done", rank, None, None))
    except Exception:
        report_q.put(("error", rank, traceback.format_exc(), None))
    finally:
        try:
            import torch.distributed as dist

            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass


class DataParallelTrainer:
    def __init__(self, train_loop_per_worker=None, *, train_loop_config=None,
                 scaling_config: ScalingConfig | None = None, datasets=None,
                 run_config: RunConfig | None = None, preprocessor=None,
                 resume_from_checkpoint=None, **_extra):
        self._train_loop = train_loop_per_worker
        self._train_loop_config = train_loop_config
        self.scaling_config = scaling_config or ScalingConfig()
        self.datasets = datasets or {}
        self.run_config = run_config or RunConfig()
        self.preprocessor = preprocessor
        self.resume_from_checkpoint = resume_from_checkpoint
        self._trial_name = ""

    # hook for subclasses (HuggingFaceTrainer wraps the user init fn)
    def _loop_fn(self):
        return self._train_loop

    def _prepare_datasets(self):
        """Fit/apply the preprocessor, then shard each dataset (M7)."""
        datasets = dict(self.datasets)
        if self.preprocessor is not None:
            for name, ds in datasets.items():
                if name == "train" and not getattr(self.preprocessor, "_is_fitted", False):
                    ds = self.preprocessor.fit_transform(ds)
                else:
                    ds = self.preprocessor.transform(ds)
                datasets[name] = ds
        n = getattr(self, "_effective_world", None) or self.scaling_config.num_workers
        shards_per_rank = []
        for rank in range(n):
            shard_tables = {}
            for name, ds in datasets.items():
                if name == "train" and n > 1:
                    shard = ds.split(n, equal=True)[rank]
                else:
                    shard = ds  # eval replicated (HF-style full eval per worker)
                # resolve object-store ref blocks in the driver: trainer
                # worker processes are not runtime workers, and per-rank
                # shards are materialized for epoch iteration anyway
                shard_tables[name] = list(shard._tables())
            shards_per_rank.append(shard_tables)
        return shards_per_rank

    def fit(self) -> Result:
        failure = self.run_config.failure_config or FailureConfig()
        attempts = failure.max_failures + 1
        last = None
        resume_dir = None
        if self.resume_from_checkpoint is not None:
            resume_dir = self.resume_from_checkpoint if isinstance(
                self.resume_from_checkpoint, str) else self.resume_from_checkpoint.path
        for attempt in range(attempts):
            last = self._fit_once(resume_dir)
            if last.error is None:
                return last
            if last.checkpoint is not None:
                resume_dir = last.checkpoint.path  # retry from best so far
        return last

    def _fit_once(self, resume_dir=None) -> Result:
        world = max(self.scaling_config.num_workers, 1)
        use_gpu = self.scaling_config.use_gpu
        if use_gpu:
            import torch

            n_gpus = torch.cuda.device_count()
            if n_gpus == 0:
                raise RuntimeError(
                    "ScalingConfig(use_gpu=True) but no GPU is visible")
            if world > n_gpus:
                # RCCL enforces one rank per device (a 2-GPU-worker job on a
                # 1-GPU box would hang at rendezvous); clamp like a
                # resource-limited scheduler rather than waiting forever on
                # resources that cannot appear on this node
                print(f"[mi355 trainer] requested {world} GPU workers but "
                      f"only {n_gpus} GPU(s) are visible — running "
                      f"{n_gpus}-worker data parallel")
                world = n_gpus
        self._effective_world = world
        run_dir = self.run_config.run_dir()
        os.makedirs(run_dir, exist_ok=True)

        shards_per_rank = self._prepare_datasets()
        ctx = mp.get_context("spawn")
        report_q = ctx.Queue()
        control_qs = [ctx.Queue() for _ in range(world)]
        port = _free_port()
        loop_bytes = cloudpickle.dumps(self._loop_fn())

        procs = []
        for rank in range(world):
            p = ctx.Process(
                target=_worker_entry,
                args=(rank, world, port, use_gpu, loop_bytes,
                      self._train_loop_config, shards_per_rank[rank], report_q,
                      control_qs[rank], resume_dir, self._trial_name),
                daemon=True,
            )
            p.start()
            procs.append(p)

        ckpt_cfg = self.run_config.checkpoint_config or CheckpointConfig()
        tracked: list[tuple[float, Checkpoint, dict]] = []
        metrics_rows = []
        last_metrics: dict = {}
        error = None
        done = set()
        n_ckpt = 0
        # synchronous report protocol: the i-th report of every rank gets the
        # decision computed at rank 0's i-th report (lockstep early stop)
        report_seq = {r: 0 for r in range(world)}
        decisions = []
        waiting = {}
        import queue as _q

        last_activity = time.monotonic()
        while len(done) < world:
            try:
                kind, rank, payload, ckpt_dir = report_q.get(timeout=1.0)
                last_activity = time.monotonic()
            except _q.Empty:
                # liveness check: a worker that died without reporting (GPU
                # fault, OOM-kill, os._exit) is surfaced as an error
                for r, p in enumerate(procs):
                    if r not in done and not p.is_alive():
                        error = error or RuntimeError(
                            f"worker {r} died unexpectedly (exitcode {p.exitcode})")
                        done.add(r)
                        for q in control_qs:
                            q.put("stop")
                if time.monotonic() - last_activity > 600:
                    error = error or RuntimeError("training workers timed out")
                    break
                continue
            except Exception:
                error = RuntimeError("training workers timed out")
                break
            if kind == "done":
                done.add(rank)
            elif kind == "error":
                error = RuntimeError(f"worker {rank} failed:\n{payload}")
                done.add(rank)
                for q in control_qs:
                    q.put("stop")
            elif kind == "report":
                seq = report_seq[rank]
                report_seq[rank] += 1
                if rank == 0:
                    last_metrics = payload
                    metrics_rows.append(payload)
                    # observability: JSONL metrics per run dir
                    # + Prometheus gauges (mi355.metrics)
                    import json as _json

                    with open(os.path.join(run_dir, "metrics.jsonl"), "a") as mf:
                        mf.write(_json.dumps(payload, default=str) + "\n")
                    from .. import metrics as _prom

                    _prom.record_train_report(self.run_config.name or "train", payload)
                    _prom.TRAIN_WORKERS.labels(
                        run=self.run_config.name or "train").set(
                        sum(p.is_alive() for p in procs))
                    if ckpt_dir:
                        n_ckpt += 1
                        kept = os.path.join(run_dir, f"checkpoint_{n_ckpt:06d}")
                        if os.path.abspath(ckpt_dir) != os.path.abspath(kept):
                            shutil.copytree(ckpt_dir, kept, dirs_exist_ok=True)
                        ckpt = Checkpoint.from_directory(kept)
                        if self.preprocessor is not None:
                            ckpt.set_preprocessor(self.preprocessor)
                        score = payload.get(ckpt_cfg.checkpoint_score_attribute,
                                            float(n_ckpt))
                        sign = -1.0 if ckpt_cfg.checkpoint_score_order == "min" else 1.0
                        tracked.append((sign * float(score), ckpt, dict(payload)))
                        tracked.sort(key=lambda t: -t[0])
                        keep = ckpt_cfg.num_to_keep
                        if keep is not None and len(tracked) > keep:
                            for _, old, _ in tracked[keep:]:
                                shutil.rmtree(old.path, ignore_errors=True)
                            tracked = tracked[:keep]
                    decisions.append(self._decide(payload))
                    for w in waiting.pop(len(decisions) - 1, []):
                        control_qs[w].put(decisions[-1])
                if seq < len(decisions):
                    control_qs[rank].put(decisions[seq])
                else:
                    waiting.setdefault(seq, []).append(rank)

        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()

        import pandas as pd

        best_ckpt = tracked[0][1] if tracked else None
        return Result(
            metrics=last_metrics,
            checkpoint=best_ckpt,
            error=error,
            path=run_dir,
            best_checkpoints=[(c, m) for _, c, m in tracked],
            metrics_dataframe=pd.DataFrame(metrics_rows) if metrics_rows else None,
            config=dict(self._train_loop_config or {}),
        )

    def _decide(self, metrics: dict) -> str:
        """Tuner/ASHA hook: continue/stop decision for this report epoch."""
        cb = getattr(self, "_report_callback", None)
        if cb is not None:
            try:
                if cb(metrics) == "stop":
                    return "stop"
            except Exception:
                pass
        return "continue"


class TorchTrainer(DataParallelTrainer):
    """API-compatible with the reference's TorchTrainer shape (BASELINE.json:
    'TorchTrainer-compatible DDP loop')."""
