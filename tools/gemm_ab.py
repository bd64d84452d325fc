#!/usr/bin/env python3
"""Per-shape A/B: hand-written MFMA GEMM (csrc/kernels/gemm.hip) vs
hipBLASLt (torch.matmul), on the T5 hot shapes (SURVEY §2.3 K1/K3/K6 and
their backward transposes), at the B=2 and B=16 train M plus decode M=256.

Writes a markdown table to stdout; redirect into profiles/r02_gemm_ab.md and
copy the winning shapes into mi355/ops/gemm_table._WINS.

Usage (GPU box): python tools/gemm_ab.py [--iters 50] [--dtype bf16]
"""
from __future__ import annotations

import argparse
import json
import sys

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from mi355 import ops  # noqa: E402

# T5-base hot GEMM set: (tag, op, rows, cols, kc)
# fwd NT: y[M,N] = x[M,K] @ W[N,K]^T
# dX  NN: dx[M,K] = dy[M,N] @ W[N,K->K,N... stored [N,K]] — contraction N
# dW  TN: dw[N,K] = dy[M,N]^T @ x[M,K] — contraction M
SHAPES = {
    # flan-t5-base: d_model 768, inner 768, d_ff 2048
    "base": [("qkv", 2304, 768), ("attn_o", 768, 768), ("cross_kv", 1536, 768),
             ("ff_wi", 4096, 768), ("ff_wo", 768, 2048), ("lm_head", 32128, 768)],
    # flan-t5-large: d_model 1024, inner 1024 (16 heads x 64), d_ff 2816
    "large": [("qkv", 3072, 1024), ("attn_o", 1024, 1024),
              ("cross_kv", 2048, 1024), ("ff_wi", 5632, 1024),
              ("ff_wo", 1024, 2816), ("lm_head", 32128, 1024)],
}


def shape_set(M: int, model: str = "base") -> list[tuple[str, str, int, int, int]]:
    out = []
    for name, N, K in SHAPES[model]:
        out.append((f"{name}.fwd", "nt", M, N, K))
        out.append((f"{name}.dx", "nn", M, K, N))
        out.append((f"{name}.dw", "tn", N, K, M))
    return out


def run_op(op: str, rows: int, cols: int, kc: int, dtype, hip: bool):
    if op == "nt":
        a = torch.randn(rows, kc, device="cuda", dtype=dtype)
        b = torch.randn(cols, kc, device="cuda", dtype=dtype)
        f = (lambda: ops.gemm_nt(a, b)) if hip else (lambda: a @ b.t())
    elif op == "nn":
        a = torch.randn(rows, kc, device="cuda", dtype=dtype)
        b = torch.randn(kc, cols, device="cuda", dtype=dtype)
        f = (lambda: ops.gemm_nn(a, b)) if hip else (lambda: a @ b)
    else:
        a = torch.randn(kc, rows, device="cuda", dtype=dtype)
        b = torch.randn(kc, cols, device="cuda", dtype=dtype)
        f = (lambda: ops.gemm_tn(a, b)) if hip else (lambda: a.t() @ b)
    return f


def time_fn(f, iters: int, graph: bool = True) -> float:
    """Median-of-3 timing.  graph=True (default) captures `iters` calls in a
    hipGraph and times the replay — the representative regime: the train/infer
    steps run graph-captured, where launch overhead (hipBLASLt's ~18us eager
    floor) vanishes and only kernel time + graph-internal boundaries remain.
    The first eager A/B (profiles/, r02 ab1-ab3) mismeasured exactly this."""
    for _ in range(5):
        f()
    torch.cuda.synchronize()
    if graph:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            for _ in range(iters):
                f()
        run = g.replay
    else:
        def run():
            for _ in range(iters):
                f()
    best = []
    for _ in range(3):
        torch.cuda.synchronize()
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        run()
        e.record()
        torch.cuda.synchronize()
        best.append(s.elapsed_time(e) / iters * 1e3)  # us
    return sorted(best)[1]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp16"])
    ap.add_argument("--ms", type=int, nargs="*", default=[1024, 8192, 256])
    ap.add_argument("--model", default="base", choices=["base", "large"])
    ap.add_argument("--json", default=None)
    args = ap.parse_args()
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float16

    rows_out = []
    print(f"# MFMA GEMM vs hipBLASLt A/B ({args.dtype}, {args.iters} iters, "
          f"{torch.cuda.get_device_name(0)})\n")
    print("| shape | op | M(rows) | N(cols) | K | hip us | blaslt us | "
          "hip TF | blaslt TF | winner |")
    print("|---|---|---|---|---|---|---|---|---|---|")
    for M in args.ms:
        for tag, op, rows, cols, kc in shape_set(M, args.model):
            flop = 2.0 * rows * cols * kc
            t_hip = time_fn(run_op(op, rows, cols, kc, dtype, True), args.iters)
            t_lib = time_fn(run_op(op, rows, cols, kc, dtype, False), args.iters)
            win = "HIP" if t_hip <= t_lib else "blaslt"
            print(f"| {tag}@M={M} | {op} | {rows} | {cols} | {kc} "
                  f"| {t_hip:.1f} | {t_lib:.1f} "
                  f"| {flop/t_hip/1e6:.0f} | {flop/t_lib/1e6:.0f} | {win} |")
            sys.stdout.flush()
            rows_out.append(dict(tag=tag, op=op, rows=rows, cols=cols, kc=kc,
                                 hip_us=t_hip, lib_us=t_lib, winner=win))
    wins = [(r["op"], r["rows"], r["cols"], r["kc"])
            for r in rows_out if r["winner"] == "HIP"]
    print("\n## _WINS entries (paste into mi355/ops/gemm_table.py)\n")
    for w in sorted(set(wins)):
        print(f"    {w!r},")
    if args.json:
        with open(args.json, "w") as f:
            json.dump(rows_out, f, indent=1)


if __name__ == "__main__":
    main()
