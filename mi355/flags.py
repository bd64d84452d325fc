"""Every environment knob the framework reads, in one place (SURVEY.md §5.6:
"env-var overrides documented in one flags module").

All knobs are read at their point of use (import order never matters); this
module is the documentation and a programmatic registry, e.g. for `mi355.cli`
or tests that want to enumerate them.
"""
from __future__ import annotations

import os
from dataclasses import dataclass


@dataclass(frozen=True)
class Flag:
    name: str
    default: str
    help: str


FLAGS = [
    # -- runtime ------------------------------------------------------------
    Flag("MI355_SESSION", "<random>",
         "Session id prefixing shared-memory object-store segment names; set "
         "by the driver before spawning workers so they attach to the same "
         "namespace."),
    Flag("MI355_VISIBLE_DEVICES", "<all>",
         "Restrict which GPUs the runtime schedules onto (comma-separated "
         "indices, like HIP_VISIBLE_DEVICES but only for the task runtime)."),
    # -- distributed training ------------------------------------------------
    Flag("MI355_BUCKET_MB", "64",
         "FlatReplica gradient all-reduce bucket size in MiB. xGMI is 7 "
         "point-to-point links x ~153 GB/s, so fewer/larger flushes beat "
         "NVSwitch-tuned 25 MB buckets."),
    Flag("MI355_EXPLICIT_FLUSH", "whole",
         "Explicit (non-hook) gradient flush granularity: whole = one "
         "collective per flat group, buckets = MI355_BUCKET_MB slices."),
    Flag("MI355_DIST_GRAPHS", "1",
         "bench.py multi-GPU path: 1 = two-graph hybrid step (hipGraph "
         "fwd+bwd and optimizer around an eager RCCL all-reduce), 0 = fully "
         "eager, full = capture the collectives inside one graph "
         "(experimental)."),
    Flag("MI355_HYBRID_GRAPH", "0",
         "Force the two-graph hybrid step at world_size 1 (validation knob)."),
    # -- kernels -------------------------------------------------------------
    Flag("MI355_CHECK_COLLECTIVES", "0",
         "1 = assert identical cross-rank collective order every step "
         "(default: first synced step only)."),
    Flag("MI355_DASHBOARD", "1",
         "Serve the web dashboard from init() (0 disables)."),
    Flag("MI355_DASHBOARD_PORT", "8265",
         "Dashboard port (falls back to an ephemeral port when taken)."),
    Flag("MI355_RESULTS_DIR", "~/mi355_results",
         "Run/trial results tree shown by `mi355 status` and the dashboard."),
    Flag("MI355_SHARE_GPU", "0",
         "Map world>1 ranks onto one GPU for smoke tests (RCCL rejects the "
         "collectives — one rank per device — but init/device plumbing is "
         "exercised; see profiles/r02_summary.md)."),
    Flag("MI355_SPILL_DIR", "<tmpdir>/mi355_<session>_spill",
         "Directory for object-store spill files."),
    Flag("MI355_SPILL_THRESHOLD_MB", "<unset>",
         "Objects larger than this spill straight to disk; unset = only on "
         "a full /dev/shm."),
    Flag("MI355_SHM_BUDGET_MB", "<unset>",
         "Per-store owned-shm footprint cap: beyond it (or on a full "
         "/dev/shm) the least-recently-read sealed segments are moved to "
         "the spill dir and unlinked; readers fall through to the spill "
         "file."),
    Flag("MI355_GEMM", "auto",
         "Projection-GEMM dispatch: auto = per-shape table from the "
         "graph-captured A/B (profiles/r02_gemm_ab.md), hip = always the "
         "hand-written MFMA kernels, blaslt = always the library."),
    Flag("MI355_GEMM_TILE", "<auto>",
         "Force the NT MFMA GEMM tile (128 or 256); auto picks 256 (8-phase "
         "counted-vmcnt kernel) when the 256-tile grid still fills the CUs."),
    Flag("MI355_BENCH_INFER", "1",
         "bench.py train mode also measures batch-256 fp16 generate and "
         "reports infer_seq_per_s in the same JSON line (GPU only)."),
    Flag("MI355_GEMM_2S", "0",
         "1 = two-stream linear backward (dW on a side stream). Off by "
         "default: measured slower at B=2 (event overhead)."),
    Flag("MI355_FORCE_REF", "0",
         "1 = route every op through the plain-PyTorch reference "
         "implementation instead of the HIP kernels (numerics baselines)."),
    Flag("MI355_ATTN_V2", "1",
         "0 = use the v1 attention forward everywhere instead of the "
         "swapped-operand in-register-softmax v2 (D=64 only)."),
    Flag("MI355_ATTN_BWD_REF", "0",
         "1 = force the GEMM-composed attention backward (recompute via "
         "hipBLASLt) instead of the hand-written kernels."),
    Flag("MI355_GEN_GRAPH", "1",
         "0 = disable the hipGraph-captured greedy-decode loop in "
         "T5.generate (falls back to the eager decode path)."),
    Flag("MI355_ATTN_BWD", "<auto>",
         "Attention backward implementation: v1 | v2 | v3. Default picks v3 "
         "when the grid fills the chip (B*H*ceil(S/128) >= 256) else v1."),
    Flag("MI355_ATTN_BWD_NQC", "<auto>",
         "q-split factor for the v1 attention backward (blocks multiply). "
         "Default auto-fills ~384 blocks."),
    Flag("MI355_ATTN_FWD_NKC", "<auto>",
         "k-split factor for the v2 attention forward (flash-decoding style "
         "partial merge). Default auto-fills ~384 blocks."),
    # -- misc ---------------------------------------------------------------
    Flag("MI355_SMALL", "<auto>",
         "examples/: tiny model + CPU (defaults to 1 when no GPU present)."),
    Flag("MI355_NUM_WORKERS", "2",
         "examples/: trainer worker count (the reference uses 2)."),
    Flag("MI355_SANITIZE", "<unset>",
         "Build the native store/tokenizer with -fsanitize=<value> "
         "(address|thread) — §5.2 sanitizer CI target."),
    Flag("MI355_JOB_NAME", "<entrypoint>",
         "mi355 job submit: override the run name recorded in job status."),
]


def snapshot() -> dict:
    """Current values of all registered flags (unset -> declared default)."""
    return {f.name: os.environ.get(f.name, f.default) for f in FLAGS}
