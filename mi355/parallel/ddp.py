"""Flat-buffer data-parallel replica with bucketed RCCL all-reduce.

MI355X-native replacement for torch DDP + NCCL (reference trains via HF
Trainer wrapped in DistributedDataParallel — SURVEY.md C8/M2/M3):

* Every parameter's storage is re-pointed into ONE contiguous low-precision
  flat buffer per (decay / no-decay) group, ordered by REVERSE registration
  order, which approximates gradient-ready order during backward.
* ``p.grad`` is pre-assigned as a view into a matching flat gradient buffer,
  so autograd accumulates straight into the communication buffer (no
  bucket-copy pass).
* Buckets are contiguous SLICES of the flat gradient buffer. As the last
  parameter of a bucket receives its gradient, the bucket slice is
  all-reduced asynchronously on the RCCL (xGMI) stream, overlapping the rest
  of backward. Bucket size is tunable (``MI355_BUCKET_MB``, default 64 MB):
  xGMI is 7 point-to-point links x ~153 GB/s per GPU, so fewer/larger flushes
  beat NVSwitch-tuned 25 MB buckets.
* The flat layout makes the optimizer step TWO fused kernel launches
  (see parallel/optim.py) and gradient-norm clipping two vectorized norms.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist

from .dist import get_world_size, is_distributed


def _no_decay(name: str, p: torch.Tensor) -> bool:
    # HF Trainer convention: biases + layer norms take no weight decay.
    return p.ndim <= 1 or name.endswith(".bias") or "layer_norm" in name


class _FlatGroup:
    def __init__(self, named_params, dtype, device, bucket_bytes,
                 assign_grads: bool = True):
        self.params = [p for _, p in named_params]
        self.names = [n for n, _ in named_params]
        numels = [p.numel() for p in self.params]
        total = sum(numels)
        self.flat_param = torch.zeros(total, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(total, dtype=dtype, device=device)
        offset = 0
        self.offsets = []
        for p in self.params:
            n = p.numel()
            self.flat_param[offset:offset + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[offset:offset + n].view(p.shape)
            if assign_grads:
                p.grad = self.flat_grad[offset:offset + n].view(p.shape)
            self.offsets.append(offset)
            offset += n
        # bucket boundaries: contiguous runs of params up to bucket_bytes
        self.buckets = []  # (start_elem, end_elem, last_param_index)
        start = 0
        cur_bytes = 0
        esize = self.flat_param.element_size()
        for i, n in enumerate(numels):
            cur_bytes += n * esize
            if cur_bytes >= bucket_bytes or i == len(numels) - 1:
                end = self.offsets[i] + n
                self.buckets.append([start, end, i])
                start = end
                cur_bytes = 0
        # param index -> bucket index (a param lies entirely in one bucket)
        self.param_bucket = {}
        self.bucket_nparams = [0] * len(self.buckets)
        for i, off in enumerate(self.offsets):
            for bi, (bs, be, _) in enumerate(self.buckets):
                if bs <= off < be:
                    self.param_bucket[i] = bi
                    self.bucket_nparams[bi] += 1
                    break
        self.bucket_ready = [0] * len(self.buckets)


class FlatReplica:
    """deferred_grads=False (default): p.grad is pre-pointed into the flat
    buffer, autograd accumulates in place, post-accumulate hooks flush
    buckets async (overlapped eager DDP).

    deferred_grads=True (the graphed bench path): p.grad stays None so the
    engine STORES each fresh grad (no accumulate add_ per param, no big
    zero-fill per step — ~0.9 ms/step at B=2); collect_grads() then batch-
    copies them into the flat buffers with _foreach_copy_, after which the
    explicit all_reduce_grads()/finish_grad_sync()/optimizer path runs as
    usual. Collective order is the deterministic bucket order on every rank.
    """

    def __init__(self, model: torch.nn.Module, bucket_mb: float | None = None,
                 process_group=None, deferred_grads: bool = False):
        self.model = model
        self.pg = process_group
        self.deferred = deferred_grads
        if bucket_mb is None:
            bucket_mb = float(os.environ.get("MI355_BUCKET_MB", "64"))
        bucket_bytes = int(bucket_mb * 1024 * 1024)
        device = next(model.parameters()).device
        dtype = next(model.parameters()).dtype

        # partition by (decay, dtype): a model can keep some params in fp32
        # (e.g. SegFormer's BatchNorm under a bf16 cast) — those get their own
        # flat group instead of being silently re-pointed into a bf16 buffer
        seen: dict[int, str] = {}
        parts: dict[tuple, list] = {}
        for name, p in model.named_parameters():
            if not p.requires_grad or id(p) in seen:
                continue
            seen[id(p)] = name
            key = ("no_decay" if _no_decay(name, p) else "decay", p.dtype)
            parts.setdefault(key, []).append((name, p))
        self.groups = []
        for decay_name in ("decay", "no_decay"):
            for (dn, dt), named in parts.items():
                if dn != decay_name:
                    continue
                named.reverse()
                self.groups.append((decay_name,
                                    _FlatGroup(named, dt, device, bucket_bytes,
                                               assign_grads=not deferred_grads)))

        self._works = []
        self._flushed = set()
        # collective-order log for the divergence assertion (VERDICT r1
        # item 2): every rank must issue the same collectives in the same
        # order or RCCL deadlocks/corrupts; checked on the first step (and
        # every step under MI355_CHECK_COLLECTIVES=1)
        self._order_log: list = []
        self._order_checked = False
        self._hooks = []
        if is_distributed():
            if not self.deferred:
                self._install_hooks()
            self.broadcast_params()

    # ------------------------------------------------------------------ #
    def _install_hooks(self):
        for gi, (_, g) in enumerate(self.groups):
            for pi, p in enumerate(g.params):
                h = p.register_post_accumulate_grad_hook(self._make_hook(gi, pi))
                self._hooks.append(h)

    def _make_hook(self, gi, pi):
        def hook(_param):
            # hipGraph capture of the backward (two-graph hybrid, bench.py):
            # collectives must NOT be recorded into the graph — the replayed
            # backward is followed by an explicit all_reduce_grads() instead
            if torch.cuda.is_available() and torch.cuda.is_current_stream_capturing():
                return
            _, g = self.groups[gi]
            bi = g.param_bucket[pi]
            g.bucket_ready[bi] += 1
            # flush only when EVERY param of the bucket has accumulated its
            # grad (ready order is not guaranteed to match layout order)
            if g.bucket_ready[bi] == g.bucket_nparams[bi]:
                self._ready_bucket(gi, bi)
        return hook

    def _ready_bucket(self, gi, bi):
        _, g = self.groups[gi]
        start, end, _ = g.buckets[bi]
        view = g.flat_grad[start:end]
        work = dist.all_reduce(view, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)
        self._works.append(work)
        self._flushed.add((gi, bi))
        self._order_log.append(("bucket", gi, bi, int(end - start)))

    def collect_grads(self):
        """Deferred mode: batch-copy the engine-stored per-param grads into
        the flat buffers (one _foreach_copy_ per group instead of ~190
        accumulate add_ kernels). No-op when grads live in the flat buffer."""
        if not self.deferred:
            return
        for _, g in self.groups:
            slices, grads = [], []
            for p, off in zip(g.params, g.offsets):
                if p.grad is None:
                    # Param unused this step: zero its flat slice, else the
                    # PREVIOUS step's gradient would be all-reduced and
                    # applied again.
                    g.flat_grad[off:off + p.numel()].zero_()
                    continue
                slices.append(g.flat_grad[off:off + p.numel()].view(p.shape))
                grads.append(p.grad)
            if grads:
                torch._foreach_copy_(slices, grads)

    def all_reduce_grads(self):
        """Explicit all-reduce of the flat grad buffers. Used by the
        two-graph hybrid step (backward replayed from a hipGraph, so the
        post-accumulate hooks never fire): g1.replay() -> all_reduce_grads()
        -> finish_grad_sync() -> g2.replay(). The flat buffers are static
        addresses, so this is safe across graph replays.

        Unlike the hook path there is no backward left to overlap with, so
        the default is ONE whole-buffer collective per group (RCCL pipelines
        a large ring message better than 8 sequential 64 MB calls);
        MI355_EXPLICIT_FLUSH=buckets restores bucketed flushes."""
        if not is_distributed():
            return
        whole = os.environ.get("MI355_EXPLICIT_FLUSH", "whole") != "buckets"
        for gi, (_, g) in enumerate(self.groups):
            if whole:
                if not any((gi, bi) in self._flushed
                           for bi in range(len(g.buckets))):
                    work = dist.all_reduce(g.flat_grad, op=dist.ReduceOp.SUM,
                                           group=self.pg, async_op=True)
                    self._works.append(work)
                    self._flushed.update((gi, bi)
                                         for bi in range(len(g.buckets)))
                    self._order_log.append(("whole", gi, -1, g.flat_grad.numel()))
                continue
            for bi, (bs, be, _last) in enumerate(g.buckets):
                if (gi, bi) in self._flushed:
                    continue
                work = dist.all_reduce(g.flat_grad[bs:be], op=dist.ReduceOp.SUM,
                                       group=self.pg, async_op=True)
                self._works.append(work)
                self._flushed.add((gi, bi))
                self._order_log.append(("bucket", gi, bi, int(be - bs)))

    # ------------------------------------------------------------------ #
    def _check_collective_order(self):
        """Assert every rank issued the SAME collectives in the SAME order
        this step (a divergence means an RCCL hang or silent corruption on
        hardware).  Runs on the first synced step, and on every step under
        MI355_CHECK_COLLECTIVES=1.  Uses all_gather_object (control-plane,
        works on gloo and nccl/RCCL alike)."""
        world = get_world_size()
        if world <= 1 or not self._order_log:
            return
        every = os.environ.get("MI355_CHECK_COLLECTIVES", "0") == "1"
        if self._order_checked and not every:
            return
        self._order_checked = True
        gathered: list = [None] * world
        dist.all_gather_object(gathered, self._order_log, group=self.pg)
        if any(g != gathered[0] for g in gathered[1:]):
            detail = "\n".join(f"rank{r}: {g[:6]}... ({len(g)} colls)"
                               for r, g in enumerate(gathered))
            raise RuntimeError(
                "collective-order divergence across ranks — every rank must "
                f"flush the same buckets in the same order:\n{detail}")

    def finish_grad_sync(self):
        """Wait for in-flight bucket all-reduces and average."""
        world = get_world_size()
        if is_distributed():
            self._check_collective_order()
            for w in self._works:
                w.wait()
            # Safety net: reduce any bucket that never flushed (e.g. a
            # parameter was unused this step so its hook never fired).
            for gi, (_, g) in enumerate(self.groups):
                for bi, (bs, be, _last) in enumerate(g.buckets):
                    if (gi, bi) not in self._flushed:
                        dist.all_reduce(g.flat_grad[bs:be], op=dist.ReduceOp.SUM,
                                        group=self.pg)
            self._works.clear()
            self._flushed.clear()
            self._order_log = []
        if world > 1:
            for _, g in self.groups:
                g.flat_grad.div_(world)

    def zero_grads(self):
        dev = self.groups[0][1].flat_grad.device if self.groups else None
        if dev is not None and dev.type == "cuda":
            from .. import ops

            ops.bump_dropout_seed(dev)  # new attention-dropout masks per step
        for _, g in self.groups:
            if self.deferred:
                # fresh store each step: the engine writes new grad tensors,
                # collect_grads overwrites the flat buffer fully — no zero-fill
                for p in g.params:
                    p.grad = None
            else:
                g.flat_grad.zero_()
            g.bucket_ready = [0] * len(g.buckets)
        self._works.clear()
        self._flushed.clear()

    def grad_norm(self) -> torch.Tensor:
        from .. import ops

        if self.groups and self.groups[0][1].flat_grad.is_cuda \
                and ops.hip_available():
            dev = self.groups[0][1].flat_grad.device
            out = torch.zeros((), dtype=torch.float32, device=dev)
            for _, g in self.groups:
                ops.sumsq_accumulate(g.flat_grad, out)
            return out.sqrt()
        sq = None
        for _, g in self.groups:
            s = g.flat_grad.float().pow(2).sum()
            sq = s if sq is None else sq + s
        return sq.sqrt()

    def broadcast_params(self):
        if not is_distributed():
            return
        for _, g in self.groups:
            dist.broadcast(g.flat_param, src=0, group=self.pg)

    # passthroughs ------------------------------------------------------ #
    def __call__(self, *a, **kw):
        return self.model(*a, **kw)

    def __getattr__(self, item):
        return getattr(self.model, item)
