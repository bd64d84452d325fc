#!/usr/bin/env python3
"""Flash-attention kernel microbenchmark (fwd and fwd+bwd TFLOP/s) at the
T5 training shapes and larger fill shapes. Output goes to profiles/."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from mi355 import ops  # noqa: E402


def flops_attn(b, h, sq, sk, d, causal):
    f = 4.0 * b * h * sq * sk * d  # QK^T + PV
    return f / 2 if causal else f


def bench(b, h, sq, sk, d, causal, bias, iters=50):
    q = torch.randn(b, h, sq, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, sk, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, sk, d, device="cuda", dtype=torch.bfloat16)
    bs = (torch.randn(1, h, sq, sk, device="cuda") * 2).to(torch.bfloat16) if bias else None

    def fwd():
        with torch.no_grad():
            return ops.attention(q, k, v, bias=bs, causal=causal)

    for _ in range(5):
        fwd()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fwd()
    torch.cuda.synchronize()
    tf_fwd = flops_attn(b, h, sq, sk, d, causal) * iters / (time.perf_counter() - t0) / 1e12

    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    vg = v.clone().requires_grad_(True)
    bg = bs.clone().requires_grad_(True) if bias else None

    def both():
        o = ops.attention(qg, kg, vg, bias=bg, causal=causal)
        o.backward(o.detach())
        for t in (qg, kg, vg, bg):
            if t is not None:
                t.grad = None

    for _ in range(3):
        both()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        both()
    torch.cuda.synchronize()
    tf_both = 3.5 * flops_attn(b, h, sq, sk, d, causal) * iters / (time.perf_counter() - t0) / 1e12
    print(f"B{b:>3} H{h:>2} Sq{sq:>5} Sk{sk:>5} D{d} causal={int(causal)} bias={int(bias)}"
          f"  fwd {tf_fwd:7.1f} TF   fwd+bwd {tf_both:7.1f} TF")


if __name__ == "__main__":
    print("# mi355 flash attention (bf16, MFMA 16x16x32, v1 + T14)")
    bench(2, 12, 512, 512, 64, False, True)    # T5-base train encoder B=2
    bench(2, 12, 512, 512, 64, True, True)     # decoder
    bench(16, 12, 512, 512, 64, False, True)   # B=16 train
    bench(256, 12, 512, 512, 64, False, True)  # batch-infer encoder
    bench(16, 16, 2048, 2048, 64, False, False)  # fill shape
    bench(16, 16, 4096, 4096, 64, True, False)   # long causal
