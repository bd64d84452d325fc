#!/usr/bin/env python3
"""AdamW kernel bandwidth microbench (28 B/param round trip)."""
import sys
import time

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import torch  # noqa: E402

from mi355 import _C  # noqa: E402


def main():
    n = 124_000_000
    dev = "cuda"
    master = torch.randn(n, device=dev, dtype=torch.float32)
    grad = torch.randn(n, device=dev, dtype=torch.bfloat16)
    m = torch.zeros(n, device=dev, dtype=torch.float32)
    v = torch.zeros(n, device=dev, dtype=torch.float32)
    lowp = torch.empty(n, device=dev, dtype=torch.bfloat16)
    gs = torch.ones((), device=dev, dtype=torch.float32)
    for _ in range(3):
        _C.adamw_multi([master], [grad], [m], [v], [lowp],
                       2e-5, 0.9, 0.999, 1e-8, 0.01, 5, gs)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 30
    for _ in range(iters):
        _C.adamw_multi([master], [grad], [m], [v], [lowp],
                       2e-5, 0.9, 0.999, 1e-8, 0.01, 5, gs)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"adamw: {dt*1e6:.1f} us for {n/1e6:.0f}M params "
          f"-> {n*28/dt/1e12:.2f} TB/s")


if __name__ == "__main__":
    main()
