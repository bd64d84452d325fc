#!/usr/bin/env python3
"""SegFormer-b0 benchmark at the reference config (512x512, batch 16
inference / batch 8 train — BASELINE.md SegFormer rows)."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from mi355.models.segformer import (SegformerConfig,
                                    SegformerForSemanticSegmentation,
                                    SegformerImageProcessor)


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--train-only", action="store_true")
    args = ap.parse_args()
    torch.manual_seed(0)
    m = SegformerForSemanticSegmentation(SegformerConfig()).to("cuda", torch.bfloat16)
    # inference: 10 batches x 16 images (Scaling_batch_inference cell-45/56)
    if args.train_only:
        _train(m := m.train(), args)
        return
    m.eval()
    pix = torch.randn(16, 3, 512, 512, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        for _ in range(5):
            m(pixel_values=pix)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            out = m(pixel_values=pix)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
    print(f"segformer-b0 inference: 10x16 images 512x512 in {dt*1000:.1f} ms "
          f"= {160/dt:.1f} img/s")
    proc = SegformerImageProcessor()
    t0 = time.perf_counter()
    segs = proc.post_process_semantic_segmentation(out, target_sizes=[(512, 512)] * 16)
    torch.cuda.synchronize()
    print(f"postprocess (fused upsample+argmax): {(time.perf_counter()-t0)*1000:.1f} ms")

    _train(m.train(), args)


def _train(m, args):
    pix8 = torch.randn(8, 3, 512, 512, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, 150, (8, 512, 512), device="cuda")
    opt = torch.optim.AdamW(m.parameters(), lr=1e-4)
    def step():
        out = m(pixel_values=pix8, labels=labels)
        opt.zero_grad(); out.loss.backward(); opt.step()
        return out.loss
    t0 = time.perf_counter()
    step()
    torch.cuda.synchronize()
    print(f"first train step (incl. any MIOpen find): "
          f"{time.perf_counter()-t0:.2f} s")
    for _ in range(5):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    n = args.steps
    print(f"segformer-b0 train: batch 8 @512x512, {dt/n*1000:.1f} ms/step = "
          f"{8*n/dt:.1f} img/s, loss={loss.item():.3f}")


if __name__ == "__main__":
    main()
