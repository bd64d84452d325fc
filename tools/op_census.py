#!/usr/bin/env python3
"""Census of aten ops in one train step via TorchDispatchMode: op name +
shapes + count, plus the innermost mi355/bench source line when the op runs
under Python (forward); backward-engine ops have no Python stack but their
shapes identify them."""
import sys
import traceback
from collections import Counter

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))

import torch  # noqa: E402
from torch.utils._python_dispatch import TorchDispatchMode  # noqa: E402

import bench  # noqa: E402
from mi355.parallel import FlatAdamW, FlatReplica  # noqa: E402

WATCH = ("copy_", "add_", "add.", "fill_", "cat", "sum", "pow", "_to_copy",
         "clone", "contiguous", "mul", "div", "zero_")


class Census(TorchDispatchMode):
    def __init__(self):
        super().__init__()
        self.counts = Counter()

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if any(w in name for w in WATCH):
            shapes = tuple(tuple(a.shape) for a in args
                           if isinstance(a, torch.Tensor))[:2]
            src = "<bwd>"
            for fr in reversed(traceback.extract_stack()):
                if ("mi355" in fr.filename or "bench" in fr.filename) \
                        and "op_census" not in fr.filename:
                    src = f"{fr.filename.split('/')[-1]}:{fr.lineno}"
                    break
            self.counts[(name, str(shapes), src)] += 1
        return func(*args, **(kwargs or {}))


def main():
    class A:
        model, batch, seq = "flan-t5-base", 2, 512

    args = A()
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    model = bench.build_model(args, device, torch.bfloat16
                              if device.type == "cuda" else torch.float32)
    replica = FlatReplica(model)
    optimizer = FlatAdamW(replica, lr=2e-5, weight_decay=0.01, max_grad_norm=1.0)
    input_ids, labels = bench.synthetic_batch(args, device, seed=1000)
    for _ in range(2):
        bench.train_step(replica, optimizer, input_ids, labels)
    if device.type == "cuda":
        torch.cuda.synchronize()

    c = Census()
    with c:
        bench.train_step(replica, optimizer, input_ids, labels)
    if device.type == "cuda":
        torch.cuda.synchronize()

    out = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/op_census.txt"
    with open(out, "w") as f:
        for (name, shapes, src), n in sorted(c.counts.items(),
                                             key=lambda x: -x[1])[:80]:
            f.write(f"{n:6d}  {name:<28} {src:<28} {shapes[:110]}\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
