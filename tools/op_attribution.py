#!/usr/bin/env python3
"""Attribute GPU kernel time of the bench train step to Python source lines
(torch.profiler with stacks). Finds where the torch-native elementwise/copy/
cat "soup" kernels come from so they can be fused or eliminated."""
import sys

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))

import torch  # noqa: E402
from torch.profiler import ProfilerActivity, profile  # noqa: E402

import bench  # noqa: E402
from mi355.parallel import FlatAdamW, FlatReplica  # noqa: E402


def main():
    class A:
        model, batch, seq = "flan-t5-base", 2, 512

    args = A()
    device = torch.device("cuda", 0)
    model = bench.build_model(args, device, torch.bfloat16)
    replica = FlatReplica(model)
    optimizer = FlatAdamW(replica, lr=2e-5, weight_decay=0.01, max_grad_norm=1.0)
    input_ids, labels = bench.synthetic_batch(args, device, seed=1000)

    for _ in range(5):
        bench.train_step(replica, optimizer, input_ids, labels)
    torch.cuda.synchronize()

    with profile(activities=[ProfilerActivity.CUDA, ProfilerActivity.CPU],
                 with_stack=True) as prof:
        for _ in range(3):
            bench.train_step(replica, optimizer, input_ids, labels)
        torch.cuda.synchronize()

    out = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/op_attribution.txt"
    with open(out, "w") as f:
        f.write("== by op, self CUDA ==\n")
        f.write(prof.key_averages().table(
            sort_by="self_cuda_time_total", row_limit=40, max_name_column_width=70))
    stacks = out.replace(".txt", "_stacks.txt")
    prof.export_stacks(stacks, "self_cuda_time_total")
    # condense: keep only soup ops, aggregate by deepest mi355/bench frame
    agg = {}
    with open(stacks) as f:
        for line in f:
            parts = line.rsplit(" ", 1)
            if len(parts) != 2:
                continue
            frames, val = parts[0].split(";"), int(parts[1])
            op = frames[-1]
            if not any(k in op for k in ("copy_", "add_", "fill_", "cat",
                                         "sum", "pow", "to ", "aten::mul",
                                         "aten::div", "contiguous")):
                continue
            src = next((fr for fr in reversed(frames)
                        if ".py(" in fr and ("mi355" in fr or "bench" in fr)),
                       "?")
            key = (op.strip(), src.strip())
            agg[key] = agg.get(key, 0) + val
    with open(out, "a") as f:
        f.write("\n\n== soup ops by source (us) ==\n")
        for (op, src), v in sorted(agg.items(), key=lambda x: -x[1])[:40]:
            f.write(f"{v/3:10.0f}us/step  {op:<24} {src}\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
