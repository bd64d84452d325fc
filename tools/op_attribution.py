#!/usr/bin/env python3
"""Attribute GPU kernel time of the bench train step to Python source lines
(torch.profiler with stacks). Finds where the torch-native elementwise/copy/
cat "soup" kernels come from so they can be fused or eliminated."""
import sys

sys.path.insert(0, ".")

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
        f.write("\n\n== by source line (group_by_stack_n=7) ==\n")
        f.write(prof.key_averages(group_by_stack_n=7).table(
            sort_by="self_cuda_time_total", row_limit=50, max_name_column_width=60))
    print("wrote", out)


if __name__ == "__main__":
    main()
