"""Fused AdamW over the FlatReplica's flat buffers.

The reference's optimizer is HF Trainer's AdamW over ~250 separate tensors
(SURVEY.md K10). Here the whole model is two contiguous buffers (decay /
no-decay), so one training step costs TWO fused HIP kernel launches that
read the bf16 flat gradient, update fp32 master + moments, and write the
bf16 flat param — no per-tensor launch storm, no optimizer.zero_grad walk.
Gradient clipping (HF default max_grad_norm=1.0) is folded into the kernel
as ``grad_scale``.
"""
from __future__ import annotations

import torch

from .. import ops
from .ddp import FlatReplica


class FlatAdamW:
    def __init__(self, replica: FlatReplica, lr: float = 2e-5, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0,
                 max_grad_norm: float | None = 1.0):
        self.replica = replica
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.max_grad_norm = max_grad_norm
        self.step_count = 0
        self.state = []
        for decay_name, g in replica.groups:
            master = g.flat_param.detach().float().clone()
            self.state.append({
                "decay": decay_name == "decay",
                "master": master,
                "exp_avg": torch.zeros_like(master),
                "exp_avg_sq": torch.zeros_like(master),
            })

    def zero_grad(self):
        self.replica.zero_grads()

    @torch.no_grad()
    def step(self, lr: float | None = None):
        self.step_count += 1
        lr = self.lr if lr is None else lr
        grad_scale = 1.0
        if self.max_grad_norm is not None and self.max_grad_norm > 0:
            total_norm = self.replica.grad_norm()
            clip = self.max_grad_norm / (total_norm + 1e-6)
            grad_scale = torch.clamp(clip, max=1.0)  # 0-dim tensor, stays on device
        for (_, g), st in zip(self.replica.groups, self.state):
            ops.adamw_step(
                [st["master"]], [g.flat_grad], [st["exp_avg"]], [st["exp_avg_sq"]],
                lr=lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                weight_decay=self.weight_decay if st["decay"] else 0.0,
                step=self.step_count, grad_scale=grad_scale,
                outs_lowp=[g.flat_param],
            )

    # minimal torch-optimizer-compatible surface ------------------------- #
    def state_dict(self):
        return {
            "step": self.step_count,
            "state": [{k: (v if not torch.is_tensor(v) else v.cpu()) for k, v in st.items()}
                      for st in self.state],
            "lr": self.lr,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.lr = sd.get("lr", self.lr)
        for st, st_in in zip(self.state, sd["state"]):
            for k in ("master", "exp_avg", "exp_avg_sq"):
                st[k].copy_(st_in[k].to(st[k].device))
