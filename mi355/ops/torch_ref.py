"""Plain-PyTorch reference implementations of every mi355 hot op.

These are the semantics contract for the hand-written HIP/CDNA4 kernels in
``mi355/csrc/kernels``: each HIP kernel is validated against the op here in
fp32 (see tests/test_ops_numerics.py). On CPU these are also the execution
path (the reference workshop runs its distributed trainer on CPU with
``no_cuda=True`` — see SURVEY.md §4.2, reference
Scaling_model_training.ipynb:cell-47).

All functions are differentiable plain-torch compositions.
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """T5LayerNorm: RMS norm, no mean subtraction, no bias.

    Matches the reference model's T5 layer norm semantics:
    accumulate the mean-square in fp32, scale in the input dtype.
    """
    dt = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    xf = xf * torch.rsqrt(var + eps)
    return (xf.to(dt)) * weight


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    """gelu_new / tanh approximation (T5 v1.1 gated act)."""
    return 0.5 * x * (1.0 + torch.tanh(math.sqrt(2.0 / math.pi) * (x + 0.044715 * x.pow(3.0))))


def geglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """Gated-GELU elementwise combine: gelu_tanh(gate) * up  (SURVEY.md K6)."""
    return gelu_tanh(gate) * up


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    bias: torch.Tensor | None = None,
    causal: bool = False,
    scale: float = 1.0,
    kpad: torch.Tensor | None = None,
    p_drop: float = 0.0,
) -> torch.Tensor:
    """Attention core with additive bias (T5 relative-position bias), unscaled
    by default (T5 absorbs 1/sqrt(d) into init — SURVEY.md K4).

    q: (B, H, Sq, D); k, v: (B, H, Sk, D); bias broadcastable to (B, H, Sq, Sk).
    """
    scores = torch.matmul(q, k.transpose(-1, -2))
    if scale != 1.0:
        scores = scores * scale
    if bias is not None:
        scores = scores + bias
    if kpad is not None:
        scores = scores + kpad.view(kpad.shape[0], 1, 1, kpad.shape[-1]).to(scores.dtype)
    if causal:
        sq, sk = scores.shape[-2], scores.shape[-1]
        idx_q = torch.arange(sq, device=scores.device).unsqueeze(-1) + (sk - sq)
        idx_k = torch.arange(sk, device=scores.device).unsqueeze(0)
        neg = torch.finfo(scores.dtype).min
        scores = scores.masked_fill(idx_k > idx_q, neg)
    probs = F.softmax(scores.float(), dim=-1).to(q.dtype)
    if p_drop > 0.0:
        probs = F.dropout(probs, p=p_drop, training=True)
    return torch.matmul(probs, v)


def cross_entropy(logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100):
    """Token-mean CE over the vocab, fp32 log-softmax."""
    return F.cross_entropy(
        logits.float().view(-1, logits.shape[-1]),
        labels.view(-1),
        ignore_index=ignore_index,
    )


def adamw_step(
    params_f32: list[torch.Tensor],
    grads: list[torch.Tensor],
    exp_avgs: list[torch.Tensor],
    exp_avg_sqs: list[torch.Tensor],
    *,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    grad_scale: float = 1.0,
    outs_lowp: list[torch.Tensor] | None = None,
):
    """Decoupled AdamW on fp32 master params.

    ``outs_lowp`` optionally receives the updated params cast to its dtype
    (bf16 model weights while masters stay fp32).
    """
    bc1 = 1.0 - beta1**step
    bc2 = 1.0 - beta2**step
    for i, (p, g, m, v) in enumerate(zip(params_f32, grads, exp_avgs, exp_avg_sqs)):
        gf = g.float()
        if grad_scale != 1.0:
            gf = gf * grad_scale
        m.mul_(beta1).add_(gf, alpha=1.0 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1.0 - beta2)
        denom = (v / bc2).sqrt_().add_(eps)
        p.mul_(1.0 - lr * weight_decay)
        p.addcdiv_(m / bc1, denom, value=-lr)
        if outs_lowp is not None:
            outs_lowp[i].copy_(p)


def softmax_bias(scores: torch.Tensor, bias: torch.Tensor | None) -> torch.Tensor:
    out = scores if bias is None else scores + bias
    return F.softmax(out.float(), dim=-1).to(scores.dtype)


def seg_postprocess(logits: torch.Tensor, size) -> torch.Tensor:
    """Bilinear-upsample class logits to ``size`` and argmax over channels
    (SURVEY.md K17 — reference post_process_semantic_segmentation)."""
    up = F.interpolate(logits.float(), size=size, mode="bilinear",
                       align_corners=False)
    return up.argmax(dim=1).to(torch.int32)
