"""Numerics: every HIP/CDNA4 kernel vs the plain-PyTorch fp32 reference
(mi355/ops/torch_ref.py), per SURVEY.md §4's test strategy."""
import math

import pytest
import torch

from mi355 import ops
from mi355.ops import torch_ref

gpu = pytest.mark.gpu


def _relerr(a, b):
    a, b = a.float(), b.float()
    return ((a - b).abs().max() / (b.abs().max() + 1e-6)).item()


# --------------------------------------------------------------------------- #
# CPU reference self-checks (run everywhere)                                  #
# --------------------------------------------------------------------------- #
def test_rms_norm_ref_matches_manual():
    x = torch.randn(4, 8, dtype=torch.float64)
    w = torch.randn(8, dtype=torch.float64)
    y = torch_ref.rms_norm(x.float(), w.float(), 1e-6)
    man = (x / torch.sqrt((x * x).mean(-1, keepdim=True) + 1e-6)) * w
    assert torch.allclose(y, man.float(), atol=1e-5)


def test_attention_ref_causal_mask():
    q = torch.randn(1, 1, 4, 8)
    k = torch.randn(1, 1, 4, 8)
    v = torch.randn(1, 1, 4, 8)
    o = torch_ref.attention(q, k, v, causal=True)
    # row 0 can only attend to key 0
    o0 = v[0, 0, 0]
    assert torch.allclose(o[0, 0, 0], o0, atol=1e-5)


def test_cross_entropy_ref_matches_torch():
    logits = torch.randn(10, 50)
    labels = torch.randint(0, 50, (10,))
    labels[3] = -100
    ref = torch.nn.functional.cross_entropy(logits, labels, ignore_index=-100)
    assert torch.allclose(torch_ref.cross_entropy(logits, labels), ref, atol=1e-6)


# --------------------------------------------------------------------------- #
# HIP kernel vs fp32 reference (GPU)                                          #
# --------------------------------------------------------------------------- #
@gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16, torch.float16])
@pytest.mark.parametrize("shape", [(128, 768), (1024, 768), (33, 512), (64, 2048), (64, 4096)])
def test_rmsnorm_fwd_bwd(dtype, shape):
    torch.manual_seed(0)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    x = torch.randn(*shape, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(shape[-1], device="cuda", dtype=dtype, requires_grad=True)
    y = ops.rms_norm(x, w, 1e-6)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    yr = torch_ref.rms_norm(xr, wr, 1e-6)
    assert _relerr(y, yr) < tol
    dy = torch.randn_like(yr)
    y.backward(dy.to(dtype))
    yr.backward(dy)
    assert _relerr(x.grad, xr.grad) < tol
    assert _relerr(w.grad, wr.grad) < tol * 4  # atomics + fp32 accum over rows


@gpu
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("D", [768, 4096])  # 4096 covers the generic (non-1p) kernels
def test_add_rms_norm_fwd_bwd(dtype, D):
    torch.manual_seed(7)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    x = torch.randn(256, D, device="cuda", dtype=dtype, requires_grad=True)
    r = torch.randn(256, D, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(D, device="cuda", dtype=dtype, requires_grad=True)
    s, y = ops.add_rms_norm(x, r, w, 1e-6)
    xr = x.detach().float().requires_grad_(True)
    rr = r.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    sr = xr + rr
    yr = torch_ref.rms_norm(sr, wr, 1e-6)
    assert _relerr(s, sr) < tol and _relerr(y, yr) < tol
    ds = torch.randn_like(yr)
    dy = torch.randn_like(yr)
    (s.float() * ds + y.float() * dy).sum().backward()
    (sr * ds + yr * dy).sum().backward()
    assert _relerr(x.grad, xr.grad) < tol
    assert _relerr(r.grad, rr.grad) < tol
    assert _relerr(w.grad, wr.grad) < tol * 4


@gpu
def test_attention_prob_dropout():
    """T5 attention-prob dropout fused in the flash kernels: p=0 == baseline;
    deterministic per seed; fwd/bwd mask-consistent (dV zero columns where a
    key was fully dropped for a query's row set); bumping reshuffles."""
    torch.manual_seed(9)
    B, H, S, D = 1, 2, 128, 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        base = ops.attention(q, k, v)
        o_p0 = ops.attention(q, k, v, p_drop=0.0)
        assert torch.equal(base, o_p0)
        o_d1 = ops.attention(q, k, v, p_drop=0.5)
        o_d2 = ops.attention(q, k, v, p_drop=0.5)
        assert torch.equal(o_d1, o_d2), "same seed must reproduce the mask"
        ops.bump_dropout_seed(q.device)
        o_d3 = ops.attention(q, k, v, p_drop=0.5)
        assert not torch.equal(o_d1, o_d3), "bumped seed must reshuffle"
        assert not torch.equal(base, o_d1)
        # expectation preserved (1/keep scaling): means within a loose band
        assert abs(o_d1.float().mean().item() - base.float().mean().item()) < 0.05
    # bwd consistency: loss decreases along the negative gradient direction
    # measured ON THE SAME MASK (seed unchanged between probes)
    qg = q.clone().requires_grad_(True)
    o = ops.attention(qg, k, v, p_drop=0.3)
    loss = o.float().pow(2).sum()
    loss.backward()
    with torch.no_grad():
        q2 = (qg - 2e-3 * qg.grad.float()).to(torch.bfloat16)
        o2 = ops.attention(q2, k, v, p_drop=0.3)
        l2 = o2.float().pow(2).sum()
    assert l2.item() < loss.item(), (l2.item(), loss.item())


@gpu
def test_decode_attn_kpad():
    torch.manual_seed(8)
    B, H, L, D = 3, 4, 96, 64
    q = torch.randn(B, H, 1, D, device="cuda", dtype=torch.bfloat16)
    kc = torch.randn(B, H, 128, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(B, H, 128, D, device="cuda", dtype=torch.bfloat16)
    keep = torch.ones(B, L, device="cuda")
    keep[:, L - 20:] = 0
    kpad = (1 - keep) * -1e9
    with torch.no_grad():
        o = ops.decode_attention(q, kc, vc, L, kpad=kpad)
        o_ref = torch_ref.attention(q.float(), kc[:, :, :L].float(),
                                    vc[:, :, :L].float(), kpad=kpad)
    assert _relerr(o, o_ref) < 3e-2


@gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_geglu_fwd_bwd(dtype):
    torch.manual_seed(1)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    g = torch.randn(512, 2048, device="cuda", dtype=dtype, requires_grad=True)
    u = torch.randn(512, 2048, device="cuda", dtype=dtype, requires_grad=True)
    out = ops.geglu(g, u)
    gr = g.detach().float().requires_grad_(True)
    ur = u.detach().float().requires_grad_(True)
    outr = torch_ref.geglu(gr, ur)
    assert _relerr(out, outr) < tol
    d = torch.randn_like(outr)
    out.backward(d.to(dtype))
    outr.backward(d)
    assert _relerr(g.grad, gr.grad) < tol
    assert _relerr(u.grad, ur.grad) < tol


@gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("rows,V", [(64, 32128), (1024, 32128), (7, 4096)])
def test_cross_entropy_fwd_bwd(dtype, rows, V):
    torch.manual_seed(2)
    tol = 1e-5 if dtype == torch.float32 else 3e-2
    logits = (torch.randn(rows, V, device="cuda") * 4).to(dtype).requires_grad_(True)
    labels = torch.randint(0, V, (rows,), device="cuda")
    labels[::5] = -100
    loss = ops.cross_entropy(logits, labels)
    lr = logits.detach().float().requires_grad_(True)
    loss_r = torch_ref.cross_entropy(lr, labels)
    assert abs(loss.item() - loss_r.item()) / abs(loss_r.item()) < tol
    loss.backward()
    loss_r.backward()
    assert _relerr(logits.grad, lr.grad) < tol


@gpu
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
@pytest.mark.parametrize(
    "B,H,Sq,Sk,D,causal,with_bias,with_kpad",
    [
        (2, 12, 512, 512, 64, False, True, False),   # T5 encoder self-attn
        (2, 12, 512, 512, 64, True, True, False),    # T5 decoder self-attn
        (2, 12, 512, 512, 64, False, False, True),   # cross-attn w/ padding
        (1, 4, 100, 333, 64, False, True, True),     # ragged tails
        (1, 2, 64, 64, 32, True, False, False),      # SegFormer head dim
        (3, 2, 1, 128, 64, False, True, False),      # single-query path
    ],
)
def test_attention_fwd(dtype, B, H, Sq, Sk, D, causal, with_bias, with_kpad):
    torch.manual_seed(3)
    q = torch.randn(B, H, Sq, D, device="cuda", dtype=dtype)
    k = torch.randn(B, H, Sk, D, device="cuda", dtype=dtype)
    v = torch.randn(B, H, Sk, D, device="cuda", dtype=dtype)
    bias = None
    if with_bias:
        bias = (torch.randn(1, H, Sq, Sk, device="cuda") * 2).to(dtype)
    kpad = None
    if with_kpad:
        keep = torch.ones(B, Sk, device="cuda")
        keep[:, Sk - Sk // 4:] = 0
        kpad = (1 - keep) * -1e9
    with torch.no_grad():
        o = ops.attention(q, k, v, bias=bias, causal=causal, kpad=kpad)
        o_ref = torch_ref.attention(
            q.float(), k.float(), v.float(),
            bias=bias.float() if bias is not None else None,
            causal=causal, kpad=kpad,
        )
    err = _relerr(o, o_ref)
    assert err < 3e-2, f"attn fwd mismatch: {err}"


@gpu
@pytest.mark.parametrize(
    "B,H,Sq,Sk,D,causal,with_bias,with_kpad",
    [
        (2, 4, 256, 256, 64, True, True, False),     # decoder self-attn shape
        (2, 4, 256, 256, 64, False, True, False),    # encoder self-attn shape
        (2, 4, 128, 320, 64, False, False, True),    # cross-attn + padding
        (1, 2, 100, 100, 64, True, True, False),     # ragged
        (1, 2, 64, 64, 32, False, True, False),      # D=32
    ],
)
def test_attention_bwd(B, H, Sq, Sk, D, causal, with_bias, with_kpad):
    torch.manual_seed(4)
    q = torch.randn(B, H, Sq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, H, Sk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, H, Sk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    bias = None
    if with_bias:
        bias = (torch.randn(1, H, Sq, Sk, device="cuda") * 2).to(torch.bfloat16).requires_grad_(True)
    kpad = None
    if with_kpad:
        keep = torch.ones(B, Sk, device="cuda")
        keep[:, Sk - Sk // 4:] = 0
        kpad = (1 - keep) * -1e9
    o = ops.attention(q, k, v, bias=bias, causal=causal, kpad=kpad)
    do = torch.randn_like(o)
    o.backward(do)

    qr, kr, vr = [t.detach().float().requires_grad_(True) for t in (q, k, v)]
    br = bias.detach().float().requires_grad_(True) if with_bias else None
    o_ref = torch_ref.attention(qr, kr, vr, bias=br, causal=causal, kpad=kpad)
    o_ref.backward(do.float())
    pairs = [(q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"), (v.grad, vr.grad, "dv")]
    if with_bias:
        pairs.append((bias.grad, br.grad, "dbias"))
    for ours, ref, name in pairs:
        err = _relerr(ours, ref)
        assert err < 5e-2, f"{name} mismatch: {err}"


@gpu
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
@pytest.mark.parametrize("L,D", [(64, 64), (100, 64), (1, 64), (512, 64), (37, 32)])
def test_decode_attn(dtype, L, D):
    torch.manual_seed(5)
    B, H, Tmax = 4, 6, 512 + 16
    q = torch.randn(B, H, 1, D, device="cuda", dtype=dtype)
    kc = torch.randn(B, H, Tmax, D, device="cuda", dtype=dtype)
    vc = torch.randn(B, H, Tmax, D, device="cuda", dtype=dtype)
    bias = (torch.randn(1, H, 1, L, device="cuda")).to(dtype)
    with torch.no_grad():
        o = ops.decode_attention(q, kc, vc, L, bias=bias)
        o_ref = torch_ref.attention(q.float(), kc[:, :, :L].float(), vc[:, :, :L].float(),
                                    bias=bias.float())
    assert _relerr(o, o_ref) < 3e-2


@gpu
@pytest.mark.parametrize("gdtype", [torch.float32, torch.bfloat16])
def test_adamw_step(gdtype):
    torch.manual_seed(6)
    n = 12345
    master = torch.randn(n, device="cuda")
    grad = torch.randn(n, device="cuda").to(gdtype)
    m = torch.randn(n, device="cuda").abs() * 0.01
    v = torch.randn(n, device="cuda").abs() * 0.001
    lowp = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    ref = [t.clone() for t in (master, m, v)]
    ops.adamw_step([master], [grad], [m], [v], lr=1e-3, beta1=0.9, beta2=0.999,
                   eps=1e-8, weight_decay=0.01, step=3, grad_scale=0.5,
                   outs_lowp=[lowp])
    torch_ref.adamw_step([ref[0]], [grad], [ref[1]], [ref[2]], lr=1e-3, beta1=0.9,
                         beta2=0.999, eps=1e-8, weight_decay=0.01, step=3,
                         grad_scale=0.5)
    assert _relerr(master, ref[0]) < 1e-5
    assert _relerr(m, ref[1]) < 1e-5
    assert _relerr(v, ref[2]) < 1e-5
    assert _relerr(lowp.float(), ref[0]) < 1e-2  # bf16 write-back


@gpu
def test_native_ext_is_loaded():
    """Guard against silent eager fallback on GPU (round-end check)."""
    assert ops.hip_available()
    import mi355._C as C
    assert C.__file__.endswith("_C.so") and "/mi355/" in C.__file__


@gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_geglu_packed_fwd_bwd(dtype):
    torch.manual_seed(3)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    x = torch.randn(512, 4096, device="cuda", dtype=dtype, requires_grad=True)
    out = ops.geglu_packed(x)
    xr = x.detach().float().requires_grad_(True)
    outr = torch_ref.geglu(*xr.chunk(2, dim=-1))
    assert _relerr(out, outr) < tol
    d = torch.randn_like(outr)
    out.backward(d.to(dtype))
    outr.backward(d)
    assert _relerr(x.grad, xr.grad) < tol


@gpu
def test_geglu_packed_dropout_deterministic():
    # same device seed + salt -> identical mask fwd/fwd and matching bwd
    # masking (zero positions in out have zero dgate/dup grads)
    torch.manual_seed(4)
    x = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    out1 = ops.geglu_packed(x, p_drop=0.5, salt=123)
    out2 = ops.geglu_packed(x, p_drop=0.5, salt=123)
    assert torch.equal(out1, out2)
    out3 = ops.geglu_packed(x, p_drop=0.5, salt=124)
    assert not torch.equal(out1, out3)
    drop_frac = (out1 == 0).float().mean().item()
    assert 0.4 < drop_frac < 0.62  # ~p plus natural zeros
    out1.backward(torch.ones_like(out1))
    g = x.grad[:, 512:]  # dup = d * gelu(gate): zero exactly where dropped
    assert ((out1 == 0) == (g == 0)).float().mean().item() > 0.95


@gpu
def test_sumsq_accumulate():
    torch.manual_seed(5)
    out = torch.zeros((), dtype=torch.float32, device="cuda")
    a = torch.randn(1_000_003, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(4096, device="cuda", dtype=torch.float32)
    ops.sumsq_accumulate(a, out)
    ops.sumsq_accumulate(b, out)
    ref = a.float().pow(2).sum() + b.pow(2).sum()
    assert _relerr(out, ref) < 1e-3


@gpu
@pytest.mark.parametrize("causal", [False, True])
def test_attention_qkv_packed_matches_split(causal):
    """Packed self-attention (one (B,S,3,H,D) input/grad) must equal the
    split q/k/v path, including the fused qkv grad vs the cat of dq/dk/dv."""
    torch.manual_seed(11)
    B, S, H, D = 2, 192, 4, 64
    qkv = torch.randn(B, S, 3, H, D, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    bias = (torch.randn(1, H, S, S, device="cuda") * 0.5).to(torch.bfloat16)
    bias.requires_grad_(True)
    out_p = ops.attention_qkv_packed(qkv, bias=bias, causal=causal)

    qkv2 = qkv.detach().clone().requires_grad_(True)
    bias2 = bias.detach().clone().requires_grad_(True)
    q = qkv2.select(2, 0).permute(0, 2, 1, 3)
    k = qkv2.select(2, 1).permute(0, 2, 1, 3)
    v = qkv2.select(2, 2).permute(0, 2, 1, 3)
    out_s = ops.attention(q, k, v, bias=bias2, causal=causal)
    assert torch.equal(out_p, out_s)

    d = torch.randn_like(out_p)
    out_p.backward(d)
    out_s.backward(d)
    assert torch.equal(qkv.grad, qkv2.grad)
    assert torch.equal(bias.grad, bias2.grad)


@gpu
def test_attention_kv_packed_matches_split():
    torch.manual_seed(12)
    B, Sq, Sk, H, D = 2, 96, 160, 4, 64
    q = torch.randn(B, H, Sq, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    kv = torch.randn(B, Sk, 2, H, D, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    out_p = ops.attention_kv_packed(q, kv)

    q2 = q.detach().clone().requires_grad_(True)
    kv2 = kv.detach().clone().requires_grad_(True)
    k = kv2.select(2, 0).permute(0, 2, 1, 3)
    v = kv2.select(2, 1).permute(0, 2, 1, 3)
    out_s = ops.attention(q2, k, v)
    assert torch.equal(out_p, out_s)
    d = torch.randn_like(out_p)
    out_p.backward(d)
    out_s.backward(d)
    assert torch.equal(q.grad, q2.grad)
    assert torch.equal(kv.grad, kv2.grad)


@gpu
def test_attention_dropout_consistent_across_bwd_impls(monkeypatch):
    """The fused attention-prob dropout mask derives from absolute (q,k)
    indices, so the v1 (q-split) and v3 (split-kernel) backwards must see the
    SAME mask and produce matching grads for one seed."""
    torch.manual_seed(21)
    B, S, H, D = 2, 256, 4, 64
    make = lambda: torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    q0, k0, v0 = make(), make(), make()
    d = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)

    def run(impl):
        monkeypatch.setenv("MI355_ATTN_BWD", impl)
        torch.manual_seed(5)
        ops._dropout_seeds.clear()  # same device seed for both runs
        q = q0.clone().requires_grad_(True)
        k = k0.clone().requires_grad_(True)
        v = v0.clone().requires_grad_(True)
        out = ops.attention(q, k, v, p_drop=0.3)
        out.backward(d)
        return out.detach(), q.grad, k.grad, v.grad

    o1, dq1, dk1, dv1 = run("v1")
    o3, dq3, dk3, dv3 = run("v3")
    monkeypatch.delenv("MI355_ATTN_BWD")
    assert torch.equal(o1, o3)  # fwd identical (same kernel + seed)
    for a, b in ((dq1, dq3), (dk1, dk3), (dv1, dv3)):
        assert _relerr(a.float(), b.float()) < 3e-2
