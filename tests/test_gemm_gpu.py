"""Numerics for the hand-written MFMA GEMM kernels (csrc/kernels/gemm.hip)
vs a plain-PyTorch fp32 reference of the same op, plus the
_MfmaLinear autograd wrapper vs F.linear.

Shapes cover the T5-base hot set: M = B*S for B=2
(1024) and decode (256); N/K in {768, 2304, 2048, 4096}; a vocab-width slice;
and small-grid shapes that trigger the split-K path."""
import pytest
import torch

from mi355 import ops

gpu = pytest.mark.gpu


def _relerr(a, b):
    a, b = a.float(), b.float()
    return ((a - b).abs().max() / (b.abs().max() + 1e-6)).item()


SHAPES = [
    # (M, N, K) output rows/cols, contraction
    (256, 768, 768),      # decode proj, split-K grid
    (1024, 2304, 768),    # fused QKV @ B=2
    (1024, 768, 2048),    # FF wo
    (1024, 4096, 768),    # FF wi packed
    (128, 128, 64),       # single tile
    (256, 32128, 768),    # lm_head decode slice (vocab width)
]


@gpu
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
@pytest.mark.parametrize("shape", SHAPES)
def test_gemm_nt_matches_fp32_ref(shape, dtype):
    M, N, K = shape
    torch.manual_seed(0)
    a = torch.randn(M, K, device="cuda", dtype=dtype)
    b = torch.randn(N, K, device="cuda", dtype=dtype)
    got = ops.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    assert got.shape == (M, N)
    assert _relerr(got, ref) < 3e-2


@gpu
@pytest.mark.parametrize("shape", SHAPES)
def test_gemm_nn_matches_fp32_ref(shape):
    M, N, K = shape
    torch.manual_seed(1)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
    got = ops.gemm_nn(a, b)
    ref = a.float() @ b.float()
    assert _relerr(got, ref) < 3e-2


@gpu
@pytest.mark.parametrize("shape", SHAPES)
def test_gemm_tn_matches_fp32_ref(shape):
    M, N, K = shape
    torch.manual_seed(2)
    a = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
    got = ops.gemm_tn(a, b)
    ref = a.float().t() @ b.float()
    assert _relerr(got, ref) < 3e-2


@gpu
def test_gemm_splitk_explicit():
    """Forcing several split-K factors must agree with Z=1."""
    torch.manual_seed(3)
    a = torch.randn(128, 128, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(128, 768, device="cuda", dtype=torch.bfloat16).t().contiguous()
    base = ops.gemm_nt(a, b, splitk=1)
    for z in (2, 4, 12):
        got = ops.gemm_nt(a, b, splitk=z)
        assert _relerr(got, base) < 1e-3, f"splitk={z}"


@gpu
def test_mfma_linear_autograd_matches_torch():
    """_MfmaLinear fwd+bwd (MI355_GEMM=hip) vs F.linear fwd+bwd."""
    import os

    torch.manual_seed(4)
    x = torch.randn(2, 512, 768, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(2304, 768, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    old = os.environ.get("MI355_GEMM")
    os.environ["MI355_GEMM"] = "hip"
    try:
        y = ops.linear(x, w)
        gy = torch.randn_like(y)
        y.backward(gy)
        got_dx, got_dw = x.grad.clone(), w.grad.clone()
    finally:
        if old is None:
            os.environ.pop("MI355_GEMM", None)
        else:
            os.environ["MI355_GEMM"] = old
    x32 = x.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    y32 = torch.nn.functional.linear(x32, w32)
    y32.backward(gy.float())
    assert _relerr(y, y32) < 3e-2
    assert _relerr(got_dx, x32.grad) < 3e-2
    assert _relerr(got_dw, w32.grad) < 3e-2


@gpu
@pytest.mark.parametrize("shape", [
    (256, 256, 64),        # single 256-tile
    (1024, 2304, 768),     # qkv
    (1024, 32128, 768),    # lm_head: partial N tile (32128 = 125.5 * 256)
    (512, 1280, 128),      # M,N%256 != 0 -> clamped edge tiles
])
def test_gemm_nt256_matches_fp32_ref(shape, monkeypatch):
    """The 256x256 8-phase kernel, forced via MI355_GEMM_TILE=256."""
    monkeypatch.setenv("MI355_GEMM_TILE", "256")
    M, N, K = shape
    torch.manual_seed(5)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    got = ops.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    assert _relerr(got, ref) < 3e-2


@gpu
@pytest.mark.parametrize("shape", [
    (1024, 768, 768),      # attn_o fwd @ B=2
    (1024, 2304, 768),     # qkv fwd
    (512, 64, 64),         # single skinny tile
    (256, 128, 128),       # skinny edge: M=256 (one tile row)
])
def test_gemm_nt_skinny_matches_fp32_ref(shape, monkeypatch):
    """The 256x64 skinny NT kernel, forced via MI355_GEMM_TILE=skinny."""
    monkeypatch.setenv("MI355_GEMM_TILE", "skinny")
    M, N, K = shape
    torch.manual_seed(6)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    got = ops.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    assert _relerr(got, ref) < 3e-2


# ---------------------------------------------------------------------- #
# CPU-checkable dispatch-policy invariants (no GPU needed)                #
# ---------------------------------------------------------------------- #
def test_gemm_policy_dims_and_env(monkeypatch):
    from mi355.ops.gemm_table import _WINS, gemm_policy

    monkeypatch.delenv("MI355_GEMM", raising=False)
    # every tabled win satisfies the kernel's dimension constraints and
    # dispatches to the HIP kernel in auto mode
    for op, rows, cols, kc in _WINS:
        assert rows % 128 == 0 and cols % 128 == 0 and kc % 64 == 0, (op, rows, cols, kc)
        assert gemm_policy(op, rows, cols, kc)
    # a shape NOT in the table stays on the library in auto mode
    assert not gemm_policy("nt", 1024, 768, 768)
    # dims the kernels cannot run are never dispatched, even with =hip
    monkeypatch.setenv("MI355_GEMM", "hip")
    assert not gemm_policy("nt", 100, 768, 768)
    monkeypatch.setenv("MI355_GEMM", "blaslt")
    for op, rows, cols, kc in _WINS:
        assert not gemm_policy(op, rows, cols, kc)


def test_upsample_bilinear_cpu_fallback_matches_interpolate():
    import torch.nn.functional as F

    torch.manual_seed(0)
    x = torch.randn(2, 3, 16, 24)
    # CPU (and non-integer-scale) inputs take the torch fp32 path verbatim
    y = ops.upsample_bilinear(x, (64, 96))
    ref = F.interpolate(x.float(), size=(64, 96), mode="bilinear",
                        align_corners=False).to(x.dtype)
    assert torch.equal(y, ref)
    y2 = ops.upsample_bilinear(x, (40, 50))  # non-integer scale
    ref2 = F.interpolate(x.float(), size=(40, 50), mode="bilinear",
                         align_corners=False).to(x.dtype)
    assert torch.equal(y2, ref2)
