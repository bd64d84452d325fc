"""SegFormer model family tests (C19/C21, K13-K18)."""
import numpy as np
import pytest
import torch

from mi355.models.segformer import (SegformerConfig,
                                    SegformerForSemanticSegmentation,
                                    SegformerImageProcessor)

TINY = SegformerConfig(depths=[1, 1, 1, 1], hidden_sizes=[8, 16, 24, 32],
                       num_attention_heads=[1, 2, 3, 4], decoder_hidden_size=32,
                       num_labels=10)


def test_forward_shapes_and_loss():
    torch.manual_seed(0)
    m = SegformerForSemanticSegmentation(TINY)
    pix = torch.randn(2, 3, 64, 64)
    labels = torch.randint(0, 10, (2, 64, 64))
    out = m(pixel_values=pix, labels=labels)
    assert out.logits.shape == (2, 10, 16, 16)  # 1/4 resolution
    assert out.loss.item() > 0
    out.loss.backward()
    assert m.decode_head.classifier.weight.grad is not None


def test_ignore_index_in_loss():
    torch.manual_seed(0)
    m = SegformerForSemanticSegmentation(TINY).eval()
    pix = torch.randn(1, 3, 64, 64)
    labels = torch.full((1, 64, 64), 255)  # everything ignored
    labels[0, 0, 0] = 3
    out = m(pixel_values=pix, labels=labels)
    assert torch.isfinite(out.loss)


def test_image_processor_reduce_labels():
    proc = SegformerImageProcessor(size=32, do_reduce_labels=True)
    img = (np.random.rand(48, 48, 3) * 255).astype(np.uint8)
    seg = np.zeros((48, 48), dtype=np.int64)
    seg[10:20, 10:20] = 5
    enc = proc(img, segmentation_maps=seg, return_tensors="pt")
    assert enc["pixel_values"].shape == (1, 3, 32, 32)
    lab = enc["labels"][0]
    assert (lab[lab != 255] == 4).all()  # label 5 -> 4; background 0 -> 255
    assert (lab == 255).any()


def test_post_process_semantic_segmentation():
    torch.manual_seed(1)
    proc = SegformerImageProcessor()

    class Out:
        logits = torch.randn(2, 10, 16, 16)

    res = proc.post_process_semantic_segmentation(Out(), target_sizes=[(64, 64), (40, 56)])
    assert res[0].shape == (64, 64) and res[1].shape == (40, 56)
    # matches torch reference
    ref = torch.nn.functional.interpolate(Out.logits[:1].float(), size=(64, 64),
                                          mode="bilinear", align_corners=False).argmax(1)[0]
    assert (res[0].cpu() == ref.to(res[0].dtype)).float().mean() > 0.999


def test_save_load_roundtrip(tmp_path):
    torch.manual_seed(0)
    m = SegformerForSemanticSegmentation(TINY).eval()
    m.save_pretrained(str(tmp_path))
    m2 = SegformerForSemanticSegmentation.from_pretrained(str(tmp_path)).eval()
    pix = torch.randn(1, 3, 64, 64)
    with torch.no_grad():
        d = (m(pixel_values=pix).logits - m2(pixel_values=pix).logits).abs().max()
    assert d.item() == 0.0


@pytest.mark.gpu
def test_segformer_gpu_train_step():
    torch.manual_seed(0)
    m = SegformerForSemanticSegmentation(TINY).to("cuda")
    pix = torch.randn(2, 3, 64, 64, device="cuda")
    labels = torch.randint(0, 10, (2, 64, 64), device="cuda")
    out = m(pixel_values=pix, labels=labels)
    out.loss.backward()
    torch.cuda.synchronize()
    assert out.loss.item() > 0


@pytest.mark.gpu
def test_seg_postprocess_kernel_vs_ref():
    from mi355 import ops
    from mi355.ops import torch_ref

    torch.manual_seed(2)
    logits = torch.randn(2, 150, 128, 128, device="cuda")
    out = ops.seg_postprocess(logits, (512, 512))
    ref = torch_ref.seg_postprocess(logits, (512, 512))
    agree = (out == ref).float().mean().item()
    assert agree > 0.999, agree  # ties may break differently


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(8, 128, 128, 128), (8, 1024, 16, 16),
                                   (2, 64, 32, 40)])
def test_dwconv3x3_matches_fp32_ref(shape):
    """Hand-written depthwise 3x3 stencil vs fp32 F.conv2d (K15)."""
    import torch.nn.functional as F
    from mi355 import ops

    torch.manual_seed(0)
    b, c, h, w = shape
    x = torch.randn(b, c, h, w, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    wt = torch.randn(c, 1, 3, 3, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    bias = torch.randn(c, device="cuda", dtype=torch.bfloat16,
                       requires_grad=True)
    y = ops.dwconv3x3(x, wt, bias)
    gy = torch.randn_like(y)
    y.backward(gy)
    x32 = x.detach().float().requires_grad_(True)
    w32 = wt.detach().float().requires_grad_(True)
    b32 = bias.detach().float().requires_grad_(True)
    y32 = F.conv2d(x32, w32, b32, padding=1, groups=c)
    y32.backward(gy.float())

    def relerr(a, ref):
        a, ref = a.float(), ref.float()
        return ((a - ref).abs().max() / (ref.abs().max() + 1e-6)).item()

    assert relerr(y, y32) < 2e-2
    assert relerr(x.grad, x32.grad) < 2e-2
    assert relerr(wt.grad, w32.grad) < 2e-2
    assert relerr(bias.grad, b32.grad) < 2e-2


@pytest.mark.gpu
def test_patch_embed_s1_matches_fp32_ref():
    """Hand-written stage-1 patch-embed conv (7x7 s4 p3, 3->32) vs fp32
    F.conv2d, forward + dW/dbias (K13)."""
    import torch.nn.functional as F
    from mi355 import ops

    torch.manual_seed(1)
    x = torch.randn(2, 3, 512, 512, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(32, 3, 7, 7, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(32, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.patch_embed_s1(x, w, b)
    gy = torch.randn_like(y)
    y.backward(gy)
    w32 = w.detach().float().requires_grad_(True)
    b32 = b.detach().float().requires_grad_(True)
    y32 = F.conv2d(x.float(), w32, b32, stride=4, padding=3)
    y32.backward(gy.float())

    def relerr(a, ref):
        a, ref = a.float(), ref.float()
        return ((a - ref).abs().max() / (ref.abs().max() + 1e-6)).item()

    assert y.shape == y32.shape
    assert relerr(y, y32) < 2e-2
    assert relerr(w.grad, w32.grad) < 2e-2
    assert relerr(b.grad, b32.grad) < 2e-2


def test_visualization_helpers():
    """Reference display-glue surface (Semantic_segmentation/utils.py):
    ade_palette / prepare_pixels_with_segmentation / visualize_predictions /
    display_example_images / convert_image_to_rgb."""
    from PIL import Image

    from mi355.models.segformer.visualization import (
        ade_palette, convert_image_to_rgb, display_example_images,
        get_image_indices, prepare_pixels_with_segmentation,
        visualize_predictions)

    pal = ade_palette()
    assert len(pal) == 151 and pal[0] == [0, 0, 0]
    assert len({tuple(c) for c in pal}) == 151  # all distinct
    img = Image.fromarray(
        (np.random.default_rng(0).random((64, 64, 3)) * 255).astype("uint8"),
        "RGB").convert("L")  # non-RGB mode exercises convert_image_to_rgb
    assert convert_image_to_rgb(img).mode == "RGB"
    seg = np.random.default_rng(1).integers(0, 150, (64, 64))
    blended = prepare_pixels_with_segmentation(img, seg)
    assert blended.shape == (64, 64, 3) and blended.dtype == np.uint8
    ax = visualize_predictions(img, seg)
    assert ax is not None
    idx = get_image_indices(10, 4, seed=0)
    assert len(idx) == 4 and len(set(idx)) == 4
    fig = display_example_images([img] * 4, n=2)
    assert fig is not None


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(2, 150, 128, 128, 512),
                                   (1, 19, 64, 64, 256)])
def test_seg_ce_fused_matches_torch(shape):
    """Fused upsample+CE (K18) vs the materializing torch path: loss and
    dLogits, incl. ignore_index pixels."""
    import torch.nn.functional as F
    from mi355 import ops

    b, c, h, w, H = shape
    torch.manual_seed(3)
    logits = torch.randn(b, c, h, w, device="cuda", dtype=torch.bfloat16,
                         requires_grad=True)
    labels = torch.randint(0, c, (b, H, H), device="cuda")
    labels[:, :7] = 255  # ignore region
    loss = ops.seg_cross_entropy_upsampled(logits, labels, ignore_index=255)
    loss.backward()
    got_grad = logits.grad.clone()

    l32 = logits.detach().float().requires_grad_(True)
    up = F.interpolate(l32, size=(H, H), mode="bilinear", align_corners=False)
    ref = F.cross_entropy(up, labels.long(), ignore_index=255)
    ref.backward()

    assert abs(loss.item() - ref.item()) / abs(ref.item()) < 2e-3
    gerr = ((got_grad.float() - l32.grad).abs().max() /
            (l32.grad.abs().max() + 1e-9)).item()
    assert gerr < 2e-2, gerr


@pytest.mark.gpu
@pytest.mark.parametrize("shape,size", [((2, 256, 64, 64), (128, 128)),
                                        ((2, 64, 32, 32), (128, 128)),
                                        ((1, 16, 16, 16), (128, 128)),
                                        ((2, 8, 16, 24), (64, 96))])
def test_upsample_bilinear_matches_torch(shape, size):
    """Gather-backward bilinear upsample (K16) vs torch fp32 interpolate."""
    import torch.nn.functional as F
    from mi355 import ops

    torch.manual_seed(3)
    x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.upsample_bilinear(x, size)
    gy = torch.randn_like(y)
    y.backward(gy)
    x32 = x.detach().float().requires_grad_(True)
    y32 = F.interpolate(x32, size=size, mode="bilinear", align_corners=False)
    y32.backward(gy.float())

    def relerr(a, ref):
        return (a.float() - ref).norm() / ref.norm().clamp(min=1e-6)

    assert relerr(y, y32) < 2e-2
    assert relerr(x.grad, x32.grad) < 2e-2


def test_mit_family_presets():
    """MiT-b0..b5 presets match the published parameter counts (meta
    device), and a non-b0 config runs forward+loss on CPU (the custom-
    kernel guards fall back cleanly off its shapes)."""
    from mi355.models.segformer.modeling import (
        SegformerConfig, SegformerForSemanticSegmentation)

    expect = {"b0": (3.4e6, 4.2e6), "b1": (13e6, 15e6), "b2": (24e6, 30e6),
              "b3": (44e6, 50e6), "b4": (60e6, 67e6), "b5": (81e6, 90e6)}
    for name, (lo, hi) in expect.items():
        with torch.device("meta"):
            m = SegformerForSemanticSegmentation(SegformerConfig.from_preset(name))
        n = sum(p.numel() for p in m.parameters())
        assert lo < n < hi, (name, n)

    m = SegformerForSemanticSegmentation(
        SegformerConfig.from_preset("b1", num_labels=19))
    x = torch.randn(1, 3, 64, 64)
    labels = torch.randint(0, 19, (1, 64, 64))
    out = m(pixel_values=x, labels=labels)
    assert out.loss.isfinite()
    out.loss.backward()
