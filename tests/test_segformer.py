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
