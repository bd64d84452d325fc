"""SegformerImageProcessor / feature extractor (reference
Scaling_batch_inference.ipynb:cell-24 preprocessing + cell-42
post_process_semantic_segmentation; Scaling_model_training.ipynb:cell-39
SegformerImageProcessor(do_reduce_labels=True))."""
from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

IMAGENET_MEAN = np.array([0.485, 0.456, 0.406], dtype=np.float32)
IMAGENET_STD = np.array([0.229, 0.224, 0.225], dtype=np.float32)


def _resize_image(img: np.ndarray, size: tuple[int, int], nearest=False) -> np.ndarray:
    """img: (H, W, C) or (H, W). Bilinear (or nearest for labels) via torch."""
    t = torch.as_tensor(img)
    if t.ndim == 2:
        t = t[None, None].float()
        mode = "nearest" if nearest else "bilinear"
        out = F.interpolate(t, size=size, mode=mode)
        return out[0, 0].numpy().astype(img.dtype)
    t = t.permute(2, 0, 1)[None].float()
    out = F.interpolate(t, size=size, mode="nearest" if nearest else "bilinear",
                        align_corners=False if not nearest else None)
    return out[0].permute(1, 2, 0).numpy()


class SegformerImageProcessor:
    def __init__(self, do_resize=True, size=None, do_rescale=True,
                 rescale_factor=1 / 255, do_normalize=True, image_mean=None,
                 image_std=None, do_reduce_labels=False, **_):
        self.do_resize = do_resize
        if isinstance(size, int):
            size = {"height": size, "width": size}
        self.size = size or {"height": 512, "width": 512}
        self.do_rescale = do_rescale
        self.rescale_factor = rescale_factor
        self.do_normalize = do_normalize
        self.image_mean = np.asarray(image_mean if image_mean is not None
                                     else IMAGENET_MEAN, dtype=np.float32)
        self.image_std = np.asarray(image_std if image_std is not None
                                    else IMAGENET_STD, dtype=np.float32)
        self.do_reduce_labels = do_reduce_labels

    @staticmethod
    def _to_ndarray(x) -> np.ndarray:
        # arrow list<list<...>> columns round-trip as nested OBJECT arrays;
        # stack them level by level into a dense ndarray
        a = np.asarray(x)
        if a.dtype != object:
            return a
        return np.stack([SegformerImageProcessor._to_ndarray(e) for e in a])

    def _one(self, image) -> np.ndarray:
        img = self._to_ndarray(image)
        if img.ndim == 2:
            img = np.stack([img] * 3, axis=-1)
        if img.shape[-1] == 4:
            img = img[..., :3]
        if self.do_resize:
            img = _resize_image(img.astype(np.float32),
                                (self.size["height"], self.size["width"]))
        img = img.astype(np.float32)
        if self.do_rescale:
            img = img * self.rescale_factor
        if self.do_normalize:
            img = (img - self.image_mean) / self.image_std
        return img.transpose(2, 0, 1)  # CHW

    def _one_label(self, seg) -> np.ndarray:
        lab = self._to_ndarray(seg)
        if self.do_resize:
            lab = _resize_image(lab, (self.size["height"], self.size["width"]),
                                nearest=True)
        lab = lab.astype(np.int64)
        if self.do_reduce_labels:
            # ADE convention: 0 = background -> ignore(255); labels shift -1
            lab = np.where(lab == 0, 255, lab - 1)
        return lab

    def __call__(self, images, segmentation_maps=None, return_tensors=None, **_):
        if not isinstance(images, (list, tuple)):
            images = [images]
        pixel_values = np.stack([self._one(im) for im in images])
        out = {"pixel_values": pixel_values}
        if segmentation_maps is not None:
            if not isinstance(segmentation_maps, (list, tuple)):
                segmentation_maps = [segmentation_maps]
            out["labels"] = np.stack([self._one_label(s) for s in segmentation_maps])
        if return_tensors == "pt":
            out = {k: torch.as_tensor(v) for k, v in out.items()}
        return out

    def preprocess(self, images, **kw):
        return self(images, **kw)

    # reference cell-42: upsample logits to original size + argmax
    def post_process_semantic_segmentation(self, outputs, target_sizes=None):
        logits = outputs.logits if hasattr(outputs, "logits") else outputs
        from ... import ops

        results = []
        for i in range(logits.shape[0]):
            size = target_sizes[i] if target_sizes is not None else logits.shape[-2:]
            seg = ops.seg_postprocess(logits[i:i + 1], (int(size[0]), int(size[1])))
            results.append(seg[0])
        return results


SegformerFeatureExtractor = SegformerImageProcessor
