from .image_processor import SegformerFeatureExtractor, SegformerImageProcessor
from .modeling import (SegformerConfig, SegformerForImageSegmentation,
                       SegformerForSemanticSegmentation)

__all__ = ["SegformerConfig", "SegformerForSemanticSegmentation",
           "SegformerForImageSegmentation", "SegformerImageProcessor",
           "SegformerFeatureExtractor"]
