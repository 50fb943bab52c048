"""TF-style evaluation preprocessing, reimplemented on PIL/numpy.

Capability parity with reference dfd/timm/data/tf_preprocessing.py (227 LoC),
which wraps TensorFlow's MnasNet/EfficientNet eval pipeline
(`preprocess_for_eval`: aspect-preserving central crop sized
img_size/(img_size+CROP_PADDING), then bicubic resize). That file hard-imports
tensorflow; this build reproduces the same geometry with PIL so the option
works without a TF install (there is deliberately no TF dependency in the
MI355X stack).
"""

import numpy as np
import torch
from PIL import Image

IMAGE_SIZE = 224
CROP_PADDING = 32


def center_crop_tf(img, img_size):
    """TF eval crop: central fraction img_size/(img_size+CROP_PADDING) of the
    shorter side, then bicubic resize to (img_size, img_size)."""
    w, h = img.size
    padded = img_size + CROP_PADDING
    crop = int(round(img_size / padded * min(w, h)))
    top = (h - crop + 1) // 2
    left = (w - crop + 1) // 2
    img = img.crop((left, top, left + crop, top + crop))
    return img.resize((img_size, img_size), Image.BICUBIC)


def preprocess_for_eval(img, img_size=IMAGE_SIZE):
    """PIL image -> fp32 numpy HWC in [0, 255] (TF convention)."""
    img = center_crop_tf(img.convert("RGB"), img_size)
    return np.asarray(img, dtype=np.float32)


class TfPreprocessTransform:
    """Drop-in for the reference TfPreprocessTransform: returns a fp32 CHW
    tensor scaled to [0, 1] (caller applies mean/std)."""

    def __init__(self, is_training=False, size=IMAGE_SIZE, interpolation="bicubic"):
        if is_training:
            raise NotImplementedError(
                "tf_preprocessing is an eval-only path (as in the reference "
                "loader, dfd/timm/data/loader.py)")
        self.size = size[-1] if isinstance(size, (tuple, list)) else size

    def __call__(self, img):
        x = preprocess_for_eval(img, self.size) / 255.0
        return torch.from_numpy(x.transpose(2, 0, 1))
