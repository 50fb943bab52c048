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
    """Drop-in for the reference TfPreprocessTransform: returns a uint8 CHW
    numpy array in [0, 255] — the reference contract
    (dfd/timm/data/tf_preprocessing.py returns `np.rollaxis(np.uint8, 2)`).
    fast_collate packs it into the batch uint8 buffer unchanged and the
    PrefetchLoader applies the mean*255/std*255 normalization on device; a
    fp32 [0,1] return here would truncate to 0/1 in that buffer."""

    def __init__(self, is_training=False, size=IMAGE_SIZE, interpolation="bicubic"):
        if is_training:
            raise NotImplementedError(
                "tf_preprocessing is an eval-only path (as in the reference "
                "loader, dfd/timm/data/loader.py)")
        self.size = size[-1] if isinstance(size, (tuple, list)) else size

    def __call__(self, img):
        x = preprocess_for_eval(img, self.size)
        return np.rollaxis(x.astype(np.uint8), 2)
