"""PIL/numpy functional ops for the transform pipeline.

This image ships no torchvision/cv2, so the functional layer
(flip/pad/crop/resize/color adjust) is implemented directly on PIL —
semantics match torchvision.transforms.functional for the operations the
reference pipelines use.
"""

import numbers

import numpy as np
from PIL import Image, ImageEnhance, ImageOps

# interpolation codes (PIL resamples)
NEAREST = Image.NEAREST
BILINEAR = Image.BILINEAR
BICUBIC = Image.BICUBIC
LANCZOS = Image.LANCZOS
HAMMING = Image.HAMMING
BOX = Image.BOX


def hflip(img):
    return img.transpose(Image.FLIP_LEFT_RIGHT)


def _expand_padding(padding):
    if isinstance(padding, numbers.Number):
        return (int(padding),) * 4
    if len(padding) == 2:
        return (int(padding[0]), int(padding[1]), int(padding[0]), int(padding[1]))
    assert len(padding) == 4
    return tuple(int(p) for p in padding)


def pad(img, padding, fill=0, padding_mode="constant"):
    """padding: int | (lr, tb) | (l, t, r, b)."""
    left, top, right, bottom = _expand_padding(padding)
    if padding_mode == "constant":
        if isinstance(fill, (list, tuple)):
            fill = tuple(fill)
        return ImageOps.expand(img, border=(left, top, right, bottom), fill=fill)
    # reflect/edge/symmetric via numpy
    arr = np.asarray(img)
    mode = {"reflect": "reflect", "edge": "edge", "symmetric": "symmetric"}[padding_mode]
    if arr.ndim == 3:
        arr = np.pad(arr, ((top, bottom), (left, right), (0, 0)), mode=mode)
    else:
        arr = np.pad(arr, ((top, bottom), (left, right)), mode=mode)
    return Image.fromarray(arr)


def crop(img, top, left, height, width):
    return img.crop((left, top, left + width, top + height))


def center_crop(img, output_size):
    if isinstance(output_size, numbers.Number):
        output_size = (int(output_size), int(output_size))
    w, h = img.size
    th, tw = output_size
    i = int(round((h - th) / 2.0))
    j = int(round((w - tw) / 2.0))
    return crop(img, i, j, th, tw)


def resize(img, size, interpolation=BILINEAR):
    """size: [h, w] -> exact; int -> shorter side scaled (torchvision
    semantics)."""
    if isinstance(size, (list, tuple)):
        h, w = int(size[0]), int(size[1])
        return img.resize((w, h), interpolation)
    w, h = img.size
    if (w <= h and w == size) or (h <= w and h == size):
        return img
    if w < h:
        ow = size
        oh = int(size * h / w)
    else:
        oh = size
        ow = int(size * w / h)
    return img.resize((ow, oh), interpolation)


def resized_crop(img, top, left, height, width, size, interpolation=BILINEAR):
    img = crop(img, top, left, height, width)
    if isinstance(size, numbers.Number):
        size = (int(size), int(size))
    return resize(img, list(size), interpolation)


def adjust_brightness(img, factor):
    return ImageEnhance.Brightness(img).enhance(factor)


def adjust_contrast(img, factor):
    return ImageEnhance.Contrast(img).enhance(factor)


def adjust_saturation(img, factor):
    return ImageEnhance.Color(img).enhance(factor)


def adjust_hue(img, hue_factor):
    """Shift hue by hue_factor in [-0.5, 0.5] (torchvision semantics)."""
    if not -0.5 <= hue_factor <= 0.5:
        raise ValueError("hue_factor is not in [-0.5, 0.5].")
    input_mode = img.mode
    if input_mode in {"L", "1", "I", "F"}:
        return img
    h, s, v = img.convert("HSV").split()
    np_h = np.array(h, dtype=np.uint8)
    # uint8 addition wraps around, matching hue circularity
    np_h += np.uint8(hue_factor * 255)
    h = Image.fromarray(np_h, "L")
    return Image.merge("HSV", (h, s, v)).convert(input_mode)
