"""Single- and multi-image (frame-group) transforms — torchvision-free.

Capability parity with reference dfd/timm/data/transforms.py: the Multi*
transforms apply ONE random draw to all 4 frames of a group (MultiRotate
:261, MultiRandomHorizontalFlip :217, MultiRandomResize :281,
MultiRandomCrop :311, MultiBlur :243, MultiColorJitter :332, MultiFlicker
:346, MultiToNumpy :20, MultiConcate :29) plus the single-image
RandomResizedCropAndInterpolation (:73), RandomResize (:173),
ToNumpy/ToTensor (:10,:35). This image ships no torchvision, so the base
transform classes (Compose, RandomCrop, ColorJitter, Resize, CenterCrop,
Normalize, ...) are implemented here on PIL/numpy (data/pil_functional.py).
"""

import math
import numbers
import random
import warnings

import numpy as np
import torch
from PIL import Image, ImageFilter

from . import pil_functional as F


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, img):
        for t in self.transforms:
            img = t(img)
        return img

    def __repr__(self):
        return self.__class__.__name__ + "(" + ", ".join(repr(t) for t in self.transforms) + ")"


class ToNumpy:
    def __call__(self, pil_img):
        np_img = np.array(pil_img, dtype=np.uint8)
        if np_img.ndim < 3:
            np_img = np.expand_dims(np_img, axis=-1)
        return np.rollaxis(np_img, 2)  # HWC -> CHW


class MultiToNumpy:
    def __call__(self, pil_imgs):
        np_imgs = [np.array(img, dtype=np.uint8) for img in pil_imgs]
        if np_imgs[0].ndim < 3:
            np_imgs = [np.expand_dims(img, axis=-1) for img in np_imgs]
        return [np.rollaxis(img, 2) for img in np_imgs]


class MultiConcate:
    """Stack the 4 CHW frames along channels -> (12, H, W)."""

    def __call__(self, np_imgs):
        return np.concatenate(np_imgs, axis=0)


class ToTensor:
    def __init__(self, dtype=torch.float32):
        self.dtype = dtype

    def __call__(self, pil_img):
        np_img = np.array(pil_img, dtype=np.uint8)
        if np_img.ndim < 3:
            np_img = np.expand_dims(np_img, axis=-1)
        np_img = np.rollaxis(np_img, 2)
        return torch.from_numpy(np_img).to(dtype=self.dtype)


class ToTensorNormalized:
    """PIL -> float CHW in [0,1] (torchvision ToTensor semantics)."""

    def __call__(self, pil_img):
        np_img = np.array(pil_img, dtype=np.uint8)
        if np_img.ndim < 3:
            np_img = np.expand_dims(np_img, axis=-1)
        np_img = np.rollaxis(np_img, 2)
        return torch.from_numpy(np_img.astype(np.float32) / 255.0)


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.as_tensor(mean, dtype=torch.float32).view(-1, 1, 1)
        self.std = torch.as_tensor(std, dtype=torch.float32).view(-1, 1, 1)

    def __call__(self, tensor):
        return (tensor - self.mean) / self.std


class Resize:
    def __init__(self, size, interpolation=F.BILINEAR):
        self.size = size
        self.interpolation = interpolation

    def __call__(self, img):
        return F.resize(img, self.size, self.interpolation)


class CenterCrop:
    def __init__(self, size):
        self.size = size

    def __call__(self, img):
        return F.center_crop(img, self.size)


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            return F.hflip(img)
        return img


def _pil_interp(method):
    if method == "bicubic":
        return F.BICUBIC
    if method == "lanczos":
        return F.LANCZOS
    if method == "hamming":
        return F.HAMMING
    return F.BILINEAR


_RANDOM_INTERPOLATION = (F.BILINEAR, F.BICUBIC)


class RandomCrop:
    """Random crop with optional padding / pad-if-needed (torchvision
    semantics, implemented on PIL)."""

    def __init__(self, size, padding=None, pad_if_needed=False, fill=0,
                 padding_mode="constant"):
        if isinstance(size, numbers.Number):
            self.size = (int(size), int(size))
        else:
            self.size = tuple(size)
        self.padding = padding
        self.pad_if_needed = pad_if_needed
        self.fill = fill
        self.padding_mode = padding_mode

    @staticmethod
    def get_params(img, output_size):
        w, h = img.size
        th, tw = output_size
        if w == tw and h == th:
            return 0, 0, h, w
        i = random.randint(0, h - th)
        j = random.randint(0, w - tw)
        return i, j, th, tw

    def __call__(self, img):
        if self.padding is not None:
            img = F.pad(img, self.padding, self.fill, self.padding_mode)
        if self.pad_if_needed and img.size[0] < self.size[1]:
            img = F.pad(img, (self.size[1] - img.size[0], 0), self.fill, self.padding_mode)
        if self.pad_if_needed and img.size[1] < self.size[0]:
            img = F.pad(img, (0, self.size[0] - img.size[1]), self.fill, self.padding_mode)
        i, j, h, w = self.get_params(img, self.size)
        return F.crop(img, i, j, h, w)


class ColorJitter:
    """Brightness/contrast/saturation/hue jitter with a shared parameter
    draw exposed via get_params (torchvision semantics)."""

    def __init__(self, brightness=0, contrast=0, saturation=0, hue=0):
        self.brightness = self._check_input(brightness, "brightness")
        self.contrast = self._check_input(contrast, "contrast")
        self.saturation = self._check_input(saturation, "saturation")
        self.hue = self._check_input(hue, "hue", center=0, bound=(-0.5, 0.5),
                                     clip_first_on_zero=False)

    @staticmethod
    def _check_input(value, name, center=1, bound=(0, float("inf")), clip_first_on_zero=True):
        if isinstance(value, numbers.Number):
            if value < 0:
                raise ValueError(f"If {name} is a single number, it must be non negative.")
            value = [center - value, center + value]
            if clip_first_on_zero:
                value[0] = max(value[0], 0)
        elif isinstance(value, (tuple, list)) and len(value) == 2:
            value = list(value)
        else:
            raise TypeError(f"{name} should be a single number or a pair.")
        if value[0] == value[1] == center:
            return None
        return value

    @staticmethod
    def get_params(brightness, contrast, saturation, hue):
        fn_idx = list(range(4))
        random.shuffle(fn_idx)
        b = None if brightness is None else random.uniform(brightness[0], brightness[1])
        c = None if contrast is None else random.uniform(contrast[0], contrast[1])
        s = None if saturation is None else random.uniform(saturation[0], saturation[1])
        h = None if hue is None else random.uniform(hue[0], hue[1])
        return fn_idx, b, c, s, h

    def _apply(self, img, fn_idx, b, c, s, h):
        for fn_id in fn_idx:
            if fn_id == 0 and b is not None:
                img = F.adjust_brightness(img, b)
            elif fn_id == 1 and c is not None:
                img = F.adjust_contrast(img, c)
            elif fn_id == 2 and s is not None:
                img = F.adjust_saturation(img, s)
            elif fn_id == 3 and h is not None:
                img = F.adjust_hue(img, h)
        return img

    def __call__(self, img):
        params = self.get_params(self.brightness, self.contrast, self.saturation, self.hue)
        return self._apply(img, *params)


class RandomResizedCropAndInterpolation:
    """Random-area crop + resize with (optionally random) interpolation."""

    def __init__(self, size, scale=(0.08, 1.0), ratio=(3.0 / 4.0, 4.0 / 3.0),
                 interpolation="bilinear"):
        if isinstance(size, tuple):
            self.size = size
        else:
            self.size = (size, size)
        if (scale[0] > scale[1]) or (ratio[0] > ratio[1]):
            warnings.warn("range should be of kind (min, max)")
        if interpolation == "random":
            self.interpolation = _RANDOM_INTERPOLATION
        else:
            self.interpolation = _pil_interp(interpolation)
        self.scale = scale
        self.ratio = ratio

    @staticmethod
    def get_params(img, scale, ratio):
        area = img.size[0] * img.size[1]
        for _ in range(10):
            target_area = random.uniform(*scale) * area
            log_ratio = (math.log(ratio[0]), math.log(ratio[1]))
            aspect_ratio = math.exp(random.uniform(*log_ratio))
            w = int(round(math.sqrt(target_area * aspect_ratio)))
            h = int(round(math.sqrt(target_area / aspect_ratio)))
            if w <= img.size[0] and h <= img.size[1]:
                i = random.randint(0, img.size[1] - h)
                j = random.randint(0, img.size[0] - w)
                return i, j, h, w
        in_ratio = img.size[0] / img.size[1]
        if in_ratio < min(ratio):
            w = img.size[0]
            h = int(round(w / min(ratio)))
        elif in_ratio > max(ratio):
            h = img.size[1]
            w = int(round(h * max(ratio)))
        else:
            w = img.size[0]
            h = img.size[1]
        i = (img.size[1] - h) // 2
        j = (img.size[0] - w) // 2
        return i, j, h, w

    def __call__(self, img):
        i, j, h, w = self.get_params(img, self.scale, self.ratio)
        if isinstance(self.interpolation, (tuple, list)):
            interpolation = random.choice(self.interpolation)
        else:
            interpolation = self.interpolation
        return F.resized_crop(img, i, j, h, w, self.size, interpolation)


class RandomResize:
    """Random uniform rescale of the whole image."""

    def __init__(self, scale=(0.9, 1.1), interpolation="bilinear"):
        if interpolation == "random":
            self.interpolation = _RANDOM_INTERPOLATION
        else:
            self.interpolation = _pil_interp(interpolation)
        self.scale = scale

    def __call__(self, img):
        if isinstance(self.interpolation, (tuple, list)):
            interpolation = random.choice(self.interpolation)
        else:
            interpolation = self.interpolation
        random_scale = random.uniform(self.scale[0], self.scale[1])
        w, h = img.size
        target_size = [int(h * random_scale), int(w * random_scale)]
        return F.resize(img, target_size, interpolation)


class MultiRandomHorizontalFlip:
    """One coin flip for the whole group."""

    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, imgs):
        if random.random() < self.p:
            return [F.hflip(img) for img in imgs]
        return imgs


class MultiBlur:
    """Per-frame independent Gaussian blur (reference transforms.py:243)."""

    def __init__(self, p, blur_radiu):
        self.p = p
        self.blur_radiu = blur_radiu

    def __call__(self, imgs):
        return [
            img.filter(ImageFilter.GaussianBlur(radius=self.blur_radiu))
            if random.random() < self.p else img
            for img in imgs
        ]


class MultiRotate:
    """One shared random rotation (expand=True) for the whole group."""

    def __init__(self, rotate_range):
        self.rotate_range = rotate_range

    def __call__(self, imgs):
        rotate_degree = random.randint(-self.rotate_range, self.rotate_range)
        return [img.rotate(rotate_degree, expand=True) for img in imgs]


class MultiRandomResize(RandomResize):
    """One shared random scale for the whole group."""

    def __call__(self, imgs):
        if isinstance(self.interpolation, (tuple, list)):
            interpolation = random.choice(self.interpolation)
        else:
            interpolation = self.interpolation
        random_scale = random.uniform(self.scale[0], self.scale[1])
        w, h = imgs[0].size
        target_size = [int(h * random_scale), int(w * random_scale)]
        return [F.resize(img, target_size, interpolation) for img in imgs]


class MultiRandomCrop(RandomCrop):
    """One shared crop window for the whole group, pad-if-needed."""

    def __call__(self, imgs):
        if self.padding is not None:
            imgs = [F.pad(img, self.padding, self.fill, self.padding_mode) for img in imgs]
        if self.pad_if_needed and imgs[0].size[0] < self.size[1]:
            imgs = [F.pad(img, (self.size[1] - img.size[0], 0), self.fill, self.padding_mode)
                    for img in imgs]
        if self.pad_if_needed and imgs[0].size[1] < self.size[0]:
            imgs = [F.pad(img, (0, self.size[0] - img.size[1]), self.fill, self.padding_mode)
                    for img in imgs]
        i, j, h, w = self.get_params(imgs[0], self.size)
        return [F.crop(img, i, j, h, w) for img in imgs]


class MultiColorJitter(ColorJitter):
    """One shared jitter parameter draw for the whole group."""

    def __call__(self, imgs):
        params = self.get_params(self.brightness, self.contrast, self.saturation, self.hue)
        return [self._apply(img, *params) for img in imgs]


class MultiFlicker:
    """Independently replace frames with a black image (simulated flicker,
    reference transforms.py:346)."""

    def __init__(self, probability):
        self.probability = probability

    def __call__(self, imgs):
        img_size = imgs[0].size
        return [
            Image.new("RGB", img_size[:2]) if random.random() < self.probability else img
            for img in imgs
        ]
