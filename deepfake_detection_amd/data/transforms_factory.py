"""Transform pipelines.

Capability parity with reference dfd/timm/data/transforms_factory.py:
`transforms_deepfake_train_v3` (:137-183), `transforms_deepfake_eval_v3`
(:225-236), imagenet train/eval pipelines (:239,:321), `create_transform`.
"""

import math

import torch

from .constants import DEFAULT_CROP_PCT, IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from .random_erasing import RandomErasing
from . import transforms
from .transforms import (
    MultiBlur,
    MultiColorJitter,
    MultiConcate,
    MultiFlicker,
    MultiRandomCrop,
    MultiRandomHorizontalFlip,
    MultiRandomResize,
    MultiRotate,
    MultiToNumpy,
    RandomResizedCropAndInterpolation,
    ToNumpy,
    _pil_interp,
)


def transforms_deepfake_train_v3(
        img_size=600, color_jitter=0.4, use_prefetcher=True, flicker=0.0,
        rotate_range=0, re_prob=0, re_mode="pixel", re_max=0.05, re_count=1,
        re_num_splits=0, noise_std=0, noise_prob=0, blur_radiu=0, blur_prob=0):
    """The production train pipeline: rotate -> hflip -> random resize
    (2/3..3/2) -> random crop (pad-if-needed) [-> blur] [-> color jitter]
    [-> flicker] -> numpy CHW -> concat to (12,H,W). Normalization and
    random-erasing run ON DEVICE in the prefetcher."""
    primary_tfl = [
        MultiRotate(rotate_range),
        MultiRandomHorizontalFlip(),
        MultiRandomResize(scale=(2.0 / 3, 3.0 / 2.0)),
        MultiRandomCrop(img_size, pad_if_needed=True),
    ]
    if blur_prob > 0.0:
        primary_tfl.append(MultiBlur(blur_prob, blur_radiu))

    secondary_tfl = []
    if color_jitter is not None:
        if isinstance(color_jitter, (list, tuple)):
            assert len(color_jitter) in (3, 4)
        else:
            color_jitter = (float(color_jitter),) * 3
        secondary_tfl += [MultiColorJitter(*color_jitter)]
    if flicker > 0.0:
        secondary_tfl += [MultiFlicker(flicker)]

    final_tfl = [MultiToNumpy(), MultiConcate()]
    assert use_prefetcher, "deepfake v3 train path requires the device-side prefetcher"
    return transforms.Compose(primary_tfl + secondary_tfl + final_tfl)


def transforms_deepfake_eval_v3(img_size=224, use_prefetcher=True):
    assert use_prefetcher, "deepfake v3 eval path requires the device-side prefetcher"
    return transforms.Compose([
        MultiRandomCrop(img_size, pad_if_needed=True),
        MultiToNumpy(),
        MultiConcate(),
    ])


def transforms_imagenet_train(
        img_size=224, scale=(0.08, 1.0), color_jitter=0.4, interpolation="random",
        auto_augment=None, random_erasing=0.0, random_erasing_mode="const",
        use_prefetcher=False, mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD,
        separate=False):
    """`separate=True` returns the (primary, secondary, final) transform
    triple AugMixDataset consumes (reference transforms_factory.py:239-318):
    primary = crop+flip, secondary = AA/jitter, final = tensorize+normalize."""
    primary_tfl = [
        RandomResizedCropAndInterpolation(img_size, scale=scale, interpolation=interpolation),
        transforms.RandomHorizontalFlip(),
    ]
    tfl = [] if separate else primary_tfl
    if auto_augment:
        # AA/RandAugment/AugMix replace color jitter
        # (reference transforms_factory.py:269-287)
        from .auto_augment import (
            augment_and_mix_transform,
            auto_augment_transform,
            rand_augment_transform,
        )

        assert isinstance(auto_augment, str)
        img_size_min = min(img_size) if isinstance(img_size, (tuple, list)) else img_size
        aa_params = dict(
            translate_const=int(img_size_min * 0.45),
            img_mean=tuple(int(round(255 * x)) for x in mean),
        )
        if interpolation and interpolation != "random":
            aa_params["interpolation"] = _pil_interp(interpolation)
        if auto_augment.startswith("rand"):
            tfl += [rand_augment_transform(auto_augment, aa_params)]
        elif auto_augment.startswith("augmix"):
            aa_params["translate_pct"] = 0.3
            tfl += [augment_and_mix_transform(auto_augment, aa_params)]
        else:
            tfl += [auto_augment_transform(auto_augment, aa_params)]
        color_jitter = None
    if color_jitter is not None:
        if isinstance(color_jitter, (list, tuple)):
            assert len(color_jitter) in (3, 4)
        else:
            color_jitter = (float(color_jitter),) * 3
        tfl += [transforms.ColorJitter(*color_jitter)]

    final_tfl = []
    if use_prefetcher:
        final_tfl += [ToNumpy()]
    else:
        final_tfl += [
            transforms.ToTensorNormalized(),
            transforms.Normalize(mean=torch.tensor(mean), std=torch.tensor(std)),
        ]
        if random_erasing > 0.0:
            final_tfl.append(
                RandomErasing(random_erasing, mode=random_erasing_mode, device="cpu"))
    if separate:
        return (transforms.Compose(primary_tfl), transforms.Compose(tfl),
                transforms.Compose(final_tfl))
    return transforms.Compose(tfl + final_tfl)


def transforms_imagenet_eval(
        img_size=224, crop_pct=None, interpolation="bilinear", use_prefetcher=False,
        mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD):
    crop_pct = crop_pct or DEFAULT_CROP_PCT
    if isinstance(img_size, tuple):
        assert len(img_size) == 2
        if img_size[-1] == img_size[-2]:
            scale_size = int(math.floor(img_size[0] / crop_pct))
        else:
            scale_size = tuple([int(x / crop_pct) for x in img_size])
    else:
        scale_size = int(math.floor(img_size / crop_pct))

    tfl = [
        transforms.Resize(scale_size, _pil_interp(interpolation)),
        transforms.CenterCrop(img_size),
    ]
    if use_prefetcher:
        tfl += [ToNumpy()]
    else:
        tfl += [
            transforms.ToTensorNormalized(),
            transforms.Normalize(mean=torch.tensor(mean), std=torch.tensor(std)),
        ]
    return transforms.Compose(tfl)


def create_transform(
        input_size, is_training=False, use_prefetcher=False, color_jitter=0.4,
        auto_augment=None, interpolation="bilinear", mean=IMAGENET_DEFAULT_MEAN,
        std=IMAGENET_DEFAULT_STD, crop_pct=None, tf_preprocessing=False,
        separate=False):
    if isinstance(input_size, tuple):
        img_size = input_size[-2:]
    else:
        img_size = input_size

    if tf_preprocessing and use_prefetcher:
        # TF preprocessing emits uint8 CHW (reference contract,
        # transforms_factory.py:507-511): the PrefetchLoader normalizes on
        # device, so this path requires the prefetcher.
        from .tf_preprocessing import TfPreprocessTransform

        return TfPreprocessTransform(is_training=is_training, size=img_size,
                                     interpolation=interpolation)
    if is_training:
        return transforms_imagenet_train(
            img_size, color_jitter=color_jitter, auto_augment=auto_augment,
            interpolation=interpolation if interpolation != "bilinear" else "random",
            use_prefetcher=use_prefetcher, mean=mean, std=std, separate=separate)
    return transforms_imagenet_eval(
        img_size, interpolation=interpolation, use_prefetcher=use_prefetcher,
        mean=mean, std=std, crop_pct=crop_pct)
