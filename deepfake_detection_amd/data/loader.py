"""DataLoader factories + device-side prefetcher.

Capability parity with reference dfd/timm/data/loader.py: `fast_collate`
uint8 collation (:12-46), `PrefetchLoader_v3` side-stream H2D + cast +
normalize + GPU RandomErasing double-buffering one batch ahead (:213-289),
`create_deepfake_loader_v3` (:724-830), `create_loader` (:372).

MI355X-native design: the prefetcher runs on a dedicated HIP stream; the
uint8->bf16 cast + (x-mean)/std normalize + NCHW->NHWC (channels_last)
relayout is ONE fused HIP kernel (ops/hip/normalize.hip) instead of the
reference's cast/sub_/div_ chain (loader.py:246-253); output is
channels_last so the whole CNN runs NHWC.
"""

import numpy as np
import torch
import torch.utils.data

from .constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from .distributed_sampler import OrderedDistributedSampler
from .mixup import FastCollateMixup
from .random_erasing import RandomErasing
from .transforms_factory import (
    create_transform,
    transforms_deepfake_eval_v3,
    transforms_deepfake_train_v3,
)


def fast_collate(batch):
    """uint8 zero-copy collation of np arrays / tensors / tuples."""
    assert isinstance(batch[0], tuple)
    batch_size = len(batch)
    if isinstance(batch[0][0], tuple):
        inner_tuple_size = len(batch[0][0])
        flattened_batch_size = batch_size * inner_tuple_size
        targets = torch.zeros(flattened_batch_size, dtype=torch.int64)
        tensor = torch.zeros((flattened_batch_size, *batch[0][0][0].shape), dtype=torch.uint8)
        for i in range(batch_size):
            assert len(batch[i][0]) == inner_tuple_size
            for j in range(inner_tuple_size):
                targets[i + j * batch_size] = batch[i][1]
                tensor[i + j * batch_size] += torch.from_numpy(batch[i][0][j])
        return tensor, targets
    if isinstance(batch[0][0], np.ndarray):
        targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)
        tensor = torch.zeros((batch_size, *batch[0][0].shape), dtype=torch.uint8)
        for i in range(batch_size):
            tensor[i] += torch.from_numpy(batch[i][0])
        return tensor, targets
    if isinstance(batch[0][0], torch.Tensor):
        targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)
        tensor = torch.zeros((batch_size, *batch[0][0].shape), dtype=torch.uint8)
        for i in range(batch_size):
            tensor[i].copy_(batch[i][0])
        return tensor, targets
    raise AssertionError(type(batch[0][0]))


def _dtype_of(fp16, dtype):
    if dtype is not None:
        return {"float16": torch.float16, "bfloat16": torch.bfloat16,
                "float32": torch.float32}[str(dtype).replace("torch.", "")]
    return torch.float16 if fp16 else torch.float32


class PrefetchLoader_v3:
    """Device-side prefetcher for (B, 3*img_num, H, W) uint8 batches.

    Stages the next batch on a side stream: async H2D copy, fused
    uint8->dtype normalize (+channels_last), optional GPU RandomErasing.
    The consuming (main) stream waits on the side stream per batch
    (reference loader.py:242-266 semantics). On CPU machines it degrades to
    synchronous torch math so the pipeline stays testable.
    """

    def __init__(self, loader, mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD,
                 fp16=False, dtype=None, re_prob=0.0, re_mode="const", re_count=1,
                 re_num_splits=0, re_max=0.1, img_num=4, channels_last=True):
        self.loader = loader
        self.img_num = img_num
        self.dtype = _dtype_of(fp16, dtype)
        self.fp16 = self.dtype == torch.float16
        self.channels_last = channels_last
        self.use_cuda = torch.cuda.is_available()
        device = "cuda" if self.use_cuda else "cpu"

        mean_t = torch.tensor([x * 255 for x in mean] * img_num, device=device, dtype=torch.float32)
        std_t = torch.tensor([x * 255 for x in std] * img_num, device=device, dtype=torch.float32)
        self.mean = mean_t.view(1, 3 * img_num, 1, 1)
        self.std = std_t.view(1, 3 * img_num, 1, 1)

        if re_prob > 0.0:
            self.random_erasing = RandomErasing(
                probability=re_prob, max_area=re_max, mode=re_mode, max_count=re_count,
                num_splits=re_num_splits, img_num=img_num, device=device)
        else:
            self.random_erasing = None

    def _normalize(self, x_u8):
        from ..ops.prefetch_ops import normalize_uint8

        return normalize_uint8(
            x_u8, self.mean, self.std, out_dtype=self.dtype,
            channels_last=self.channels_last)

    def __iter__(self):
        if not self.use_cuda:
            for input, target in self.loader:
                input = self._normalize(input)
                if self.random_erasing is not None:
                    input = self.random_erasing(input)
                yield input, target
            return

        stream = torch.cuda.Stream()
        first = True
        input = target = None

        for next_input, next_target in self.loader:
            with torch.cuda.stream(stream):
                next_input = next_input.cuda(non_blocking=True)
                next_target = next_target.cuda(non_blocking=True)
                next_input = self._normalize(next_input)
                if self.random_erasing is not None:
                    next_input = self.random_erasing(next_input)

            if not first:
                yield input, target
            else:
                first = False

            torch.cuda.current_stream().wait_stream(stream)
            input = next_input
            target = next_target

        if input is not None:
            yield input, target

    def __len__(self):
        return len(self.loader)

    @property
    def sampler(self):
        return self.loader.sampler

    @property
    def dataset(self):
        return self.loader.dataset

    @property
    def mixup_enabled(self):
        if isinstance(self.loader.collate_fn, FastCollateMixup):
            return self.loader.collate_fn.mixup_enabled
        return False

    @mixup_enabled.setter
    def mixup_enabled(self, x):
        if isinstance(self.loader.collate_fn, FastCollateMixup):
            self.loader.collate_fn.mixup_enabled = x


# Generic single-image prefetcher (reference PrefetchLoader, loader.py:291)
class PrefetchLoader(PrefetchLoader_v3):
    def __init__(self, loader, mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD,
                 fp16=False, dtype=None, re_prob=0.0, re_mode="const", re_count=1,
                 re_num_splits=0, channels_last=True):
        super().__init__(
            loader, mean=mean, std=std, fp16=fp16, dtype=dtype, re_prob=re_prob,
            re_mode=re_mode, re_count=re_count, re_num_splits=re_num_splits,
            img_num=1, channels_last=channels_last)


# legacy name: the v1 prefetcher had the same side-stream copy/normalize
PrefetchLoader_v1 = PrefetchLoader


def create_deepfake_loader_v3(
        dataset, input_size, batch_size, is_training=False, use_prefetcher=True,
        re_prob=0.0, re_mode="const", re_count=1, re_split=False, re_max=0.02,
        color_jitter=0.4, auto_augment=None, num_aug_splits=0, interpolation="bilinear",
        mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD, num_workers=1,
        distributed=False, crop_pct=None, collate_fn=None, pin_memory=True,
        fp16=True, dtype=None, tf_preprocessing=False, has_gpu=True, flicker=0.0,
        rotate_range=0, noise_std=0, noise_prob=0, blur_radiu=0, blur_prob=0,
        persistent_workers=True):
    """Production loader: multi-image transform -> fast_collate ->
    PrefetchLoader_v3 (reference loader.py:724-830)."""
    re_num_splits = 0
    if re_split:
        re_num_splits = num_aug_splits or 2

    separate = num_aug_splits > 0
    if isinstance(input_size, (tuple, list)):
        img_size = input_size[-2:]
        if img_size[0] == img_size[1]:
            img_size = img_size[0]
    else:
        img_size = input_size

    if is_training:
        transform = transforms_deepfake_train_v3(
            img_size, color_jitter=color_jitter, use_prefetcher=use_prefetcher,
            flicker=flicker, rotate_range=rotate_range, re_prob=re_prob,
            re_mode=re_mode, re_count=re_count, re_num_splits=re_num_splits,
            noise_std=noise_std, noise_prob=noise_prob, blur_radiu=blur_radiu,
            blur_prob=blur_prob)
    else:
        assert not separate, "Separate transforms not supported for validation preprocessing"
        transform = transforms_deepfake_eval_v3(img_size, use_prefetcher=use_prefetcher)

    dataset.set_transform(transform)

    sampler = None
    if distributed:
        if is_training:
            sampler = torch.utils.data.distributed.DistributedSampler(dataset)
        else:
            sampler = OrderedDistributedSampler(dataset)

    if collate_fn is None:
        collate_fn = fast_collate if use_prefetcher else torch.utils.data.dataloader.default_collate

    loader = torch.utils.data.DataLoader(
        dataset,
        batch_size=batch_size,
        shuffle=sampler is None and is_training,
        num_workers=num_workers,
        sampler=sampler,
        collate_fn=collate_fn,
        pin_memory=pin_memory,
        drop_last=is_training,
        persistent_workers=persistent_workers and num_workers > 0,
    )

    if use_prefetcher:
        img_num = int(input_size[0] / 3) if isinstance(input_size, (tuple, list)) else 4
        loader = PrefetchLoader_v3(
            loader, mean=mean, std=std, fp16=fp16, dtype=dtype,
            re_prob=re_prob if is_training else 0.0, re_mode=re_mode,
            re_count=re_count, re_num_splits=re_num_splits, re_max=re_max,
            img_num=img_num)

    return loader


def create_loader(
        dataset, input_size, batch_size, is_training=False, use_prefetcher=True,
        re_prob=0.0, re_mode="const", re_count=1, re_split=False, color_jitter=0.4,
        interpolation="bilinear", mean=IMAGENET_DEFAULT_MEAN, std=IMAGENET_DEFAULT_STD,
        num_workers=1, distributed=False, crop_pct=None, collate_fn=None,
        pin_memory=True, fp16=False, dtype=None, tf_preprocessing=False,
        auto_augment=None, num_aug_splits=0, persistent_workers=True):
    """Generic single-image loader (reference loader.py:372). With
    num_aug_splits>1 the dataset must be an AugMixDataset: the train
    transform is built as a (primary, secondary, final) triple."""
    re_num_splits = 0
    if re_split:
        re_num_splits = num_aug_splits or 2

    transform = create_transform(
        input_size, is_training=is_training, use_prefetcher=use_prefetcher,
        color_jitter=color_jitter, auto_augment=auto_augment,
        interpolation=interpolation, mean=mean, std=std,
        crop_pct=crop_pct, tf_preprocessing=tf_preprocessing,
        separate=num_aug_splits > 1 and is_training)
    if hasattr(dataset, "set_transform"):
        dataset.set_transform(transform)
    else:
        dataset.transform = transform

    sampler = None
    if distributed:
        if is_training:
            sampler = torch.utils.data.distributed.DistributedSampler(dataset)
        else:
            sampler = OrderedDistributedSampler(dataset)

    if collate_fn is None:
        collate_fn = fast_collate if use_prefetcher else torch.utils.data.dataloader.default_collate

    loader = torch.utils.data.DataLoader(
        dataset,
        batch_size=batch_size,
        shuffle=sampler is None and is_training,
        num_workers=num_workers,
        sampler=sampler,
        collate_fn=collate_fn,
        pin_memory=pin_memory,
        drop_last=is_training,
        persistent_workers=persistent_workers and num_workers > 0,
    )
    if use_prefetcher:
        loader = PrefetchLoader(
            loader, mean=mean, std=std, fp16=fp16, dtype=dtype,
            re_prob=re_prob if is_training else 0.0, re_mode=re_mode,
            re_count=re_count, re_num_splits=re_num_splits)
    return loader


def fast_collate_v1(batch):
    """Pair-interleave collate for DeepFakeDataset_v1_bak items
    (fake_img, real_img, fake_rot, real_rot) -> shuffled (2B, ...) uint8
    batch with alternating labels (reference loader.py:48-99)."""
    assert isinstance(batch[0], tuple)
    batch_size = len(batch)
    first = batch[0][0]
    if isinstance(first, np.ndarray):
        targets = torch.tensor([i % 2 for i in range(2 * batch_size)], dtype=torch.int64)
        tensor = torch.zeros((batch_size * 2, *first.shape), dtype=torch.uint8)
        for i in range(batch_size):
            tensor[2 * i] += torch.from_numpy(batch[i][0])
            tensor[2 * i + 1] += torch.from_numpy(batch[i][1])
        perm = torch.randperm(tensor.size(0))
        return tensor[perm], targets[perm]
    if isinstance(first, torch.Tensor):
        targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)
        tensor = torch.zeros((batch_size, *first.shape), dtype=torch.uint8)
        for i in range(batch_size):
            tensor[i].copy_(batch[i][0])
        return tensor, targets
    raise TypeError(f"fast_collate_v1: unsupported item type {type(first)}")


def _create_deepfake_loader_generic(
        dataset, input_size, batch_size, is_training=False, use_prefetcher=True,
        re_prob=0.0, re_mode="const", re_count=1, color_jitter=0.4,
        interpolation="bilinear", mean=IMAGENET_DEFAULT_MEAN,
        std=IMAGENET_DEFAULT_STD, num_workers=1, distributed=False,
        crop_pct=None, collate_fn=None, pin_memory=True, fp16=False, dtype=None):
    """Shared body of the legacy single-image loader factories
    (reference loader.py:457,543,633): imagenet-style transform +
    fast_collate + single-image PrefetchLoader."""
    from .transforms_factory import transforms_imagenet_eval, transforms_imagenet_train

    img_size = input_size[-1] if isinstance(input_size, (tuple, list)) else input_size
    if is_training:
        transform = transforms_imagenet_train(
            img_size, color_jitter=color_jitter, interpolation=interpolation,
            use_prefetcher=use_prefetcher, mean=mean, std=std)
    else:
        transform = transforms_imagenet_eval(
            img_size, interpolation=interpolation, use_prefetcher=use_prefetcher,
            mean=mean, std=std, crop_pct=crop_pct)
    if hasattr(dataset, "set_transform"):
        dataset.set_transform(transform)
    else:
        dataset.transform = transform

    sampler = None
    if distributed:
        if is_training:
            sampler = torch.utils.data.distributed.DistributedSampler(dataset)
        else:
            sampler = OrderedDistributedSampler(dataset)
    if collate_fn is None:
        collate_fn = fast_collate if use_prefetcher else torch.utils.data.dataloader.default_collate

    loader = torch.utils.data.DataLoader(
        dataset, batch_size=batch_size, shuffle=sampler is None and is_training,
        num_workers=num_workers, sampler=sampler, collate_fn=collate_fn,
        pin_memory=pin_memory, drop_last=is_training)
    if use_prefetcher:
        loader = PrefetchLoader(
            loader, mean=mean, std=std, fp16=fp16, dtype=dtype,
            re_prob=re_prob if is_training else 0.0, re_mode=re_mode,
            re_count=re_count, img_num=1)
    return loader


def create_deepfake_loader(dataset, input_size, batch_size, **kwargs):
    """Legacy v0 factory (reference loader.py:457)."""
    return _create_deepfake_loader_generic(dataset, input_size, batch_size, **kwargs)


def create_deepfake_loader_v1(dataset, input_size, batch_size, **kwargs):
    """Legacy paired factory (reference loader.py:543): fast_collate_v1
    deinterleaves (fake, real) pairs into a shuffled 2B batch."""
    kwargs.setdefault("collate_fn", fast_collate_v1)
    return _create_deepfake_loader_generic(dataset, input_size, batch_size, **kwargs)


def create_deepfake_loader_v2(dataset, input_size, batch_size, **kwargs):
    """Legacy v2 factory (reference loader.py:633): single-image dataset with
    per-root list files (DeepFakeDataset_v2)."""
    return _create_deepfake_loader_generic(dataset, input_size, batch_size, **kwargs)
