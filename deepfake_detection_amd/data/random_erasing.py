"""GPU batch RandomErasing, img_num-aware.

Parity: reference dfd/timm/data/random_erasing.py — erases each 3-channel
frame slice of the (B, 3*img_num, H, W) batch independently (:96-100);
const/rand/pixel fill modes. Runs post-normalization on the prefetch
stream. Rectangle selection is host-side RNG (cheap); the fills are device
tensor writes.

Structure (this implementation): a single `_rects` generator yields the
accepted (top, left, h, w) rectangles for one image; `__call__` iterates
the frame slices of the batch and fills each rectangle with the mode's
fill tensor.
"""

import math
import random

import torch


class RandomErasing:
    """Random-erase rectangles per image (per frame slice when img_num>1)."""

    def __init__(self, probability=0.5, min_area=0.02, max_area=1 / 3,
                 min_aspect=0.3, max_aspect=None, mode="const", min_count=1,
                 max_count=None, num_splits=0, device="cuda", img_num=1):
        self.probability = probability
        self.min_area = min_area
        self.max_area = max_area
        max_aspect = max_aspect or 1 / min_aspect
        self.log_aspect_ratio = (math.log(min_aspect), math.log(max_aspect))
        self.min_count = min_count
        self.max_count = max_count or min_count
        self.num_splits = num_splits
        mode = mode.lower()
        assert mode in ("rand", "pixel", "const", "")
        self.mode = mode or "const"
        self.device = device
        self.img_num = img_num

    def _fill(self, chan, h, w, dtype, device):
        if self.mode == "pixel":
            return torch.empty((chan, h, w), dtype=dtype, device=device).normal_()
        if self.mode == "rand":
            return torch.empty((chan, 1, 1), dtype=dtype, device=device).normal_()
        return torch.zeros((chan, 1, 1), dtype=dtype, device=device)

    def _rects(self, img_h, img_w):
        """Accepted rectangles for one image (possibly none)."""
        if random.random() > self.probability:
            return
        count = (self.min_count if self.min_count == self.max_count
                 else random.randint(self.min_count, self.max_count))
        budget = img_h * img_w / count
        for _ in range(count):
            for _attempt in range(10):
                area = random.uniform(self.min_area, self.max_area) * budget
                aspect = math.exp(random.uniform(*self.log_aspect_ratio))
                h = int(round(math.sqrt(area * aspect)))
                w = int(round(math.sqrt(area / aspect)))
                if h < img_h and w < img_w:
                    yield (random.randint(0, img_h - h),
                           random.randint(0, img_w - w), h, w)
                    break

    def _erase(self, img, chan, img_h, img_w, dtype):
        for top, left, h, w in self._rects(img_h, img_w):
            img[:, top:top + h, left:left + w] = self._fill(
                chan, h, w, dtype, img.device)

    def _frame_slices(self, sample, chan):
        """The independent erase targets of one sample: the whole image, or
        each img_num-th channel group (one video frame) separately."""
        if self.img_num == 1:
            yield sample, chan
        else:
            per = chan // self.img_num
            for f in range(self.img_num):
                yield sample[f * per:(f + 1) * per], per

    def __call__(self, input):
        if input.dim() == 3:
            self._erase(input, *input.size(), input.dtype)
            return input
        batch_size, chan, img_h, img_w = input.size()
        # with aug-splits the first (clean) split is left unerased
        first = batch_size // self.num_splits if self.num_splits > 1 else 0
        for i in range(first, batch_size):
            for view, vchan in self._frame_slices(input[i], chan):
                self._erase(view, vchan, img_h, img_w, input.dtype)
        return input
