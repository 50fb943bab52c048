"""Mixup augmentation.

Capability parity with reference dfd/timm/data/mixup.py: smoothed one-hot
target construction, tensor-level batch mixing against the flipped batch,
and a collate-time uint8 variant for the prefetcher path (which owns
normalization on the GPU side, so collate must stay in uint8).

Design notes (this implementation): the pairing is always `i <-> B-1-i`
(flip), one lambda per batch drawn from Beta(alpha, alpha); the collate
variant mixes whole stacked arrays vectorized in fp32 and rounds back to
uint8 in a single pass rather than per-sample.
"""

import numpy as np
import torch


def one_hot(x, num_classes, on_value=1.0, off_value=0.0, device="cuda"):
    """Smoothed one-hot rows: off_value everywhere, on_value at the label."""
    idx = x.long().view(-1, 1)
    out = torch.full((idx.shape[0], num_classes), off_value, device=device)
    return out.scatter_(1, idx, on_value)


def _smooth_values(num_classes, smoothing):
    off = smoothing / num_classes
    return 1.0 - smoothing + off, off


def mixup_target(target, num_classes, lam=1.0, smoothing=0.0, device="cuda"):
    """lam-weighted blend of the smoothed one-hots of target and its flip."""
    on, off = _smooth_values(num_classes, smoothing)
    fwd = one_hot(target, num_classes, on, off, device)
    rev = one_hot(target.flip(0), num_classes, on, off, device)
    return fwd.mul(lam).add_(rev, alpha=1.0 - lam)


def mixup_batch(input, target, alpha=0.2, num_classes=1000, smoothing=0.1, disable=False):
    """Device-tensor mixup: x <- lam*x + (1-lam)*flip(x), soft targets."""
    lam = 1.0 if disable else float(np.random.beta(alpha, alpha))
    input = input.mul(lam).add_(input.flip(0), alpha=1.0 - lam)
    return input, mixup_target(target, num_classes, lam, smoothing, device=input.device)


class FastCollateMixup:
    """Collate-time mixup over uint8 numpy frames (prefetcher path).

    Keeps the batch in uint8 — the PrefetchLoader casts/normalizes on a side
    HIP stream — so mixing happens in fp32 here and is rounded back.
    """

    def __init__(self, mixup_alpha=1.0, label_smoothing=0.1, num_classes=1000):
        self.mixup_alpha = mixup_alpha
        self.label_smoothing = label_smoothing
        self.num_classes = num_classes
        self.mixup_enabled = True

    def _draw_lam(self):
        if not self.mixup_enabled:
            return 1.0
        return float(np.random.beta(self.mixup_alpha, self.mixup_alpha))

    def __call__(self, batch):
        lam = self._draw_lam()
        labels = torch.tensor([sample[1] for sample in batch], dtype=torch.int64)
        soft = mixup_target(labels, self.num_classes, lam, self.label_smoothing,
                            device="cpu")

        frames = np.stack([np.asarray(sample[0], dtype=np.float32) for sample in batch])
        mixed = frames * lam + frames[::-1] * (1.0 - lam)
        np.round(mixed, out=mixed)
        return torch.from_numpy(mixed.astype(np.uint8)), soft
