"""Inference preprocessing + model wrapper.

Parity: reference dfd/params.py — `img_num=4` (:31), `DeepFakeModel`
softmax wrapper (:34-42), `resize` aspect-preserving fit into 600x600
(:45-55), `padding_image` center zero-pad (:58-68), ImageNet mean/std x255
constants (:24-27). cv2 is not in this image; resize uses PIL bilinear.
"""

import numpy as np
import torch
import torch.nn as nn
from PIL import Image

img_size = 600
img_num = 4

mean = [0.485, 0.456, 0.406]
std = [0.229, 0.224, 0.225]
mean_255 = np.array([x * 255 for x in mean], dtype=np.float32).reshape(3, 1, 1)
std_255 = np.array([x * 255 for x in std], dtype=np.float32).reshape(3, 1, 1)


class DeepFakeModel(nn.Module):
    """base model + softmax(-1); scores[:, 0] = fake probability."""

    def __init__(self, basemodel):
        super().__init__()
        self.model = basemodel
        self.softmax = nn.Softmax(dim=-1)

    def forward(self, x):
        x = self.model(x)
        return self.softmax(x)


def resize(img, target_size=img_size):
    """Aspect-preserving resize so the longer side == target_size.

    img: HWC uint8 numpy array. Returns HWC uint8 numpy array.
    """
    h, w = img.shape[:2]
    scale = target_size / max(h, w)
    new_h, new_w = int(round(h * scale)), int(round(w * scale))
    pil = Image.fromarray(img)
    pil = pil.resize((new_w, new_h), Image.BILINEAR)
    return np.asarray(pil)


def padding_image(img, target_size=img_size):
    """Center zero-pad an HWC image up to (target_size, target_size)."""
    h, w = img.shape[:2]
    c = img.shape[2] if img.ndim == 3 else 1
    out = np.zeros((target_size, target_size, c), dtype=img.dtype)
    top = (target_size - h) // 2
    left = (target_size - w) // 2
    out[top:top + h, left:left + w] = img if img.ndim == 3 else img[..., None]
    return out


def preprocess_image(img_hwc_uint8, device="cuda", dtype=torch.float16):
    """Full inference preprocessing of ONE image to a (1, 12, 600, 600)
    tensor (reference test.py:49-58): resize -> center pad -> CHW ->
    normalize -> stack x4 frames into 12 channels."""
    arr = padding_image(resize(img_hwc_uint8))
    chw = np.rollaxis(arr, 2).astype(np.float32)
    chw = (chw - mean_255) / std_255
    t = torch.from_numpy(chw).to(device=device, dtype=dtype)
    t = torch.cat([t] * img_num, dim=0)
    return t.unsqueeze(0)
