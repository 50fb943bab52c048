"""Filesystem + GPU-selection helpers.

Parity: reference dfd/utils.py — `get_proper_gpu` picks the GPU with most
free memory (reference uses `nvidia-smi -x -q` + xmltodict, :14-54; here
rocm-smi / torch.cuda.mem_get_info on ROCm), `new_dir`/`check_file`/
`del_file` (:57-80).
"""

import json
import os
import shutil
import subprocess

__all__ = ["get_gpu_free_memory", "get_proper_gpu", "new_dir", "check_file",
           "del_file", "copy_file"]


def get_gpu_free_memory():
    """Return {gpu_index: free_bytes}; prefers torch, falls back to
    rocm-smi JSON output."""
    try:
        import torch

        if torch.cuda.is_available():
            out = {}
            for i in range(torch.cuda.device_count()):
                free, _total = torch.cuda.mem_get_info(i)
                out[i] = free
            return out
    except Exception:
        pass
    try:
        res = subprocess.run(
            ["rocm-smi", "--showmeminfo", "vram", "--json"],
            capture_output=True, text=True, timeout=30)
        data = json.loads(res.stdout)
        out = {}
        for card, fields in data.items():
            if not card.startswith("card"):
                continue
            idx = int(card.replace("card", ""))
            total = int(fields.get("VRAM Total Memory (B)", 0))
            used = int(fields.get("VRAM Total Used Memory (B)", 0))
            out[idx] = total - used
        return out
    except Exception:
        return {}


def get_proper_gpu(num=1):
    """Indices of the `num` GPUs with the most free memory."""
    free = get_gpu_free_memory()
    ranked = sorted(free, key=free.get, reverse=True)
    return ranked[:num]


def new_dir(path, renew=False):
    if os.path.exists(path) and renew:
        shutil.rmtree(path)
    os.makedirs(path, exist_ok=True)
    return path


def check_file(path):
    return os.path.isfile(path)


def del_file(path):
    if os.path.isfile(path):
        os.remove(path)
        return True
    return False


def copy_file(src, dst):
    shutil.copyfile(src, dst)
