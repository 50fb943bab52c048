"""Experimental: 1x1 (pointwise) conv forward as an MFMA GEMM
(ops/hip/pwconv.hip). Off by default — MIOpen's igemm kernels serve the 1x1
convs in the main path; set DFD_AMD_PW_MFMA=1 to A/B this kernel. The
BASELINE.json north star names this op as one to own natively; enable it
per-shape once it measures faster than MIOpen (tools/bench_kernels.py).
"""

import os

import torch

from .extension import load_extension


def pw_mfma_enabled() -> bool:
    return os.environ.get("DFD_AMD_PW_MFMA", "0") == "1"


def pw_conv2d_fwd(x, weight):
    """bf16 NHWC 1x1 conv forward on matrix cores. No autograd (A/B use)."""
    ext = load_extension()
    x = x.contiguous(memory_format=torch.channels_last)
    return ext.pw_conv2d_fwd_mfma(x, weight)
