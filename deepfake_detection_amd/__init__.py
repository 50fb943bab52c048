"""deepfake_detection_amd — MI355X-native distributed deepfake-detection stack.

A from-scratch AMD CDNA4 (gfx950) re-design of the capabilities of
TARTRL/Deepfake_Detection (reference: /root/reference): PyTorch-ROCm for the
framework layer, hand-written HIP kernels for the CNN hot path (NHWC /
channels_last), and RCCL over xGMI for distributed data parallelism.

Public API mirrors the reference `dfd` package (reference setup.py:29):
model factory (`create_deepfake_model_v4`), dataset/loader
(`DeepFakeDataset_v3`, `create_deepfake_loader_v3`), train/test runners, and
`.pth.tar` checkpoint layout.
"""

__version__ = "0.1.0"

from .models import (  # noqa: F401
    create_model,
    create_deepfake_model,
    create_deepfake_model_v3,
    create_deepfake_model_v4,
    is_model,
    list_models,
    model_entrypoint,
    register_model,
    load_checkpoint,
    resume_checkpoint,
)
