from .cross_entropy import LabelSmoothingCrossEntropy, SoftTargetCrossEntropy  # noqa: F401
from .jsd import JsdCrossEntropy  # noqa: F401
