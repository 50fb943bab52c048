"""HIP/CDNA4 kernel package.

`deepfake_detection_amd.ops` owns every device-side hot op of the CNN path
(reference native surface inventory: SURVEY.md §2.6). The kernels live in
ops/hip/*.hip, are compiled for gfx950 in-tree (setup.py / __graft_entry__
build()), and are loaded here. On a machine WITH a GPU the extension is
mandatory — ops raise if it is missing (no silent eager fallback); on a
CPU-only machine the pure-PyTorch reference implementations in
ops/reference.py serve the unit tests.
"""

from . import functional  # noqa: F401
from .extension import has_extension, load_extension  # noqa: F401
