"""Model zoo (registry + factory + architectures)."""

from .registry import (  # noqa: F401
    register_model,
    list_models,
    is_model,
    model_entrypoint,
    list_modules,
    is_model_in_modules,
)
from .helpers import (  # noqa: F401
    load_state_dict,
    load_checkpoint,
    resume_checkpoint,
    load_pretrained,
)
from . import efficientnet  # noqa: F401  (registers entrypoints)
from . import mobilenetv3  # noqa: F401
from . import resnet  # noqa: F401
from . import xception  # noqa: F401
from . import densenet  # noqa: F401
from . import inception_v3  # noqa: F401
from . import senet  # noqa: F401
from . import dpn  # noqa: F401
from . import res2net  # noqa: F401
from . import sknet  # noqa: F401
from . import inception_v4  # noqa: F401
from . import gluon_resnet  # noqa: F401
from . import selecsls  # noqa: F401
from . import inception_resnet_v2  # noqa: F401
from . import gluon_xception  # noqa: F401
from . import dla  # noqa: F401
from . import hrnet  # noqa: F401
from . import nasnet  # noqa: F401
from . import pnasnet  # noqa: F401
from .factory import (  # noqa: F401
    create_model,
    create_deepfake_model,
    create_deepfake_model_v3,
    create_deepfake_model_v4,
)
from .efficientnet import EfficientNet, EfficientNetFeatures  # noqa: F401
