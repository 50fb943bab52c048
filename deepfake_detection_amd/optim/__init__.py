from .adamw import AdamW  # noqa: F401
from .extras import (Lookahead, Nadam, NovoGrad, NvNovoGrad,  # noqa: F401
                     PlainRAdam, RAdam)
from .optim_factory import add_weight_decay, create_optimizer  # noqa: F401
from .rmsprop_tf import RMSpropTF  # noqa: F401
