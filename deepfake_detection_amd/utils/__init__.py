from .checkpoint_saver import CheckpointSaver  # noqa: F401
from .ema import ModelEma  # noqa: F401
from .logging_utils import setup_default_logging  # noqa: F401
from .meters import AverageMeter, accuracy  # noqa: F401
from .model import get_state_dict, unwrap_model  # noqa: F401
from .summary import update_summary  # noqa: F401
