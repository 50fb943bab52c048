from .distributed import (  # noqa: F401
    distribute_bn,
    init_distributed,
    is_primary,
    reduce_tensor,
    wrap_ddp,
)
from .server_json import load_server_json, parse_server  # noqa: F401
