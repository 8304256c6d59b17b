from .dist import setup_distributed, is_main, get_rank, get_world_size, barrier  # noqa: F401
from .ddp import GradReducer, broadcast_params  # noqa: F401
