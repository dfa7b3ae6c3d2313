from .logging import get_logger, log_rank0
from .env import env_int, env_str, env_bool

__all__ = ["get_logger", "log_rank0", "env_int", "env_str", "env_bool"]
