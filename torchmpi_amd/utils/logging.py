"""Per-rank prefixed logging (replaces the reference's rank-0 gated prints,
TorchMPI init.lua verbose mode — SURVEY.md section 5 observability)."""
import logging
import os
import sys

_LOGGERS = {}


def get_logger(name: str = "torchmpi_amd") -> logging.Logger:
    if name in _LOGGERS:
        return _LOGGERS[name]
    logger = logging.getLogger(name)
    rank = os.environ.get("RANK", "?")
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(
        logging.Formatter(f"[%(name)s r{rank}] %(levelname)s: %(message)s")
    )
    logger.addHandler(handler)
    logger.propagate = False
    level = os.environ.get("TORCHMPI_LOG", "warning").upper()
    logger.setLevel(getattr(logging, level, logging.WARNING))
    _LOGGERS[name] = logger
    return logger


def log_rank0(msg: str, level: int = logging.INFO) -> None:
    if os.environ.get("RANK", "0") == "0":
        get_logger().log(level, msg)
