"""Environment tunables (reference: lib/constants.cpp chunk sizes / cutovers,
SURVEY.md section 5 config system).

Recognized variables:
  TORCHMPI_CHUNK_BYTES          pipelined-chain chunk size (default 4 MiB GPU / 1 MiB CPU)
  TORCHMPI_NUM_RINGS            number of concurrent xGMI rings (default: all link-disjoint)
  TORCHMPI_MULTIRING_MIN_BYTES  below this, allreduce uses a single ring
  TORCHMPI_BIND_ADDR            IP for the TCP transports (default 127.0.0.1)
  TORCHMPI_LOG                  logger level (debug/info/warning/error)
"""
import os


def env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v else default


def env_str(name: str, default: str) -> str:
    return os.environ.get(name, default)


def env_bool(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() in ("1", "true", "yes", "on")
