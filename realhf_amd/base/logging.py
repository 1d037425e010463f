"""Logging helpers (reference: realhf/base/logging.py).

Named loggers with a compact format; a dedicated "bench" logger emits
greppable `${name:value}$` markers that the profiling tools parse.
"""
import logging
import os
import sys

_FMT = "%(asctime)s.%(msecs)03d [%(levelname).1s] %(name)s: %(message)s"
_DATEFMT = "%H:%M:%S"

_configured = False


def _configure_root():
    global _configured
    if _configured:
        return
    level = os.environ.get("REALHF_AMD_LOG_LEVEL", "INFO").upper()
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(_FMT, _DATEFMT))
    root = logging.getLogger("realhf_amd")
    root.setLevel(level)
    root.addHandler(handler)
    root.propagate = False
    _configured = True


def getLogger(name: str = "") -> logging.Logger:
    _configure_root()
    if not name:
        return logging.getLogger("realhf_amd")
    return logging.getLogger(f"realhf_amd.{name}")


blogger = getLogger("bench")


def mark(key: str, value) -> str:
    """Greppable benchmark marker, e.g. ``${e2e:1.234}$``."""
    s = f"${{{key}:{value}}}$"
    blogger.info(s)
    return s
