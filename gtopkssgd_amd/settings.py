"""Global settings and the shared logger.

Reference parity: `upstream:settings.py` — a module-level
logger plus a handful of compile-time flags. Kept deliberately small.
"""
import logging
import os

DEBUG = bool(int(os.environ.get("GTOPK_DEBUG", "0")))

# Iterations of dense warm-up before sparsification kicks in (the reference
# family warms up with dense all-reduce for a few iterations; 0 disables).
WARMUP_ITERS = int(os.environ.get("GTOPK_WARMUP_ITERS", "0"))

# Maximum flat-gradient element count addressable with int32 indices.
MAX_INT32_NUMEL = 2**31 - 1

_FMT = "%(asctime)s [%(levelname)s] %(name)s: %(message)s"


def _build_logger() -> logging.Logger:
    logger = logging.getLogger("gtopkssgd_amd")
    if not logger.handlers:
        handler = logging.StreamHandler()
        handler.setFormatter(logging.Formatter(_FMT))
        logger.addHandler(handler)
    logger.setLevel(logging.DEBUG if DEBUG else logging.INFO)
    return logger


logger = _build_logger()
