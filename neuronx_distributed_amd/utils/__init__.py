from .logger import get_logger

import os


def cpu_mode() -> bool:
    """True when forced onto the CPU/gloo path (reference
    utils/__init__.py:6-8 NXD_CPU_MODE)."""
    return os.environ.get("NXDA_CPU_MODE", os.environ.get("NXD_CPU_MODE", "0")) == "1"
