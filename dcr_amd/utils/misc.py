"""Small parity helpers (reference utils_ret.py)."""
from __future__ import annotations

import argparse


def bool_flag(s: str) -> bool:
    """Parse boolean CLI flags (reference: utils_ret.py:463-474)."""
    FALSY = {"off", "false", "0"}
    TRUTHY = {"on", "true", "1"}
    if s.lower() in FALSY:
        return False
    if s.lower() in TRUTHY:
        return True
    raise argparse.ArgumentTypeError("invalid value for a boolean flag")
