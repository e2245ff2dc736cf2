"""py/nezha_harness — ctypes test harness.

Kept as the SURVEY-named entry point; the implementation lives in the
nezha_amd package (same ctypes binding over include/nezha.h)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nezha_amd import *  # noqa: F401,F403
