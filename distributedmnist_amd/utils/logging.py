"""Logging setup: INFO lines in a format the reference's log scrapers accept
(benchmark.py matches '.*step ([0-9]*),.*', '.*ELAPSED TIMES (.*)', etc. —
any prefix is fine, message bodies must be exact)."""

from __future__ import annotations

import logging
import sys


def setup(rank: int = 0, level=logging.INFO):
    root = logging.getLogger()
    if root.handlers:
        return root
    h = logging.StreamHandler(sys.stdout)
    h.setFormatter(logging.Formatter("INFO:dmnist:%(message)s"))
    root.addHandler(h)
    root.setLevel(level)
    return root
