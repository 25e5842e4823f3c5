"""Per-payload zstd decode probe (words/text/random) — quick A/B.

Run from anywhere: python scripts/zstd_probe.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from gpu_probe import zstd_bench  # noqa: E402

if __name__ == "__main__":
    for payload in ("words", "text", "random"):
        zstd_bench(payload=payload)
