"""Per-payload zstd decode probe (words/text/random) — quick A/B."""
import sys

sys.path.insert(0, ".")
from scripts.gpu_probe import zstd_bench  # noqa: E402

if __name__ == "__main__":
    for payload in ("words", "text", "random"):
        zstd_bench(payload=payload)
