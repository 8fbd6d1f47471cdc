"""Wall-clock benchmark harness (parity: the reference's benchmarks/
benchmark.py): times a full CLI training run of any *_benchmarks experiment.

    python benchmarks/benchmark.py exp=ppo_benchmarks [overrides...]
    python benchmarks/benchmark.py exp=dreamer_v3_benchmarks runtime=gpu-bf16
"""

from __future__ import annotations

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from sheeprl_amd.cli import run


def main() -> None:
    args = sys.argv[1:]
    if not any(a.startswith("exp=") for a in args):
        raise SystemExit("usage: python benchmarks/benchmark.py exp=<name>_benchmarks [overrides...]")
    t0 = time.perf_counter()
    run(args)
    elapsed = time.perf_counter() - t0
    print(f"\n[benchmark] {' '.join(args)} -> {elapsed:.2f} s")


if __name__ == "__main__":
    main()
