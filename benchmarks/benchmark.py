"""Throughput benchmark driver (parity: benchmarks/benchmark.py): run one of
the *_benchmarks experiment presets through the CLI and report wall time.

    python benchmarks/benchmark.py exp=ppo_benchmarks
    python benchmarks/benchmark.py exp=dreamer_v3_benchmarks

The flagship MI355X benchmark (DreamerV3 env-frames/s vs BASELINE.md) lives
in `bench.py` at the repo root; this script is for quick cross-algorithm
comparisons on the reference's own benchmark presets.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from sheeprl_amd.cli import run

if __name__ == "__main__":
    args = sys.argv[1:] or ["exp=ppo_benchmarks"]
    tic = time.perf_counter()
    run(args)
    print(f"\nbenchmark wall time: {time.perf_counter() - tic:.2f} s")
