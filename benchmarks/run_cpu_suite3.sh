#!/bin/bash
# 4-thread CPU rows: the reference's published numbers come from a 4-CPU
# Lightning Studio; its OMP_NUM_THREADS routing leaves torch's default
# intra-op pool in place, so num_threads=4 is the comparable setting.
set -u
cd "$(dirname "$0")/.."
for exp in dreamer_v3_benchmarks dreamer_v2_benchmarks dreamer_v1_benchmarks sac_benchmarks; do
  echo "=== $exp 1-device num_threads=4 ==="
  timeout 7200 python benchmarks/benchmark.py exp=$exp num_threads=4 2>&1 | tail -2
done
