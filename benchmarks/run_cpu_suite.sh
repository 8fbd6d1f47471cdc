#!/bin/bash
# CPU wall-clock parity suite (BASELINE.md "Wall-clock benchmarks" rows).
# One run per cell (reference reports mean of 5); 1- and 2-device (gloo).
set -u
cd "$(dirname "$0")/.."
out=benchmarks/out
for exp in ppo_benchmarks a2c_benchmarks sac_benchmarks; do
  echo "=== $exp 1-device ==="
  timeout 3600 python benchmarks/benchmark.py exp=$exp 2>&1 | tail -2
  echo "=== $exp 2-device ==="
  timeout 3600 python benchmarks/benchmark.py exp=$exp runtime.devices=2 2>&1 | tail -2
done
