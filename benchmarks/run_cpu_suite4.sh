#!/bin/bash
set -u
cd "$(dirname "$0")/.."
echo "=== dreamer_v2 1-device num_threads=4 (post ratio fix) ==="
timeout 7200 python benchmarks/benchmark.py exp=dreamer_v2_benchmarks num_threads=4 2>&1 | tail -2
echo "=== sac 2-device num_threads=4 ==="
timeout 3600 python benchmarks/benchmark.py exp=sac_benchmarks num_threads=4 runtime.devices=2 2>&1 | tail -2
