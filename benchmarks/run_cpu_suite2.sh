#!/bin/bash
set -u
cd "$(dirname "$0")/.."
echo "=== sac 1-device (foreach adam) ==="
timeout 3600 python benchmarks/benchmark.py exp=sac_benchmarks 2>&1 | tail -2
echo "=== sac 2-device (foreach adam) ==="
timeout 3600 python benchmarks/benchmark.py exp=sac_benchmarks runtime.devices=2 2>&1 | tail -2
echo "=== dreamer_v3 1-device ==="
timeout 7200 python benchmarks/benchmark.py exp=dreamer_v3_benchmarks 2>&1 | tail -2
echo "=== dreamer_v2 1-device ==="
timeout 7200 python benchmarks/benchmark.py exp=dreamer_v2_benchmarks 2>&1 | tail -2
echo "=== dreamer_v1 1-device ==="
timeout 7200 python benchmarks/benchmark.py exp=dreamer_v1_benchmarks 2>&1 | tail -2
