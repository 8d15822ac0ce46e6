#!/bin/bash
# Round-2 GPU call A: full GPU test suite (fold CRC + concurrency now in),
# CRC impl A/B sweep, wide-k mixed-radix A/B.
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu ==="
  timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -15
  echo "=== crc_ab ==="
  timeout 420 python scripts/crc_ab.py
  echo "=== bench_variants ec(16,4) ==="
  timeout 420 ./lizardfs_amd/csrc/bench_variants 16 4 1024 6
  echo "=== bench_variants ec(32,6) ==="
  timeout 420 ./lizardfs_amd/csrc/bench_variants 32 6 512 6
  echo "=== done ==="
} > gpurun_out/r2a.log 2>&1
tail -100 gpurun_out/r2a.log
