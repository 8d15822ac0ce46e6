#!/bin/bash
# Round-2 GPU call B: CRC C1/NT sweep, MR at ec(8,2), streaming replicate
# pipeline test + bench, quick bench.py crc/encode sanity with new defaults.
mkdir -p gpurun_out
{
  echo "=== crc_ab v2 ==="
  timeout 420 python scripts/crc_ab.py
  echo "=== bench_variants ec(8,2) [MR check] ==="
  timeout 420 ./lizardfs_amd/csrc/bench_variants 8 2 1024 6
  echo "=== replicate tests ==="
  timeout 420 python -m pytest tests/test_replicate.py tests/test_concurrency.py -m gpu -q 2>&1 | tail -5
  echo "=== replicate pipeline bench ==="
  timeout 600 python scripts/replicate_pipeline_bench.py 48
  echo "=== bench.py --op crc (new default) ==="
  timeout 420 python bench.py --op crc --steps 10 --warmup 3 --stripes 256 --skip-cpu-baseline 2>/dev/null
  echo "=== done ==="
} > gpurun_out/r2b.log 2>&1
tail -60 gpurun_out/r2b.log
