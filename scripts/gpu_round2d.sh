#!/bin/bash
# Round-2 GPU call D: PF A/B + footprint effect on CRC; decode ordering
# check; ec(32,6) CH1 variant.
mkdir -p gpurun_out
{
  echo "=== decode FIRST (fresh box) ==="
  timeout 500 python bench.py --op decode --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null
  echo "=== encode ==="
  timeout 500 python bench.py --op encode --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null
  echo "=== decode again ==="
  timeout 500 python bench.py --op decode --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null
  echo "=== crc_ab 16 GiB ==="
  timeout 420 python scripts/crc_ab.py 16
  echo "=== crc_ab 64 GiB ==="
  timeout 420 python scripts/crc_ab.py 64
  echo "=== bench.py crc 1024 stripes (default impl) ==="
  timeout 500 python bench.py --op crc --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null
  echo "=== bench_variants ec(32,6) CH1 probe ==="
  timeout 420 ./lizardfs_amd/csrc/bench_variants 32 6 512 6 2>&1 | grep -E "mr|ql|stripes"
  echo "=== done ==="
} > gpurun_out/r2d.log 2>&1
tail -60 gpurun_out/r2d.log
