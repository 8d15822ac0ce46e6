#!/bin/bash
# Round-1 profiling pass: parity, full bench, kernel stats, HBM counters.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. GPU parity (v2 kernel, incl. ragged-tail path)
timeout 400 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  > gpurun_out/pytest_v2.log 2>&1
echo "pytest rc=$?"

# 2. full-size headline bench (1024x64MiB stripes, CPU baseline leg)
timeout 600 python bench.py --steps 10 --warmup 3 \
  > gpurun_out/bench_full.json 2> gpurun_out/bench_full.log
echo "bench rc=$?"

# 3. kernel-trace stats (short run)
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_stats \
  -o stats -- python /root/repo/bench.py --stripes 256 --steps 5 --warmup 2 \
  --skip-cpu-baseline > /root/repo/gpurun_out/bench_prof.json 2>/dev/null
echo "rocprof stats rc=$?"

# 4. HBM counters, separate passes (FETCH_SIZE=3 TCC slots, WRITE_SIZE=2)
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/prof_fetch \
  -o fetch -- python /root/repo/bench.py --stripes 64 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "pmc fetch rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d /root/repo/gpurun_out/prof_write \
  -o write -- python /root/repo/bench.py --stripes 64 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "pmc write rc=$?"

tail -2 /root/repo/gpurun_out/pytest_v2.log
cat /root/repo/gpurun_out/bench_full.json
find /root/repo/gpurun_out/prof_stats /root/repo/gpurun_out/prof_fetch \
  /root/repo/gpurun_out/prof_write -type f | head -20
