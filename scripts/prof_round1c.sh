#!/bin/bash
# Final round-1 measurement: all ops, stats, PMC on tuned kernels.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 400 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_c.log 2>&1
echo "pytest rc=$?"
for op in encode decode crc encode_crc mixed; do
  timeout 600 python bench.py --op $op --steps 20 --warmup 5 \
    > gpurun_out/bench_c_$op.json 2> gpurun_out/bench_c_$op.log
  echo "$op rc=$?"
done

cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profc \
  -o stats -- python /root/repo/bench.py --stripes 256 --steps 5 --warmup 2 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "stats rc=$?"
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/profc \
  -o fetch_enc -- python /root/repo/bench.py --stripes 256 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "fetch_enc rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d /root/repo/gpurun_out/profc \
  -o write_enc -- python /root/repo/bench.py --stripes 256 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "write_enc rc=$?"
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/profc \
  -o fetch_crc -- python /root/repo/bench.py --op crc --stripes 256 --steps 3 \
  --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
echo "fetch_crc rc=$?"
timeout 300 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT -d /root/repo/gpurun_out/profc \
  -o lds_crc -- python /root/repo/bench.py --op crc --stripes 256 --steps 3 \
  --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
echo "lds_crc rc=$?"

tail -1 /root/repo/gpurun_out/pytest_c.log
for op in encode decode crc encode_crc mixed; do
  echo "=== $op"; cat /root/repo/gpurun_out/bench_c_$op.json
done
