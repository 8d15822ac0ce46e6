#!/bin/bash
# Round-2 GPU call E: PF-default validation + BV sweep + final 30-step
# records + rocprof evidence on the adopted CRC config.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 500 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_r2e.log 2>&1
echo "pytest rc=$?"
tail -2 gpurun_out/pytest_r2e.log

timeout 420 python scripts/crc_ab.py 64 > gpurun_out/crc_ab_r2e.log 2>&1
tail -8 gpurun_out/crc_ab_r2e.log

for op in encode decode crc encode_crc mixed; do
  extra="--skip-cpu-baseline"
  if [ "$op" = encode ] || [ "$op" = crc ]; then extra=""; fi
  timeout 700 python bench.py --op $op --steps 30 --warmup 5 $extra \
    > gpurun_out/final_r2_$op.json 2> gpurun_out/final_r2_$op.log
  echo "$op rc=$?"
  cat gpurun_out/final_r2_$op.json
done

timeout 420 python scripts/replicate_pipeline_bench.py 48 > gpurun_out/repl_r2e.log 2>&1
cat gpurun_out/repl_r2e.log

cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profr2e \
  -o stats_crc_pf -- python /root/repo/bench.py --op crc --stripes 256 \
  --steps 5 --warmup 2 --skip-cpu-baseline > /dev/null 2>&1
echo "stats_crc rc=$?"
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/profr2e \
  -o fetch_crc_pf -- python /root/repo/bench.py --op crc --stripes 256 \
  --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
echo "fetch_crc rc=$?"
echo done
