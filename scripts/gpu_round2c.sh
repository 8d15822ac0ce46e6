#!/bin/bash
# Round-2 GPU call C: full suite on the MR product path, torchrun 2-rank
# gloo mixed validation, bench ops with reference CPU baseline, rocprof
# kernel-trace + PMC evidence for the new kernels.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 500 python -m pytest tests/ -q -m gpu > gpurun_out/pytest_r2c.log 2>&1
echo "pytest rc=$?"
tail -2 gpurun_out/pytest_r2c.log

# the one real exchange (mixed all-gather) executed multi-rank on hardware
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29531 bench.py --op mixed --gpus 2 \
  --backend gloo --stripes 64 --steps 5 --warmup 2 --validate \
  --skip-cpu-baseline > gpurun_out/bench_r2c_mixed2.json \
  2> gpurun_out/bench_r2c_mixed2.log
echo "mixed2 rc=$?"
cat gpurun_out/bench_r2c_mixed2.json

# full single-GPU op set with the new kernels; encode also times the
# reference AVX2 CPU baseline on this box's cores
for op in encode decode crc encode_crc mixed; do
  extra="--skip-cpu-baseline"
  if [ "$op" = encode ] || [ "$op" = crc ]; then extra=""; fi
  timeout 700 python bench.py --op $op --steps 20 --warmup 5 $extra \
    > gpurun_out/bench_r2c_$op.json 2> gpurun_out/bench_r2c_$op.log
  echo "$op rc=$?"
  cat gpurun_out/bench_r2c_$op.json
done

cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profr2 \
  -o stats_enc -- python /root/repo/bench.py --stripes 256 --steps 5 --warmup 2 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "stats_enc rc=$?"
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profr2 \
  -o stats_crc -- python /root/repo/bench.py --op crc --stripes 256 --steps 5 \
  --warmup 2 --skip-cpu-baseline > /dev/null 2>&1
echo "stats_crc rc=$?"
for pmc in FETCH_SIZE WRITE_SIZE; do
  timeout 300 rocprofv3 --pmc $pmc -d /root/repo/gpurun_out/profr2 \
    -o ${pmc}_enc -- python /root/repo/bench.py --stripes 256 --steps 3 \
    --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
  echo "${pmc}_enc rc=$?"
  timeout 300 rocprofv3 --pmc $pmc -d /root/repo/gpurun_out/profr2 \
    -o ${pmc}_crc -- python /root/repo/bench.py --op crc --stripes 256 \
    --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
  echo "${pmc}_crc rc=$?"
done
# VALU pressure on the new kernels
for pmc in SQ_INSTS_VALU SQ_WAIT_INST_ANY SQ_BUSY_CYCLES; do
  timeout 300 rocprofv3 --pmc $pmc -d /root/repo/gpurun_out/profr2 \
    -o ${pmc}_crc -- python /root/repo/bench.py --op crc --stripes 256 \
    --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
  echo "${pmc}_crc rc=$?"
done
timeout 300 rocprofv3 --pmc SQ_INSTS_VALU -d /root/repo/gpurun_out/profr2 \
  -o SQ_INSTS_VALU_enc16 -- python /root/repo/bench.py --op encode_crc \
  --stripes 128 --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
echo "valu_enc16 rc=$?"

ls /root/repo/gpurun_out/profr2/ 2>/dev/null | head
echo done
