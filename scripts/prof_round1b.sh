#!/bin/bash
# Final round-1 measurement pass: parity, 3 bench ops, profiles, torchrun.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 400 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  > gpurun_out/pytest_final.log 2>&1
echo "pytest rc=$?"

timeout 600 python bench.py --steps 20 --warmup 5 \
  > gpurun_out/bench_encode.json 2> gpurun_out/bench_encode.log
echo "encode rc=$?"
timeout 600 python bench.py --op decode --steps 20 --warmup 5 \
  > gpurun_out/bench_decode.json 2> gpurun_out/bench_decode.log
echo "decode rc=$?"
timeout 600 python bench.py --op crc --steps 20 --warmup 5 \
  > gpurun_out/bench_crc.json 2> gpurun_out/bench_crc.log
echo "crc rc=$?"

# torchrun single-rank sanity (the driver's N>1 launch shape)
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29531 bench.py --gpus 1 --steps 3 \
  --warmup 1 --stripes 64 --skip-cpu-baseline \
  > gpurun_out/bench_torchrun.json 2> gpurun_out/bench_torchrun.log
echo "torchrun rc=$?"

cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof2 \
  -o stats -- python /root/repo/bench.py --stripes 256 --steps 5 --warmup 2 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "stats rc=$?"
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/prof2 \
  -o fetch -- python /root/repo/bench.py --stripes 256 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "fetch rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d /root/repo/gpurun_out/prof2 \
  -o write -- python /root/repo/bench.py --stripes 256 --steps 3 --warmup 1 \
  --skip-cpu-baseline > /dev/null 2>&1
echo "write rc=$?"
# CRC kernel counters too
timeout 300 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/prof2 \
  -o crcfetch -- python /root/repo/bench.py --op crc --stripes 256 --steps 3 \
  --warmup 1 --skip-cpu-baseline > /dev/null 2>&1
echo "crcfetch rc=$?"

tail -2 /root/repo/gpurun_out/pytest_final.log
echo ===; cat /root/repo/gpurun_out/bench_encode.json
echo ===; cat /root/repo/gpurun_out/bench_decode.json
echo ===; cat /root/repo/gpurun_out/bench_crc.json
echo ===; cat /root/repo/gpurun_out/bench_torchrun.json
