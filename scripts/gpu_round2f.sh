#!/bin/bash
# Round-2 GPU call F: final records with the robust CRC config + soak +
# kernel stats on the adopted shape.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 420 python scripts/crc_ab.py 64 > gpurun_out/crc_ab_r2f.log 2>&1
tail -8 gpurun_out/crc_ab_r2f.log

for op in encode decode crc encode_crc mixed; do
  extra="--skip-cpu-baseline"
  if [ "$op" = encode ] || [ "$op" = crc ]; then extra=""; fi
  timeout 700 python bench.py --op $op --steps 30 --warmup 5 $extra \
    > gpurun_out/final_r2f_$op.json 2> gpurun_out/final_r2f_$op.log
  echo "$op rc=$?"
  cat gpurun_out/final_r2f_$op.json
done

timeout 500 python scripts/soak.py 360 > gpurun_out/soak_r2f.log 2>&1
echo "soak rc=$?"
tail -3 gpurun_out/soak_r2f.log

cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profr2f \
  -o stats_crc -- python /root/repo/bench.py --op crc --stripes 256 --steps 5 \
  --warmup 2 --skip-cpu-baseline > /dev/null 2>&1
echo "stats rc=$?"
echo done
