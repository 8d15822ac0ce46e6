#!/bin/bash
# Round-2 GPU call H: round-end dress rehearsal + extended fuzz.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 500 python -m pytest tests -x -q -m gpu > gpurun_out/pytest_r2h.log 2>&1
echo "pytest rc=$?"
tail -2 gpurun_out/pytest_r2h.log

timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()"
echo "smoke rc=$?"

timeout 600 python bench.py > gpurun_out/bench_r2h_default.json 2> gpurun_out/bench_r2h_default.log
echo "bench rc=$?"
cat gpurun_out/bench_r2h_default.json

LIZEC_SWEEP_SEED=777 LIZEC_SWEEP_TRIALS=60 timeout 600 \
  python -m pytest tests/test_gpu_sweep.py -x -q > gpurun_out/sweep777.log 2>&1
echo "sweep777 rc=$?"; tail -1 gpurun_out/sweep777.log
LIZEC_SWEEP_SEED=424242 LIZEC_SWEEP_TRIALS=60 timeout 600 \
  python -m pytest tests/test_gpu_sweep.py -x -q > gpurun_out/sweep424242.log 2>&1
echo "sweep424242 rc=$?"; tail -1 gpurun_out/sweep424242.log

timeout 420 python scripts/replicate_pipeline_bench.py 64 > gpurun_out/repl_r2h.log 2>&1
cat gpurun_out/repl_r2h.log
echo done
