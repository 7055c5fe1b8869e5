#!/bin/bash
# First-GPU-call checkup: tests, smoke, bench eager+graph, rocprof stats.
# Run via: gpurun --timeout 1500 -- 'bash tools/gpu_checkup.sh'
set -x
cd "$(dirname "$0")/.."
REPO=$(pwd)
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== GPU info ==" > gpurun_out/checkup.log
rocm-smi --showproductname >> gpurun_out/checkup.log 2>&1

echo "== pytest -m gpu ==" | tee -a gpurun_out/checkup.log
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" | tee -a gpurun_out/checkup.log
tail -20 gpurun_out/pytest_gpu.log

echo "== smoke ==" | tee -a gpurun_out/checkup.log
timeout 300 python __graft_entry__.py smoke > gpurun_out/smoke.log 2>&1
echo "smoke exit: $?" | tee -a gpurun_out/checkup.log
tail -5 gpurun_out/smoke.log

echo "== bench eager ==" | tee -a gpurun_out/checkup.log
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_eager.json 2> gpurun_out/bench_eager.err
echo "bench eager exit: $?" | tee -a gpurun_out/checkup.log
cat gpurun_out/bench_eager.json

echo "== bench graph ==" | tee -a gpurun_out/checkup.log
timeout 300 python bench.py --steps 30 --warmup 10 --graph > gpurun_out/bench_graph.json 2> gpurun_out/bench_graph.err
echo "bench graph exit: $?" | tee -a gpurun_out/checkup.log
cat gpurun_out/bench_graph.json

echo "== rocprofv3 stats ==" | tee -a gpurun_out/checkup.log
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof" -- \
  python "$REPO/bench.py" --steps 10 --warmup 5 > "$REPO/gpurun_out/prof_bench.log" 2>&1
echo "rocprof exit: $?" | tee -a "$REPO/gpurun_out/checkup.log"
ls -la "$REPO/gpurun_out/prof" >> "$REPO/gpurun_out/checkup.log" 2>&1
tail -40 "$REPO/gpurun_out/prof_bench.log"
echo DONE
