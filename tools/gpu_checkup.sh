#!/bin/bash
# Standard GPU-box checkup: tests, smoke, bench (eager/graph/fp16), epoch,
# rocprof stats. Run via:
#   gpurun --timeout 1800 -- 'bash tools/gpu_checkup.sh'
set -x
cd "$(dirname "$0")/.."
REPO=$(pwd)
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 200 python __graft_entry__.py smoke 2>&1 | tail -1
timeout 200 python bench.py --steps 30 --warmup 10 2>/dev/null
timeout 200 python bench.py --steps 30 --warmup 10 --graph 2>/dev/null
timeout 200 python bench.py --steps 20 --warmup 5 --dtype fp16 2>/dev/null
timeout 300 python bench.py --model bert-large --seq-len 512 --batch-size 16 \
    --steps 10 --warmup 3 2>/dev/null
timeout 300 python single-gpu-cls.py 2>&1 | grep 耗时

cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof" -- \
  python "$REPO/bench.py" --steps 10 --warmup 5 \
  > "$REPO/gpurun_out/prof_bench.log" 2>&1
echo "rocprof rc=$?"
echo DONE
