#!/bin/bash
# GPU validation: tests, ZeRO-3 + ZeRO-2 bench, rocprof profile.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
export TMPDIR=/tmp
export HSA_ENABLE_IPC_MODE_LEGACY=0

rocm-smi --showproductname > gpurun_out/smi.log 2>&1
python -c "import torch; print(torch.__version__, torch.cuda.get_device_name(0))" >> gpurun_out/smi.log 2>&1

echo "=== pytest gpu ===" | tee gpurun_out/pytest_gpu.log
timeout 900 python -m pytest tests -m gpu -x -q >> gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
tail -5 gpurun_out/pytest_gpu.log

echo "=== bench zero3 1gpu ==="
timeout 600 python bench.py --gpus 1 --steps 5 --warmup 2 --zero-stage 3 > gpurun_out/bench_z3.log 2>&1
echo "bench exit: $?" >> gpurun_out/bench_z3.log
tail -2 gpurun_out/bench_z3.log

echo "=== bench zero2 1gpu ==="
timeout 600 python bench.py --gpus 1 --steps 5 --warmup 2 --zero-stage 2 > gpurun_out/bench_z2.log 2>&1
echo "bench exit: $?" >> gpurun_out/bench_z2.log
tail -2 gpurun_out/bench_z2.log

echo "=== rocprof stats (llama3-8b z3, 2 steps) ==="
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -- \
  python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 2 --warmup 1 --zero-stage 3 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
echo "rocprof exit: $?" >> "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log"
tail -3 "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log"
# keep merged output small: extract kernel stats csv, drop huge traces
find "$GRAFT_REPO_ROOT/gpurun_out/prof" -name "*.db" -size +20M -delete 2>/dev/null
find "$GRAFT_REPO_ROOT/gpurun_out/prof" -name "*kernel_trace*" -size +5M -delete 2>/dev/null
ls -laR "$GRAFT_REPO_ROOT/gpurun_out/prof" 2>/dev/null | head -30
echo DONE
