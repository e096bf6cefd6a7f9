set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/prof
python -c "import torch; print(torch.__version__, torch.cuda.is_available(), torch.cuda.get_device_name(0))" > gpurun_out/env.log 2>&1
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu_exit=$?" >> gpurun_out/pytest_gpu.log
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" > gpurun_out/smoke.log 2>&1
echo "smoke_exit=$?" >> gpurun_out/smoke.log
timeout 600 python bench.py --steps 10 --warmup 3 --phases > gpurun_out/bench1.json 2> gpurun_out/bench1.err
echo "bench_exit=$?" >> gpurun_out/bench1.err
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench.log" 2>&1
echo "prof_exit=$?" >> "$GRAFT_REPO_ROOT/gpurun_out/prof_bench.log"
tail -3 "$GRAFT_REPO_ROOT/gpurun_out/bench1.json"
