set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/prof4
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu_exit=$?" >> gpurun_out/pytest_gpu.log
timeout 600 python bench.py --steps 10 --warmup 3 --phases > gpurun_out/bench4.json 2> gpurun_out/bench4.err
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench4b.json 2>&1
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof4" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_bench4.log" 2>&1
tail -1 "$GRAFT_REPO_ROOT/gpurun_out/pytest_gpu.log"
tail -2 "$GRAFT_REPO_ROOT/gpurun_out/bench4.err"
tail -1 "$GRAFT_REPO_ROOT/gpurun_out/bench4b.json"
