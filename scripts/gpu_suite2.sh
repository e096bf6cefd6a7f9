set -x
cd "$GRAFT_REPO_ROOT"
timeout 600 python bench_suite.py all --rows 1000000 --reps 3 2>&1 | grep "^{" > gpurun_out/suite2_1gpu.json
timeout 420 python bench.py --steps 3 --warmup 1 --rows 16000000 > gpurun_out/bench_16m.json 2>&1
echo "exit16m=$?" >> gpurun_out/bench_16m.json
cat gpurun_out/suite2_1gpu.json
tail -2 gpurun_out/bench_16m.json
