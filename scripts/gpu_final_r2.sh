#!/usr/bin/env bash
# Round-2 closing measurements: full GPU tier, flagship bench + soak, all
# bench_suite configs, API read/write, shape sweep, and a rocprofv3 stats
# capture of the round-2 kernels (wave codec + inflater).
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

python -m pytest tests -m gpu -q > gpurun_out/r2_pytest_final.log 2>&1
echo "PYTEST_RC=$?" | tee -a gpurun_out/r2_pytest_final.log
grep -E "passed|failed" gpurun_out/r2_pytest_final.log | tail -1

timeout 300 python bench.py --steps 20 --warmup 5 2>/dev/null | tee gpurun_out/r2_bench_final.json
timeout 300 python bench.py --steps 5 --warmup 2 --rows 4000000 2>/dev/null | tee gpurun_out/r2_bench_4m.json

for CFG in plumbing partitionby infer gzip_bytearray; do
  timeout 300 python bench_suite.py $CFG --rows 1000000 --reps 3 2>/dev/null | tail -1
done | tee gpurun_out/r2_suite_final.json

timeout 300 python exp/exp_apiread.py 2>&1 | tail -6 | tee gpurun_out/r2_apiread.txt
timeout 300 python exp/bench_shapes.py 2>&1 | tail -5 | tee gpurun_out/r2_shapes_final.txt

# kernel stats: wave codec (huge shape) and the inflater
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/r2_prof_huge" -o huge -- \
  python "$GRAFT_REPO_ROOT/exp/bench_huge.py" > "$GRAFT_REPO_ROOT/gpurun_out/r2_prof_huge.log" 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/r2_prof_inf" -o inf -- \
  python "$GRAFT_REPO_ROOT/exp/exp_inflate_only.py" > "$GRAFT_REPO_ROOT/gpurun_out/r2_prof_inf.log" 2>&1
cd "$GRAFT_REPO_ROOT"
find gpurun_out/r2_prof_huge gpurun_out/r2_prof_inf -name "*stats*" | head
for f in $(find gpurun_out/r2_prof_huge gpurun_out/r2_prof_inf -name "*kernel_stats*.csv" 2>/dev/null); do
  echo "== $f"; head -15 "$f"
done
echo DONE
