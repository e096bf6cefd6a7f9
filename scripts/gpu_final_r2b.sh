#!/usr/bin/env bash
# Round-2 late wrap: full GPU tier, flagship bench + soak, all bench_suite
# configs, API read/write probes, shape sweep, and rocprofv3 stats of the
# round-2 kernels (wave codec + root-table inflater).
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

python -m pytest tests -m gpu -q > gpurun_out/w_pytest.log 2>&1
echo "PYTEST_RC=$?" >> gpurun_out/w_pytest.log
grep -E "passed|failed" gpurun_out/w_pytest.log | tail -1

timeout 300 python bench.py --steps 20 --warmup 5 2>/dev/null | tail -1 | tee gpurun_out/w_bench.json
timeout 300 python bench.py --steps 5 --warmup 2 --rows 4000000 2>/dev/null | tail -1 | tee gpurun_out/w_bench_4m.json

for CFG in plumbing partitionby infer gzip_bytearray; do
  timeout 300 python bench_suite.py $CFG --rows 1000000 --reps 3 2>/dev/null | tail -1
done | tee gpurun_out/w_suite.json

timeout 300 python exp/exp_apiread.py 2>&1 | tail -6 | tee gpurun_out/w_apiread.txt
timeout 200 python exp/exp_apiwrite.py 2>&1 | tail -8 | tee gpurun_out/w_apiwrite.txt
timeout 300 python exp/bench_shapes.py 2>&1 | tail -5 | tee gpurun_out/w_shapes.txt
timeout 200 python exp/exp_gzread_phases.py 2>&1 | tail -8 | tee gpurun_out/w_gzphases.txt

export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/w_prof_inf" -o inf -- \
  python "$GRAFT_REPO_ROOT/exp/exp_inflate_only.py" > "$GRAFT_REPO_ROOT/gpurun_out/w_prof_inf.log" 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/w_prof_bench" -o bch -- \
  python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/w_prof_bench.log" 2>&1
cd "$GRAFT_REPO_ROOT"
for f in $(find gpurun_out/w_prof_inf gpurun_out/w_prof_bench -name "*kernel_stats*.csv" 2>/dev/null); do
  echo "== $f"; head -18 "$f"
done
echo DONE
# PMC counters for the routed inflate kernels (separate pass, counters only)
cd /tmp
timeout 200 rocprofv3 --pmc SQ_WAVES SQ_INSTS_VALU SQ_INSTS_SALU SQ_INSTS_LDS SQ_BUSY_CYCLES -d "$GRAFT_REPO_ROOT/gpurun_out/w_pmc_inf" -- python "$GRAFT_REPO_ROOT/exp/exp_inflate_only.py" > "$GRAFT_REPO_ROOT/gpurun_out/w_pmc_inf.log" 2>&1
cd "$GRAFT_REPO_ROOT"
echo PMC_DONE
