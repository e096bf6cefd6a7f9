set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/pmc
export TMPDIR=/tmp
rocprofv3 --list-avail > gpurun_out/pmc_avail.txt 2>&1
PMC=""
for c in SQ_WAVES SQ_INSTS_VALU SQ_INSTS_VMEM SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_BUSY_CYCLES GRBM_GUI_ACTIVE; do
  grep -qw "$c" gpurun_out/pmc_avail.txt && PMC="$PMC $c"
done
echo "PMC set:$PMC" | tee gpurun_out/pmc_set.txt
cd /tmp
timeout 600 rocprofv3 --pmc $PMC --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/pmc" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 > "$GRAFT_REPO_ROOT/gpurun_out/pmc_run.log" 2>&1
echo "pmc_exit=$?" >> "$GRAFT_REPO_ROOT/gpurun_out/pmc_run.log"
cd "$GRAFT_REPO_ROOT"
timeout 300 python bench.py --steps 5 --warmup 2 --rows 4000000 > gpurun_out/bench_4m.json 2>&1
echo "soak_exit=$?" >> gpurun_out/bench_4m.json
tail -2 gpurun_out/pmc_run.log
tail -2 gpurun_out/bench_4m.json
