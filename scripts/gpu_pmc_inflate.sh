set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/pmc_inf2
export TMPDIR=/tmp
rocprofv3 --list-avail > gpurun_out/pmc_avail2.txt 2>&1
PMC=""
for c in SQ_WAVES SQ_INSTS_VALU SQ_INSTS_SALU SQ_INSTS_LDS SQ_INSTS_VMEM SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_BUSY_CYCLES; do
  grep -qw "$c" gpurun_out/pmc_avail2.txt && PMC="$PMC $c"
done
echo "PMC set:$PMC" | tee gpurun_out/pmc_set2.txt
cd /tmp
timeout 420 rocprofv3 --pmc $PMC -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_inf2" -- python "$GRAFT_REPO_ROOT/exp/exp_inflate_only.py" > "$GRAFT_REPO_ROOT/gpurun_out/pmc_inf2/run.log" 2>&1
echo "exit=$?" >> "$GRAFT_REPO_ROOT/gpurun_out/pmc_inf2/run.log"
cd "$GRAFT_REPO_ROOT"
tail -4 gpurun_out/pmc_inf2/run.log
CSV=$(find gpurun_out/pmc_inf2 -name "*counter*.csv" | head -1)
python - <<'PY'
import csv, glob, collections
fs = glob.glob("gpurun_out/pmc_inf2/**/*counter*.csv", recursive=True)
print("csv files:", fs)
agg = collections.defaultdict(float)
for fn in fs:
    for row in csv.DictReader(open(fn)):
        k = (row.get("Kernel_Name") or row.get("kernel_name",""), row.get("Counter_Name") or row.get("counter_name",""))
        agg[k] += float(row.get("Counter_Value") or row.get("counter_value") or 0)
for (kn, cn), v in sorted(agg.items()):
    if "inflate" in kn:
        print(f"{kn.split('(')[0][:50]:52s} {cn:24s} {v:,.0f}")
PY
