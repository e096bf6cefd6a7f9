cd "$GRAFT_REPO_ROOT"
echo "=== pytest gpu ==="
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -1
echo "=== smoke ==="
timeout 180 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -1
echo "=== bench x2 ==="
for i in 1 2; do timeout 200 python bench.py --steps 10 --warmup 3 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['value']/1e6,2),'M rows/s,',round(d['ms_per_step'],2),'ms/step,',round(d['config']['mb_per_sec_roundtrip']/1000,2),'GB/s')"; done
timeout 200 python bench.py --steps 10 --warmup 3 2>/dev/null | tail -1 > gpurun_out/bench_wrap.json
echo "=== suite ==="
timeout 600 python bench_suite.py all --rows 1000000 --reps 3 2>&1 | grep "^{" > gpurun_out/suite_wrap.json
python3 - <<'PY'
import json
for line in open("gpurun_out/suite_wrap.json"):
    d = json.loads(line)
    print(f"  {d['metric'][:50]:52s} {d['value']:,.0f} {d['unit']}")
PY
echo "=== rocprof final ==="
mkdir -p gpurun_out/proffinal
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/proffinal" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 5 --warmup 2 >/dev/null 2>&1
echo done
