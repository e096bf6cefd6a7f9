cd "$GRAFT_REPO_ROOT"
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -1
timeout 600 python bench_suite.py all --rows 1000000 --reps 3 2>&1 | grep "^{" > gpurun_out/suite_final.json
timeout 300 python bench.py --steps 5 --warmup 2 --rows 4000000 2>/dev/null | tail -1 > gpurun_out/bench4m_final.json
timeout 200 python bench.py --steps 10 --warmup 3 2>/dev/null | tail -1 > gpurun_out/bench1m_final.json
timeout 120 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -1
cat gpurun_out/suite_final.json
python3 -c "import json; d=json.load(open('gpurun_out/bench1m_final.json')); print('1M:', round(d['value']/1e6,1), 'M rows/s')"
python3 -c "import json; d=json.load(open('gpurun_out/bench4m_final.json')); print('4M:', round(d['value']/1e6,1), 'M rows/s')"
