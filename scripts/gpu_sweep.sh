cd "$GRAFT_REPO_ROOT"
for rs in 25165824 50331648 100663296; do
  for ws in 2 3 5; do
    v=$(TFREC_READ_SLICE=$rs TFREC_WRITE_SLICES=$ws timeout 200 python bench.py --steps 8 --warmup 2 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print(f'{d[\"ms_per_step\"]:.2f}')")
    echo "read_slice=$((rs>>20))MB write_slices=$ws -> $v ms/step"
  done
done
