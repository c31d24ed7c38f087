cd $GRAFT_REPO_ROOT
for q in 8 16 32; do
  for d in 32 64; do
    echo "== queues=$q depth=$d"
    GPU_MAX_HW_QUEUES=$q timeout 200 python bench.py --mode inference --steps 200 --warmup 40 --depth $d 2>/dev/null | python3 -c "import json,sys; r=json.load(sys.stdin); print(f'{r[\"value\"]:.0f} bases/s  {r[\"ms_per_step\"]:.3f} ms/step')"
  done
done
