cd $GRAFT_REPO_ROOT
run() {
  echo "== queues=$1 depth=$2 warmup=$3"
  GPU_MAX_HW_QUEUES=$1 timeout 300 python bench.py --mode inference --steps 300 --warmup $3 --depth $2 2>/dev/null | python3 -c "import json,sys; r=json.load(sys.stdin); print(f'{r[\"value\"]:.0f} bases/s  {r[\"ms_per_step\"]:.3f} ms/step')"
}
run 16 64 80
run 16 96 112
run 16 128 150
run 20 96 112
run 24 96 112
run 12 64 80
