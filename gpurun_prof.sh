set -x
cd /tmp && export TMPDIR=/tmp
cd $GRAFT_REPO_ROOT
mkdir -p gpurun_out
# current numbers first (fast)
timeout 240 python bench.py --mode inference --steps 300 --warmup 50 > gpurun_out/bench_inf_now.log 2>&1
timeout 240 python bench.py --mode train --steps 200 --warmup 30 > gpurun_out/bench_tr_now.log 2>&1
# kernel stats, inference (depth 32 pipelined)
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_inf -o inf -- python $GRAFT_REPO_ROOT/bench.py --mode inference --steps 60 --warmup 10 > $GRAFT_REPO_ROOT/gpurun_out/prof_inf.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_tr -o tr -- python $GRAFT_REPO_ROOT/bench.py --mode train --steps 40 --warmup 10 > $GRAFT_REPO_ROOT/gpurun_out/prof_tr.log 2>&1
tail -2 $GRAFT_REPO_ROOT/gpurun_out/bench_inf_now.log $GRAFT_REPO_ROOT/gpurun_out/bench_tr_now.log
grep -h "gru_layer_fwd\|embed_mlp\|head_fwd" $GRAFT_REPO_ROOT/gpurun_out/prof_inf.log | head -5
