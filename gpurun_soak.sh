cd $GRAFT_REPO_ROOT
echo "=== inference soak (3000 steps)"
timeout 420 python bench.py --steps 3000 --warmup 80 2>/dev/null | python3 -c "import json,sys; r=json.load(sys.stdin); print(f'{r[\"value\"]/1e6:.2f} M bases/s over 3000 steps')"
echo "=== train soak (1500 steps)"
timeout 420 python bench.py --mode train --steps 1500 --warmup 50 2>/dev/null | python3 -c "import json,sys; r=json.load(sys.stdin); print(f'{r[\"value\"]:.0f} windows/s over 1500 steps')"
cd /tmp && export TMPDIR=/tmp
cat > /tmp/grub.py <<'PY'
import sys; sys.path.insert(0,'/root/repo')
import torch
from roko_amd.model import RokoModel
from roko_amd.ops.train import fused_train_step, FusedAdam
from roko_amd import config as C
m = RokoModel().cuda().train()
opt = FusedAdam(list(m.parameters()), lr=C.LR)
x = torch.randint(0,12,(128,200,90),dtype=torch.uint8,device='cuda')
y = torch.randint(0,5,(128,90),device='cuda')
for _ in range(15): fused_train_step(m, x, y, opt)
torch.cuda.synchronize()
PY
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE -d $GRAFT_REPO_ROOT/gpurun_out/pmc_tr -o tr -- python /tmp/grub.py > $GRAFT_REPO_ROOT/gpurun_out/pmc_tr.log 2>&1
echo pmc rc=$?
