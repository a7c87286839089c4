#!/bin/bash
cd /root/repo
mkdir -p gpurun_out/prof
export TMPDIR=/tmp
OUT=gpurun_out/pmc.txt
: > $OUT
cat > /tmp/gb.py <<'PYEOF'
import sys, torch
sys.path.insert(0, "/root/repo")
from tensorflowonspark_amd.ops import get_ext
ext = get_ext(required=True)
M = N = K = 4096
a = (torch.randn(M, K, device="cuda") / 8).bfloat16()
b = (torch.randn(N, K, device="cuda") / 8).bfloat16()
for _ in range(30):
    ext.gemm_bt(a, b, True)
torch.cuda.synchronize()
print("done")
PYEOF
(cd /tmp && timeout 300 rocprofv3 --pmc MfmaUtil VALUBusy OccupancyPercent --kernel-trace --output-format csv -d /root/repo/gpurun_out/prof -o pmcgemm -- python /tmp/gb.py) >> $OUT 2>&1
echo "exit: $?" >> $OUT
(cd /tmp && timeout 300 rocprofv3 --pmc MfmaUtil VALUBusy --kernel-trace --output-format csv -d /root/repo/gpurun_out/prof -o pmcrn -- python /root/repo/bench.py --steps 2 --warmup 1 --batch 512 --feed device) >> $OUT 2>&1
echo "exit: $?" >> $OUT
tail -6 $OUT
ls gpurun_out/prof/ | grep pmc
