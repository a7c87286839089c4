#!/bin/bash
cd /root/repo
mkdir -p gpurun_out/prof
OUT=gpurun_out/bench5.txt
: > $OUT
echo "=== pytest gpu all ===" >> $OUT
timeout 700 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -4 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b256" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 256 --feed device
run "b512" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed device
run "b512 spark" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed spark
run "unet b64" python bench.py --model unet --steps 10 --warmup 3 --batch 64 --feed device
run "deeplab b32" python bench.py --model deeplabv3 --steps 10 --warmup 3 --batch 32 --feed device
echo "=== b256 find1 ===" >> $OUT
MIOPEN_FIND_MODE=1 timeout 900 python bench.py --model resnet50 --steps 10 --warmup 15 --batch 256 --feed device >> $OUT 2>&1
echo "exit: $?" >> $OUT
export TMPDIR=/tmp
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o b512h -- python /root/repo/bench.py --model resnet50 --steps 5 --warmup 2 --batch 512 --feed device) >> $OUT 2>&1
grep -E '"value"|passed|failed|exit' $OUT
