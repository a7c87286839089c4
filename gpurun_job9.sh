#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench9.txt
: > $OUT
echo "=== pytest gpu full ===" >> $OUT
timeout 700 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -4 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b1024 mask" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed device
run "b1024 spark" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed spark
export TMPDIR=/tmp
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o b1024j -- python /root/repo/bench.py --model resnet50 --steps 5 --warmup 2 --batch 1024 --feed device) >> $OUT 2>&1
grep -E '"value"|passed|failed|exit' $OUT
