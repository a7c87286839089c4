#!/bin/bash
cd /root/repo
mkdir -p gpurun_out/prof
OUT=gpurun_out/bench2.txt
: > $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b256 device channels_last" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 256 --feed device
run "b256 device NCHW" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 256 --feed device --no-channels-last
run "b256 spark" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 256 --feed spark
run "b512 device" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed device
export TMPDIR=/tmp
echo "=== rocprof stats ===" >> $OUT
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o b256 -- python /root/repo/bench.py --model resnet50 --steps 5 --warmup 2 --batch 256 --feed device) >> $OUT 2>&1
echo "exit: $?" >> $OUT
grep -E '"value"|exit' $OUT
ls -la gpurun_out/prof | head
