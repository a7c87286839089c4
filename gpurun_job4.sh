#!/bin/bash
cd /root/repo
mkdir -p gpurun_out/prof
OUT=gpurun_out/bench4.txt
: > $OUT
echo "=== pytest bn ops ===" >> $OUT
timeout 600 python -m pytest tests/test_gpu_ops.py -m gpu -q -k "bn or resnet" 2>&1 | tail -6 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b256 fused2" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 256 --feed device
run "b512 fused2" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed device
export TMPDIR=/tmp
echo "=== rocprof stats ===" >> $OUT
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o b256g -- python /root/repo/bench.py --model resnet50 --steps 5 --warmup 2 --batch 256 --feed device) >> $OUT 2>&1
echo "exit: $?" >> $OUT
grep -E '"value"|passed|failed|exit' $OUT
