#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench7.txt
: > $OUT
echo "=== conv1x1 test ===" >> $OUT
timeout 400 python -m pytest tests/test_gpu_ops.py -m gpu -q -k conv1x1 2>&1 | tail -3 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b512 mfma+miopenwrw" TFOS_CONV1X1=mfma python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed device
run "b512 miopen" TFOS_CONV1X1=miopen python bench.py --model resnet50 --steps 10 --warmup 3 --batch 512 --feed device
run "b768 mfma" TFOS_CONV1X1=mfma python bench.py --model resnet50 --steps 10 --warmup 3 --batch 768 --feed device
export TMPDIR=/tmp
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o b512i -- env TFOS_CONV1X1=mfma python /root/repo/bench.py --model resnet50 --steps 5 --warmup 2 --batch 512 --feed device) >> $OUT 2>&1
grep -E '"value"|passed|failed|exit' $OUT
