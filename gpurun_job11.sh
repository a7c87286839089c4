#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench11.txt
: > $OUT
echo "=== conv3x3 tests ===" >> $OUT
timeout 500 python -m pytest tests/test_gpu_ops.py -m gpu -q -k "conv3x3" 2>&1 | tail -4 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 420 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b1024 conv3x3=mfma" TFOS_CONV3X3=mfma python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed device
run "b1024 conv3x3=miopen" TFOS_CONV3X3=miopen python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed device
cat $OUT | grep -E '"value"|passed|failed|exit'
