#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench14.txt
: > $OUT
echo "=== conv tests (wrw=mfma) ===" >> $OUT
timeout 400 python -m pytest tests/test_gpu_ops.py -m gpu -q -k "conv1x1 or conv3x3" 2>&1 | tail -3 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 360 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b1024 wrw=mfma" TFOS_WRW=mfma python bench.py --steps 8 --warmup 3 --batch 1024 --feed device
run "b1024 wrw=miopen" TFOS_WRW=miopen python bench.py --steps 8 --warmup 3 --batch 1024 --feed device
cat $OUT | grep -E '"value"|passed|failed|exit'
