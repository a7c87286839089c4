#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench10.txt
: > $OUT
echo "=== gemm microbench (256 kernel on) ===" >> $OUT
timeout 300 python tools/gemm_bench.py >> $OUT 2>&1
echo "exit: $?" >> $OUT
echo "=== gemm microbench (256 kernel off) ===" >> $OUT
TFOS_GEMM256=off timeout 300 python tools/gemm_bench.py >> $OUT 2>&1
echo "=== conv1x1+gemm tests ===" >> $OUT
timeout 400 python -m pytest tests/test_gpu_ops.py -m gpu -q -k "conv1x1 or gemm or mfma" 2>&1 | tail -3 >> $OUT
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b1024" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed device
cat $OUT
