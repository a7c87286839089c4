#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench15.txt
: > $OUT
echo "=== pytest gpu FULL ===" >> $OUT
timeout 700 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -3 >> $OUT
echo "=== smoke ===" >> $OUT
timeout 240 python __graft_entry__.py smoke >> $OUT 2>&1
run() { echo "=== $1 ===" >> $OUT; shift; timeout 300 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "bench default" python bench.py --steps 10 --warmup 3
cat $OUT | grep -E '"value"|passed|failed|smoke OK|exit'
