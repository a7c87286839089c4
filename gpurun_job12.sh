#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench12.txt
: > $OUT
echo "=== pytest gpu FULL ===" >> $OUT
timeout 800 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -3 >> $OUT
echo "=== smoke ===" >> $OUT
timeout 300 python __graft_entry__.py smoke >> $OUT 2>&1
run() { echo "=== $1 ===" >> $OUT; shift; timeout 420 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "bench default(b1024)" python bench.py --steps 10 --warmup 3
run "bench spark b1024" python bench.py --steps 10 --warmup 3 --feed spark
export TMPDIR=/tmp
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o final -- python /root/repo/bench.py --steps 5 --warmup 2 --batch 1024 --feed device) >> $OUT 2>&1
cat $OUT | grep -E '"value"|passed|failed|smoke OK|exit'
