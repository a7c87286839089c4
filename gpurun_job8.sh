#!/bin/bash
cd /root/repo
OUT=gpurun_out/bench8.txt
: > $OUT
echo "=== pytest gpu full ===" >> $OUT
timeout 700 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -4 >> $OUT
echo "=== smoke ===" >> $OUT
timeout 300 python __graft_entry__.py smoke >> $OUT 2>&1
run() { echo "=== $1 ===" >> $OUT; shift; timeout 400 env "$@" >> $OUT 2>&1; echo "exit: $?" >> $OUT; }
run "b768" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 768 --feed device
run "b1024" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 1024 --feed device
run "b768 spark" python bench.py --model resnet50 --steps 10 --warmup 3 --batch 768 --feed spark
echo "=== torchrun world1 rccl ===" >> $OUT
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29511 bench.py --gpus 1 --steps 5 --warmup 2 --batch 256 >> $OUT 2>&1
echo "exit: $?" >> $OUT
grep -E '"value"|passed|failed|smoke|exit' $OUT
