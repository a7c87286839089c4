#!/bin/bash
cd /root/repo
mkdir -p gpurun_out
echo "=== amd-smi list ===" > gpurun_out/smi.txt
amd-smi list --json >> gpurun_out/smi.txt 2>&1 || true
echo "=== amd-smi process ===" >> gpurun_out/smi.txt
amd-smi process --json >> gpurun_out/smi.txt 2>&1 || true
echo "=== rocm-smi showid ===" >> gpurun_out/smi.txt
rocm-smi --showid --json >> gpurun_out/smi.txt 2>&1 || true
echo "=== pytest gpu ===" > gpurun_out/pytest_gpu.txt
timeout 600 python -m pytest tests/test_gpu_ops.py -m gpu -q 2>&1 | tail -40 >> gpurun_out/pytest_gpu.txt
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.txt
echo "=== bench device feed ===" > gpurun_out/bench1.txt
timeout 300 python bench.py --model resnet50 --steps 10 --warmup 3 --batch 64 --feed device >> gpurun_out/bench1.txt 2>&1
echo "exit: $?" >> gpurun_out/bench1.txt
echo "=== bench spark feed ===" >> gpurun_out/bench1.txt
timeout 300 python bench.py --model resnet50 --steps 10 --warmup 3 --batch 64 --feed spark >> gpurun_out/bench1.txt 2>&1
echo "exit: $?" >> gpurun_out/bench1.txt
tail -5 gpurun_out/pytest_gpu.txt
tail -8 gpurun_out/bench1.txt
