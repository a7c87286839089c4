# Build/test entry points (CI shape parity with the reference's tox targets).

.PHONY: build test test-gpu bench bench-kernels lint clean

build:
	python __graft_entry__.py

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 20 --warmup 5

# per-kernel microbenchmarks (GPU box): conv shapes, weight-grad routing
# table, BN bandwidth, DDP bucket-size sweep
bench-kernels:
	python tools/conv_bench.py --batch 1024
	python tools/conv_bench.py --wrw --batch 1024
	python tools/bn_bench.py --batch 1024
	python tools/bucket_sweep.py --steps 6 --warmup 2

lint:
	python -m pycodestyle --max-line-length=160 tensorflowonspark_amd || true

clean:
	rm -rf build tensorflowonspark_amd/ops/tfosr_hip_ops.so tools/bin
