#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 sync-SGD images/sec, InputMode.SPARK feed.

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``; for N>1
launched under ``torch.distributed.run`` with one rank per GPU (RANK/LOCAL_RANK/
WORLD_SIZE/MASTER_* in env). W untimed warmup steps, then exactly K timed steps
bracketed by barrier + torch.cuda.synchronize on both sides; rank 0 prints one
JSON line with the whole-job aggregate.

The default feed is the framework's real InputMode.SPARK ingest path: a feeder
process per rank pushes synthetic ImageNet-shaped uint8 blocks through the
shared-memory ring + manager (exactly what Spark feeder tasks do), and the
training loop consumes them via TFNode.DataFeed, packs NHWC->NCHW on GPU, and
overlaps H2D with compute. ``--feed device`` generates batches on-GPU instead
(isolates compute from ingest for profiling).
"""

import argparse
import json
import multiprocessing
import os
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=0,
                   help="per-GPU batch size (0 = auto: 1024 on GPU, 8 on CPU)")
    p.add_argument("--model", default="resnet50",
                   choices=["resnet50", "resnet56_cifar", "mnist_cnn", "mnist_mlp",
                            "unet", "deeplabv3"])
    p.add_argument("--feed", default="spark", choices=["spark", "device"],
                   help="spark = real shm-ring ingest path; device = on-GPU synthetic")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--channels-last", action="store_true", default=True)
    p.add_argument("--no-channels-last", dest="channels_last", action="store_false")
    p.add_argument("--bucket-mb", type=float, default=25.0)
    return p.parse_args()


def _feeder_main(mgr_addr, authkey, shape, batch, nclasses, y_shape, stop_evt):
    """Feeder process: emulates the Spark feeder task — synthetic uint8 image
    blocks + int64 labels through the manager/ring (one block per batch)."""
    from tensorflowonspark_amd import TFManager
    from tensorflowonspark_amd.utils import shmring
    mgr = TFManager.connect(mgr_addr, authkey)
    ring = shmring.BlockRing(mgr.get("ring_name"), mgr.get("ring_slots"),
                             mgr.get("ring_slot_bytes"),
                             data_queue=mgr.get_queue("input"),
                             free_queue=mgr.get_queue("free"), create=False)
    rng = np.random.default_rng(1234)
    # pre-generate a pool of blocks; cycle through (generation must not be
    # the bottleneck we measure — this emulates rows arriving from Spark)
    pool = []
    for _ in range(4):
        x = rng.integers(0, 256, size=(batch,) + shape, dtype=np.uint8)
        y = rng.integers(0, nclasses, size=(batch,) + y_shape, dtype=np.int64)
        pool.append({"x": x, "y": y})
    i = 0
    while not stop_evt.is_set():
        try:
            ring.put_arrays(pool[i % len(pool)], timeout=1)
            i += 1
        except Exception:
            continue
    ring.close()


class SparkFeed:
    """Worker-side consumer: DataFeed -> pinned staging -> async H2D."""

    def __init__(self, shape, batch, nclasses, device, slot_bytes, y_shape=()):
        from tensorflowonspark_amd import TFManager, TFNode
        authkey = b"benchkey"
        self.mgr = TFManager.start(authkey, ["input", "output", "error", "free"],
                                   "local")
        from tensorflowonspark_amd.utils import shmring
        ring_name = "tfosr_bench_{}".format(os.getpid())
        # bound /dev/shm use: 8 ranks x big batches must not exhaust shm
        # 4 slots x ~150MB x 8 ranks stays well inside /dev/shm
        slots = max(2, min(4, (1 << 30) // max(1, slot_bytes)))
        while True:
            try:
                self.ring = shmring.BlockRing(
                    ring_name, slots, slot_bytes,
                    data_queue=self.mgr.get_queue("input"),
                    free_queue=self.mgr.get_queue("free"), create=True)
                break
            except OSError:
                if slots <= 2:
                    raise
                slots //= 2
        self.mgr.set("ring_name", ring_name)
        self.mgr.set("ring_slots", slots)
        self.mgr.set("ring_slot_bytes", slot_bytes)
        self.mgr.set("state", "running")
        self.stop_evt = multiprocessing.Event()
        self.proc = multiprocessing.Process(
            target=_feeder_main,
            args=(self.mgr.address, authkey, shape, batch, nclasses, y_shape, self.stop_evt),
            daemon=True)
        self.proc.start()
        self.feed = TFNode.DataFeed(self.mgr, train_mode=True)
        self.device = device
        self.use_cuda = device.type == "cuda"
        self._stop = False
        if self.use_cuda:
            # double-buffered pinned staging + background prefetch thread:
            # shm -> pinned (thread) overlaps compute; pinned -> HBM is an
            # async hipMemcpyAsync on a side stream
            import queue as pyq
            import threading
            self.nbuf = 3
            self.x_pin = [torch.empty((batch,) + shape, dtype=torch.uint8,
                                      pin_memory=True) for _ in range(self.nbuf)]
            self.y_pin = [torch.empty((batch,) + y_shape, dtype=torch.int64,
                                      pin_memory=True) for _ in range(self.nbuf)]
            self.copy_stream = torch.cuda.Stream()
            self._ready = pyq.Queue(maxsize=self.nbuf)
            self._free = pyq.Queue()
            for i in range(self.nbuf):
                self._free.put((i, None))
            self._thread = threading.Thread(target=self._prefetch_loop, daemon=True)
            self._thread.start()

    def _prefetch_loop(self):
        while not self._stop:
            i, evt = self._free.get()
            if i is None:
                break
            if evt is not None:
                evt.synchronize()  # prior H2D from this buffer must be done
            dest = {"x": self.x_pin[i].numpy(), "y": self.y_pin[i].numpy()}
            if not self.feed.next_arrays_into(dest):
                self._ready.put(None)
                break
            self._ready.put(i)

    def next(self):
        if self.use_cuda:
            i = self._ready.get()
            if i is None:
                raise StopIteration("feed ended")
            with torch.cuda.stream(self.copy_stream):
                x = self.x_pin[i].to(self.device, non_blocking=True)
                y = self.y_pin[i].to(self.device, non_blocking=True)
                evt = torch.cuda.Event()
                evt.record(self.copy_stream)
            torch.cuda.current_stream().wait_stream(self.copy_stream)
            self._free.put((i, evt))
            return x, y
        arrs = self.feed.next_arrays()
        return torch.from_numpy(arrs["x"]), torch.from_numpy(arrs["y"])

    def close(self):
        self._stop = True
        self.stop_evt.set()
        if self.use_cuda:
            self._free.put((None, None))
        self.proc.join(timeout=10)
        if self.proc.is_alive():
            self.proc.terminate()
        self.ring.close()
        self.ring.unlink()
        self.mgr.shutdown()


def build_model(name, device):
    from tensorflowonspark_amd import models
    if name == "resnet50":
        return models.resnet50(num_classes=1000).to(device), (224, 224, 3), 1000
    if name == "resnet56_cifar":
        return models.resnet56_cifar().to(device), (32, 32, 3), 10
    if name == "mnist_cnn":
        return models.MNISTNet().to(device), (28, 28, 1), 10
    if name == "mnist_mlp":
        # BASELINE config 1: world_size=2 CPU/gloo plumbing model
        return models.MNISTMLP().to(device), (28, 28, 1), 10
    if name == "unet":
        from tensorflowonspark_amd.models.segmentation import unet_mobilenet
        return unet_mobilenet(num_classes=3).to(device), (128, 128, 3), 3
    if name == "deeplabv3":
        from tensorflowonspark_amd.models.segmentation import deeplabv3_resnet50
        return deeplabv3_resnet50(num_classes=21).to(device), (512, 512, 3), 21
    raise ValueError(name)


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    # WORLD_SIZE comes from torch.distributed.run; a bare `--gpus N` without
    # a launcher must NOT wait for peers that will never arrive
    world = int(os.environ.get("WORLD_SIZE", "1"))
    use_cuda = torch.cuda.is_available()
    distributed = world > 1

    if distributed:
        import torch.distributed as dist
        backend = "nccl" if use_cuda else "gloo"
        if use_cuda:
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        dist.init_process_group(backend=backend)
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) \
        if use_cuda else torch.device("cpu")

    batch = args.batch or (1024 if use_cuda else 8)
    model, shape, nclasses = build_model(args.model, device)
    if args.channels_last and use_cuda:
        model = model.to(memory_format=torch.channels_last)
    model.train()

    from tensorflowonspark_amd.ops.modules import BucketSGD, nhwc_pack, \
        softmax_cross_entropy
    from tensorflowonspark_amd.parallel import DDPEngine
    engine = DDPEngine(model, bucket_mb=args.bucket_mb)
    # linear-scaling LR, capped so synthetic runs at huge global batch
    # stay numerically sane
    opt = BucketSGD(engine, lr=0.1 * min(world * batch, 2048) / 256,
                    momentum=0.9, weight_decay=1e-4)

    amp_dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    amp = torch.autocast(device_type=device.type, dtype=amp_dtype,
                         enabled=args.dtype == "bf16")

    seg = args.model in ("unet", "deeplabv3")
    y_shape = tuple(shape[:2]) if seg else ()
    slot_bytes = (int(np.prod((batch,) + shape))
                  + int(np.prod((batch,) + y_shape)) * 8 + (1 << 16))
    feed = None
    if args.feed == "spark":
        feed = SparkFeed(shape, batch, nclasses, device, slot_bytes,
                         y_shape=y_shape)
    else:
        rng = torch.Generator(device="cpu").manual_seed(rank)
        x_dev = torch.randint(0, 256, (batch,) + shape, dtype=torch.uint8,
                              generator=rng).to(device)
        y_dev = torch.randint(0, nclasses, (batch,) + y_shape, dtype=torch.int64,
                              generator=rng).to(device)

    def get_batch():
        if feed is not None:
            return feed.next()
        return x_dev, y_dev

    def step():
        x_u8, y = get_batch()
        opt.zero_grad()
        with amp:
            x = nhwc_pack(x_u8, out_dtype=amp_dtype if use_cuda else torch.float32,
                          channels_last=args.channels_last and use_cuda)
            if args.channels_last and use_cuda:
                x = x.contiguous(memory_format=torch.channels_last)
            logits = model(x)
            if seg:
                c = logits.shape[1]
                flat = logits.permute(0, 2, 3, 1).reshape(-1, c)
                loss = softmax_cross_entropy(flat, y.reshape(-1))
            else:
                loss = softmax_cross_entropy(logits, y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        return loss

    def barrier_sync():
        if distributed:
            torch.distributed.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.time()
    for _ in range(args.steps):
        loss = step()
    barrier_sync()
    elapsed = time.time() - t0

    # MAX over ranks
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    if feed is not None:
        feed.close()

    if rank == 0:
        images_per_sec = world * batch * args.steps / elapsed
        ms_per_step = elapsed / args.steps * 1000
        result = {
            "metric": "images/sec (whole node) ResNet-50 sync-SGD InputMode.SPARK"
                      if args.model == "resnet50" else
                      "images/sec {} sync-SGD".format(args.model),
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no number
            "dtype": args.dtype,
            "data": "synthetic ({} feed)".format(args.feed),
            "config": {
                "model": args.model,
                "global_batch": world * batch,
                "per_gpu_batch": batch,
                "image_shape": list(shape),
                "parallelism": "dp{}".format(world),
                "loss_final": round(float(loss.detach().float().cpu()), 4),
            },
        }
        print(json.dumps(result))

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
