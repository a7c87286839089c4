"""Multi-rank DDP engine self-check on ONE GPU.

gpurun leases are single-GPU, so world_size=2 over RCCL is impossible there
(two ranks cannot share one device); instead both ranks run on cuda:0 with the
gloo backend — the full DDPEngine path (bucket hooks, comm stream, flat
buffers, fused optimizer) executes on GPU tensors with a real 2-rank
all-reduce, so the driver's 8-GPU RCCL run is not the first time the
distributed code meets a GPU. Gradient correctness is asserted against the
single-process average.
"""

import multiprocessing as mp
import os

import pytest
import torch

gpu = pytest.mark.gpu
requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


def _rank_main(rank, world, port, ret):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "RANK": str(rank), "WORLD_SIZE": str(world),
        })
        torch.distributed.init_process_group("gloo", rank=rank,
                                             world_size=world)
        torch.manual_seed(100 + rank)
        from tensorflowonspark_amd.models import MNISTNet
        from tensorflowonspark_amd.ops.modules import BucketSGD
        from tensorflowonspark_amd.parallel import DDPEngine

        dev = torch.device("cuda:0")
        model = MNISTNet().to(dev)
        engine = DDPEngine(model, bucket_mb=1)   # broadcast syncs rank seeds
        opt = BucketSGD(engine, lr=0.1, momentum=0.9)

        torch.manual_seed(7 + rank)             # different data per rank
        x = torch.randn(16, 1, 28, 28, device=dev)
        y = model(x.to(torch.bfloat16) if False else x)
        loss = y.square().mean()
        loss.backward()
        engine.finalize_backward()

        # after all-reduce every rank's bucket grads must be identical:
        # report a checksum
        g = torch.cat([b.buffer.float().flatten() for b in engine._buckets])
        opt.step()
        p = torch.cat([b.param_flat.float().flatten()
                       for b in engine._buckets])
        ret[rank] = (float(g.sum()), float(g.abs().sum()),
                     float(p.sum()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        ret[rank] = ("error", repr(e), "")


@gpu
@requires_gpu
def test_two_rank_engine_on_one_gpu():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    ret = ctx.Manager().dict()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, ret))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    assert ret.get(0) and ret.get(1), ret
    assert ret[0][0] != "error", ret[0]
    assert ret[1][0] != "error", ret[1]
    # identical reduced grads and identical updated params on both ranks
    assert abs(ret[0][0] - ret[1][0]) < 1e-3 * max(1, abs(ret[0][0]))
    assert abs(ret[0][1] - ret[1][1]) < 1e-3 * max(1, abs(ret[0][1]))
    assert abs(ret[0][2] - ret[1][2]) < 1e-3 * max(1, abs(ret[0][2]))


def _rank_resnet(rank, world, port, ret):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "RANK": str(rank), "WORLD_SIZE": str(world),
        })
        torch.distributed.init_process_group("gloo", rank=rank,
                                             world_size=world)
        torch.manual_seed(200 + rank)
        from tensorflowonspark_amd.models import resnet50
        from tensorflowonspark_amd.ops.modules import (BucketSGD,
                                                       softmax_cross_entropy)
        from tensorflowonspark_amd.parallel import DDPEngine

        dev = torch.device("cuda:0")
        model = resnet50(num_classes=100).to(dev) \
            .to(memory_format=torch.channels_last)
        model.train()
        engine = DDPEngine(model, bucket_mb=8)
        opt = BucketSGD(engine, lr=0.05, momentum=0.9)
        torch.manual_seed(31 + rank)
        x = torch.randn(8, 3, 64, 64, device=dev).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        yl = torch.randint(0, 100, (8,), device=dev)
        for _ in range(2):
            opt.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = softmax_cross_entropy(model(x), yl)
            loss.backward()
            engine.finalize_backward()
            opt.step()
        torch.cuda.synchronize()
        p = torch.cat([b.param_flat.float().flatten()
                       for b in engine._buckets])
        ret[rank] = (float(p.sum()), float(p.abs().sum()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        ret[rank] = ("error", repr(e))


@gpu
@requires_gpu
def test_two_rank_resnet50_fused_blocks():
    """ResNet-50 with the fused-block backward under a real 2-rank DDP
    all-reduce (one GPU, gloo over CUDA tensors): both ranks must hold
    identical parameters after two optimizer steps — the exact composition
    the driver's 8-GPU RCCL run exercises."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    ret = ctx.Manager().dict()
    procs = [ctx.Process(target=_rank_resnet, args=(r, 2, port, ret))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=600)
    assert ret.get(0) and ret.get(1), dict(ret)
    assert ret[0][0] != "error", ret[0]
    assert ret[1][0] != "error", ret[1]
    assert abs(ret[0][0] - ret[1][0]) < 1e-3 * max(1, abs(ret[0][0])), ret
    assert abs(ret[0][1] - ret[1][1]) < 1e-3 * max(1, abs(ret[0][1])), ret
