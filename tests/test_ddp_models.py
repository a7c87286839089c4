"""DDPEngine + models: single-process numerics and world_size=2 gloo sync."""

import multiprocessing
import os

import pytest
import torch

from tensorflowonspark_amd.models import MNISTMLP, MNISTNet, resnet50, resnet56_cifar
from tensorflowonspark_amd.ops.modules import BucketSGD, softmax_cross_entropy
from tensorflowonspark_amd.parallel import DDPEngine


def _train_vanilla(model, data, target, lr, steps):
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    for _ in range(steps):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(data), target)
        loss.backward()
        opt.step()
    return loss.item()


def _train_engine(model, data, target, lr, steps):
    engine = DDPEngine(model, bucket_mb=1)
    opt = BucketSGD(engine, lr=lr, momentum=0.9)
    for _ in range(steps):
        opt.zero_grad()
        loss = softmax_cross_entropy(model(data), target)
        loss.backward()
        engine.finalize_backward()
        opt.step()
    return loss.item()


def test_engine_matches_vanilla_sgd():
    torch.manual_seed(0)
    data = torch.randn(32, 1, 28, 28)
    target = torch.randint(0, 10, (32,))
    m1 = MNISTMLP()
    m2 = MNISTMLP()
    m2.load_state_dict(m1.state_dict())
    l1 = _train_vanilla(m1, data, target, 0.05, 5)
    l2 = _train_engine(m2, data, target, 0.05, 5)
    assert abs(l1 - l2) < 1e-5
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), "flat-param training diverged"


def test_models_forward_shapes():
    assert MNISTNet()(torch.randn(2, 1, 28, 28)).shape == (2, 10)
    assert resnet56_cifar()(torch.randn(2, 3, 32, 32)).shape == (2, 10)
    assert resnet50()(torch.randn(1, 3, 64, 64)).shape == (1, 1000)


def test_resnet50_param_count():
    n = sum(p.numel() for p in resnet50().parameters())
    assert abs(n - 25.55e6) < 0.2e6, n  # canonical ResNet-50 ~25.5M params


def _ddp_worker(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(42 + rank)  # different init per rank: broadcast must fix it
    model = MNISTMLP(hidden=32)
    engine = DDPEngine(model, bucket_mb=1)
    opt = BucketSGD(engine, lr=0.1, momentum=0.9)
    # each rank trains on a *different* shard
    g = torch.Generator().manual_seed(100 + rank)
    data = torch.randn(16, 784, generator=g)
    target = torch.randint(0, 10, (16,), generator=g)
    for _ in range(3):
        opt.zero_grad()
        loss = softmax_cross_entropy(model(data), target)
        loss.backward()
        engine.finalize_backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    result_q.put((rank, flat.sum().item(), flat[:5].tolist()))
    torch.distributed.destroy_process_group()


def test_world2_gloo_params_stay_in_sync():
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, total, head = q.get(timeout=120)
        results[rank] = (total, head)
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    assert results[0] == pytest.approx(results[1]), \
        "ranks diverged: {} vs {}".format(results[0], results[1])


def _uneven_worker(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    model = MNISTMLP(hidden=16)
    engine = DDPEngine(model, bucket_mb=1)
    opt = BucketSGD(engine, lr=0.05)
    # rank 0 has 5 batches, rank 1 only 3: without the guard rank 0's 4th
    # all-reduce would hang forever
    nbatches = 5 if rank == 0 else 3
    data = [torch.randn(8, 784) for _ in range(nbatches)]
    steps = 0
    it = iter(data)
    while True:
        batch = next(it, None)
        if not engine.all_ranks_ready(batch is not None):
            break
        opt.zero_grad()
        loss = softmax_cross_entropy(model(batch), torch.randint(0, 10, (8,)))
        loss.backward()
        engine.finalize_backward()
        opt.step()
        steps += 1
    result_q.put((rank, steps))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_uneven_partition_guard():
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_uneven_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = dict(q.get(timeout=120) for _ in range(2))
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    # both ranks stopped together after the shorter feed (3 steps)
    assert results == {0: 3, 1: 3}


def test_no_sync_gradient_accumulation():
    """no_sync() skips bucket reduction; grads accumulate across micro-steps
    and one synced step matches a single large-batch step."""
    torch.manual_seed(3)
    data = torch.randn(32, 784)
    target = torch.randint(0, 10, (32,))

    m1 = MNISTMLP(hidden=16)
    m2 = MNISTMLP(hidden=16)
    m2.load_state_dict(m1.state_dict())

    # big-batch reference
    e1 = DDPEngine(m1, bucket_mb=1)
    o1 = BucketSGD(e1, lr=0.1, momentum=0.0)
    o1.zero_grad()
    loss = softmax_cross_entropy(m1(data), target, reduction="sum") / 32
    loss.backward()
    e1.finalize_backward()
    o1.step()

    # two accumulated micro-batches
    e2 = DDPEngine(m2, bucket_mb=1)
    o2 = BucketSGD(e2, lr=0.1, momentum=0.0)
    o2.zero_grad()
    with e2.no_sync():
        l1 = softmax_cross_entropy(m2(data[:16]), target[:16], reduction="sum") / 32
        l1.backward()
    l2 = softmax_cross_entropy(m2(data[16:]), target[16:], reduction="sum") / 32
    l2.backward()
    e2.finalize_backward()
    o2.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_bucket_sgd_nesterov():
    torch.manual_seed(4)
    m1 = MNISTMLP(hidden=8)
    m2 = MNISTMLP(hidden=8)
    m2.load_state_dict(m1.state_dict())
    ref = torch.optim.SGD(m1.parameters(), lr=0.05, momentum=0.9, nesterov=True)
    e = DDPEngine(m2, bucket_mb=1)
    opt = BucketSGD(e, lr=0.05, momentum=0.9, nesterov=True)
    x = torch.randn(8, 784)
    y = torch.randint(0, 10, (8,))
    for _ in range(3):
        ref.zero_grad()
        torch.nn.functional.cross_entropy(m1(x), y).backward()
        ref.step()
        opt.zero_grad()
        loss = softmax_cross_entropy(m2(x), y)
        loss.backward()
        e.finalize_backward()
        opt.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()
