"""Small remaining surfaces: extension dispatch policy, DataFeed array blocks,
PS client sharding, gemm CPU fallback."""

import numpy as np
import pytest
import torch

from tensorflowonspark_amd import TFManager, TFNode
from tensorflowonspark_amd.ops import modules
from tensorflowonspark_amd.utils import shmring


def test_get_ext_required_raises_without_so(monkeypatch):
    """On a GPU box a missing extension must fail loudly, not fall back."""
    import tensorflowonspark_amd.ops as ops
    monkeypatch.setattr(ops, "_ext", None)
    monkeypatch.setattr(ops, "_ext_checked", True)
    monkeypatch.delenv("TFOS_ALLOW_EAGER_FALLBACK", raising=False)
    with pytest.raises(RuntimeError, match="not built"):
        ops.get_ext(required=True)
    monkeypatch.setenv("TFOS_ALLOW_EAGER_FALLBACK", "1")
    assert ops.get_ext(required=True) is None  # explicit debug escape hatch


def test_gemm_bf16_cpu_fallback():
    a = torch.randn(5, 8).bfloat16()
    b = torch.randn(8, 3).bfloat16()
    out = modules.gemm_bf16(a, b)
    ref = a.float() @ b.float()
    assert torch.allclose(out, ref, atol=1e-2, rtol=1e-2)


def test_datafeed_next_arrays_roundtrip():
    mgr = TFManager.start(b"k", ["input", "output", "error", "free"], "local")
    try:
        ring = shmring.BlockRing("tfosr_arr_test", 3, 1 << 20,
                                 data_queue=mgr.get_queue("input"),
                                 free_queue=mgr.get_queue("free"), create=True)
        mgr.set("ring_name", ring.name)
        mgr.set("ring_slots", 3)
        mgr.set("ring_slot_bytes", 1 << 20)
        x = np.arange(24, dtype=np.uint8).reshape(2, 3, 4)
        y = np.array([7, 9], dtype=np.int64)
        ring.put_arrays({"x": x, "y": y})
        mgr.get_queue("input").put(None)

        feed = TFNode.DataFeed(mgr, train_mode=True)
        out = feed.next_arrays()
        assert np.array_equal(out["x"], x) and np.array_equal(out["y"], y)
        assert feed.next_arrays() is None
        assert feed.should_stop()
        # next_arrays_into variant
        ring2 = shmring.BlockRing("tfosr_arr_test2", 3, 1 << 20,
                                  data_queue=mgr.get_queue("input"),
                                  free_queue=mgr.get_queue("free"), create=True)
        mgr.set("ring_name", ring2.name)
        feed2 = TFNode.DataFeed(mgr, train_mode=True)
        ring2.put_arrays({"x": x, "y": y})
        dst = {"x": np.zeros_like(x), "y": np.zeros_like(y)}
        assert feed2.next_arrays_into(dst)
        assert np.array_equal(dst["x"], x) and np.array_equal(dst["y"], y)
        ring.close(); ring.unlink(); ring2.close(); ring2.unlink()
    finally:
        mgr.shutdown()


def test_ps_client_sharding():
    from tensorflowonspark_amd.parallel.ps import PSClient
    c = PSClient(["h1:1", "h2:2", "h3:3"])
    assert [c._shard(i) for i in range(6)] == [0, 1, 2, 0, 1, 2]


def test_piecewise_and_cosine_schedules():
    from tensorflowonspark_amd.utils.schedule import CosineLR, PiecewiseLR
    s = PiecewiseLR(0.4, warmup_epochs=2, boundaries=[10], decays=[0.5])
    assert s(1.0) == pytest.approx(0.2)
    assert s(5) == pytest.approx(0.4)
    assert s(11) == pytest.approx(0.2)
    c = CosineLR(1.0, 10)
    assert c(5) == pytest.approx(0.5, abs=1e-6)


def test_bucket_adamw_matches_torch():
    from tensorflowonspark_amd.ops.modules import BucketAdam
    from tensorflowonspark_amd.parallel import DDPEngine
    torch.manual_seed(5)
    m1 = torch.nn.Linear(6, 3)
    m2 = torch.nn.Linear(6, 3)
    m2.load_state_dict(m1.state_dict())
    ref = torch.optim.AdamW(m1.parameters(), lr=0.01, weight_decay=0.05)
    e = DDPEngine(m2, bucket_mb=1)
    opt = BucketAdam(e, lr=0.01, weight_decay=0.05, decoupled=True)
    x = torch.randn(12, 6)
    y = torch.randn(12, 3)
    for _ in range(4):
        ref.zero_grad()
        torch.nn.functional.mse_loss(m1(x), y).backward()
        ref.step()
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(m2(x), y)
        loss.backward()
        e.finalize_backward()
        opt.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_dilated_resnet_output_stride_16():
    """DeepLabV3 backbone: stage-4 stride must become dilation even when the
    convs are the routed MFMA modules (round-2 regression: the original
    surgery only matched nn.Conv2d and silently left output stride 32)."""
    import torch

    from tensorflowonspark_amd.models.segmentation import DilatedResNet50
    m = DilatedResNet50()
    m.eval()
    with torch.no_grad():
        f = m(torch.randn(1, 3, 64, 64))
    assert f.shape[-1] == 64 // 16, f.shape
    # fused-block path must be off for the dilated blocks
    assert all(not b._block_fusable for b in m.stages[3])


def test_steptimer_rate():
    from tensorflowonspark_amd.utils.metrics import StepTimer
    t = StepTimer(batch_size=10, log_every=2)
    assert t.rate() == 0.0
    t.step()  # starts the window
    t.step()
    t.step()  # completes a window of 2 -> rate set
    assert t.rate() > 0.0
