"""PackPlan descriptor math vs the eager torch transforms (CPU).

The batched pack kernel is a pure affine gather driven by these descriptors;
simulating the gather here proves each spec (permute, flip-via-negative-
strides, parity tap classes, zero-padded stem) produces exactly what the
eager packing code produced — without needing a GPU.
"""

import torch

from tensorflowonspark_amd.models.resnet import Bottleneck, conv1x1
from tensorflowonspark_amd.ops import packplan
from tensorflowonspark_amd.ops.modules import FusedBN, StemConv7x7


def _simulate(spec):
    """Replicate pack_bf16_kernel for one descriptor."""
    p, shape, strides, soff, valid = spec
    src = p.detach().reshape(-1)
    out = torch.zeros(*shape, dtype=torch.bfloat16)
    for c0 in range(shape[0]):
        for c1 in range(shape[1]):
            for c2 in range(shape[2]):
                for c3 in range(shape[3]):
                    if c0 < valid[0] and c1 < valid[1] and c2 < valid[2] \
                            and c3 < valid[3]:
                        idx = (soff + c0 * strides[0] + c1 * strides[1]
                               + c2 * strides[2] + c3 * strides[3])
                        out[c0, c1, c2, c3] = src[idx].to(torch.bfloat16)
    return out


def _build_block(cin=32, width=16, stride=1, down=True):
    import torch.nn as nn
    ds = None
    if down:
        ds = nn.Sequential(conv1x1(cin, width * 4, stride),
                           FusedBN(width * 4))
    torch.manual_seed(3)
    return Bottleneck(cin, width, stride, ds)


def test_gemm_row_specs():
    plan = packplan.PackPlan("cpu")
    w = torch.randn(8, 12, 1, 1)
    i1 = packplan._gemm_row_spec(plan, w)
    i2 = packplan._gemm_rowT_spec(plan, w)
    got1 = _simulate(plan.specs[i1]).reshape(8, 12)
    got2 = _simulate(plan.specs[i2]).reshape(12, 8)
    assert torch.equal(got1, w.view(8, 12).to(torch.bfloat16))
    assert torch.equal(got2, w.view(8, 12).t().contiguous().to(torch.bfloat16))


def test_w9_and_w9p_specs():
    plan = packplan.PackPlan("cpu")
    w = torch.randn(6, 4, 3, 3)
    i1 = packplan._w9_spec(plan, w)
    i2 = packplan._w9p_spec(plan, w)
    got_w9 = _simulate(plan.specs[i1]).reshape(6, 9 * 4)
    ref_w9 = w.permute(0, 2, 3, 1).reshape(6, 36).to(torch.bfloat16)
    assert torch.equal(got_w9, ref_w9)
    got_w9p = _simulate(plan.specs[i2]).reshape(4, 9 * 6)
    ref_w9p = w.flip(2, 3).permute(1, 2, 3, 0).reshape(4, 54) \
        .to(torch.bfloat16)
    assert torch.equal(got_w9p, ref_w9p)


def test_parity_class_specs():
    plan = packplan.PackPlan("cpu")
    w = torch.randn(6, 4, 3, 3)
    wperm = w.flip(2, 3).permute(1, 2, 3, 0)  # [ci][r][s][co]
    for (ph, pw), (rl, sl) in packplan._class_taps(3, 1).items():
        i = packplan._w9p_class_spec(plan, w, rl, sl)
        got = _simulate(plan.specs[i])            # [ci, nr, ns, co]
        ref = torch.stack(
            [wperm[:, r, s, :] for r in rl for s in sl], dim=1) \
            .reshape(4, len(rl), len(sl), 6).to(torch.bfloat16)
        assert torch.equal(got, ref), (ph, pw)


def test_stem_spec():
    plan = packplan.PackPlan("cpu")
    w = torch.randn(5, 3, 7, 7)
    i = packplan._stem_spec(plan, w)
    got = _simulate(plan.specs[i])               # [co, 7, 8, 4]
    ref = torch.zeros(5, 7, 8, 4, dtype=torch.bfloat16)
    ref[:, :, :7, :3] = w.permute(0, 2, 3, 1).to(torch.bfloat16)
    assert torch.equal(got, ref)


def test_build_resnet_plan_layout():
    """Plan over a tiny ResNet covers every fused block and the stem, with
    disjoint arena regions and correct view shapes."""
    from tensorflowonspark_amd.models import resnet50
    m = resnet50(num_classes=10)
    # force-build on CPU: only descriptor construction, no kernel launch
    plan = packplan.build_resnet_plan(m, "cpu")
    assert plan is not None
    fused = [b for b in m.modules()
             if isinstance(b, Bottleneck) and b._block_fusable]
    stems = [b for b in m.modules() if isinstance(b, StemConv7x7)]
    assert fused and stems
    for b in fused:
        packs = b._tfos_packs
        for k in ("w1b", "w9", "w3b", "w3bT", "w1bT"):
            assert k in packs
        if b.stride == 2:
            assert "w9p_00" in packs and "w9p_11" in packs
        else:
            assert "w9p" in packs
        if b.downsample is not None:
            assert "wdb" in packs and "wdbT" in packs
    assert "w224" in stems[0]._tfos_packs
    # arena slices are disjoint and cover the arena exactly
    total = sum(int(torch.tensor(sh).prod())
                for _p, sh, _s, _o, _v in plan.specs)
    assert total == plan._total == plan._arena.numel()
