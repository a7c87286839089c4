"""Batched per-step weight packing (one kernel launch per step).

The fused ResNet path consumes per-conv packed bf16 weights (GEMM rows,
im2col-permuted 3x3 filters, flipped backward-data filters, parity-class tap
subsets, the zero-padded stem filter). Packing them with eager torch ops costs
~300 tiny permute/cast launches per step (~3-4 ms at b1024 — profiles/README
backlog item 3). Every one of those transforms is an affine gather from the
fp32 parameter storage (flips = negative strides, zero padding = source-extent
clamps, parity tap subsets = Cartesian (r, s) products), so a single
descriptor-table kernel (``pack_bf16`` in ``elementwise.hip``) produces all of
them into one bf16 arena in one launch; the plan re-runs only when a parameter
``_version`` changes (i.e. once per optimizer step, not per micro-batch).
"""

import struct

import torch

from . import get_ext

_DESC_FMT = "<qqqi4i4i4i4x"  # src_ptr, dst_off, soff, n, od[4], ss[4], sv[4]
assert struct.calcsize(_DESC_FMT) == 8 * 3 + 4 + 48 + 4


class PackPlan:
    def __init__(self, device):
        self.device = device
        self.specs = []          # (param, out_shape, strides, soff, valid)
        self.views = []          # filled by finalize(): arena views
        self._arena = None
        self._descbuf = None
        self._cumbuf = None
        self._total = 0
        self._params = []
        self._versions = None

    def add(self, param, out_shape, strides, soff, valid=None):
        """Register one packed tensor; returns its index (view after
        finalize()). out_shape/strides/valid are padded to 4 dims (leading
        1s / 0-strides)."""
        out_shape = list(out_shape)
        strides = list(strides)
        valid = list(valid) if valid is not None else list(out_shape)
        while len(out_shape) < 4:
            out_shape.insert(0, 1)
            strides.insert(0, 0)
            valid.insert(0, 1)
        assert len(out_shape) == 4 and param.dtype == torch.float32
        self.specs.append((param, out_shape, strides, soff, valid))
        return len(self.specs) - 1

    def finalize(self):
        offs, cum = [], [0]
        total = 0
        for _p, shape, _s, _o, _v in self.specs:
            n = 1
            for d in shape:
                n *= d
            offs.append(total)
            total += n
            cum.append(total)
        self._total = total
        self._arena = torch.empty(total, dtype=torch.bfloat16,
                                  device=self.device)
        blobs = []
        for (p, shape, strides, soff, valid), off in zip(self.specs, offs):
            n = 1
            for d in shape:
                n *= d
            blobs.append(struct.pack(
                _DESC_FMT, p.data_ptr(), off, soff, n, *shape, *strides,
                *valid))
        self._descbuf = torch.frombuffer(
            bytearray(b"".join(blobs)), dtype=torch.uint8).to(self.device)
        self._cumbuf = torch.tensor(cum, dtype=torch.int64,
                                    device=self.device)
        self.views = [
            self._arena[off:off + int(torch.tensor(shape).prod())]
            .view(*[d for d in shape if True])
            for (p, shape, _s, _o, _v), off in zip(self.specs, offs)]
        self._params = [p for p, *_ in self.specs]
        self._ptrs = [p.data_ptr() for p in self._params]
        self._versions = None

    def stale_pointers(self):
        """True if any parameter's storage moved since finalize() (e.g. a
        DDPEngine flattened params after the plan was built) — the
        descriptors' raw pointers would then be dangling."""
        return any(p.data_ptr() != q for p, q in zip(self._params, self._ptrs))

    def run_if_stale(self):
        vers = [p._version for p in self._params]
        if vers == self._versions:
            return False
        ext = get_ext(required=True)
        ext.pack_bf16(self._descbuf, self._cumbuf, len(self.specs),
                      self._arena, self._total)
        self._versions = vers
        return True


def _gemm_row_spec(plan, w):
    """[Cout, Cin(,1,1)] fp32 -> bf16 [Cout, Cin] rows (cast only)."""
    co, ci = w.shape[0], w.shape[1]
    return plan.add(w, (co, ci), (ci, 1), 0)


def _gemm_rowT_spec(plan, w):
    """transposed rows: bf16 [Cin, Cout]."""
    co, ci = w.shape[0], w.shape[1]
    return plan.add(w, (ci, co), (1, ci), 0)


def _w9_spec(plan, w):
    """[C2, C1, 3, 3] -> [C2, 3, 3, C1] (im2col fwd layout)."""
    c2, c1 = w.shape[0], w.shape[1]
    return plan.add(w, (c2, 3, 3, c1), (9 * c1, 3, 1, 9), 0)


def _w9p_spec(plan, w):
    """[C2, C1, 3, 3] -> flipped/swapped [C1, 3, 3, C2] for backward-data:
    out[ci][r][s][co] = w[co][ci][2-r][2-s] (negative strides + offset)."""
    c2, c1 = w.shape[0], w.shape[1]
    return plan.add(w, (c1, 3, 3, c2), (9, -3, -1, 9 * c1), 8)


def _w9p_class_spec(plan, w, r_list, s_list):
    """One stride-2 parity class: out[ci][ri][si][co] =
    w[co][ci][2-r_list[ri]][2-s_list[si]]; r/s lists are arithmetic (step 2)
    so the gather is affine."""
    c2, c1 = w.shape[0], w.shape[1]
    r0, s0 = r_list[0], s_list[0]
    rstep = (r_list[1] - r_list[0]) if len(r_list) > 1 else 1
    sstep = (s_list[1] - s_list[0]) if len(s_list) > 1 else 1
    soff = (2 - r0) * 3 + (2 - s0)
    return plan.add(w, (c1, len(r_list), len(s_list), c2),
                    (9, -rstep * 3, -sstep, 9 * c1), soff)


def _stem_spec(plan, w):
    """[64, 3, 7, 7] -> zero-padded NHWC4 row layout [64, 7, 8, 4]:
    out[co][r][s][c] = w[co][c][r][s] (s<7, c<3; else 0)."""
    co = w.shape[0]
    return plan.add(w, (co, 7, 8, 4), (147, 7, 1, 49), 0,
                    valid=(co, 7, 7, 3))


def _class_taps(k, pp):
    """Valid (r_list, s_list) per output parity class (ph, pw)."""
    out = {}
    for ph in (0, 1):
        for pw in (0, 1):
            rl = [r for r in range(k) if (ph - pp + r) % 2 == 0]
            sl = [s for s in range(k) if (pw - pp + s) % 2 == 0]
            out[(ph, pw)] = (rl, sl)
    return out


def build_resnet_plan(model, device):
    """Walk a ResNet's fused-eligible blocks + stem; returns (plan, attach)
    where attach() stores each block's packed views on the module."""
    from ..models.resnet import Bottleneck
    from .modules import StemConv7x7

    plan = PackPlan(device)
    entries = []  # (module, {name: idx})
    for m in model.modules():
        if isinstance(m, Bottleneck) and getattr(m, "_block_fusable", False):
            w1, w2, w3 = m.conv1.weight, m.conv2.weight, m.conv3.weight
            e = {
                "w1b": _gemm_row_spec(plan, w1),
                "w9": _w9_spec(plan, w2),
                "w3b": _gemm_row_spec(plan, w3),
                "w3bT": _gemm_rowT_spec(plan, w3),
                "w1bT": _gemm_rowT_spec(plan, w1),
            }
            if m.stride == 1:
                e["w9p"] = _w9p_spec(plan, w2)
            else:
                for (ph, pw), (rl, sl) in _class_taps(3, 1).items():
                    e["w9p_{}{}".format(ph, pw)] = _w9p_class_spec(
                        plan, w2, rl, sl)
            if m.downsample is not None:
                wd = m.downsample[0].weight
                e["wdb"] = _gemm_row_spec(plan, wd)
                e["wdbT"] = _gemm_rowT_spec(plan, wd)
            entries.append((m, e))
        elif isinstance(m, StemConv7x7):
            entries.append((m, {"w224": _stem_spec(plan, m.weight)}))
    if not entries:
        return None
    plan.finalize()
    for m, e in entries:
        m._tfos_packs = {k: plan.views[i] for k, i in e.items()}
    return plan


def ensure_packed(model, x):
    """Model-level pre-forward hook: lazily build the plan and (re)pack when
    any parameter version changed. Returns the packs availability flag."""
    if not (model.training and x.is_cuda and x.dtype == torch.bfloat16):
        return False
    import os
    if os.environ.get("TFOS_PACK_PLAN", "on") == "off" \
            or os.environ.get("TFOS_FUSED_BLOCK", "on") == "off":
        return False
    plan = getattr(model, "_tfos_packplan", None)
    if plan is None:
        if getattr(model, "_tfos_packplan_failed", False):
            return False
        try:
            plan = build_resnet_plan(model, x.device)
        except Exception:
            model._tfos_packplan_failed = True
            return False
        if plan is None:
            model._tfos_packplan_failed = True
            return False
        model._tfos_packplan = plan
    if plan.stale_pointers():  # e.g. params re-flattened: rebuild descriptors
        try:
            plan = build_resnet_plan(model, x.device)
        except Exception:
            model._tfos_packplan_failed = True
            return False
        model._tfos_packplan = plan
    plan.run_if_stale()
    return True
