"""Flattened-time execution helpers for 3D convnets on MI355X.

MIOpen lowers NCDHW conv3d through Im3d2Col + GEMM (see profiles/) — an
im2col materialization up to 27x the input.  Instead the I3D and R(2+1)D
backbones run on (B*T, C, H, W) channels_last tensors:

* a (1,k,k) conv is a plain conv2d over all frames;
* a (kt,k,k) conv becomes ONE merged conv2d with kt*O output channels (the
  temporal taps) followed by :func:`temporal_merge` — a shifted strided add
  over t (zero temporal padding ≡ skipping out-of-range taps);
* a separable pooling window splits exactly into a spatial 2D pool and
  :func:`temporal_max` (max over a separable window commutes).

All helpers keep channels_last layout and support any (kt, stride, pad).
"""
from __future__ import annotations

import torch

from .. import ops


def cached_cl_weight(mod, name: str, src: torch.Tensor, build):
    """Per-module cache of transformed conv weights for the flat path.

    The flat decomposition slices/permutes the stored Conv3d weights; the
    resulting views are NOT channels_last-contiguous, and F.conv2d with a
    non-CL weight silently drops MIOpen's NHWC igemm path (measured: I3D at
    17 TF/s effective vs RAFT's 911 with module convs).  Build once, store
    CL-contiguous, invalidate on weight version/dtype/device change."""
    key = (src._version, src.dtype, src.device, src.data_ptr())
    cache = mod.__dict__.setdefault('_flat_w_cache', {})
    ent = cache.get(name)
    if ent is None or ent[0] != key:
        val = build()
        if isinstance(val, torch.Tensor) and val.dim() == 4:
            val = val.contiguous(memory_format=torch.channels_last)
        cache[name] = (key, val)
        return val
    return ent[1]


def flatten_time(x: torch.Tensor) -> torch.Tensor:
    """(B, C, T, H, W) → (B*T, C, H, W) channels_last, one copy.

    The explicit .contiguous matters: at B=1 the reshape can alias the
    NCDHW storage and hand back a non-CL VIEW, which silently dropped
    every downstream op to its strided slow path."""
    b, c, t, h, w = x.shape
    y = x.permute(0, 2, 3, 4, 1).reshape(b * t, h, w, c).contiguous()
    return y.permute(0, 3, 1, 2)


def unflatten_time(xf: torch.Tensor, b: int) -> torch.Tensor:
    """(B*T, C, H, W) channels_last → (B, C, T, H, W) contiguous."""
    bt, c, h, w = xf.shape
    t = bt // b
    return xf.permute(0, 2, 3, 1).reshape(b, t, h, w, c) \
        .permute(0, 4, 1, 2, 3).contiguous()


def cl_empty(bt: int, c: int, h: int, w: int, like: torch.Tensor):
    return torch.empty(bt, c, h, w, device=like.device, dtype=like.dtype,
                       memory_format=torch.channels_last)


def temporal_merge(y: torch.Tensor, b: int, kt: int, st: int = 1,
                   p0: int = 1, bias_tap: int = None,
                   relu: bool = False, p1: int = None) -> torch.Tensor:
    """y (B*T, kt*O, H, W): temporal-tap conv outputs stacked along channels
    → (B*T', O, H, W) with out[to] = Σ_dt y_dt[to*st - p0 + dt] (zero
    temporal padding).  ``bias_tap``: the tap that already carries the conv
    bias — it must be valid for every output position (callers put the bias
    on a middle tap); the accumulator is initialized from it."""
    bt, ckt, h, w = y.shape
    o = ckt // kt
    t = bt // b
    if p1 is None:
        p1 = p0
    to = (t + p0 + p1 - kt) // st + 1
    if bias_tap is None:
        bias_tap = kt // 2
    fused = ops.temporal_merge_fused(y, b, kt, st, p0, relu, p1)
    if fused is not None:
        return fused
    out = cl_empty(b * to, o, h, w, y)
    y5 = y.view(b, t, ckt, h, w)
    o5 = out.view(b, to, o, h, w)
    # init from the always-valid bias tap
    s_lo = 0 * st - p0 + bias_tap
    o5.copy_(y5[:, s_lo:s_lo + (to - 1) * st + 1:st,
                bias_tap * o:(bias_tap + 1) * o])
    for dt in range(kt):
        if dt == bias_tap:
            continue
        # out index j valid where 0 <= j*st - p0 + dt < t
        j_lo = max(0, -(-(p0 - dt) // st))
        j_hi = min(to - 1, (t - 1 + p0 - dt) // st)
        if j_lo > j_hi:
            continue
        s0 = j_lo * st - p0 + dt
        o5[:, j_lo:j_hi + 1] += \
            y5[:, s0:s0 + (j_hi - j_lo) * st + 1:st, dt * o:(dt + 1) * o]
    return torch.relu_(out) if relu else out


def temporal_select(xf: torch.Tensor, b: int, st: int) -> torch.Tensor:
    """Take every st-th frame: (B*T, C, H, W) → (B*ceil(T/st), C, H, W)."""
    if st == 1:
        return xf
    bt, c, h, w = xf.shape
    t = bt // b
    sel = xf.view(b, t, c, h, w)[:, ::st]
    to = sel.shape[1]
    out = cl_empty(b * to, c, h, w, xf)
    out.view(b, to, c, h, w).copy_(sel)
    return out


def temporal_max(xf: torch.Tensor, b: int, kt: int, st: int,
                 p0: int, p1: int, pad_zero: bool = True) -> torch.Tensor:
    """Shifted maximum along t: (B*T, C, H, W) → (B*T', C, H, W).
    ``pad_zero``: padding positions contribute the value 0 to the max
    (F.pad(0) + maxpool semantics, as TF-SAME I3D uses)."""
    bt, c, h, w = xf.shape
    t = bt // b
    to = (t + p0 + p1 - kt) // st + 1
    out = cl_empty(b * to, c, h, w, xf)
    out.fill_(float('-inf'))
    x5 = xf.view(b, t, c, h, w)
    o5 = out.view(b, to, c, h, w)
    for dt in range(kt):
        j_lo = max(0, -(-(p0 - dt) // st))
        j_hi = min(to - 1, (t - 1 + p0 - dt) // st)
        if j_lo > j_hi:
            continue
        s_lo = j_lo * st - p0 + dt
        src = x5[:, s_lo:s_lo + (j_hi - j_lo) * st + 1:st]
        dst = o5[:, j_lo:j_hi + 1]
        torch.maximum(dst, src, out=dst)
    if pad_zero:
        head = -(-p0 // st)
        if head > 0:
            o5[:, :head].clamp_(min=0)
        j_tail = -(-(t + p0 - kt + 1) // st)
        if j_tail < to:
            o5[:, j_tail:].clamp_(min=0)
    return out
