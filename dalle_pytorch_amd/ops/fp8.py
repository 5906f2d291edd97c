"""fp8 (e4m3) forward path for the projection GEMMs.

gfx950 runs e4m3 MFMA at 2x the bf16 rate; measured on MI355X the raw
ff1-shaped GEMM goes 1297 -> 2134 TF/s (scripts/probe_fp8.py). Master
weights stay bf16/fp32 — fp8 exists only on the wire into the GEMM:

* activations: one torch amax reduce + the one-pass quant_fp8 HIP kernel
  (the eager cast chain costs more than the fp8 GEMM saves);
* weights: quantized once per optimizer step (cached on the parameter's
  version counter, so the reversible recompute pass reuses it);
* backward: standard bf16 dgrad/wgrad — fp8 is forward-only, which under
  reversible execution still covers 2 of the 4 GEMM passes per step.

Enable with DALLE_AMD_FP8=1 (or ``set_fp8_enabled(True)``; bench.py
exposes ``--fp8``). Default OFF: the headline BASELINE numbers are bf16.
"""

import os

import torch
import torch.nn.functional as F

from dalle_pytorch_amd.ops.dispatch import hip_module, using_eager_fallback

_FORCED = None
FP8_MAX = 448.0


def set_fp8_enabled(flag):
    global _FORCED
    _FORCED = bool(flag) if flag is not None else None


def fp8_enabled():
    if _FORCED is not None:
        return _FORCED
    return os.environ.get('DALLE_AMD_FP8', '0') == '1'


_w_cache = {}
_generation = 0


def fp8_mark_step():
    """Invalidate the per-step weight-quant cache. Call after optimizer
    steps — parameter version counters are not reliably bumped by every
    fused/foreach CUDA optimizer path."""
    global _generation
    _generation += 1


def _quant(t, ext):
    t = t.detach().contiguous()
    scale = ext.amax_bf16(t)   # outputs amax/448 directly (one pass)
    return ext.quant_fp8(t, scale), scale.squeeze()


def _quant_weight(w, ext):
    """Per-step cache: the reversible recompute re-runs every forward with
    unchanged weights, so quantize once per (param, version). Keyed by
    storage+shape so fresh VIEWS of the same slice (the split-vocab head
    re-slices its weight every step) hit the cache too."""
    key = (w.data_ptr(), tuple(w.shape))
    hit = _w_cache.get(key)
    stamp = (w._version, _generation)
    if hit is not None and hit[0] == stamp:
        return hit[1], hit[2]
    wb = w.detach()
    if wb.dtype != torch.bfloat16:
        wb = wb.to(torch.bfloat16)
    wq, ws = _quant(wb, ext)
    _w_cache[key] = (stamp, wq, ws)
    return wq, ws


_wt_cache = {}
_cast_cache = {}


def _cached_bf16(w):
    """Per-step cached bf16 copy of a master weight (what autocast's
    per-autocast-region weight cache would provide)."""
    if w.dtype == torch.bfloat16:
        return w
    key = (w.data_ptr(), tuple(w.shape))
    stamp = (w._version, _generation)
    hit = _cast_cache.get(key)
    if hit is not None and hit[0] == stamp:
        return hit[1]
    wb = w.detach().to(torch.bfloat16)
    if len(_cast_cache) > 512:
        _cast_cache.clear()
    _cast_cache[key] = (stamp, wb)
    return wb


def _transposed_weight(w):
    """Per-step cached contiguous bf16 W^T: hipBLASLt runs dgrad ~15%
    faster fed a transposed-view B operand (scripts/probe_dgrad.py:
    ff1-dgrad 1261 -> 1460 TF/s). Keyed on the MASTER parameter (stable
    storage), never on cast temporaries."""
    key = (w.data_ptr(), tuple(w.shape))
    stamp = (w._version, _generation)
    hit = _wt_cache.get(key)
    if hit is not None and hit[0] == stamp:
        return hit[1]
    wt = w.detach()
    if wt.dtype != torch.bfloat16:
        wt = wt.to(torch.bfloat16)
    wt = wt.t().contiguous()
    if len(_wt_cache) > 512:
        _wt_cache.clear()
    _wt_cache[key] = (stamp, wt)
    return wt


def _linear_backward(ctx, dout):
    """Shared backward: transposed-weight dgrad, standard wgrad/bias."""
    x2, weight = ctx.saved_tensors
    do2 = dout.reshape(-1, dout.shape[-1]).contiguous()
    wt = _transposed_weight(weight)
    dx = (do2 @ wt.t()).reshape(*dout.shape[:-1], weight.shape[1])
    dw = do2.t() @ x2.to(do2.dtype)
    db = do2.sum(0) if ctx.has_bias else None
    return dx, dw.to(weight.dtype), db


class _FastDgradLinearFn(torch.autograd.Function):
    """bf16 F.linear with the transposed-dgrad backward. Receives MASTER
    weights (any float dtype) and casts inside, so the per-step caches key
    on stable parameter storage."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        wb = _cached_bf16(weight)
        bb = bias if (bias is None or bias.dtype == torch.bfloat16) \
            else bias.to(torch.bfloat16)
        ctx.save_for_backward(x.reshape(-1, x.shape[-1]), weight)
        ctx.has_bias = bias is not None
        return F.linear(x, wb, bb)

    @staticmethod
    def backward(ctx, dout):
        return _linear_backward(ctx, dout)


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = hip_module()
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        xq, xs = _quant(x2, ext)
        wq, ws = _quant_weight(weight, ext)
        out = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                               bias=bias, out_dtype=torch.bfloat16)
        ctx.save_for_backward(x2, weight)
        ctx.has_bias = bias is not None
        return out.reshape(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dout):
        return _linear_backward(ctx, dout)


def _fp8_ok(x, weight):
    if not fp8_enabled() or using_eager_fallback(x):
        return False
    if x.dtype != torch.bfloat16:
        return False
    rows = x.numel() // x.shape[-1]
    k, n = weight.shape[1], weight.shape[0]
    if rows % 16 or k % 16 or n % 16 or rows < 256:
        return False
    if rows * k < (32 << 20):
        # quantization passes don't amortize on small inputs (configs B/D
        # measured net-negative: 421k -> 379k and 8.3k -> 7.4k)
        return False
    # profitability gate, measured per-shape on MI355X (probe_fp8b + end-to-
    # end A/B): ff1 (N=8192,K=1024) +375 us/call and qkv (N=3072) +123 win;
    # ff2 (K=4096) and the square out-proj are net NEUTRAL in isolation and
    # LOSE ~1%% end-to-end (launch-gap overhead of the extra quant kernels),
    # so only clearly-profitable shapes pass
    return n * 2 >= k * 3


def _fast_dgrad_ok(x, w):
    if using_eager_fallback(x) or x.dtype != torch.bfloat16:
        return False
    rows = x.numel() // x.shape[-1]
    return (rows >= 1024 and rows % 16 == 0 and w.shape[0] % 16 == 0
            and w.shape[1] % 16 == 0 and torch.is_grad_enabled())


def fp8_linear(linear_module, x):
    """F.linear through the fp8 forward path when profitable; else a bf16
    linear with the transposed-dgrad backward; else the module itself.
    Drop-in for ``linear_module(x)``."""
    w = linear_module.weight
    if _fp8_ok(x, w):
        bias = linear_module.bias
        if bias is not None and bias.dtype != torch.bfloat16:
            bias = bias.to(torch.bfloat16)
        return _Fp8LinearFn.apply(x.contiguous(), w, bias)
    if _fast_dgrad_ok(x, w):
        return _FastDgradLinearFn.apply(x.contiguous(), w, linear_module.bias)
    return linear_module(x)


def fp8_linear_raw(x, weight, bias):
    """F.linear(x, weight, bias) with the fp8 forward path when profitable
    (raw-tensor form for weight slices, e.g. the split-vocab CE head)."""
    if _fp8_ok(x, weight) and weight.is_contiguous():
        if bias is not None and bias.dtype != torch.bfloat16:
            bias = bias.to(torch.bfloat16)
        return _Fp8LinearFn.apply(x.contiguous(), weight, bias)
    if _fast_dgrad_ok(x, weight) and weight.is_contiguous():
        return _FastDgradLinearFn.apply(x.contiguous(), weight, bias)
    return F.linear(x, weight, bias)
