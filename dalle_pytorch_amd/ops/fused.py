"""Fused elementwise ops (memory-bound; reference kernel rows K6/K7).

On gfx950 these are single HBM-bound passes (vectorized bf16x8 loads per the
CDNA4 guide); on CPU the eager equivalents run.
"""

import torch
import torch.nn.functional as F

from dalle_pytorch_amd.ops.dispatch import hip_module, using_eager_fallback


class _GegluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_module()
        x = x.contiguous()
        out = ext.geglu_fwd(x)
        ctx.save_for_backward(x)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_module()
        (x,) = ctx.saved_tensors
        return ext.geglu_bwd(x, dout.contiguous())


def geglu(x: torch.Tensor) -> torch.Tensor:
    """GEGLU gate: split the last dim in half, return value * gelu(gate).

    Matches reference transformer.py:106-109 (erf-based exact gelu).
    """
    if using_eager_fallback(x) or x.shape[-1] % 16:
        a, b = x.chunk(2, dim=-1)
        return a * F.gelu(b)
    return _GegluFn.apply(x)


class _TokenShiftFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, text_len, image_size):
        ext = hip_module()
        ctx.text_len, ctx.image_size = text_len, image_size
        return ext.token_shift(x.contiguous(), text_len, image_size, False)

    @staticmethod
    def backward(ctx, dout):
        ext = hip_module()
        dx = ext.token_shift(dout.contiguous(), ctx.text_len, ctx.image_size, True)
        return dx, None, None


def token_shift(x, text_len, image_size):
    """Fused training-path token shift (reference transformer.py:165-186):
    text halves shift one token; image quarters shift from the grid row
    above / left neighbor. One gather kernel each way."""
    return _TokenShiftFn.apply(x, text_len, image_size)


def token_shift_supported(x):
    return (x.is_cuda and hip_module() is not None
            and x.shape[-1] * x.element_size() % 64 == 0)


class _AddScaledFn(torch.autograd.Function):
    """out = x + gamma * y with per-channel gamma — the residual + LayerScale
    fusion (reference transformer.py:74-88 + reversible.py:138-140 adds).
    Backward: dx aliases dout (no kernel), dy/dgamma in one fused pass."""

    @staticmethod
    def forward(ctx, x, y, gamma):
        ext = hip_module()
        y = y.contiguous()
        gf = gamma.detach().reshape(-1).float().contiguous()
        out = ext.resls_fwd(x.contiguous(), y, gf)
        ctx.save_for_backward(y, gf)
        ctx.gamma_meta = (gamma.shape, gamma.dtype)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_module()
        y, gf = ctx.saved_tensors
        dout = dout.contiguous()
        dy, dgamma = ext.resls_bwd(dout, y, gf)
        shape, dtype = ctx.gamma_meta
        return dout, dy, dgamma.reshape(shape).to(dtype)


def add_scaled(x, y, gamma):
    """x + gamma * y (gamma broadcast over the last dim)."""
    if (x.is_cuda and x.dtype == torch.bfloat16 and y.dtype == torch.bfloat16
            and not using_eager_fallback(x) and x.shape[-1] % 8 == 0
            and x.shape[-1] // 8 <= 256 and 256 % (x.shape[-1] // 8) == 0):
        return _AddScaledFn.apply(x, y, gamma)
    return x + y * gamma.to(y.dtype)


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = hip_module()
        y, mean, rstd = ext.ln_fwd(x.contiguous(), weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_module()
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.ln_bwd(x, dy.contiguous(), weight, mean, rstd)
        return dx, dgamma.to(weight.dtype), dbeta.to(weight.dtype), None


_LN_DIMS = (512, 1024, 1536, 2048)


def layer_norm(x, weight, bias, eps=1e-5):
    """LayerNorm over the last dim; fused bf16 kernel on GPU for the
    transformer widths, F.layer_norm elsewhere. Matches autocast semantics
    (fp32 statistics and affine) with a single bf16 rounding at the output.
    """
    if (x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] in _LN_DIMS
            and not using_eager_fallback(x)):
        return _LayerNormFn.apply(x, weight, bias, eps)
    if weight is not None and weight.dtype != x.dtype and x.dtype != torch.float32:
        weight = weight.to(x.dtype)
        bias = bias.to(x.dtype) if bias is not None else None
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)
