"""Layer-stack executors: plain sequential and reversible (O(1) activations).

Native re-implementation of the reference's RevNet executor
(reversible.py:20-157). The reversible path runs the forward under
``no_grad`` and reconstructs inputs during backward, replaying recorded RNG
state so dropout patterns match between forward and recompute — on ROCm the
device generator state captured via ``torch.cuda`` maps to the HIP Philox
generator, so replay is exact on MI355X. Memory cost is depth-independent:
with 288 GB HBM3E the 64-layer flagship fits without checkpointing.
"""

import torch
import torch.nn as nn
from torch.autograd.function import Function
from torch.utils.checkpoint import get_device_states, set_device_states


def route_args(router, args, depth):
    """Distribute call kwargs to the (attn, ff) pair of each layer.

    ``router[key]`` is a depth-long tuple of (to_attn, to_ff) booleans;
    mirrors reference reversible.py:8-17.
    """
    routed = [({}, {}) for _ in range(depth)]
    for key, per_layer in router.items():
        if key not in args:
            continue
        val = args[key]
        for i, (f_on, g_on) in enumerate(per_layer):
            if f_on:
                routed[i][0][key] = val
            if g_on:
                routed[i][1][key] = val
    return routed


class Deterministic(nn.Module):
    """Wraps a module so a later replay reproduces its forward exactly
    (reference reversible.py:20-50): RNG state for dropout patterns AND the
    autocast state. The latter goes beyond the reference — its recompute ran
    in whatever precision the backward thread had, which on a bf16-autocast
    MI355X run would silently recompute in fp32 (no bf16 MFMA, ~8x slower
    GEMMs, measured in rocprof as fp32 Tensile kernels)."""

    def __init__(self, net):
        super().__init__()
        self.net = net
        self._cpu_state = None
        self._gpu_devices = None
        self._gpu_states = None
        self._had_gpu = False
        self._autocast = None   # (enabled, dtype) for 'cuda'

    def record_rng(self, *tensors):
        self._cpu_state = torch.get_rng_state()
        if torch.cuda._initialized:
            self._had_gpu = True
            self._gpu_devices, self._gpu_states = get_device_states(*tensors)
        self._autocast = (torch.is_autocast_enabled('cuda'),
                          torch.get_autocast_dtype('cuda'))

    def forward(self, *args, record_rng=False, set_rng=False, **kwargs):
        if record_rng:
            self.record_rng(*args)
        if not set_rng:
            return self.net(*args, **kwargs)
        devices = self._gpu_devices if self._had_gpu else []
        with torch.random.fork_rng(devices=devices, enabled=True):
            torch.set_rng_state(self._cpu_state)
            if self._had_gpu:
                set_device_states(self._gpu_devices, self._gpu_states)
            if self._autocast is not None and self._autocast[0] and torch.cuda.is_available():
                with torch.autocast('cuda', dtype=self._autocast[1], enabled=True):
                    return self.net(*args, **kwargs)
            return self.net(*args, **kwargs)


def _residual_branch(det, inp, residual, record_rng, args):
    """residual + det(inp): fused into one pass when the wrapped module is a
    LayerScale (its `residual=` path routes through ops.fused.add_scaled)."""
    if getattr(det.net, 'supports_residual', False):
        return det(inp, residual=residual, record_rng=record_rng, **args)
    return residual + det(inp, record_rng=record_rng, **args)


class ReversibleBlock(nn.Module):
    """y1 = x1 + f(x2); y2 = x2 + g(y1). Backward reconstructs x from y and
    re-runs f,g once each (RevNet decomposition; role of reference
    reversible.py:54-106, re-derived). The two streams stay separate tensors
    end to end — per-block cat/chunk round-trips would be pure copy traffic
    (~12 GB/step at the flagship shape) — and the residual adds are fused
    with LayerScale."""

    def __init__(self, f, g):
        super().__init__()
        self.f = Deterministic(f)
        self.g = Deterministic(g)

    def forward(self, x1, x2, f_args={}, g_args={}):
        rec = self.training
        with torch.no_grad():
            y1 = _residual_branch(self.f, x2, x1, rec, f_args)
            y2 = _residual_branch(self.g, y1, x2, rec, g_args)
        return y1, y2

    @staticmethod
    def _replay_grad(det, inp, upstream, args):
        """Re-run the branch on a fresh leaf under its recorded RNG/autocast
        state, push `upstream` through it (accumulating parameter grads), and
        hand back (branch output detached, gradient w.r.t. the input)."""
        leaf = inp.detach().requires_grad_(True)
        with torch.enable_grad():
            out = det(leaf, set_rng=True, **args)
        torch.autograd.backward(out, upstream)
        return out.detach(), leaf.grad

    def backward_pass(self, y1, y2, dy1, dy2, f_args={}, g_args={}):
        # invert g: x2 = y2 - g(y1); the replay also yields dL/dy1 through g
        gy1, dy1_via_g = self._replay_grad(self.g, y1, dy2, g_args)
        x2 = y2 - gy1
        dx1 = dy1 + dy1_via_g
        del y2, gy1, dy1, dy1_via_g
        # invert f: x1 = y1 - f(x2); replay yields dL/dx2 through f
        fx2, dx2_via_f = self._replay_grad(self.f, x2, dx1, f_args)
        x1 = y1 - fx2
        dx2 = dy2 + dx2_via_f
        del y1, fx2, dy2, dx2_via_f
        return x1, x2, dx1, dx2


class _ReversibleFunction(Function):
    @staticmethod
    def forward(ctx, x1, x2, blocks, args):
        ctx.args = args
        for block, kw in zip(blocks, args):
            x1, x2 = block(x1, x2, **kw)
        ctx.y1 = x1.detach()
        ctx.y2 = x2.detach()
        ctx.blocks = blocks
        return x1, x2

    @staticmethod
    def backward(ctx, dy1, dy2):
        y1, y2 = ctx.y1, ctx.y2
        for block, kw in zip(ctx.blocks[::-1], ctx.args[::-1]):
            y1, y2, dy1, dy2 = block.backward_pass(y1, y2, dy1, dy2, **kw)
        return dy1, dy2, None, None


class SequentialSequence(nn.Module):
    """x = x + attn(x); x = x + ff(x) per layer (reference reversible.py:126-141)."""

    def __init__(self, layers, args_route={}, layer_dropout=0.):
        super().__init__()
        assert all(len(r) == len(layers) for r in args_route.values())
        self.layers = layers
        self.args_route = args_route

    def forward(self, x, **kwargs):
        args = route_args(self.args_route, kwargs, len(self.layers))
        for (f, g), (f_args, g_args) in zip(self.layers, args):
            for branch, branch_args in ((f, f_args), (g, g_args)):
                if getattr(branch, 'supports_residual', False):
                    x = branch(x, residual=x, **branch_args)
                else:
                    x = x + branch(x, **branch_args)
        return x


class ReversibleSequence(nn.Module):
    """Duplicates the stream into two halves, runs reversible blocks, and
    averages the halves at the end (reference reversible.py:143-157)."""

    def __init__(self, blocks, args_route={}):
        super().__init__()
        self.args_route = args_route
        self.blocks = nn.ModuleList([ReversibleBlock(f=f, g=g) for f, g in blocks])

    def forward(self, x, **kwargs):
        args = route_args(self.args_route, kwargs, len(self.blocks))
        args = [{'f_args': fa, 'g_args': ga} for fa, ga in args]
        # duplicated input stream (reference reversible.py:150) without the
        # cat: the two streams are independent tensors throughout
        y1, y2 = _ReversibleFunction.apply(x, x.clone(), self.blocks, args)
        return (y1 + y2) / 2
