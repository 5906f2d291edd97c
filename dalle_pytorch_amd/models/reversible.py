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


class ReversibleBlock(nn.Module):
    """y1 = x1 + f(x2); y2 = x2 + g(y1). Backward reconstructs x from y and
    re-runs f,g once each (reference reversible.py:54-106). The two streams
    stay separate tensors end to end — the reference's per-block
    cat/chunk round-trips are pure copy traffic (~12 GB/step at the
    flagship shape)."""

    def __init__(self, f, g):
        super().__init__()
        self.f = Deterministic(f)
        self.g = Deterministic(g)

    def forward(self, x1, x2, f_args={}, g_args={}):
        with torch.no_grad():
            y1 = x1 + self.f(x2, record_rng=self.training, **f_args)
            y2 = x2 + self.g(y1, record_rng=self.training, **g_args)
        return y1, y2

    def backward_pass(self, y1, y2, dy1, dy2, f_args={}, g_args={}):
        with torch.enable_grad():
            y1.requires_grad = True
            gy1 = self.g(y1, set_rng=True, **g_args)
            torch.autograd.backward(gy1, dy2)

        with torch.no_grad():
            x2 = y2 - gy1
            del y2, gy1
            dx1 = dy1 + y1.grad
            del dy1
            y1.grad = None

        with torch.enable_grad():
            x2.requires_grad = True
            fx2 = self.f(x2, set_rng=True, **f_args)
            torch.autograd.backward(fx2, dx1, retain_graph=True)

        with torch.no_grad():
            x1 = y1 - fx2
            del y1, fx2
            dx2 = dy2 + x2.grad
            del dy2
            x2.grad = None
        return x1, x2.detach(), dx1, dx2


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
            x = x + f(x, **f_args)
            x = x + g(x, **g_args)
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
