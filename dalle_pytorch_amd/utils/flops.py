"""Analytic FLOPs profiler for the DALLE training step.

Replaces the reference's DeepSpeed flops profiler hook
(train_dalle.py:91,492-499): at the chosen profile step the trainer calls
:func:`profile_step`, which prints a per-component forward-FLOP table, the
whole-step (fwd + bwd ~ 3x fwd; reversible ~5x) estimate, and achieved
TFLOP/s from the measured step time.
"""


def _attn_flops(kind, b, heads, n, d, text_len, fmap):
    """Score+PV matmul flops for one attention layer (fwd)."""
    if kind in ('full', 'sparse'):
        pairs = n * n  # block-sparse skips tiles at runtime; count dense bound
    elif kind in ('axial_row', 'axial_col'):
        img = fmap * fmap
        pairs = text_len * text_len + img * (text_len + fmap)
    elif kind == 'conv_like':
        img = fmap * fmap
        pairs = text_len * text_len + img * (text_len + 25)
    else:
        pairs = n * n
    return 2 * 2 * b * heads * pairs * d   # QK^T + PV, 2 flops per MAC


def dalle_forward_flops(dalle, batch_size):
    """Component table of forward FLOPs for one step at this batch size."""
    n = dalle.total_seq_len
    dim = dalle.transformer.layers.layers[0][0].fn.norm.normalized_shape[0] \
        if hasattr(dalle.transformer.layers, 'layers') else \
        dalle.transformer.layers.blocks[0].f.net.norm.normalized_shape[0]
    fmap = dalle.transformer.image_fmap_size
    text_len = n - fmap * fmap + 1
    b = batch_size

    rows = []
    total = 0

    def add(name, flops):
        nonlocal total
        rows.append((name, flops))
        total += flops

    # transformer layers
    layers = dalle.transformer.layers
    pairs = layers.layers if hasattr(layers, 'layers') else \
        [(blk.f.net, blk.g.net) for blk in layers.blocks]
    for i, (attn_wrap, ff_wrap) in enumerate(pairs):
        leaf = attn_wrap
        while hasattr(leaf, 'fn'):
            leaf = leaf.fn
        heads = leaf.heads
        d_head = getattr(leaf, 'dim_head', 64)
        kind = type(leaf).__name__
        kind_key = {'Attention': 'full', 'SparseAxialCausalAttention':
                    'axial_row' if getattr(leaf, 'axis', 0) == 0 else 'axial_col',
                    'SparseConvCausalAttention': 'conv_like',
                    'SparseAttention': 'sparse'}.get(kind, 'full')
        qkv = 2 * b * n * dim * 3 * heads * d_head
        proj = 2 * b * n * heads * d_head * dim
        core = _attn_flops(kind_key, b, heads, n, d_head, text_len, fmap)
        add(f'layer{i}.attn({kind_key})', qkv + core + proj)
        # GEGLU FF: dim -> 8*dim -> (gate) -> 4*dim -> dim
        add(f'layer{i}.ff', 2 * b * n * dim * 8 * dim + 2 * b * n * 4 * dim * dim)

    add('to_logits', 2 * b * n * dim * dalle.total_tokens)
    vae = dalle.vae
    if hasattr(vae, 'encoder') and hasattr(vae, 'image_size'):
        # rough conv-stack estimate: 2 * K^2 * Cin * Cout * H * W per conv
        conv_fl = 0
        import torch.nn as nn_mod
        size = vae.image_size
        for m in vae.encoder.modules():
            if isinstance(m, nn_mod.Conv2d):
                stride = m.stride[0]
                size_out = max(size // stride, 1)
                conv_fl += 2 * m.kernel_size[0] * m.kernel_size[1] * \
                    m.in_channels * m.out_channels * size_out * size_out * b
                if stride > 1:
                    size = size_out
        add('vae.encode(frozen)', conv_fl)
    return rows, total


def profile_step(dalle, batch_size, step_time_s, reversible=False, out=print):
    rows, fwd = dalle_forward_flops(dalle, batch_size)
    mult = 5.0 if reversible else 3.0   # bwd ~2x fwd; reversible re-runs fwd
    step_flops = fwd * mult
    out('--- flops profile (analytic) ---')
    for name, fl in rows:
        out(f'  {name:32s} {fl / 1e9:10.2f} GFLOP')
    out(f'  forward total                  {fwd / 1e12:10.3f} TFLOP')
    out(f'  step estimate (x{mult:.0f})           {step_flops / 1e12:10.3f} TFLOP')
    out(f'  measured step time             {step_time_s * 1e3:10.1f} ms')
    out(f'  achieved                       {step_flops / step_time_s / 1e12:10.1f} TFLOP/s')
    return step_flops
