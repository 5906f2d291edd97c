"""MI355X decode engine: static-shape autoregressive generation.

The model-side cached decode (reference semantics: a dict cache of growing
k/v tensors, NonCached full-sequence recompute for sparse layers) launches
hundreds of small kernels per token and reallocates every step. This
engine runs the same math with fully static shapes:

* preallocated per-layer K/V caches [b, h, N, d], written at ``offset``;
* single-token attention over all N key slots with a length mask (the
  wasted dot products are negligible; shapes never change) — the sparse
  variants decode through their dense pattern masks, so axial/conv/
  block-sparse models get O(1)-shape decode instead of the reference's
  O(n^2) NonCached recompute;
* the token-shift deque replaced by an S-slot ring buffer of channel
  quarters, indexed modulo S with tensor ops;
* rotary rows and the logits-mask row gathered by a device-side offset
  tensor.

Because every step is shape-static it can be captured as a HIP graph
(``torch.cuda.CUDAGraph`` on ROCm): one graph replay per generated token,
INCLUDING sampling — the fused top-k+gumbel kernel and the out-buffer
write are captured, so a token step is a single replay with zero eager
kernels (the graph-safe philox state advances the noise each replay).

Correctness is pinned by tests/test_decode_engine.py: engine generation ==
the model's dict-cache generation (itself verified bitwise-equal to
uncached recomputation) across attention types, shift, stable, reversible.
"""

import torch
import torch.nn.functional as F
from torch import nn

from dalle_pytorch_amd.ops.fused import layer_norm as fused_layer_norm

from dalle_pytorch_amd.models import attention as attn_mod
from dalle_pytorch_amd.models.transformer import CachedAs, NonCached, PreShiftToken
from dalle_pytorch_amd.models.dalle import gumbel_sample
from dalle_pytorch_amd.ops import attention_core


def _unwrap(module):
    """Peel LayerScale/PreNorm/CachedAs/NonCached/PreShiftToken wrappers."""
    info = {'scale': None, 'norm': None, 'norm_out': None, 'shift': None}
    m = module
    while True:
        name = type(m).__name__
        if name == 'LayerScale':
            info['scale'] = m.scale
            m = m.fn
        elif name == 'PreNorm':
            info['norm'] = m.norm
            if isinstance(m.norm_out, nn.LayerNorm):
                info['norm_out'] = m.norm_out
            m = m.fn
        elif isinstance(m, (CachedAs, NonCached)):
            m = m.fn
        elif isinstance(m, PreShiftToken):
            info['shift'] = m
            m = m.fn
        else:
            return m, info


def _pattern_mask(leaf, seq_len, device):
    """Dense [N, N] bool attend-pattern excluding causality (the length
    mask supplies it at decode time; attention_core applies it in prefill)."""
    name = type(leaf).__name__
    if name == 'Attention':
        return leaf.static_mask.to(device) if leaf.static_mask is not None else None
    if name == 'SparseAxialCausalAttention':
        return attn_mod.axial_mask(seq_len, leaf.text_len, leaf.image_size,
                                   leaf.axis).to(device)
    if name == 'SparseConvCausalAttention':
        return attn_mod.conv_mask(seq_len, leaf.text_len, leaf.image_size,
                                  leaf.kernel_size, leaf.dilation).to(device)
    if name == 'SparseAttention':
        return leaf._build_mask()[:seq_len, :seq_len].to(device)
    raise ValueError(f'unsupported attention {name}')


class _LayerState:
    __slots__ = ('leaf', 'info', 'pattern', 'k', 'v', 'ring', 'is_attn', 'w',
                 'live', 'live_cnt')


class FastDecoder:
    """Static-shape decoder bound to one DALLE instance and batch size."""

    def __init__(self, dalle, batch_size, device=None, dtype=None,
                 use_graph=False):
        self.dalle = dalle
        self.b = batch_size
        self.device = device if device is not None else next(dalle.parameters()).device
        self.dtype = dtype if dtype is not None else \
            (torch.bfloat16 if self.device.type == 'cuda' else torch.float32)
        self.N = dalle.total_seq_len
        self.use_graph = use_graph and self.device.type == 'cuda'
        tr = dalle.transformer
        self.rotary = tr.pos_emb is not None
        if self.rotary:
            ang = tr.pos_emb.squeeze(0).float().to(self.device)
            self.cos = ang.cos().contiguous()
            self.sin = ang.sin().contiguous()

        layers = tr.layers
        self.reversible = not hasattr(layers, 'layers')
        pairs = layers.layers if not self.reversible else \
            [(blk.f.net, blk.g.net) for blk in layers.blocks]

        self.states = []
        for (attn_wrap, ff_wrap) in pairs:
            for wrap, is_attn in ((attn_wrap, True), (ff_wrap, False)):
                st = _LayerState()
                st.leaf, st.info = _unwrap(wrap)
                st.is_attn = is_attn
                st.pattern = None
                st.k = st.v = st.ring = None
                st.live = st.live_cnt = None
                if is_attn:
                    leaf = st.leaf
                    st.pattern = _pattern_mask(leaf, self.N, self.device)
                    h, d = leaf.heads, leaf.dim_head
                    st.k = torch.zeros(self.b, h, self.N, d, device=self.device,
                                       dtype=self.dtype)
                    st.v = torch.zeros_like(st.k)
                if st.info['shift'] is not None:
                    S = st.info['shift'].image_size
                    dim = st.info['norm'].normalized_shape[0]
                    st.ring = torch.zeros(self.b, S, dim // 2,
                                          device=self.device, dtype=self.dtype)
                st.w = self._materialize(st)
                self.states.append(st)

        # head weights, pre-cast once (the decode loop never re-casts)
        cast = lambda t: None if t is None else \
            t.detach().to(self.device, self.dtype)
        castf = lambda t: None if t is None else \
            t.detach().to(self.device, torch.float32)
        head_ln, head_lin = dalle.to_logits[0], dalle.to_logits[1]
        ntt, nit = dalle.num_text_tokens, dalle.num_image_tokens
        w_full = cast(head_lin.weight)
        b_full = cast(head_lin.bias)
        self.head_w = {
            'ln_w': cast(head_ln.weight), 'ln_b': cast(head_ln.bias),
            'w': w_full, 'b': b_full,
            # decode steps are always image positions: the logits mask there
            # allows exactly the image-vocab rows, so generation projects
            # against this slice only (~7x less head GEMM + sampling work)
            'w_img': w_full[ntt:ntt + nit].contiguous(),
            'b_img': None if b_full is None else b_full[ntt:ntt + nit].contiguous(),
            'b_img32': castf(None if head_lin.bias is None
                             else head_lin.bias[ntt:ntt + nit]),
        }
        if self.head_w['w_img'].dtype == torch.bfloat16:
            self.head_w['w_img_pk'] = self._sk2_pack(self.head_w['w_img'])
        else:
            self.head_w['w_img_pk'] = None

        from dalle_pytorch_amd.ops.dispatch import hip_available
        self.offset_t = torch.zeros(1, dtype=torch.long, device=self.device)
        self.key_arange = torch.arange(self.N, device=self.device)
        # fused single-token kernels need bf16 caches, d=64 and the extension
        dims_ok = all((not st.is_attn) or st.leaf.dim_head == 64
                      for st in self.states)
        self._fused_decode = (self.device.type == 'cuda'
                              and self.dtype == torch.bfloat16
                              and self.N <= 4096 and dims_ok
                              and hip_available())
        self._graph = None
        self._g_token = None
        self._g_logits = None
        if self._fused_decode:
            self._build_live_tables()
        import os as _os
        self._sk2_on = (self._fused_decode
                        and _os.environ.get('DALLE_AMD_SK2', '1') == '1')
        # the K=4096 ff2 is the one decode shape where hipBLASLt still wins
        # (14 us in-situ F.linear vs 17.9 us sk2 — sk2's ring drains on HBM
        # latency at only 64 blocks); default keeps ff2 on hipBLASLt
        self._sk2_maxk = int(_os.environ.get('DALLE_AMD_SK2_MAXK', '2048'))
        self._fused_prelude = self._fused_decode and all(
            st.info['norm_out'] is None
            and st.info['scale'] is not None
            and st.info['norm'].normalized_shape[0] in (512, 1024, 2048)
            for st in self.states)

    def _build_live_tables(self):
        """Per attention layer, per offset row: the indices of the keys the
        pattern + causality actually allow. The decode kernel then iterates
        listed keys only — under the flagship axial patterns ~288 of 1281
        slots are live, so scanning the full range wastes ~75% of the K/V
        row loads (measured: the scan-form part kernel spent 78 us/dispatch
        at ~288 live keys)."""
        ar = torch.arange(self.N, device=self.device)
        causal = ar.unsqueeze(0) <= ar.unsqueeze(1)      # [q, k]
        for st in self.states:
            if not st.is_attn:
                continue
            allow = causal if st.pattern is None else causal & st.pattern
            cnt = allow.sum(1, dtype=torch.int32)
            lmax = int(cnt.max())
            idx = torch.where(allow, ar.unsqueeze(0).expand_as(allow), self.N)
            idx, _ = idx.sort(dim=1)
            idx = idx[:, :lmax]
            st.live = torch.where(idx == self.N, torch.zeros_like(idx),
                                  idx).to(torch.int32).contiguous()
            st.live_cnt = cnt.contiguous()

    @staticmethod
    def _sk2_pack(w):
        """Pack [N, K] bf16 into MFMA A-fragment order [N/16][K/32][lane][8]
        (lane = kgroup*16 + col) so the sk2 decode GEMM streams each weight
        tile as one contiguous region. None if the shape doesn't qualify."""
        N, K = w.shape
        if N % 32 or K % 1024 or not w.is_cuda or w.dtype != torch.bfloat16:
            return None
        return (w.reshape(N // 16, 16, K // 32, 4, 8)
                 .permute(0, 2, 3, 1, 4).contiguous())

    @staticmethod
    def _sk2_pack_fp8(w):
        """e4m3 variant of _sk2_pack: per-tensor scale (amax/448), packed
        as bytes in the same fragment order. Halves the weight stream of
        the weight-bound decode GEMMs; opt-in via DALLE_AMD_FP8_DECODE=1
        (a quality trade — logits move ~0.1-1%% per element)."""
        N, K = w.shape
        if N % 32 or K % 1024 or not w.is_cuda or w.dtype != torch.bfloat16:
            return None
        amax = w.detach().abs().amax().float().clamp(min=1e-12)
        scale = amax / 448.0
        q = (w.detach().float() / scale).clamp(-448., 448.) \
            .to(torch.float8_e4m3fn)
        pk = (q.view(torch.uint8).reshape(N // 16, 16, K // 32, 4, 8)
               .permute(0, 2, 3, 1, 4).contiguous())
        return pk, float(scale)

    def _materialize(self, st):
        """Pre-cast this branch's weights to the engine dtype once: under
        autocast the casts would otherwise be captured into the decode graph
        and replayed for every token (measured ~30% of replay time)."""
        cast = lambda t: None if t is None else \
            t.detach().to(self.device, self.dtype)
        castf = lambda t: None if t is None else \
            t.detach().to(self.device, torch.float32)
        # LN weights stay fp32: the fused kernel takes fp32 affine directly
        # (a bf16 copy would be re-cast on every graph replay)
        w = {'ln_w': castf(st.info['norm'].weight),
             'ln_b': castf(st.info['norm'].bias),
             'scale': cast(st.info['scale']),
             'scale32': None if st.info['scale'] is None else
             castf(st.info['scale']).reshape(-1).contiguous()}
        if st.info['norm_out'] is not None:
            w['lno_w'] = castf(st.info['norm_out'].weight)
            w['lno_b'] = castf(st.info['norm_out'].bias)
        castT = lambda t: None if t is None else \
            t.detach().to(self.device, self.dtype).t().contiguous()
        if st.is_attn:
            w['qkv'] = cast(st.leaf.to_qkv.weight)
            w['qkv_T'] = castT(st.leaf.to_qkv.weight)
            w['qkv_pk'] = self._sk2_pack(w['qkv'])
            w['out_w'] = cast(st.leaf.to_out[0].weight)
            w['out_T'] = castT(st.leaf.to_out[0].weight)
            w['out_pk'] = self._sk2_pack(w['out_w'])
            if self._want_fp8_decode():
                w['qkv_pk8'] = self._sk2_pack_fp8(w['qkv'])
                w['out_w_pk8'] = self._sk2_pack_fp8(w['out_w'])
            w['out_b'] = cast(st.leaf.to_out[0].bias)
            w['out_b32'] = castf(st.leaf.to_out[0].bias)
        else:
            net = st.leaf.net
            w['ff1_w'] = cast(net[0].weight)
            w['ff1_pk'] = self._sk2_pack(w['ff1_w'])
            # the geglu-fused epilogue (mode 1) needs N = 2 * ff2-in width
            if w['ff1_pk'] is not None and net[0].weight.shape[0] % 64:
                w['ff1_pk'] = None
            w['ff1_b'] = cast(net[0].bias)
            w['ff1_b32'] = castf(net[0].bias)
            w['ff2_w'] = cast(net[3].weight)
            w['ff2_pk'] = self._sk2_pack(w['ff2_w'])
            if self._want_fp8_decode():
                w['ff1_w_pk8'] = self._sk2_pack_fp8(w['ff1_w'])
                w['ff2_w_pk8'] = self._sk2_pack_fp8(w['ff2_w'])
            w['ff2_b'] = cast(net[3].bias)
            w['ff2_b32'] = castf(net[3].bias)
        return w

    @staticmethod
    def _want_fp8_decode():
        import os
        return os.environ.get('DALLE_AMD_FP8_DECODE', '0') == '1'

    def _sk2_ok(self, rows, pk, mode=0, k_dim=0):
        # measured on-box (scripts/bench_sk2.py): at 128 rows the MT=8
        # variants tie or lose to hipBLASLt except the geglu-fused ff1;
        # at <= 64 rows sk2 wins every decode shape
        if not (self._sk2_on and pk is not None) or k_dim > self._sk2_maxk:
            return False
        if rows in (16, 32, 64):
            return True
        return rows == 128 and mode == 1

    def _sk2(self, x, pk, bias32, N, K, mode, wscale=1.0):
        """Weights-streaming decode GEMM on a pre-packed tile layout with
        the epilogue (bias / geglu / fp32 head) fused — one dispatch where
        hipBLASLt + eager epilogues took two or three (see sk2_kernel).
        A uint8/e4m3 pack streams fp8 weights (wscale = its amax/448)."""
        from dalle_pytorch_amd.ops.dispatch import hip_module
        rows = x.numel() // K
        out = hip_module().sk2(x.reshape(rows, K), pk, bias32, N, K, mode,
                               wscale)
        no = N // 2 if mode == 1 else N
        return out.view(*x.shape[:-1], no)

    def _lin_t(self, x, wT, bias):
        """Decode linear with a pre-transposed weight: hipBLASLt picks a
        plain-NN kernel instead of the slow transposed-B one at M<=128
        (decode weights are static, so the transposed copy is free)."""
        rows = x.numel() // x.shape[-1]
        x2 = x.reshape(rows, x.shape[-1])
        out = torch.addmm(bias, x2, wT) if bias is not None else x2 @ wT
        return out.view(*x.shape[:-1], wT.shape[1])

    def _lin(self, x, w, bias32, bias):
        """Decode-step linear. The skinny-M weights-streaming kernel is kept
        behind DALLE_AMD_SKINNY=1: measured on-box, the decode step is bound
        by a ~5 us per-kernel execution floor (gen_kernel_stats profile), so
        the 3-kernel skinny path (zero + gemm + cast) loses to one hipBLASLt
        dispatch even though neither is near the weight-bandwidth floor."""
        import os
        rows = x.numel() // x.shape[-1]
        if (self._fused_decode and rows <= 128 and x.shape[-1] % 32 == 0
                and w.shape[0] % 4 == 0
                and os.environ.get('DALLE_AMD_SKINNY', '0') == '1'):
            from dalle_pytorch_amd.ops.dispatch import hip_module
            out = hip_module().skinny_gemm(
                x.reshape(rows, x.shape[-1]), w, bias32)
            return out.view(*x.shape[:-1], w.shape[0])
        return F.linear(x, w, bias)

    # ----------------------------------------------------------- branches

    def _rot3(self, q, k, v, cos, sin):
        rot = cos.shape[-1]

        def one(t):
            head, tail = t[..., :rot], t[..., rot:]
            h2 = head.reshape(*head.shape[:-1], -1, 2)
            a, b2 = h2.unbind(-1)
            rh = torch.stack((-b2, a), -1).reshape(head.shape)
            out = head * cos + rh * sin
            return torch.cat((out, tail), dim=-1) if tail.shape[-1] else out
        return one(q), one(k), one(v)

    def _proj(self, x, st_w, key, bias32_key, bias_key, mode=0):
        """Single-token projection: sk2 on the packed weight when profitable,
        else the measured-best hipBLASLt form for the shape."""
        w = st_w[key]
        rows = x.numel() // x.shape[-1]
        pk8 = st_w.get(key + '_pk8')
        if pk8 is not None and self._sk2_ok(rows, pk8[0], mode):
            return self._sk2(x, pk8[0], st_w.get(bias32_key), w.shape[0],
                             w.shape[1], mode, pk8[1])
        pk = st_w.get(key[:-2] + '_pk') if key.endswith('_w') else \
            st_w.get(key + '_pk')
        if self._sk2_ok(rows, pk, mode, w.shape[1]):
            return self._sk2(x, pk, st_w.get(bias32_key), w.shape[0],
                             w.shape[1], mode)
        return None

    def _ff_decode(self, st, y):
        """Single-token feed-forward: geglu-fused sk2 ff1 when available,
        hipBLASLt ff2 (measured faster than sk2 at K=4096). fp8 packs
        (opt-in) take both — at half the weight bytes sk2 wins ff2 too."""
        rows = y.numel() // y.shape[-1]
        pk8_1, pk8_2 = st.w.get('ff1_w_pk8'), st.w.get('ff2_w_pk8')
        if pk8_1 is not None and pk8_2 is not None \
                and self._sk2_ok(rows, pk8_1[0], 1):
            y = self._sk2(y, pk8_1[0], st.w['ff1_b32'],
                          st.w['ff1_w'].shape[0], st.w['ff1_w'].shape[1],
                          1, pk8_1[1])
            return self._sk2(y, pk8_2[0], st.w['ff2_b32'],
                             st.w['ff2_w'].shape[0], st.w['ff2_w'].shape[1],
                             0, pk8_2[1])
        if self._sk2_ok(rows, st.w['ff1_pk'], 1):
            y = self._sk2(y, st.w['ff1_pk'], st.w['ff1_b32'],
                          st.w['ff1_w'].shape[0], st.w['ff1_w'].shape[1], 1)
        else:
            from dalle_pytorch_amd.ops import geglu
            y = geglu(self._lin(y, st.w['ff1_w'], st.w['ff1_b32'],
                                st.w['ff1_b']))
        if self._sk2_ok(rows, st.w['ff2_pk'], 0, st.w['ff2_w'].shape[1]):
            return self._sk2(y, st.w['ff2_pk'], st.w['ff2_b32'],
                             st.w['ff2_w'].shape[0], st.w['ff2_w'].shape[1],
                             0)
        return self._lin(y, st.w['ff2_w'], st.w['ff2_b32'], st.w['ff2_b'])

    def _attn(self, st, x, offset_t, n):
        """x [b, n, dim] at positions offset..offset+n-1 (prefill has
        offset 0; decode has n == 1)."""
        leaf = st.leaf
        h, d = leaf.heads, leaf.dim_head
        qkv = self._proj(x, st.w, 'qkv', None, None) if n == 1 else None
        if qkv is None:
            qkv = (self._lin_t(x, st.w['qkv_T'], None) if n == 1
                   else F.linear(x, st.w['qkv']))
        if n == 1 and self._fused_decode:
            from dalle_pytorch_amd.ops.dispatch import hip_module
            out = hip_module().fa_decode(
                qkv.view(self.b, -1), st.k, st.v,
                self.cos if self.rotary else None,
                self.sin if self.rotary else None,
                offset_t, st.pattern, leaf.scale,
                st.live, st.live_cnt).view(self.b, 1, h * d)
            y = self._proj(out, st.w, 'out_w', 'out_b32', 'out_b')
            return y if y is not None else \
                self._lin_t(out, st.w['out_T'], st.w['out_b'])
        q, k, v = (t.reshape(self.b, n, h, d).permute(0, 2, 1, 3)
                   for t in qkv.chunk(3, dim=-1))
        if self.rotary:
            if n == 1:
                cos = self.cos.index_select(0, offset_t).unsqueeze(0).to(x.dtype)
                sin = self.sin.index_select(0, offset_t).unsqueeze(0).to(x.dtype)
            else:
                cos = self.cos[:n].to(x.dtype)
                sin = self.sin[:n].to(x.dtype)
            q, k, v = self._rot3(q, k, v, cos, sin)

        if n == 1:
            idx = offset_t.view(1, 1, 1, 1).expand(self.b, h, 1, d)
            st.k.scatter_(2, idx, k.to(self.dtype))
            st.v.scatter_(2, idx, v.to(self.dtype))
            scores = torch.matmul(q * leaf.scale,
                                  st.k.to(q.dtype).transpose(-1, -2))
            neg = -torch.finfo(scores.dtype).max
            allow = self.key_arange.unsqueeze(0) <= offset_t.unsqueeze(1)
            if st.pattern is not None:
                allow = allow & st.pattern.index_select(0, offset_t)
            scores = scores.masked_fill(~allow.view(1, 1, 1, self.N), neg)
            out = torch.matmul(scores.softmax(-1), st.v.to(q.dtype))
        else:
            st.k[:, :, :n] = k.to(self.dtype)
            st.v[:, :, :n] = v.to(self.dtype)
            sm = st.pattern[:n, :n].contiguous() if st.pattern is not None else None
            out = attention_core(q, k, v, leaf.scale, causal=True,
                                 static_mask=sm)
        out = out.permute(0, 2, 1, 3).reshape(self.b, n, h * d)
        return F.linear(out, st.w['out_w'], st.w['out_b'])

    def _shift_prefill(self, st, x, n):
        """Training-style token shift over the prompt + ring seeding
        (mirrors PreShiftToken's non-cached branch + deque seeding,
        reference transformer.py:155-198)."""
        shift = st.info['shift']
        S, text_len = shift.image_size, shift.text_len
        dim = x.shape[-1]
        if n < text_len:
            return x
        x_text, x_img = x[:, :text_len], x[:, text_len:]
        t_shift, t_pass = x_text.chunk(2, dim=-1)
        t_shift = F.pad(t_shift, (0, 0, 1, -1))
        x_text = torch.cat((t_shift, t_pass), dim=-1)

        n_img = x_img.shape[1]
        if n_img:
            grid = F.pad(x_img, (0, 0, 0, S * S - n_img))
            grid = grid.reshape(self.b, S, S, dim)
            top, left, *rest = grid.chunk(4, dim=-1)
            left = F.pad(left, (0, 0, 1, -1))
            top = F.pad(top, (0, 0, 0, 0, 1, -1))
            grid = torch.cat((top, left, *rest), dim=-1)
            x_img = grid.reshape(self.b, S * S, dim)[:, :n_img]
            # seed the ring with the SHIFTED last-row quarters — exactly what
            # the model's deque seeding records (transformer.py:188-198)
            last = x_img[:, -min(S, n_img):]
            for j in range(last.shape[1]):
                g = n_img - last.shape[1] + j
                st.ring[:, g % S] = last[:, j, :dim // 2].to(self.dtype)
        return torch.cat((x_text, x_img), dim=1)

    def _shift_decode(self, st, x, offset_t):
        shift = st.info['shift']
        S, text_len = shift.image_size, shift.text_len
        if self._fused_decode:
            from dalle_pytorch_amd.ops.dispatch import hip_module
            return hip_module().shift_decode(
                x.reshape(self.b, -1).contiguous(), offset_t, st.ring,
                text_len).view(self.b, 1, -1)
        dim = x.shape[-1]
        qdim = dim // 4
        g = (offset_t - text_len).clamp(min=0)
        pos = torch.remainder(g, S)
        prev = torch.remainder(g - 1, S)
        half = x[..., : dim // 2]                          # [b, 1, dim/2]

        gather_idx = pos.view(1, 1, 1).expand(self.b, 1, dim // 2)
        top = st.ring.gather(1, gather_idx).to(x.dtype)
        left = st.ring.gather(1, prev.view(1, 1, 1).expand(self.b, 1, dim // 2)
                              ).to(x.dtype)
        top_q = top[..., :qdim]
        left_q = left[..., qdim:]
        left_q = torch.where((g % S == 0).view(1, 1, 1),
                             torch.zeros_like(left_q), left_q)
        st.ring.scatter_(1, gather_idx, half.to(self.dtype))
        return torch.cat((top_q, left_q, x[..., dim // 2:]), dim=-1)

    def _branch(self, st, x, offset_t, n):
        dim = x.shape[-1]
        y = fused_layer_norm(x, st.w['ln_w'], st.w['ln_b'],
                             st.info['norm'].eps)
        if st.info['shift'] is not None:
            y = self._shift_prefill(st, y, n) if n > 1 else \
                self._shift_decode(st, y, offset_t)
        if st.is_attn:
            y = self._attn(st, y, offset_t, n)
        elif n == 1:
            y = self._ff_decode(st, y)
        else:
            from dalle_pytorch_amd.ops import geglu
            y = F.linear(y, st.w['ff1_w'], st.w['ff1_b'])
            y = geglu(y)
            y = F.linear(y, st.w['ff2_w'], st.w['ff2_b'])
        if st.info['norm_out'] is not None:
            y = F.layer_norm(y, (dim,), st.w['lno_w'], st.w['lno_b'],
                             st.info['norm_out'].eps)
        return y

    def _residual(self, x, st, y):
        # x + y*scale in one fused kernel (addcmul) instead of mul + add
        if st.w['scale'] is not None:
            return torch.addcmul(x, y, st.w['scale'])
        return x + y

    def _body(self, st, z, offset_t):
        """Branch compute after the (fused) LN+shift prelude: attention or FF."""
        if st.is_attn:
            return self._attn(st, z, offset_t, 1)
        return self._ff_decode(st, z)

    def _prelude(self, ext, stream, pend, st, off):
        """One fused kernel: apply the previous branch's residual to the
        stream, LayerNorm it for this branch, token-shift if wrapped."""
        y, scale32 = pend
        shift = st.info['shift']
        xn, z = ext.dec_prelude(
            stream.view(self.b, -1),
            None if y is None else y.view(self.b, -1),
            scale32, st.w['ln_w'], st.w['ln_b'],
            st.ring if shift is not None else None,
            self.offset_t, st.info['norm'].eps,
            shift.image_size if shift is not None else 1,
            shift.text_len if shift is not None else 0)
        return xn.view(self.b, 1, -1), z.view(self.b, 1, -1)

    def _run_stack_decode_fused(self, x, off):
        """Decode step with fused preludes (the step is bound by a ~5 us
        per-kernel floor: residual+LN+shift as one kernel per branch)."""
        from dalle_pytorch_amd.ops.dispatch import hip_module
        ext = hip_module()
        it = iter(self.states)
        if not self.reversible:
            pend = (None, None)
            last_scale = None
            for st in self.states:
                x, z = self._prelude(ext, x, pend, st, off)
                pend = (self._body(st, z, off), st.w['scale32'])
                last_scale = st.w['scale']
            y = pend[0]
            return torch.addcmul(x, y, last_scale) if last_scale is not None \
                else x + y
        x1, x2 = x, x.clone()
        pend2 = (None, None)
        g_st = None
        for f_st in it:
            g_st = next(it)
            x2, zf = self._prelude(ext, x2, pend2, f_st, off)
            f_out = self._body(f_st, zf, off)
            x1, zg = self._prelude(ext, x1, (f_out, f_st.w['scale32']), g_st, off)
            g_out = self._body(g_st, zg, off)
            pend2 = (g_out, g_st.w['scale32'])
        x2 = torch.addcmul(x2, pend2[0], g_st.w['scale']) \
            if g_st.w['scale'] is not None else x2 + pend2[0]
        return (x1 + x2) / 2

    def _run_stack(self, x, offset_t, n):
        if n == 1 and getattr(self, '_fused_prelude', False):
            return self._run_stack_decode_fused(x, offset_t)
        if not self.reversible:
            it = iter(self.states)
            for attn_st in it:
                ff_st = next(it)
                x = self._residual(x, attn_st, self._branch(attn_st, x, offset_t, n))
                x = self._residual(x, ff_st, self._branch(ff_st, x, offset_t, n))
            return x
        x1, x2 = x, x.clone()
        it = iter(self.states)
        for f_st in it:
            g_st = next(it)
            x1 = self._residual(x1, f_st, self._branch(f_st, x2, offset_t, n))
            x2 = self._residual(x2, g_st, self._branch(g_st, x1, offset_t, n))
        return (x1 + x2) / 2

    def _head(self, x, position_mask_rows):
        d = self.dalle
        if d.stable:
            x = x / x.amax(dim=-1, keepdim=True)
        x = F.layer_norm(x, (x.shape[-1],), self.head_w['ln_w'],
                         self.head_w['ln_b'], d.to_logits[0].eps)
        logits = F.linear(x, self.head_w['w'], self.head_w['b'])
        lm = d.logits_mask[0].to(self.device).index_select(0, position_mask_rows)
        return logits.masked_fill(lm.unsqueeze(0), -torch.finfo(logits.dtype).max)

    def _head_img(self, x):
        """Image-vocab-only head for decode steps: an image position's logits
        mask allows exactly the image-vocab rows, so projecting against that
        slice is equivalent and ~7x cheaper (plus top-k/gumbel shrink with it)."""
        d = self.dalle
        if d.stable:
            x = x / x.amax(dim=-1, keepdim=True)
        x = F.layer_norm(x, (x.shape[-1],), self.head_w['ln_w'],
                         self.head_w['ln_b'], d.to_logits[0].eps)
        y = self._proj(x, self.head_w, 'w_img', 'b_img32', 'b_img', mode=2)
        return y if y is not None else \
            self._lin(x, self.head_w['w_img'], self.head_w['b_img32'],
                      self.head_w['b_img'])

    # ----------------------------------------------------------- prefill

    @torch.no_grad()
    def prefill(self, text, image_tokens=None):
        """Embed the prompt (BOS + text [+ primed image tokens]), run the
        stack batched while writing the static caches, return last-position
        logits. Mirrors DALLE.forward's embedding path (:589-624)."""
        d = self.dalle
        device = self.device
        t_range = torch.arange(d.text_seq_len, device=device) + \
            (d.num_text_tokens - d.text_seq_len)
        text = torch.where(text == 0, t_range, text)
        text = F.pad(text, (1, 0), value=0)
        tokens = d.text_emb(text)
        tokens = tokens + d.text_pos_emb(torch.arange(text.shape[1], device=device))
        if image_tokens is not None and image_tokens.numel():
            iemb = d.image_emb(image_tokens)
            iemb = iemb + d.image_pos_emb(iemb)
            tokens = torch.cat((tokens, iemb), dim=1)
        n = tokens.shape[1]
        x = tokens.to(self.dtype)
        x = self._run_stack(x, None, n)
        rows = torch.arange(n, device=device)
        logits = self._head(x, rows)
        self.offset_t.fill_(n)
        return logits[:, -1]

    # ------------------------------------------------------------- step

    def step(self, token):
        """One decode step: image-token ids [b] at position offset ->
        logits [b, total_tokens] (or [b, num_image_tokens] in the generate
        loop's image-head mode); advances the offset."""
        d = self.dalle
        off = self.offset_t
        emb = d.image_emb(token).unsqueeze(1)
        if not self.rotary:
            g = (off - d.text_seq_len - 1).clamp(min=0)
            full = (d.image_pos_emb.weights[0] + d.image_pos_emb.weights[1]) \
                .reshape(1, -1, emb.shape[-1])
            emb = emb + full.index_select(1, g)
        x = self._run_stack(emb.to(self.dtype), off, 1)
        if getattr(self, '_img_head', False):
            logits = self._head_img(x)[:, 0]
        else:
            logits = self._head(x, off)[:, 0]
        self.offset_t += 1
        return logits

    # --------------------------------------------------------- generate

    @torch.no_grad()
    def generate(self, text, filter_thres=0.9, temperature=1.0, cond_scale=1.0):
        """Batch image generation; with ``cond_scale != 1`` classifier-free
        guidance runs the conditioned and null-conditioned streams as ONE
        doubled batch (one static graph, one kernel sequence — the reference
        pays a second full forward per step, dalle_pytorch.py:564-574).
        Requires a decoder built with batch_size == 2 * text batch then."""
        d = self.dalle
        guided = cond_scale != 1
        nb = text.shape[0]
        expect = 2 * nb if guided else nb
        assert expect == self.b, \
            f'decoder built for batch {self.b}, got {nb} (guided={guided})'
        was_training = d.training
        d.eval()
        self._img_head = True
        text = text[:, :d.text_seq_len]
        if guided:
            # null-conditioned half: zeroed text (forward's null_cond path
            # maps zeros to the unique padding ids)
            text = torch.cat((text, torch.zeros_like(text)), dim=0)
        try:
            ntt, nit = d.num_text_tokens, d.num_image_tokens
            # reference top-k semantics: k is a fraction of the FULL vocab
            k = max(int((1 - filter_thres) * d.total_tokens), 1)
            k = min(k, nit)
            sample_args = (nb, guided, cond_scale, k, temperature)
            # the captured graph is REUSED across generate() calls: all its
            # state (offset, gen pointer, out buffer, k/v caches, rings)
            # lives in persistent device tensors reset in place. Recapture
            # only when a capture-baked constant changes — per-call
            # recapture leaked the old graph pool (~74 MB/call, soak-tested)
            # and cost ~100 ms of capture per batch.
            if getattr(self, '_graph_key', None) != sample_args:
                self._graph = None
                self._g_token = self._g_next = None
                self._out_buf = torch.empty(nb, d.image_seq_len,
                                            dtype=torch.long, device=self.device)
                self._gen_ptr = torch.zeros(1, dtype=torch.long,
                                            device=self.device)
                self._graph_key = sample_args
            else:
                self._gen_ptr.zero_()
            logits = self.prefill(text)[:, ntt:ntt + nit]

            # token 0 from the prefill logits (outside any graph)
            li = logits.float()
            if guided:
                cond, null = li[:nb], li[nb:]
                li = null + (cond - null) * cond_scale
            vals, idx = li.topk(k, dim=-1)
            filtered = torch.full_like(li, -torch.finfo(li.dtype).max)
            filtered.scatter_(1, idx, vals)
            token = gumbel_sample(filtered, temperature=temperature)
            self._out_buf.index_copy_(1, self._gen_ptr, token.unsqueeze(1))
            self._gen_ptr += 1

            step_fn = self._graph_token_step if self.use_graph else self._token_step
            for _ in range(d.image_seq_len - 1):
                token = step_fn(token, *sample_args)
            images = d.vae.decode(self._out_buf)
        finally:
            self._img_head = False
            d.train(was_training)
        return images

    def _snapshot(self):
        return ([st.k.clone() if st.k is not None else None for st in self.states],
                [st.v.clone() if st.v is not None else None for st in self.states],
                [st.ring.clone() if st.ring is not None else None for st in self.states],
                self.offset_t.clone(),
                self._gen_ptr.clone() if getattr(self, '_gen_ptr', None) is not None else None)

    def _restore(self, snap):
        ks, vs, rings, off, ptr = snap
        for st, k, v, r in zip(self.states, ks, vs, rings):
            if k is not None:
                st.k.copy_(k)
                st.v.copy_(v)
            if r is not None:
                st.ring.copy_(r)
        self.offset_t.copy_(off)
        if ptr is not None:
            self._gen_ptr.copy_(ptr)

    # ------------------------------------------------- fully in-graph step

    def _token_step(self, token, nb, guided, cond_scale, k, temperature):
        """token ids [nb] -> next token ids [nb], sampling included — the
        whole thing is graph-capturable (RNG advances via the graph-safe
        philox state), so a generation step replays with ZERO eager kernels."""
        feed = torch.cat((token, token), dim=0) if guided else token
        logits = self.step(feed).float()
        if guided:
            cond, null = logits[:nb], logits[nb:]
            logits = null + (cond - null) * cond_scale
        if self._fused_decode and logits.shape[-1] <= 8192:
            from dalle_pytorch_amd.ops.dispatch import hip_module
            noise = torch.rand_like(logits)   # graph-capture-safe philox
            # the sampler writes the next-token feed buffer IN PLACE and
            # appends to the sequence buffer itself: no index_copy_ kernel
            # and no eager feed copy between graph replays
            nxt = hip_module().sample_topk_gumbel(
                logits.contiguous(), noise, k, max(temperature, 1e-10),
                out_tok=token, seq=self._out_buf, seq_ptr=self._gen_ptr)
        else:
            vals, idx = logits.topk(k, dim=-1)
            filtered = torch.full_like(logits, -torch.finfo(logits.dtype).max)
            filtered.scatter_(1, idx, vals)
            nxt = gumbel_sample(filtered, temperature=temperature)
            self._out_buf.index_copy_(1, self._gen_ptr, nxt.unsqueeze(1))
        self._gen_ptr += 1
        return nxt

    def _graph_token_step(self, token, *args):
        if self._graph is None:
            self._g_token = token.clone()
            snap = self._snapshot()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self._g_next = self._token_step(self._g_token, *args)
            torch.cuda.current_stream().wait_stream(s)
            self._restore(snap)
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._g_next = self._token_step(self._g_token, *args)
            self._graph.replay()
            return self._g_next
        if token is not self._g_token:   # the sampler writes the feed
            self._g_token.copy_(token)   # buffer in place, so steady-state
        self._graph.replay()             # replays skip this copy entirely
        return self._g_next

    def _graph_step(self, token):
        if self._graph is None:
            self._g_token = token.clone()
            snap = self._snapshot()     # warmup mutates caches/rings/offset
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self._g_logits = self.step(self._g_token)
            torch.cuda.current_stream().wait_stream(s)
            self._restore(snap)
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._g_logits = self.step(self._g_token)
            # capture records without executing: run the real step once
            self._graph.replay()
            return self._g_logits
        self._g_token.copy_(token)
        self._graph.replay()
        return self._g_logits
