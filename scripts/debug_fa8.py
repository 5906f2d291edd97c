import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import os, torch
import dalle_pytorch_amd._hip as ext

def run(q, k, v, scale=0.125, causal=False):
    out, lse = ext.fa_fwd(q, k, v, scale, causal, None, None, None, False)
    return out.float(), lse

def oracle(q, k, v, scale=0.125, causal=False):
    s = (q.float() * scale) @ k.float().transpose(-1, -2)
    if causal:
        n = s.shape[-1]
        s = s.masked_fill(torch.ones(n, n, device=s.device, dtype=torch.bool).triu_(1), float('-inf'))
    return s.softmax(-1) @ v.float()

torch.manual_seed(0)
b, h, n = 1, 1, 64
q = torch.randn(b, h, n, 64, device='cuda').bfloat16()
k = torch.randn(b, h, n, 64, device='cuda').bfloat16()
v = torch.randn(b, h, n, 64, device='cuda').bfloat16()

# case C: Q=0 -> P uniform -> out = mean(V)  (tests PV/V-tile side only)
qz = torch.zeros_like(q)
oc, _ = run(qz, k, v)
rc = v.float().mean(2, keepdim=True).expand_as(v)
print('C (V-side) max err:', (oc - rc).abs().max().item())

# case B: V = identity -> out = P  (tests QK^T + softmax + P redistribution)
vi = torch.eye(64, device='cuda').bfloat16().view(1, 1, 64, 64)
ob, _ = run(q, k, vi)
s = (q.float() * 0.125) @ k.float().transpose(-1, -2)
rb = s.softmax(-1)
err = (ob - rb).abs()
print('B (P path) max err:', err.abs().max().item())
if err.max() > 1e-2:
    bad = (err[0,0] > 1e-2)
    print('  bad q rows:', bad.any(1).nonzero().flatten().tolist()[:12])
    print('  bad k cols:', bad.any(0).nonzero().flatten().tolist()[:20])
    # find permutation: for each q row, which ref col best matches got col 0..
    g, r = ob[0,0], rb[0,0]
    # guess column permutation by correlating
    perm = []
    for c in range(16):
        d = (g[:, c:c+1] - r).abs().sum(0)
        perm.append(int(d.argmin()))
    print('  got col -> ref col (first 16):', perm)
    permr = []
    for rr in range(8):
        d = (g[rr:rr+1, :] - r).abs().sum(1)
        permr.append(int(d.argmin()))
    print('  got row -> ref row (first 8):', permr)

# case A: full random
oa, lse = run(q, k, v)
ra = oracle(q, k, v)
print('A (full) max err:', (oa - ra).abs().max().item())
sl = torch.logsumexp(s, -1)
print('lse err:', (lse.float()[0,0] - sl[0,0]).abs().max().item())

# case D: causal
od, _ = run(q, k, v, causal=True)
rd = oracle(q, k, v, causal=True)
print('D (causal) max err:', (od - rd).abs().max().item())

# case E: nq=192 multi-tile online softmax
n2 = 192
q2 = torch.randn(b, h, n2, 64, device='cuda').bfloat16()
k2 = torch.randn(b, h, n2, 64, device='cuda').bfloat16()
v2 = torch.randn(b, h, n2, 64, device='cuda').bfloat16()
oe, _ = run(q2, k2, v2, causal=True)
re = oracle(q2, k2, v2, causal=True)
print('E (192 causal) max err:', (oe - re).abs().max().item())
