"""RCCL DP engine: multi-process (gloo, CPU) correctness of broadcast,
bucketed gradient all-reduce, no_sync accumulation, scalar averaging."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dalle_pytorch_amd.parallel import DataParallelEngine, average_scalar


def _run_worker(rank, world, port, fn_name):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    dist.init_process_group('gloo', rank=rank, world_size=world)
    try:
        globals()[fn_name](rank, world)
    finally:
        dist.destroy_process_group()


def _spawn(fn_name, world=2, port=29611):
    mp.spawn(_run_worker, args=(world, port, fn_name), nprocs=world, join=True)


def _model(seed):
    torch.manual_seed(seed)
    return torch.nn.Sequential(torch.nn.Linear(8, 32), torch.nn.Tanh(),
                               torch.nn.Linear(32, 1))


def _check_broadcast(rank, world):
    model = _model(seed=rank)          # deliberately different weights
    DataParallelEngine(model, bucket_bytes=1 << 10)
    ref = _model(seed=0)               # rank 0's init
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p.data, q.data)


def _check_grad_allreduce(rank, world):
    torch.manual_seed(0)
    model = _model(seed=0)
    engine = DataParallelEngine(model, bucket_bytes=1 << 10, wire_dtype=None)
    torch.manual_seed(100 + rank)
    x = torch.randn(4, 8)
    model(x).sum().backward()
    engine.finish_gradient_sync()
    got = [p.grad.clone() for p in model.parameters()]

    # oracle: average of per-rank grads computed independently
    ref_model = _model(seed=0)
    acc = [torch.zeros_like(p) for p in ref_model.parameters()]
    for r in range(world):
        m = _model(seed=0)
        torch.manual_seed(100 + r)
        xr = torch.randn(4, 8)
        m(xr).sum().backward()
        for a, p in zip(acc, m.parameters()):
            a += p.grad / world
    for g, a in zip(got, acc):
        assert torch.allclose(g, a, atol=1e-6), (g - a).abs().max()


def _check_no_sync_accumulation(rank, world):
    model = _model(seed=0)
    engine = DataParallelEngine(model, bucket_bytes=1 << 10, wire_dtype=None)
    torch.manual_seed(200 + rank)
    x1, x2 = torch.randn(4, 8), torch.randn(4, 8)
    with engine.no_sync():
        model(x1).sum().backward()
    model(x2).sum().backward()
    engine.finish_gradient_sync()

    # oracle: per-rank sum of both microbatch grads, then world-average
    acc = [torch.zeros_like(p) for p in model.parameters()]
    for r in range(world):
        m = _model(seed=0)
        torch.manual_seed(200 + r)
        a1, a2 = torch.randn(4, 8), torch.randn(4, 8)
        (m(a1).sum() + m(a2).sum()).backward()
        for a, p in zip(acc, m.parameters()):
            a += p.grad / world
    for p, a in zip(model.parameters(), acc):
        assert torch.allclose(p.grad, a, atol=1e-6)


def _check_average_scalar(rank, world):
    v = average_scalar(torch.tensor(float(rank)))
    assert torch.allclose(v, torch.tensor((world - 1) / 2))


def _check_distributed_clip(rank, world):
    """Clip after all-reduce == torch clip on the averaged-grad oracle."""
    model = _model(seed=0)
    engine = DataParallelEngine(model, bucket_bytes=1 << 10, wire_dtype=None)
    torch.manual_seed(300 + rank)
    x = torch.randn(4, 8)
    (model(x) ** 2).sum().backward()
    engine.finish_gradient_sync()
    total = engine.clip_grad_norm_(0.02)

    ref = _model(seed=0)
    acc = [torch.zeros_like(p) for p in ref.parameters()]
    for r in range(world):
        m = _model(seed=0)
        torch.manual_seed(300 + r)
        xr = torch.randn(4, 8)
        (m(xr) ** 2).sum().backward()
        for a, p in zip(acc, m.parameters()):
            a += p.grad / world
    for a, p in zip(acc, ref.parameters()):
        p.grad = a
    ref_total = torch.nn.utils.clip_grad_norm_(ref.parameters(), 0.02)
    assert torch.allclose(total, ref_total, rtol=1e-5)
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.grad, q.grad, rtol=1e-5, atol=1e-8)


def _check_bf16_wire_allreduce(rank, world):
    """Default bf16-wire buckets: halved payload; grads
    match the fp32 oracle within one bf16 rounding step."""
    model = _model(seed=0)
    engine = DataParallelEngine(model, bucket_bytes=1 << 10)   # default wire
    assert engine.wire_dtype == torch.bfloat16
    torch.manual_seed(100 + rank)
    x = torch.randn(4, 8)
    model(x).sum().backward()
    engine.finish_gradient_sync()

    acc = [torch.zeros_like(p) for p in model.parameters()]
    for r in range(world):
        m = _model(seed=0)
        torch.manual_seed(100 + r)
        xr = torch.randn(4, 8)
        m(xr).sum().backward()
        for a, p in zip(acc, m.parameters()):
            a += p.grad / world
    for p, a in zip(model.parameters(), acc):
        scale = a.abs().max().clamp(min=1e-6)
        assert ((p.grad - a).abs().max() / scale) < 2e-2


def _check_convergence_equivalence(rank, world):
    """DP loss curve == single-process full-batch loss curve:
    each rank trains on its half of a fixed global batch; with fp32 wire the
    curves match to float tolerance, with bf16 wire to bf16 tolerance."""
    torch.manual_seed(42)
    data = torch.randn(8, 8)
    target = torch.randn(8, 1)

    def run(world_slice=None, engine_kw=None, steps=6):
        model = _model(seed=0)
        engine = DataParallelEngine(model, bucket_bytes=1 << 10,
                                    **(engine_kw or {}))
        opt = torch.optim.Adam(model.parameters(), lr=1e-2)
        losses = []
        for _ in range(steps):
            x, y = (data, target) if world_slice is None else (
                data.chunk(world, 0)[rank], target.chunk(world, 0)[rank])
            loss = ((model(x) - y) ** 2).mean()
            opt.zero_grad(set_to_none=False)
            engine.zero_grad()
            loss.backward()
            engine.finish_gradient_sync()
            opt.step()
            losses.append(engine.average_all(loss.detach()).item())
        return losses

    dp = run(world_slice=True, engine_kw=dict(wire_dtype=None))
    dp_bf16 = run(world_slice=True)

    # single-process oracle computed identically on every rank
    model = _model(seed=0)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    ref = []
    for _ in range(6):
        loss = ((model(data) - target) ** 2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        ref.append(loss.item())

    for a, b in zip(dp, ref):
        assert abs(a - b) < 1e-5, (dp, ref)
    for a, b in zip(dp_bf16, ref):
        assert abs(a - b) < 5e-2, (dp_bf16, ref)
    assert dp[-1] < dp[0]   # it actually trains


@pytest.mark.parametrize('fn,port', [
    ('_check_broadcast', 29611),
    ('_check_grad_allreduce', 29612),
    ('_check_no_sync_accumulation', 29613),
    ('_check_average_scalar', 29614),
    ('_check_distributed_clip', 29615),
    ('_check_bf16_wire_allreduce', 29616),
    ('_check_convergence_equivalence', 29617),
])
def test_distributed_gloo(fn, port):
    _spawn(fn, world=2, port=port)


def test_single_process_noop():
    model = _model(seed=0)
    engine = DataParallelEngine(model, bucket_bytes=1 << 10, wire_dtype=None)
    x = torch.randn(4, 8)
    model(x).sum().backward()
    engine.finish_gradient_sync()
    assert engine.is_root and engine.world_size == 1
    g0 = [p.grad.clone() for p in model.parameters()]
    engine.zero_grad()
    assert all((p.grad == 0).all() for p in model.parameters())

    m2 = _model(seed=0)
    m2(x).sum().backward()
    for a, b in zip(g0, (p.grad for p in m2.parameters())):
        assert torch.allclose(a, b)


def test_bucket_clip_grad_norm_matches_torch():
    """engine.clip_grad_norm_ (per-bucket norms) == torch's per-param clip."""
    torch.manual_seed(7)
    x = torch.randn(4, 8)

    m1 = _model(seed=3)
    eng = DataParallelEngine(m1, bucket_bytes=1 << 10)
    (m1(x) ** 2).sum().backward()
    eng.finish_gradient_sync()
    total = eng.clip_grad_norm_(0.05)

    m2 = _model(seed=3)
    (m2(x) ** 2).sum().backward()
    ref_total = torch.nn.utils.clip_grad_norm_(m2.parameters(), 0.05)

    assert torch.allclose(total, ref_total, rtol=1e-6)
    for a, b in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(a.grad, b.grad, rtol=1e-6, atol=1e-7)

    # above the max norm nothing is scaled
    m3 = _model(seed=3)
    eng3 = DataParallelEngine(m3, bucket_bytes=1 << 10)
    (m3(x) ** 2).sum().backward()
    g0 = [p.grad.clone() for p in m3.parameters()]
    eng3.clip_grad_norm_(1e9)
    for a, p in zip(g0, m3.parameters()):
        assert torch.allclose(a, p.grad)


def test_distributed_utils_facade():
    """Reference-surface facade maps onto the RCCL engine."""
    import argparse
    from dalle_pytorch_amd import distributed_utils as du

    parser = argparse.ArgumentParser()
    parser = du.wrap_arg_parser(parser)
    args = parser.parse_args([])
    backend = du.set_backend_from_args(args)
    backend.initialize()
    assert backend.get_world_size() == 1 and backend.is_root_worker()
    assert du.using_backend(type(backend))
    backend.check_batch_size(4)

    model = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    m2, o2, _, _ = backend.distribute(model=model, optimizer=opt)
    assert m2 is model and o2 is opt
    v = backend.average_all(torch.tensor(3.0))
    assert float(v) == 3.0

    args = parser.parse_args(['--distributed_backend', 'deepspeed'])
    backend = du.set_backend_from_args(args)
    assert du.using_backend(du.DeepSpeedBackend)
