"""Fake-world DP tests over gloo on CPU: bucketing,
overlap hooks, averaging and broadcast, without any GPU."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn

from yet_another_mobilenet_series_amd.parallel.ddp import DataParallelEngine


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))


def _worker(rank, world, tmpfile, overlap, out_path):
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmpfile}", rank=rank, world_size=world)
    try:
        torch.manual_seed(100 + rank)  # different init per rank: broadcast must fix
        model = _make_model(seed=100 + rank)
        eng = DataParallelEngine(model, bucket_mb=0.0001, overlap=overlap)
        torch.manual_seed(0)
        xs = torch.randn(world * 4, 8)
        ys = torch.randn(world * 4, 4)
        x = xs[rank * 4:(rank + 1) * 4]
        y = ys[rank * 4:(rank + 1) * 4]
        for _ in range(3):
            eng.zero_grad()
            loss = ((eng(x) - y) ** 2).mean()
            loss.backward()
            eng.finish()
            with torch.no_grad():
                for p in model.parameters():
                    p -= 0.1 * p.grad
        if rank == 0:
            torch.save([p.detach().clone() for p in model.parameters()],
                       out_path)
    finally:
        dist.destroy_process_group()


def _single_process_reference():
    model = _make_model(seed=100)  # rank-0 init is what broadcast spreads
    torch.manual_seed(0)
    xs = torch.randn(8, 8)
    ys = torch.randn(8, 4)
    for _ in range(3):
        for p in model.parameters():
            p.grad = None
        # mean over per-rank mean losses == mean of bucket averages
        loss = 0.5 * (((model(xs[:4]) - ys[:4]) ** 2).mean() +
                      ((model(xs[4:]) - ys[4:]) ** 2).mean())
        loss.backward()
        with torch.no_grad():
            for p in model.parameters():
                p -= 0.1 * p.grad
    return [p.detach() for p in model.parameters()]


@pytest.mark.parametrize("overlap", [False, True])
def test_dp_matches_single_process(tmp_path, overlap):
    world = 2
    # gloo TCP setup is occasionally flaky under parallel test load: retry
    for attempt in range(3):
        ctx = mp.get_context("spawn")
        tmpfile = str(tmp_path / f"init_{overlap}_{attempt}")
        out_path = str(tmp_path / f"out_{overlap}_{attempt}.pt")
        ps = [ctx.Process(target=_worker,
                          args=(r, world, tmpfile, overlap, out_path))
              for r in range(world)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
        if all(p.exitcode == 0 for p in ps) and os.path.exists(out_path):
            got = torch.load(out_path)
            break
        if attempt == 2:
            assert False, f"workers failed: {[p.exitcode for p in ps]}"
    want = _single_process_reference()
    for a, b in zip(got, want):
        assert torch.allclose(a, b, atol=1e-5)


def test_bucket_layout_world1():
    model = _make_model()
    eng = DataParallelEngine(model, bucket_mb=0.0001, force_bucketed=True)
    total = sum(p.numel() for p in model.parameters())
    assert sum(b.flat.numel() for b in eng.buckets) == total
    assert len(eng.buckets) >= 2  # tiny bucket_mb forces multiple buckets
    # grads are views into flat buffers
    eng.zero_grad()
    loss = eng(torch.randn(4, 8)).sum()
    loss.backward()
    for b in eng.buckets:
        off = 0
        for p in b.params:
            assert p.grad.data_ptr() == b.flat[off:off + p.numel()].data_ptr()
            off += p.numel()


def test_grad_order_reverse_registration():
    model = _make_model()
    eng = DataParallelEngine(model, bucket_mb=1e-9, force_bucketed=True)
    first_bucket_params = set(id(p) for p in eng.buckets[0].params)
    last_param = list(model.parameters())[-1]
    assert id(last_param) in first_bucket_params


def _dying_worker(rank, world, tmpfile, out_path):
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    import datetime
    dist.init_process_group(
        "gloo", init_method=f"file://{tmpfile}", rank=rank, world_size=world,
        timeout=datetime.timedelta(seconds=5))
    try:
        if rank == 1:
            os._exit(7)  # simulate a dead rank mid-step
        t = torch.ones(4)
        try:
            dist.all_reduce(t)  # must NOT hang forever: timeout or error
            result = "completed"
        except Exception:
            result = "raised"
        torch.save(result, out_path)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_dead_rank_does_not_hang(tmp_path):
    """Failure detection: a dead rank must surface as a
    timeout/error on the survivors within the configured window, not a
    silent hang."""
    world = 2
    ctx = mp.get_context("spawn")
    tmpfile = str(tmp_path / "init_dead")
    out_path = str(tmp_path / "survivor.pt")
    ps = [ctx.Process(target=_dying_worker,
                      args=(r, world, tmpfile, out_path))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(60)
        if p.is_alive():
            p.terminate()
            assert False, "survivor hung past the collective timeout"
    assert os.path.exists(out_path)
    assert torch.load(out_path) == "raised"

def _accum_worker(rank, world, tmpfile, out_path):
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmpfile}", rank=rank, world_size=world)
    try:
        model = _make_model(seed=7)
        eng = DataParallelEngine(model, bucket_mb=0.0001, overlap=True)
        torch.manual_seed(1)
        xs = torch.randn(world * 8, 8)
        ys = torch.randn(world * 8, 4)
        # 2 micro-batches of 4 per rank, one optimizer boundary
        for micro in range(2):
            lo = rank * 8 + micro * 4
            eng.set_no_sync(micro < 1)
            loss = ((eng(xs[lo:lo + 4]) - ys[lo:lo + 4]) ** 2).mean() / 2
            loss.backward()
        eng.finish()
        if rank == 0:
            torch.save([p.grad.detach().clone()
                        for p in model.parameters()], out_path)
    finally:
        dist.destroy_process_group()


def test_grad_accumulation_matches_big_batch(tmp_path):
    """k=2 accumulation over gloo world-2 == one global 16-sample step."""
    world = 2
    for attempt in range(3):
        ctx = mp.get_context("spawn")
        tmpfile = str(tmp_path / f"init_acc_{attempt}")
        out_path = str(tmp_path / f"out_acc_{attempt}.pt")
        ps = [ctx.Process(target=_accum_worker,
                          args=(r, world, tmpfile, out_path))
              for r in range(world)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
        if all(p.exitcode == 0 for p in ps) and os.path.exists(out_path):
            got = torch.load(out_path)
            break
        if attempt == 2:
            assert False, f"workers failed: {[p.exitcode for p in ps]}"
    model = _make_model(seed=7)
    torch.manual_seed(1)
    xs = torch.randn(16, 8)
    ys = torch.randn(16, 4)
    # mean over the 4 micro-losses (2 ranks x 2 micros), matching /2 + /world
    loss = sum(((model(xs[i * 4:(i + 1) * 4]) - ys[i * 4:(i + 1) * 4]) ** 2
                ).mean() for i in range(4)) / 4
    loss.backward()
    for a, p in zip(got, model.parameters()):
        assert torch.allclose(a, p.grad, atol=1e-5), (a - p.grad).abs().max()


def test_auto_bucket_sizing():
    """bucket_mb=None targets ~7 in-flight reduces (one per xGMI link) for
    big models and a 1 MB floor for small ones."""
    from yet_another_mobilenet_series_amd.parallel.ddp import (
        MIN_BUCKET_MB, XGMI_LINKS, auto_bucket_mb)
    v3l_bytes = 5_483_000 * 4  # V3-Large fp32 grads ≈ 21.9 MB
    mb = auto_bucket_mb(v3l_bytes)
    assert abs(mb - 21.9 / XGMI_LINKS) / mb < 0.05
    assert auto_bucket_mb(100_000 * 4) == MIN_BUCKET_MB
    # engine path: world-1 stays unbucketed; forced bucketing honors auto
    model = _make_model()
    eng = DataParallelEngine(model, bucket_mb=None, force_bucketed=True)
    assert eng.bucket_mb == MIN_BUCKET_MB
    assert len(eng.buckets) == 1  # tiny model -> one bucket


def _bf16_reduce_worker(rank, world, tmpfile, out_path):
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmpfile}", rank=rank, world_size=world)
    try:
        model = _make_model(seed=100)
        eng = DataParallelEngine(model, bucket_mb=0.0001, overlap=True,
                                 reduce_dtype=torch.bfloat16)
        torch.manual_seed(0)
        xs = torch.randn(world * 4, 8)
        ys = torch.randn(world * 4, 4)
        eng.zero_grad()
        loss = ((eng(xs[rank * 4:(rank + 1) * 4]) -
                 ys[rank * 4:(rank + 1) * 4]) ** 2).mean()
        loss.backward()
        eng.finish()
        if rank == 0:
            torch.save([p.grad.detach().clone()
                        for p in model.parameters()], out_path)
    finally:
        dist.destroy_process_group()


def test_bf16_wire_reduction(tmp_path):
    """reduce_dtype=bf16: gradients reduce over a bf16 wire buffer and land
    back in the fp32 flat grads within bf16 rounding of the fp32 result."""
    world = 2
    ctx = mp.get_context("spawn")
    tmpfile = str(tmp_path / "init_bf16")
    out_path = str(tmp_path / "grads.pt")
    ps = [ctx.Process(target=_bf16_reduce_worker,
                      args=(r, world, tmpfile, out_path))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(120)
        assert p.exitcode == 0
    got = torch.load(out_path)
    # fp32 reference grads of the same global batch
    model = _make_model(seed=100)
    torch.manual_seed(0)
    xs = torch.randn(8, 8)
    ys = torch.randn(8, 4)
    loss = 0.5 * (((model(xs[:4]) - ys[:4]) ** 2).mean() +
                  ((model(xs[4:]) - ys[4:]) ** 2).mean())
    loss.backward()
    for g, p in zip(got, model.parameters()):
        assert torch.allclose(g.float(), p.grad, rtol=2e-2, atol=1e-3)


def test_dp_world8_gloo(tmp_path):
    """World-8 fake-world parity: the bucket/overlap machinery at the real
    node's rank count (VERDICT round-1 item 3)."""
    world = 8
    ctx = mp.get_context("spawn")
    tmpfile = str(tmp_path / "init_w8")
    out_path = str(tmp_path / "params_w8.pt")
    ps = [ctx.Process(target=_worker,
                      args=(r, world, tmpfile, True, out_path))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(180)
        assert p.exitcode == 0
    got = torch.load(out_path)
    # single-process reference over the same global batch
    model = _make_model(seed=100)
    torch.manual_seed(0)
    xs = torch.randn(world * 4, 8)
    ys = torch.randn(world * 4, 4)
    for _ in range(3):
        for p in model.parameters():
            p.grad = None
        loss = sum(((model(xs[r * 4:(r + 1) * 4]) -
                     ys[r * 4:(r + 1) * 4]) ** 2).mean()
                   for r in range(world)) / world
        loss.backward()
        with torch.no_grad():
            for p in model.parameters():
                p -= 0.1 * p.grad
    for a, b in zip(got, model.parameters()):
        assert torch.allclose(a, b.detach(), rtol=1e-4, atol=1e-5)


def _broadcast_worker(rank, world, tmpfile, out_path):
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmpfile}", rank=rank, world_size=world)
    try:
        from yet_another_mobilenet_series_amd.parallel.distributed import (
            broadcast_module)
        torch.manual_seed(123 + rank)  # divergent init AND buffers
        model = nn.Sequential(nn.Linear(8, 16), nn.BatchNorm1d(16),
                              nn.ReLU(), nn.Linear(16, 4))
        model(torch.randn(4 + rank, 8))  # divergent BN running stats
        for b in model.buffers():
            b.add_(rank)
        broadcast_module(model, src=0)
        state = {k: v.clone() for k, v in model.state_dict().items()}
        torch.save(state, f"{out_path}.{rank}")
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_broadcast_module_syncs_params_and_buffers(tmp_path):
    """broadcast_module makes every rank bit-identical to rank 0 —
    params AND buffers (BN running stats, counters)."""
    for attempt in range(3):
        ctx = mp.get_context("spawn")
        tmpfile = str(tmp_path / f"bcast_init_{attempt}")
        out = str(tmp_path / "bcast_state")
        ps = [ctx.Process(target=_broadcast_worker,
                          args=(r, 2, tmpfile, out)) for r in range(2)]
        [p.start() for p in ps]
        [p.join(120) for p in ps]
        if all(p.exitcode == 0 for p in ps):
            break
    assert all(p.exitcode == 0 for p in ps)
    s0 = torch.load(out + ".0", weights_only=True)
    s1 = torch.load(out + ".1", weights_only=True)
    assert set(s0) == set(s1)
    for k in s0:
        assert torch.equal(s0[k], s1[k]), k
