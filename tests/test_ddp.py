"""Multi-process (gloo, CPU) tests of the RCCL-shaped data-parallel
engine: parameter broadcast, deterministic bucketed all-reduce, and
equivalence with single-process large-batch training."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from alphafold2_amd.models.evoformer import FeedForward

WORLD = 2


def _free_port():
    """OS-assigned free port (fixed ports collide with TIME_WAIT when
    suites run back-to-back)."""
    import socket
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        return s.getsockname()[1]


def _setup(rank, world_size, port):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world_size)
    os.environ['RANK'] = str(rank)
    dist.init_process_group('gloo', rank=rank, world_size=world_size)


def _ddp_worker(rank, port, q):
    from alphafold2_amd.parallel import DataParallelEngine
    _setup(rank, WORLD, port)
    torch.manual_seed(100 + rank)  # intentionally different init per rank
    model = FeedForward(dim=16)
    engine = DataParallelEngine(model, bucket_cap_mb=0.0001)  # many buckets

    # after broadcast all ranks hold rank-0 weights
    psum = sum(p.sum().item() for p in model.parameters())

    torch.manual_seed(rank)  # different data per rank
    x = torch.randn(4, 16)
    out = model(x)
    out.pow(2).sum().backward()
    engine.finalize()

    grads = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
    q.put((rank, psum, grads.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_grad_allreduce():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_ddp_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, psum, grads = q.get()
        results[rank] = (psum, grads)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    # parameter broadcast: identical initial weights
    assert abs(results[0][0] - results[1][0]) < 1e-6

    # both ranks end with identical (averaged) gradients
    g0 = torch.as_tensor(results[0][1])
    g1 = torch.as_tensor(results[1][1])
    assert torch.allclose(g0, g1, atol=1e-6)

    # and they equal the single-process average of per-rank gradients
    torch.manual_seed(100)
    model = FeedForward(dim=16)
    expected = None
    for rank in range(WORLD):
        model.zero_grad()
        torch.manual_seed(rank)
        x = torch.randn(4, 16)
        model(x).pow(2).sum().backward()
        g = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
        expected = g if expected is None else expected + g
    expected = expected / WORLD
    assert torch.allclose(g0, expected, atol=1e-5), \
        (g0 - expected).abs().max()


def _no_sync_worker(rank, port, q):
    from alphafold2_amd.parallel import DataParallelEngine
    _setup(rank, WORLD, port)
    torch.manual_seed(7)
    model = FeedForward(dim=8)
    engine = DataParallelEngine(model, bucket_cap_mb=64)

    torch.manual_seed(rank * 13)
    x1 = torch.randn(2, 8)
    x2 = torch.randn(2, 8)
    with engine.no_sync():
        model(x1).pow(2).sum().backward()
    model(x2).pow(2).sum().backward()
    engine.finalize()
    grads = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
    q.put((rank, grads.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_no_sync_accumulation():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_no_sync_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, grads = q.get()
        results[rank] = grads
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert torch.allclose(torch.as_tensor(results[0]),
                          torch.as_tensor(results[1]), atol=1e-6)


class _TwoPath(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(8, 8)
        self.b = torch.nn.Linear(8, 8)

    def forward(self, x, use_b=True):
        out = self.a(x)
        if use_b:
            out = out + self.b(x)
        return out


def _unused_param_worker(rank, port, q):
    """A param that gets its grad in a no_sync micro-batch but is NOT
    used in the final (sync) micro-batch must still be reduced."""
    from alphafold2_amd.parallel import DataParallelEngine
    _setup(rank, WORLD, port)
    torch.manual_seed(5)
    model = _TwoPath()
    engine = DataParallelEngine(model, bucket_cap_mb=64)

    torch.manual_seed(rank * 31)
    x1 = torch.randn(2, 8)
    x2 = torch.randn(2, 8)
    with engine.no_sync():
        model(x1, use_b=True).pow(2).sum().backward()
    model(x2, use_b=False).pow(2).sum().backward()  # b unused here
    engine.finalize()
    grads = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
    q.put((rank, x1.numpy(), x2.numpy(), grads.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_unused_param_in_sync_micro():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_unused_param_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, x1, x2, grads = q.get()
        results[rank] = (x1, x2, grads)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    assert torch.allclose(torch.as_tensor(results[0][2]),
                          torch.as_tensor(results[1][2]), atol=1e-6)

    # equals the all-rank average of locally accumulated gradients
    torch.manual_seed(5)
    model = _TwoPath()
    expected = None
    for rank in range(WORLD):
        model.zero_grad()
        x1 = torch.as_tensor(results[rank][0])
        x2 = torch.as_tensor(results[rank][1])
        model(x1, use_b=True).pow(2).sum().backward()
        model(x2, use_b=False).pow(2).sum().backward()
        g = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
        expected = g if expected is None else expected + g
    expected = expected / WORLD
    g0 = torch.as_tensor(results[0][2])
    assert torch.allclose(g0, expected, atol=1e-5), \
        (g0 - expected).abs().max()


def _model_worker(rank, port, q):
    """Full Alphafold2 step under DDP: template/extra-MSA modules receive
    no grads (exercises partial-bucket reduction on the real model)."""
    from alphafold2_amd import Alphafold2
    from alphafold2_amd.parallel import DataParallelEngine
    import torch.nn.functional as F
    _setup(rank, WORLD, port)
    torch.manual_seed(42 + rank)
    model = Alphafold2(dim=32, depth=1, heads=2, dim_head=16)
    model.train()
    engine = DataParallelEngine(model, bucket_cap_mb=0.5)

    torch.manual_seed(7 * (rank + 1))
    seq = torch.randint(0, 21, (1, 12))
    msa = torch.randint(0, 21, (1, 3, 12))
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask)
    loss = ret.distance.pow(2).mean() + ret.msa_mlm_loss
    loss.backward()
    engine.finalize()

    gsum = torch.cat([p.grad.reshape(-1) for p in model.parameters()
                      if p.grad is not None])
    # pickle by value (numpy) — torch tensors ride shared memory that
    # vanishes when the child exits before the parent reads the queue
    q.put((rank, gsum.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_full_model_step():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_model_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, gsum = q.get()
        results[rank] = gsum
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    # averaged grads identical on both ranks
    g0 = torch.as_tensor(results[0])
    g1 = torch.as_tensor(results[1])
    assert g0.shape == g1.shape
    assert torch.allclose(g0, g1, atol=1e-6)


def _never_used_worker(rank, port, q):
    """With skip_unused_buckets=True (uniform-graph contract), a bucket
    whose params never produce grads must skip its all-reduce (flat
    never allocated) while used buckets still average."""
    from alphafold2_amd.parallel import DataParallelEngine
    _setup(rank, WORLD, port)
    torch.manual_seed(3)
    model = _TwoPath()
    engine = DataParallelEngine(model, bucket_cap_mb=0.0001,  # per-param
                                skip_unused_buckets=True)

    torch.manual_seed(rank)
    x = torch.randn(2, 8)
    for _ in range(2):
        model.zero_grad(set_to_none=True)
        model(x, use_b=False).pow(2).sum().backward()
        engine.finalize()
    b_buckets = [engine._param_bucket[p] for p in model.b.parameters()]
    q.put((rank,
           all(bk.flat is None for bk in b_buckets),
           model.a.weight.grad.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_never_used_bucket_skipped():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_never_used_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, b_skipped, a_grad = q.get()
        results[rank] = (b_skipped, a_grad)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    for r in range(WORLD):
        assert results[r][0], 'unused bucket must never allocate/reduce'
    assert torch.allclose(torch.as_tensor(results[0][1]),
                          torch.as_tensor(results[1][1]), atol=1e-6)


def _divergent_graph_worker(rank, port, q):
    """Ranks running DIFFERENT graphs in the same step (rank 0 uses the
    `b` branch, rank 1 does not — the realistic per-rank-data case from
    the round-1 advisory).  Default engine settings must neither hang
    nor mix buckets: the non-user contributes zeros and both ranks end
    with the same average."""
    from alphafold2_amd.parallel import DataParallelEngine
    _setup(rank, WORLD, port)
    torch.manual_seed(11)
    model = _TwoPath()
    engine = DataParallelEngine(model, bucket_cap_mb=0.0001)  # per-param

    torch.manual_seed(rank * 17)
    x = torch.randn(2, 8)
    model(x, use_b=(rank == 0)).pow(2).sum().backward()
    engine.finalize()
    grads = {n: (p.grad.numpy().copy() if p.grad is not None else None)
             for n, p in model.named_parameters()}
    q.put((rank, x.numpy(), grads))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_divergent_graphs_across_ranks():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_divergent_graph_worker, args=(r, port, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, x, grads = q.get()
        results[rank] = (x, grads)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    # expected: average over ranks, zeros where a rank skipped the branch
    torch.manual_seed(11)
    model = _TwoPath()
    expected = {n: torch.zeros_like(p) for n, p in model.named_parameters()}
    for rank in range(WORLD):
        model.zero_grad()
        x = torch.as_tensor(results[rank][0])
        model(x, use_b=(rank == 0)).pow(2).sum().backward()
        for n, p in model.named_parameters():
            if p.grad is not None:
                expected[n] += p.grad / WORLD

    for rank in range(WORLD):
        for n, g in results[rank][1].items():
            if g is None:
                continue  # rank never produced a local grad for n
            assert torch.allclose(torch.as_tensor(g), expected[n],
                                  atol=1e-6), (rank, n)
