"""RCCL-on-hardware tests (every collective path had
only ever run under gloo on CPU — the NCCL(=RCCL) backend itself was never
initialised on a GPU).  These exercise torch.distributed backend "nccl" on
gfx950 with device tensors through every helper in parallel/dist.py.

World size 1 is a real RCCL communicator (ncclCommInitRank with one rank):
the collectives run the RCCL code path, device buffers, stream semantics —
exactly what a multi-rank run does per rank, minus the xGMI traffic, so it
kills the "designed but never executed on hardware" risk before the
driver's 8-GPU scaling run."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("MI355X required", allow_module_level=True)


@pytest.fixture(scope="module")
def nccl_group():
    import torch.distributed as td

    from traffic_classifier_sdn_amd.parallel import dist

    assert not td.is_initialized()
    torch.cuda.set_device(0)
    td.init_process_group(
        backend="nccl",
        init_method="tcp://127.0.0.1:29731",
        rank=0,
        world_size=1,
    )
    assert td.get_backend() == "nccl"
    yield dist
    td.destroy_process_group()


def test_rccl_allreduce_device(nccl_group):
    d = nccl_group
    assert d.is_initialized() and d.world_size() == 1
    t = torch.arange(1024, dtype=torch.float32, device="cuda")
    d.allreduce_(t)
    torch.cuda.synchronize()
    assert torch.equal(t.cpu(), torch.arange(1024, dtype=torch.float32))


def test_rccl_allreduce_flat_device(nccl_group):
    d = nccl_group
    ts = [
        torch.full((6, 12), 2.0, device="cuda"),
        torch.full((6,), 3.0, device="cuda"),
        torch.arange(15, dtype=torch.float32, device="cuda"),
    ]
    d.allreduce_flat(ts)
    torch.cuda.synchronize()
    assert torch.equal(ts[0].cpu(), torch.full((6, 12), 2.0))
    assert torch.equal(ts[2].cpu(), torch.arange(15, dtype=torch.float32))


def test_rccl_allgather_broadcast_barrier(nccl_group):
    d = nccl_group
    t = torch.randn(128, 5, device="cuda")
    out = d.allgather(t)
    torch.cuda.synchronize()
    assert len(out) == 1 and torch.equal(out[0], t)
    b = torch.randn(64, device="cuda")
    d.broadcast_(b, src=0)
    d.barrier()
    torch.cuda.synchronize()


def test_rccl_sharded_knn_predict(nccl_group):
    """The sharded KNN merge path (all-gather of per-shard top-k) on the
    RCCL backend with device tensors — config #4's collective."""
    from traffic_classifier_sdn_amd.models import KNeighborsClassifier

    rng = np.random.default_rng(0)
    X = rng.normal(size=(4096, 12))
    y = rng.integers(0, 6, size=4096)
    m = KNeighborsClassifier(n_neighbors=5, device="cuda")
    m.fit(X, y, sharded=True)  # world 1: shard == full set, merge path runs
    Q = torch.from_numpy(rng.normal(size=(512, 12))).float().cuda()
    got = m.predict_index(Q)
    m2 = KNeighborsClassifier(n_neighbors=5, device="cuda").fit(X, y)
    want = m2.predict_index(Q)
    assert torch.equal(got.cpu(), want.cpu())


def test_rccl_dp_logistic_grad_allreduce(nccl_group):
    """One data-parallel LR fit step with the gradient all-reduce on RCCL
    (config #3's collective pattern: fused flat gradient buffer)."""
    from traffic_classifier_sdn_amd import ops

    rng = np.random.default_rng(1)
    X = torch.from_numpy(rng.normal(size=(2048, 12))).double().cuda()
    yv = torch.from_numpy(rng.integers(0, 6, size=2048)).cuda()
    coef = torch.zeros(6, 12, dtype=torch.float64, device="cuda")
    b = torch.zeros(6, dtype=torch.float64, device="cuda")
    loss, g_coef, g_b = ops.logistic_loss_grad(X, yv, coef, b)
    nccl_group.allreduce_flat([g_coef, g_b])
    torch.cuda.synchronize()
    assert torch.isfinite(loss).all()
    assert torch.isfinite(g_coef).all() and g_coef.abs().sum() > 0
