"""Multi-process data-parallel fit tests (gloo backend, world_size=2) —
the CPU-side contract of the RCCL paths (single-node
multi-rank collective tests, no multi-node mocking needed)."""

import os
import pickle

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from traffic_classifier_sdn_amd.utils.datasets import synthetic_flow_rows

WORLD = 2


def _run_ranks(fn, world=WORLD, args=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = []
    port = 29000 + (os.getpid() % 500)
    for r in range(world):
        p = ctx.Process(target=_worker, args=(fn, r, world, port, q, args))
        p.start()
        procs.append(p)
    results = {}
    for _ in range(world):
        r, val = q.get(timeout=600)
        if isinstance(val, str) and val.startswith("ERROR"):
            for p in procs:
                p.terminate()
            raise RuntimeError(val)
        results[r] = val
    for p in procs:
        p.join(timeout=60)
    return results


def _worker(fn, rank, world, port, q, args):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as td

        td.init_process_group(backend="gloo", rank=rank, world_size=world)
        out = fn(rank, world, *args)
        q.put((rank, out))
        td.barrier()
        td.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"ERROR rank{rank}: {e}\n{traceback.format_exc()}"))


# ---- worker fns (module-level for spawn pickling) ---------------------


def _allreduce_flat_worker(rank, world):
    from traffic_classifier_sdn_amd.parallel import dist

    a = torch.full((3,), float(rank + 1))
    b = torch.full((2, 2), float(10 * (rank + 1)))
    dist.allreduce_flat([a, b])
    return a.numpy(), b.numpy()


def _gnb_sharded_worker(rank, world):
    from traffic_classifier_sdn_amd.models import GaussianNB
    from traffic_classifier_sdn_amd.parallel.dist import shard_range

    X = synthetic_flow_rows(400, seed=7).astype(np.float64)
    y = np.asarray(["a", "b", "c", "d"] * 100, dtype=object)
    lo, hi = shard_range(len(X), rank, world)
    m = GaussianNB(device="cpu").fit(X[lo:hi], y[lo:hi], sharded=True)
    return m.theta_.numpy(), m.var_.numpy(), m.predict(X[:50])


def _kmeans_sharded_worker(rank, world):
    from traffic_classifier_sdn_amd.models import KMeans
    from traffic_classifier_sdn_amd.parallel.dist import shard_range

    X = synthetic_flow_rows(600, seed=8).astype(np.float64)
    lo, hi = shard_range(len(X), rank, world)
    m = KMeans(n_clusters=4, n_init=2, seed=0, device="cpu").fit(X[lo:hi], sharded=True)
    return m.cluster_centers_.numpy(), m.inertia_


def _knn_sharded_worker(rank, world):
    from traffic_classifier_sdn_amd.models import KNeighborsClassifier
    from traffic_classifier_sdn_amd.parallel.dist import shard_range

    R = synthetic_flow_rows(500, seed=9).astype(np.float64)
    y = np.asarray(["x", "y"] * 250, dtype=object)
    Q = synthetic_flow_rows(40, seed=10).astype(np.float64)
    lo, hi = shard_range(len(R), rank, world)
    m = KNeighborsClassifier(device="cpu").fit(R[lo:hi], y[lo:hi], sharded=True)
    return m.predict(Q)


def _lr_sharded_worker(rank, world):
    from traffic_classifier_sdn_amd.models import LogisticRegression
    from traffic_classifier_sdn_amd.parallel.dist import shard_range

    rng = np.random.default_rng(11)
    X = rng.normal(size=(300, 12))
    y = np.asarray(["p", "q", "r"])[rng.integers(0, 3, 300)]
    lo, hi = shard_range(len(X), rank, world)
    m = LogisticRegression(device="cpu").fit(X[lo:hi], y[lo:hi], sharded=True)
    return m.coef_.numpy(), m.intercept_.numpy()


def _rf_treeparallel_worker(rank, world):
    from traffic_classifier_sdn_amd.models import RandomForestClassifier

    X = synthetic_flow_rows(300, seed=12).astype(np.float64)
    y = np.asarray(["u", "v"] * 150, dtype=object)
    m = RandomForestClassifier(n_estimators=8, seed=0, device="cpu").fit(X, y)
    return len(m.trees_), m.predict(X[:20])


# ---- tests ------------------------------------------------------------


def test_allreduce_flat():
    res = _run_ranks(_allreduce_flat_worker)
    for r in range(WORLD):
        a, b = res[r]
        np.testing.assert_allclose(a, np.full(3, 3.0))  # 1+2
        np.testing.assert_allclose(b, np.full((2, 2), 30.0))


def test_gnb_sharded_fit_equals_full():
    from traffic_classifier_sdn_amd.models import GaussianNB

    res = _run_ranks(_gnb_sharded_worker)
    X = synthetic_flow_rows(400, seed=7).astype(np.float64)
    y = np.asarray(["a", "b", "c", "d"] * 100, dtype=object)
    full = GaussianNB(device="cpu").fit(X, y)
    for r in range(WORLD):
        theta, var, pred = res[r]
        np.testing.assert_allclose(theta, full.theta_.numpy(), rtol=1e-8)
        np.testing.assert_allclose(var, full.var_.numpy(), rtol=1e-6)
        np.testing.assert_array_equal(pred, full.predict(X[:50]))


def test_kmeans_sharded_fit():
    res = _run_ranks(_kmeans_sharded_worker)
    c0, i0 = res[0]
    c1, i1 = res[1]
    np.testing.assert_allclose(c0, c1)  # all ranks converge identically
    assert i0 == pytest.approx(i1)


def test_knn_sharded_predict_equals_full():
    from traffic_classifier_sdn_amd.models import KNeighborsClassifier

    res = _run_ranks(_knn_sharded_worker)
    R = synthetic_flow_rows(500, seed=9).astype(np.float64)
    y = np.asarray(["x", "y"] * 250, dtype=object)
    Q = synthetic_flow_rows(40, seed=10).astype(np.float64)
    full = KNeighborsClassifier(device="cpu").fit(R, y)
    expect = full.predict(Q)
    for r in range(WORLD):
        np.testing.assert_array_equal(res[r], expect)


def test_lr_sharded_fit_close_to_full():
    from traffic_classifier_sdn_amd.models import LogisticRegression

    res = _run_ranks(_lr_sharded_worker)
    rng = np.random.default_rng(11)
    X = rng.normal(size=(300, 12))
    y = np.asarray(["p", "q", "r"])[rng.integers(0, 3, 300)]
    full = LogisticRegression(device="cpu").fit(X, y)
    coef0, b0 = res[0]
    coef1, b1 = res[1]
    np.testing.assert_allclose(coef0, coef1, atol=1e-10)  # ranks agree
    np.testing.assert_allclose(coef0, full.coef_.numpy(), atol=2e-2)


def test_rf_tree_parallel_fit():
    res = _run_ranks(_rf_treeparallel_worker)
    n0, p0 = res[0]
    n1, p1 = res[1]
    assert n0 == n1 == 8  # every rank ends with the full forest
    np.testing.assert_array_equal(p0, p1)


def _svc_sharded_worker(rank, world):
    from traffic_classifier_sdn_amd.models import SVC
    from traffic_classifier_sdn_amd.parallel.dist import shard_range

    rng = np.random.default_rng(21)
    X = rng.normal(size=(240, 12)) * 3
    y = np.asarray(["m", "n", "o"])[rng.integers(0, 3, 240)]
    # make it learnable
    X[y == "m", 0] += 8
    X[y == "n", 1] += 8
    lo, hi = shard_range(len(X), rank, world)
    m = SVC(device="cpu", max_iter=5000).fit(X[lo:hi], y[lo:hi], sharded=True)
    return (
        m.support_vectors_.shape[0],
        m.intercept_.numpy(),
        m.predict(X[:40]),
    )


def test_svc_sharded_fit_matches_full():
    from traffic_classifier_sdn_amd.models import SVC

    res = _run_ranks(_svc_sharded_worker)
    rng = np.random.default_rng(21)
    X = rng.normal(size=(240, 12)) * 3
    y = np.asarray(["m", "n", "o"])[rng.integers(0, 3, 240)]
    X[y == "m", 0] += 8
    X[y == "n", 1] += 8
    full = SVC(device="cpu", max_iter=5000).fit(X, y)
    nsv0, b0, p0 = res[0]
    nsv1, b1, p1 = res[1]
    np.testing.assert_array_equal(p0, p1)  # ranks agree exactly
    np.testing.assert_allclose(b0, b1)
    # sharded solution is the SAME optimisation problem: predictions match
    # the single-process fit almost everywhere
    assert (p0 == full.predict(X[:40])).mean() > 0.9
    acc_sh = (p0 == y[:40]).mean()
    acc_full = (full.predict(X[:40]) == y[:40]).mean()
    assert acc_sh >= acc_full - 0.1


def test_knn_sharded_world4():
    """Sharded KNN merge at world_size=4 (uneven shards)."""
    res = _run_ranks(_knn_sharded_worker4, world=4)
    ref = res[0]["full"]
    for r in range(4):
        np.testing.assert_array_equal(res[r]["sharded"], ref)


def _knn_sharded_worker4(rank, world):
    from traffic_classifier_sdn_amd.models import KNeighborsClassifier
    from traffic_classifier_sdn_amd.parallel import dist as d

    X = synthetic_flow_rows(4003, seed=1)  # uneven split across 4 ranks
    y = np.random.default_rng(0).integers(0, 6, 4003)
    lo, hi = d.shard_range(len(X))
    m = KNeighborsClassifier(n_neighbors=5)
    m.fit(X[lo:hi], y[lo:hi], sharded=True)
    Q = synthetic_flow_rows(257, seed=2)
    out = {"sharded": m.predict(Q), "full": None}
    if rank == 0:  # unsharded reference on the full rows
        f = KNeighborsClassifier(n_neighbors=5)
        f.fit(X, y, sharded=False)
        out["full"] = f.predict(Q)
    return out
