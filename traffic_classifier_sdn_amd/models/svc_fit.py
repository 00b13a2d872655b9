"""Scalable RBF-SVC dual fit: libsvm WSS-1 SMO with on-the-fly kernel rows.

No kernel matrix is ever materialised — each iteration selects the maximal
violating pair, solves the 2-variable subproblem analytically, and updates
the dual gradient with ONE fused kernel pass over the local row shard
(csrc smo_update: two RBF rows + gradient update in one sweep).  Rows are
sharded across ranks (BASELINE config #3: 1M rows, 1/2/4/8 MI355X); per
iteration the ranks exchange one small all-gather of pair candidates
(value, row payload) — latency-bound, so everything is packed into a
single collective.

Device-generic: CUDA path uses the HIP kernels, CPU path the torch ops —
identical control flow, so the multi-process logic is testable on gloo.
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from ..parallel import dist


def _local_select_cpu(y, alpha, grad, C):
    myg = -(y.double() * grad)
    up = ((y > 0) & (alpha < C)) | ((y < 0) & (alpha > 0))
    low = ((y > 0) & (alpha > 0)) | ((y < 0) & (alpha < C))
    ninf = torch.tensor(-math.inf, dtype=torch.float64)
    pinf = torch.tensor(math.inf, dtype=torch.float64)
    up_vals = torch.where(up, myg, ninf)
    low_vals = torch.where(low, myg, pinf)
    i = int(torch.argmax(up_vals))
    j = int(torch.argmin(low_vals))
    return i, float(up_vals[i]), j, float(low_vals[j])


def _local_select_gpu(y, alpha, grad, C, _buf):
    from ..ops import gpu as og

    _buf.zero_()
    og._ext.smo_select(y, alpha, grad, float(C), _buf)
    packed = _buf.cpu().numpy().view(np.uint64)
    i = int(packed[0] & 0xFFFFFFFF)
    j = int(packed[1] & 0xFFFFFFFF)
    if packed[0] == 0:
        i = -1
    if packed[1] == 0:
        j = -1
    up_val = float(-(float(y[i]) * float(grad[i]))) if i >= 0 else -math.inf
    low_val = float(-(float(y[j]) * float(grad[j]))) if j >= 0 else math.inf
    return i, up_val, j, low_val


def _grad_update_cpu(X, y, grad, xi, xj, yidai, yjdaj, gamma):
    di = ((X - xi) ** 2).sum(dim=1)
    dj = ((X - xj) ** 2).sum(dim=1)
    ki = torch.exp(-gamma * di).double()
    kj = torch.exp(-gamma * dj).double()
    grad += y.double() * (yidai * ki + yjdaj * kj)


def _reconstruct_grad(X, y, alpha, grad, gamma):
    """Rebuild grad_i = y_i·Σ_j α_j y_j K(i,j) − 1 from scratch (f64).

    The fused iteration updates the gradient incrementally with two f32
    kernel rows per step; after tens of thousands of rank-1 updates the
    accumulated rounding drift corrupts the working-set selection (measured:
    a 60K-iteration non-separable pair dragged 6-class held-out accuracy
    0.998 → 0.761, profiles/svc_grad_reconstruct_r02.md).  Periodic full
    reconstruction — libsvm does the same when its cache makes it cheap —
    bounds the drift.  Cost: one chunked f64 RBF block per SV set, ~10² ms
    at 333K rows × tens of thousands of SVs, amortised over thousands of
    iterations."""
    sv = alpha > 1e-12
    g = torch.full_like(grad, -1.0)
    m = int(sv.sum())
    if m > 0:
        Xs = X[sv].double()
        coef = (alpha[sv] * y[sv].double())
        xs_sq = (Xs * Xs).sum(1)
        n = X.shape[0]
        step = max(1, 48_000_000 // m)  # ≤ ~384 MB per f64 kernel block
        y64 = y.double()
        for lo in range(0, n, step):
            hi = min(n, lo + step)
            Xb = X[lo:hi].double()
            d2 = (Xb * Xb).sum(1, keepdim=True) + xs_sq.unsqueeze(0) - 2.0 * (Xb @ Xs.T)
            Kb = torch.exp(-gamma * d2.clamp_min_(0))
            g[lo:hi] += (Kb @ coef) * y64[lo:hi]
    grad.copy_(g)


def _smo_fused_gpu(X, y, alpha, grad, C, gamma, tol, max_iter, chunk=128,
                   recompute_every=8192, wss2=False):
    """Single-GPU fast path: the whole select→solve→update iteration runs on
    device (csrc smo_solve updates alpha in place and re-arms the select
    buffer), so the host only polls the convergence status once per
    ``chunk`` iterations instead of 5 round-trips per iteration.

    ``wss2=True`` switches to libsvm's second-order working-set selection
    (smo_row / smo_select2 / smo_solve2 / smo_update_dev2): i's kernel row
    is computed once, drives the second-order j choice, and serves as the
    i-half of the gradient update.  MEASURED OFF by default: on flow-stat
    rows with gamma='scale' the RBF kernel values sit near 1 (tiny gamma),
    the curvature a_t = 2 − 2 y_i y_t K_it degenerates to ~tau for
    same-class candidates, and WSS-2 cut iterations only ~11% while the two
    extra n-row passes cost ~35%/iteration — a net loss
    (profiles/svc_wss2_r02.md).  The path stays built, tested and available
    for kernels with real curvature spread."""
    from ..ops import gpu as og

    device = X.device
    sel = torch.zeros(3, dtype=torch.int64, device=device)
    rows = torch.zeros(24, dtype=torch.float32, device=device)
    sol = torch.zeros(4, dtype=torch.float64, device=device)
    krow = (
        torch.zeros(X.shape[0], dtype=torch.float32, device=device)
        if wss2 else None
    )

    def one_iter():
        if wss2:
            og._ext.smo_select(y, alpha, grad, float(C), sel[:2])
            og._ext.smo_row(X, sel, sol, krow, float(gamma))
            og._ext.smo_select2(y, alpha, grad, krow, sel, sol, float(C))
            og._ext.smo_solve2(X, y, alpha, grad, sel, rows, sol, float(C), float(tol), float(gamma))
            og._ext.smo_update_dev2(X, y, grad, rows, sol, krow, float(gamma))
            return
        og._ext.smo_select(y, alpha, grad, float(C), sel[:2])
        og._ext.smo_solve(X, y, alpha, grad, sel[:2], rows, sol, float(C), float(tol), float(gamma))
        og._ext.smo_update_dev(X, y, grad, rows, sol, float(gamma))

    # capture `chunk` iterations in one hipGraph: every kernel argument is a
    # device buffer, so the captured sequence is exact; a converged solve
    # turns the remaining replayed iterations into no-ops (status latch)
    graph = None
    if max_iter < 8 * chunk:
        # capture cost (~chunk x 3 launches) only amortises on long fits
        it = 0
        while it < max_iter:
            for _ in range(min(chunk, max_iter - it)):
                one_iter()
                it += 1
            if float(sol[2]) != 0.0:
                break
        return it
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            one_iter()  # warmup outside capture
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            for _ in range(chunk):
                one_iter()
        it = 1  # the warmup iteration is real work
    except Exception:
        graph = None
        it = 0

    since_recompute = it
    while it < max_iter:
        if graph is not None and (max_iter - it) >= chunk:
            graph.replay()
            it += chunk
            since_recompute += chunk
        else:
            for _ in range(min(chunk, max_iter - it)):
                one_iter()
                it += 1
                since_recompute += 1
        if float(sol[2]) != 0.0:  # one D2H per chunk
            break
        if since_recompute >= recompute_every:
            _reconstruct_grad(X, y, alpha, grad, gamma)
            since_recompute = 0
    if it >= recompute_every // 2:
        # fresh gradient for the intercept/duality-gap readout (short fits
        # skip it so small-shape parity vs the CPU oracle stays bit-stable)
        _reconstruct_grad(X, y, alpha, grad, gamma)
    return it


def smo_fit_pair(
    X: torch.Tensor,
    y_pm: torch.Tensor,
    C: float = 1.0,
    gamma: float = 1.0,
    tol: float = 1e-3,
    max_iter: int = 200_000,
    check_every: int = 1,
) -> Tuple[torch.Tensor, float, int]:
    """SMO on the local shard (X[n_local,12] f32, y_pm in {-1,+1} f32).

    Returns (alpha_local f64, intercept b, iterations).  With an initialised
    process group the pair search is global (one fused all-gather/iter).
    """
    device = X.device
    n = X.shape[0]
    y = y_pm.to(torch.float32)
    alpha = torch.zeros(n, dtype=torch.float64, device=device)
    grad = -torch.ones(n, dtype=torch.float64, device=device)
    is_gpu = device.type == "cuda"
    sel_buf = torch.zeros(2, dtype=torch.int64, device=device) if is_gpu else None
    rows_buf = torch.zeros(24, dtype=torch.float32, device=device) if is_gpu else None
    world = dist.world_size()
    it = 0
    if is_gpu and world == 1:
        it = _smo_fused_gpu(X.contiguous(), y.contiguous(), alpha, grad, C, gamma, tol, max_iter)
        return alpha, _smo_intercept(y, alpha, grad, C, device), it
    for it in range(1, max_iter + 1):
        if is_gpu:
            i, up_val, j, low_val = _local_select_gpu(y, alpha, grad, C, sel_buf)
        else:
            i, up_val, j, low_val = _local_select_cpu(y, alpha, grad, C)

        # candidate payload: [up_val, low_val, xi(12), yi, ai, gi, xj(12), yj, aj, gj]
        def payload(idx, val):
            if idx < 0:
                return [val] + [0.0] * 15
            return (
                [val]
                + X[idx].double().tolist()
                + [float(y[idx]), float(alpha[idx]), float(grad[idx])]
            )

        if world > 1:
            local = torch.tensor(
                payload(i, up_val) + payload(j, low_val), dtype=torch.float64,
                device=device,  # NCCL collectives reject CPU tensors
            )
            gathered = dist.allgather(local)
            up_rank = int(np.argmax([float(g[0]) for g in gathered]))
            low_rank = int(np.argmin([float(g[16]) for g in gathered]))
            gu = gathered[up_rank]
            gl = gathered[low_rank]
            up_val = float(gu[0])
            low_val = float(gl[16])
            xi = gu[1:13]
            yi, ai, gi = float(gu[13]), float(gu[14]), float(gu[15])
            xj = gl[17:29]
            yj, aj, gj = float(gl[29]), float(gl[30]), float(gl[31])
            i_here = i if up_rank == dist.rank() else -1
            j_here = j if low_rank == dist.rank() else -1
        else:
            if i < 0 or j < 0:
                break
            xi = X[i].double().cpu()
            xj = X[j].double().cpu()
            yi, ai, gi = float(y[i]), float(alpha[i]), float(grad[i])
            yj, aj, gj = float(y[j]), float(alpha[j]), float(grad[j])
            i_here, j_here = i, j

        if up_val - low_val < tol:
            break

        # analytic 2-variable solve (K_ii = K_jj = 1 for RBF)
        d2 = float(((xi - xj) ** 2).sum())
        kij = math.exp(-gamma * d2)
        a = 2.0 - 2.0 * yi * yj * kij
        if a <= 0:
            a = 1e-12
        d = (up_val - low_val) / a
        ai_new = ai + yi * d
        aj_new = aj - yj * d
        s = yi * ai + yj * aj
        ai_new = min(max(ai_new, 0.0), C)
        aj_new = yj * (s - yi * ai_new)
        aj_new = min(max(aj_new, 0.0), C)
        ai_new = yi * (s - yj * aj_new)
        ai_new = min(max(ai_new, 0.0), C)
        dai, daj = ai_new - ai, aj_new - aj
        if abs(dai) < 1e-16 and abs(daj) < 1e-16:
            break
        if i_here >= 0:
            alpha[i_here] = ai_new
        if j_here >= 0:
            alpha[j_here] = aj_new

        xif = torch.as_tensor(xi, dtype=torch.float32)
        xjf = torch.as_tensor(xj, dtype=torch.float32)
        if is_gpu:
            from ..ops import gpu as og

            rows_buf[:12] = xif.to(device)
            rows_buf[12:] = xjf.to(device)
            og._ext.smo_update(X, y, grad, rows_buf, yi * dai, yj * daj, gamma)
        else:
            _grad_update_cpu(X, y, grad, xif, xjf, yi * dai, yj * daj, gamma)

    return alpha, _smo_intercept(y, alpha, grad, C, device), it


def _smo_intercept(y, alpha, grad, C, device) -> float:
    """Intercept from free vectors (global across ranks)."""
    myg = -(y.double() * grad)
    free = (alpha > 1e-12) & (alpha < C - 1e-12)
    ssum = torch.tensor([float(myg[free].sum()), float(free.sum())],
                        dtype=torch.float64, device=device)
    dist.allreduce_(ssum)
    if float(ssum[1]) > 0:
        return float(ssum[0] / ssum[1])
    up = ((y > 0) & (alpha < C)) | ((y < 0) & (alpha > 0))
    low = ((y > 0) & (alpha > 0)) | ((y < 0) & (alpha < C))
    hi = torch.tensor(
        [float(torch.where(up, myg, torch.tensor(-math.inf, dtype=torch.float64, device=device)).max())],
        device=device,
    )
    lo = torch.tensor(
        [float(torch.where(low, myg, torch.tensor(math.inf, dtype=torch.float64, device=device)).min())],
        device=device,
    )
    dist.allreduce_(hi, op=torch.distributed.ReduceOp.MAX if dist.is_initialized() else None)
    dist.allreduce_(lo, op=torch.distributed.ReduceOp.MIN if dist.is_initialized() else None)
    return float((hi[0] + lo[0]) / 2.0)
