"""GPU random-forest fit: level-synchronous histogram tree builder.

The reference's RF fit runs sklearn's depth-first exact-gini Cython builder
(SURVEY.md §2.2 N4); the exact CPU equivalent lives in random_forest.py.
This module is the MI355X-native builder for large row counts: features are
quantised once to a 256-bin per-feature grid, and each tree grows
breadth-first — one fused HIP histogram scatter per level
(csrc rf_hist_kernel: hist[node][feature][bin][class] over all live rows),
then a fully vectorised split search (cumulative class counts -> weighted
gini) and a gather-based row partition, all device tensors.  Per-level node
bookkeeping is vectorised numpy; level histograms are processed in node
chunks so memory stays bounded on adversarial (incompressible-label) data.
Trees come out in the same dict format as the exact builder, renumbered to
depth-first preorder via per-level subtree-size computation so
left_child == index+1 holds for the packed traversal layout.

Split semantics: candidate thresholds are the quantile edges; with
``torch.bucketize(x, edges)`` (right=False), ``bin(x) <= b`` is EXACTLY
``x <= edges[b]`` (both f32), so serve-time traversal reproduces the
training-time partition bit-for-bit.  max_features=3 random features per
node, bootstrap resampling, min_samples_split=2, strict impurity decrease —
sklearn defaults; tree SHAPES differ from the exact builder (binned
thresholds) but accuracy parity is the test (published RF: 99.87%).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from .. import ops

import os as _os
import time as _time

_TIMING = _os.environ.get("TCSDN_RF_TIMING", "") not in ("", "0")
_tacc = {}


def _tick(name, t0, device):
    if not _TIMING:
        return 0.0
    if device is not None and str(device).startswith("cuda"):
        torch.cuda.synchronize()
    now = _time.perf_counter()
    _tacc[name] = _tacc.get(name, 0.0) + (now - t0)
    return now


N_BINS = 256
_CHUNK_NODES = 8192  # per rf_hist pass: 8192 * 73.7 KB ~ 600 MB of histogram (each extra chunk re-scans ALL rows, so size the chunk to hold a whole combined multi-tree frontier)


def quantize(X: torch.Tensor, max_sample: int = 262_144, seed: int = 0):
    """Per-feature quantile edges (255 interior cut points) + u8 bin codes."""
    n, F = X.shape
    Xs = X
    if n > max_sample:
        g = torch.Generator(device="cpu").manual_seed(seed)
        idx = torch.randint(0, n, (max_sample,), generator=g).to(X.device)
        Xs = X[idx]
    qs = torch.linspace(0, 1, N_BINS + 1, device=X.device, dtype=torch.float32)[1:-1]
    edges = torch.quantile(Xs.float(), qs, dim=0).T.contiguous()  # [F, 255]
    bins = torch.empty(n, F, dtype=torch.uint8, device=X.device)
    for f in range(F):
        bins[:, f] = torch.bucketize(X[:, f].float().contiguous(), edges[f]).to(torch.uint8)
    return bins, edges


def _level_split_search(B, Y, nid, L, C, mf, g, device):
    """Chunked histogram + split search over the (combined) frontier.

    Returns numpy arrays over the L frontier nodes:
    (counts [L,C], best_imp [L], best_f [L], best_b [L]).
    """
    F = B.shape[1]
    cnt_all = np.empty((L, C), dtype=np.float64)
    imp_all = np.empty(L, dtype=np.float64)
    f_all = np.empty(L, dtype=np.int64)
    b_all = np.empty(L, dtype=np.int64)
    tf = _time.perf_counter() if _TIMING else 0.0
    featsel = torch.argsort(torch.rand(L, F, generator=g), dim=1)[:, :mf]
    if _TIMING:
        _tick("ls_featsel", tf, None)
    if B.is_cuda:
        return _level_split_search_gpu(B, Y, nid, L, C, featsel, device)
    for lo in range(0, L, _CHUNK_NODES):
        hi = min(L, lo + _CHUNK_NODES)
        Lc = hi - lo
        if L <= _CHUNK_NODES:
            nidw = nid
        else:
            nidw = torch.where((nid >= lo) & (nid < hi), nid - lo, torch.full_like(nid, -1))
        hist = ops.rf_hist(B, Y, nidw, Lc, C).float()  # [Lc, F, 256, C]
        cnt_node = hist[:, 0].sum(dim=1)               # [Lc, C]
        n_node = cnt_node.sum(dim=1)
        cum = hist.cumsum(dim=2)
        del hist
        nl = cum.sum(dim=3)
        nr = n_node[:, None, None] - nl
        pl = cum / nl.clamp(min=1.0).unsqueeze(-1)
        gl = 1.0 - (pl * pl).sum(dim=3)
        del pl
        right = cnt_node[:, None, None, :] - cum
        del cum
        pr = right / nr.clamp(min=1.0).unsqueeze(-1)
        del right
        gr = 1.0 - (pr * pr).sum(dim=3)
        del pr
        imp = (nl * gl + nr * gr) / n_node[:, None, None].clamp(min=1.0)
        imp = torch.where((nl < 1) | (nr < 1), torch.full_like(imp, float("inf")), imp)
        del nl, nr, gl, gr
        fmask = torch.ones(Lc, F, dtype=torch.bool)
        fmask.scatter_(1, featsel[lo:hi], False)
        imp = imp.masked_fill(fmask.to(device).unsqueeze(-1), float("inf"))
        flat = imp.reshape(Lc, F * 256)
        best_imp, best_fb = flat.min(dim=1)
        del imp, flat
        cnt_all[lo:hi] = cnt_node.cpu().numpy()
        imp_all[lo:hi] = best_imp.cpu().numpy()
        f_all[lo:hi] = (best_fb // 256).cpu().numpy()
        b_all[lo:hi] = (best_fb % 256).cpu().numpy()
    return cnt_all, imp_all, f_all, b_all


def _dec_key(u):
    """Inverse of the kernel's order-preserving f32 encode."""
    hi = (u >> 32).astype(np.uint32)
    pos = hi >= 0x80000000
    bits = np.where(pos, hi & 0x7FFFFFFF, ~hi)
    return bits.astype(np.uint32).view(np.float32)


def _level_split_search_gpu(B, Y, nid, L, C, featsel, device):  # noqa: C901
    """Fused-kernel split search (csrc rf_split_kernel): one thread per
    (node, feature), packed u64 atomicMin per node — replaces the torch
    cumsum chain whose GB-scale intermediates dominated the level cost."""
    from ..ops import gpu as og

    t0 = _time.perf_counter() if _TIMING else 0.0
    F = B.shape[1]
    mf = featsel.shape[1]
    # frank[node, f] = slot of f among the node's mtry candidates (0xff:
    # not a candidate); fidx[node, slot] = real feature id of the slot —
    # together they drive the COMPACT [L, mf, 256, C] histogram whose
    # atomic working set stays Infinity-Cache-resident (the full 12-plane
    # buffer reached 600 MB and fell to HBM-latency atomics)
    frank = torch.full((L, F), 0xFF, dtype=torch.uint8)
    slots = torch.arange(mf, dtype=torch.uint8).expand(L, mf)
    frank.scatter_(1, featsel, slots)
    frank_dev = frank.to(device)
    fidx_dev = featsel.to(torch.uint8).contiguous().to(device)
    t0 = _tick("ls_fsel", t0, device)
    cnt_all = np.empty((L, C), dtype=np.float64)
    imp_all = np.empty(L, dtype=np.float64)
    f_all = np.empty(L, dtype=np.int64)
    b_all = np.empty(L, dtype=np.int64)
    for lo in range(0, L, _CHUNK_NODES):
        hi = min(L, lo + _CHUNK_NODES)
        Lc = hi - lo
        if L <= _CHUNK_NODES:
            nidw = nid
        else:
            nidw = torch.where((nid >= lo) & (nid < hi), nid - lo, torch.full_like(nid, -1))
        best, cnt = og._ext.rf_level_compact(
            B, Y, nidw, frank_dev[lo:hi].contiguous(),
            fidx_dev[lo:hi].contiguous(), Lc, C,
        )
        t0 = _tick("ls_hist", t0, device)
        u = best.cpu().numpy().view(np.uint64)
        valid = u != np.uint64(0xFFFFFFFFFFFFFFFF)
        imp = np.where(valid, _dec_key(u).astype(np.float64), np.inf)
        imp_all[lo:hi] = imp
        f_all[lo:hi] = np.where(valid, (u >> np.uint64(16)) & np.uint64(0xFFFF), 0).astype(np.int64)
        b_all[lo:hi] = np.where(valid, u & np.uint64(0xFFFF), 0).astype(np.int64)
        cnt_all[lo:hi] = cnt.cpu().numpy()
        t0 = _tick("ls_decode", t0, device)
    return cnt_all, imp_all, f_all, b_all


def build_forest_hist(
    X: torch.Tensor,
    y_idx: torch.Tensor,
    n_classes: int,
    n_estimators: int,
    max_features: int = 3,
    seed: Optional[int] = 0,
    bootstrap: bool = True,
    max_depth: int = 40,
    min_samples_split: int = 2,
    tree_range=None,
) -> List[Dict[str, np.ndarray]]:
    """Grow ALL trees of the (local) forest level-synchronously.

    Round-2 rewrite: the first version built trees one at a time — 25 trees
    x ~12 levels = ~300 sequential {hist kernel + split search + partition}
    iterations, so the fit was bound by per-level Python/launch/D2H glue,
    not the scatter kernel (9% of step time, profiles/pmc_counters_r02.md).
    Now every tree's frontier lives in ONE combined node numbering: each
    level runs ONE rf_hist scatter over the T*n bootstrap rows, one
    vectorised split search, and one partition pass — ~12 iterations total
    for any tree count.  The combined frontier stays sorted by tree (stable
    child emission), so per-tree records fall out by masking at the end.
    """
    device = X.device
    _ts = _time.perf_counter() if _TIMING else 0.0
    n, F = X.shape
    seed0 = 0 if seed is None else seed
    bins, edges = quantize(X, seed=seed0)
    y8 = y_idx.to(torch.uint8).contiguous().to(device)
    edges_np = edges.cpu().numpy()
    tree_list = list(tree_range) if tree_range is not None else list(range(n_estimators))
    T = len(tree_list)
    if T == 0:
        return []

    # per-tree bootstrap resample (same per-tree generators/seeds as the
    # sequential builder so resamples are reproducible per tree id)
    if bootstrap:
        if str(device).startswith("cuda"):
            # generate the T*n resample indices on device (the CPU randint
            # path measured 100+ ms of the 290 ms step); per-tree generator
            # seeds keep resamples deterministic per tree id, so the
            # tree-parallel distributed fit stays rank-independent
            rows = torch.cat(
                [
                    torch.randint(
                        0, n, (n,), device=device,
                        generator=torch.Generator(device=device).manual_seed(
                            seed0 + 1000003 * t
                        ),
                    )
                    for t in tree_list
                ]
            )
        else:
            rows = torch.cat(
                [
                    torch.randint(
                        0, n, (n,),
                        generator=torch.Generator(device="cpu").manual_seed(
                            seed0 + 1000003 * t
                        ),
                    )
                    for t in tree_list
                ]
            ).to(device)
        Ball = bins[rows].contiguous()
        Yall = y8[rows].contiguous()
    else:
        Ball = bins.repeat(T, 1).contiguous()
        Yall = y8.repeat(T).contiguous()

    gsel = torch.Generator(device="cpu").manual_seed(seed0 * 7919 + 13)
    # combined-frontier-LOCAL node id per row; tree t's root is slot t
    nid = torch.repeat_interleave(
        torch.arange(T, dtype=torch.int32), n
    ).to(device)

    f_tree = np.arange(T, dtype=np.int64)       # tree of each frontier node
    f_gid = np.zeros(T, dtype=np.int64)         # per-tree global node id
    tot = np.ones(T, dtype=np.int64)            # per-tree node counts

    # per-level combined records (split per tree at the end)
    lvl_tree: List[np.ndarray] = []
    lvl_ids: List[np.ndarray] = []
    lvl_vals: List[np.ndarray] = []
    lvl_feat: List[np.ndarray] = []
    lvl_thr: List[np.ndarray] = []
    lvl_left: List[np.ndarray] = []
    lvl_right: List[np.ndarray] = []

    t0 = _tick("setup", _ts, device) if _TIMING else 0.0
    for depth in range(max_depth):
        L = len(f_tree)
        if L == 0:
            break
        cnt, best_imp, best_f, best_b = _level_split_search(
            Ball, Yall, nid, L, n_classes, max_features, gsel, device
        )
        t0 = _tick("split_search", t0, device)
        n_node = cnt.sum(axis=1)
        pp = cnt / np.maximum(n_node, 1.0)[:, None]
        parent_gini = 1.0 - (pp * pp).sum(axis=1)
        can = (
            (n_node >= min_samples_split)
            & np.isfinite(best_imp)
            & (best_imp < parent_gini - 1e-12)
            & (depth < max_depth - 1)
        )
        n_split = int(can.sum())
        feat = np.where(can, best_f, -1).astype(np.int64)
        thr = np.where(can, edges_np[np.minimum(best_f, F - 1), best_b], 0.0)
        # per-tree child global ids: frontier is tree-sorted, so the
        # within-tree split rank is positional
        ctree = f_tree[can]
        counts_t = np.bincount(ctree, minlength=T)
        starts_t = np.concatenate([[0], np.cumsum(counts_t)[:-1]])
        rank_t = np.arange(n_split, dtype=np.int64) - starts_t[ctree]
        kids = tot[ctree] + 2 * rank_t
        left = np.full(L, -1, dtype=np.int64)
        right = np.full(L, -1, dtype=np.int64)
        left[can] = kids
        right[can] = kids + 1
        lvl_tree.append(f_tree)
        lvl_ids.append(f_gid)
        lvl_vals.append(cnt)
        lvl_feat.append(feat)
        lvl_thr.append(thr)
        lvl_left.append(left)
        lvl_right.append(right)
        if n_split == 0:
            f_tree = np.array([], dtype=np.int64)
            break
        # row partition: combined local id -> new local id of the LEFT child
        lmap = np.full(L, -1, dtype=np.int64)
        lmap[can] = 2 * np.arange(n_split, dtype=np.int64)
        if Ball.is_cuda:
            # one fused pass (csrc rf_partition_kernel) instead of ~8 torch
            # index/gather/where kernels over the T*n rows
            from ..ops import gpu as og

            og._ext.rf_partition(
                Ball,
                nid,
                torch.from_numpy(lmap.astype(np.int32)).to(device),
                torch.from_numpy(np.maximum(feat, 0).astype(np.int32)).to(device),
                torch.from_numpy(best_b.astype(np.int32)).to(device),
            )
        else:
            lmap_t = torch.from_numpy(lmap).to(device)
            feat_t = torch.from_numpy(np.maximum(feat, 0)).to(device)
            bin_t = torch.from_numpy(best_b).to(device)
            live = nid >= 0
            nid_l = nid.long().clamp(min=0)
            splitting = live & (lmap_t[nid_l] >= 0)
            vals = Ball.gather(1, feat_t[nid_l].unsqueeze(1)).squeeze(1).long()
            go_left = vals <= bin_t[nid_l]
            child = lmap_t[nid_l] + torch.where(go_left, 0, 1)
            nid = torch.where(splitting, child.to(torch.int32), torch.full_like(nid, -1))
        # next frontier, combined-local order [l0, r0, l1, r1, ...]
        nxt_gid = np.empty(2 * n_split, dtype=np.int64)
        nxt_gid[0::2] = kids
        nxt_gid[1::2] = kids + 1
        nxt_tree = np.empty(2 * n_split, dtype=np.int64)
        nxt_tree[0::2] = ctree
        nxt_tree[1::2] = ctree
        f_gid = nxt_gid
        f_tree = nxt_tree
        tot += 2 * counts_t
        t0 = _tick("partition+bookkeeping", t0, device)

    # split the combined records per tree and flatten each
    t0 = _tick("levels_tail", t0, device) if _TIMING else 0.0
    trees: List[Dict[str, np.ndarray]] = []
    for t in range(T):
        ids_l, vals_l, feat_l, thr_l, left_l, right_l = [], [], [], [], [], []
        for d in range(len(lvl_ids)):
            m = lvl_tree[d] == t
            if not m.any():
                continue
            ids_l.append(lvl_ids[d][m])
            vals_l.append(lvl_vals[d][m])
            feat_l.append(lvl_feat[d][m])
            thr_l.append(lvl_thr[d][m])
            left_l.append(lvl_left[d][m])
            right_l.append(lvl_right[d][m])
        trees.append(
            _flatten_tree(int(tot[t]), n_classes, ids_l, vals_l, feat_l, thr_l, left_l, right_l)
        )
    if _TIMING:
        _tick("flatten", t0, device)
        import sys as _sys

        print("RF_TIMING " + " ".join(f"{k}={v*1e3:.1f}ms" for k, v in _tacc.items()),
              file=_sys.stderr, flush=True)
        _tacc.clear()
    return trees


def _flatten_tree(tot, C, lvl_ids, lvl_vals, lvl_feat, lvl_thr, lvl_left, lvl_right) -> Dict[str, np.ndarray]:
    """Per-level records -> id-indexed arrays -> depth-first preorder."""
    feature = np.full(tot, -1, dtype=np.int64)
    thr = np.zeros(tot, dtype=np.float64)
    left = np.full(tot, -1, dtype=np.int64)
    right = np.full(tot, -1, dtype=np.int64)
    values = np.zeros((tot, C), dtype=np.float64)
    for ids, v, f, th, l, r in zip(lvl_ids, lvl_vals, lvl_feat, lvl_thr, lvl_left, lvl_right):
        feature[ids] = f
        thr[ids] = th
        left[ids] = l
        right[ids] = r
        values[ids] = v

    # preorder renumber: pos(left) = pos(parent)+1,
    # pos(right) = pos(parent)+1+size(left); sizes bottom-up per level
    size = np.ones(tot, dtype=np.int64)
    for ids, l, r in zip(reversed(lvl_ids), reversed(lvl_left), reversed(lvl_right)):
        has = l >= 0
        size[ids[has]] = 1 + size[l[has]] + size[r[has]]
    pos = np.zeros(tot, dtype=np.int64)
    for ids, l, r in zip(lvl_ids, lvl_left, lvl_right):
        has = l >= 0
        p = pos[ids[has]]
        pos[l[has]] = p + 1
        pos[r[has]] = p + 1 + size[l[has]]
    inv = np.empty(tot, dtype=np.int64)
    inv[pos] = np.arange(tot)
    out_feat = feature[inv].astype(np.int32)
    out_thr = thr[inv]
    out_vals = values[inv]
    out_left = np.where(left[inv] >= 0, pos[np.maximum(left[inv], 0)], -1).astype(np.int32)
    out_right = np.where(right[inv] >= 0, pos[np.maximum(right[inv], 0)], -1).astype(np.int32)
    return {
        "left": out_left,
        "right": out_right,
        "feature": out_feat,
        "threshold": out_thr,
        "values": out_vals,
    }
