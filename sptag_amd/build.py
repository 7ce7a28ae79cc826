"""GPU-accelerated index builder (torch tooling over the GPU for the heavy
linear algebra; rocBLAS GEMMs via torch.matmul).

Mirrors the reference build SEMANTICS (citations to /root/reference/AnnService):
  - cosine pre-normalization of base vectors to norm=base in place
    (BKTIndex.cpp:749-756, CommonUtils.h:62 — C-cast truncation for int8);
  - BKT: hierarchical k-means tree, node children contiguous, leaf nodes are
    the vectors themselves, node centerid is a member vector
    (BKTree.h:547-625). Build is not deterministic in the reference either
    (std::mt19937 shuffles — SURVEY.md §8c), so equivalence is judged by
    search recall/QPS on the produced index, not bytes.
  - KNN graph: TP-tree partitions, exact KNN inside each leaf, candidates
    merged over trees (NeighborhoodGraph.h:301-360);
  - RNG prune: candidates ascending by distance, accept c iff for every
    already-accepted b: RNGFactor*dist(b,c) >= dist(q,c), up to degree, pad
    -1 (RelativeNeighborhoodGraph.h:18-35 RebuildNeighbors).

Output arrays load via AnnIndex.FromArrays / sptag_amd_create_index and
save to the reference's byte format via AnnIndex.Save.
"""
import numpy as np
import torch


def _dev():
    return "cuda" if torch.cuda.is_available() else "cpu"


def normalize_base(vectors, distmethod):
    """Reference cosine build normalization (BKTIndex.cpp:749-756)."""
    if distmethod != "Cosine":
        return vectors
    x = vectors.astype(np.float64)
    norms = np.sqrt((x * x).sum(axis=1, keepdims=True))
    norms[norms == 0] = 1.0
    if vectors.dtype == np.int8:
        out = np.trunc(x * 127.0 / norms)  # C-cast truncation (CommonUtils.h:62)
        return out.astype(np.int8)
    return (x / norms).astype(np.float32)


# ------------------------------------------------------------------ #
# BKT tree
# ------------------------------------------------------------------ #

def _ragged_arange(counts):
    """[0..c0), [0..c1), ... concatenated (numpy)."""
    counts = np.asarray(counts, dtype=np.int64)
    total = int(counts.sum())
    if total == 0:
        return np.zeros(0, dtype=np.int64)
    out = np.ones(total, dtype=np.int64)
    out[0] = 0
    starts = np.cumsum(counts)[:-1]
    out[starts] = 1 - counts[:-1]
    return np.cumsum(out)


def _kmeans_split(xf, members, K, iters=8, sample=4096, gen=None):
    """k-means split of one cluster. Returns (groups: list of int64 tensors,
    reps: list of representative vector ids)."""
    m = members.numel()
    K = min(K, max(2, m // 8))
    if m > sample:
        sel = torch.randperm(m, generator=gen, device=members.device)[:sample]
        smp = members[sel]
    else:
        smp = members
    pts = xf[smp]
    perm = torch.randperm(smp.numel(), generator=gen, device=members.device)[:K]
    centers = pts[perm].clone()
    for _ in range(iters):
        lab = torch.cdist(pts, centers).argmin(1)
        sums = torch.zeros_like(centers)
        sums.index_add_(0, lab, pts)
        cnt = torch.bincount(lab, minlength=centers.shape[0]).clamp(min=1)
        newc = sums / cnt[:, None].float()
        keep = torch.bincount(lab, minlength=centers.shape[0]) > 0
        centers = torch.where(keep[:, None], newc, centers)
    xm = xf[members]
    d2 = torch.cdist(xm, centers)          # [m, K]
    lab = d2.argmin(1)
    Kc = centers.shape[0]
    # representative per cluster = member closest to the centroid (vectorized)
    big = torch.finfo(d2.dtype).max
    dmask = d2.clone()
    onehot = torch.nn.functional.one_hot(lab, Kc).bool()
    dmask[~onehot] = big
    rep_local = dmask.argmin(0)            # [K]
    counts = torch.bincount(lab, minlength=Kc)
    groups, reps = [], []
    order = lab.argsort(stable=True)
    start = 0
    for k in range(Kc):
        c = int(counts[k].item())
        if c == 0:
            continue
        groups.append(members[order[start:start + c]])
        reps.append(int(members[rep_local[k]].item()))
        start += c
    if len(groups) <= 1:
        groups, reps = [], []
        per = (m + K - 1) // K
        for k in range(0, m, per):
            g = members[k:k + per]
            groups.append(g)
            reps.append(int(g[0].item()))
    return groups, reps


def build_bkt_tree(vectors, *, kmeans_k=32, leaf_size=32, big_cluster=4096,
                   seed=2016, device=None, verbose=False):
    """Returns (tree_start int32[ntrees], tree_nodes int32[N,3]) in the
    reference layout (BKTree.h:25 BKTNode {centerid,childStart,childEnd},
    children contiguous, childEnd exclusive, leaves childStart=-1)."""
    device = device or _dev()
    n = vectors.shape[0]
    xf = torch.as_tensor(vectors, device=device)
    if xf.dtype != torch.float32:
        xf = xf.float()
    gen = torch.Generator(device=device)
    gen.manual_seed(seed)

    cent, cs, ce = [n], [-1], [-1]   # root: centerid=n (unused; as reference)
    small = []                        # (node_idx, members) clusters <= big_cluster
    stack = [(0, torch.arange(n, device=device, dtype=torch.int64))]
    while stack:
        node, members = stack.pop()
        groups, reps = _kmeans_split(xf, members, kmeans_k, gen=gen)
        cs[node] = len(cent)
        for g, r in zip(groups, reps):
            child = len(cent)
            cent.append(r)
            cs.append(-1)
            ce.append(-1)
            if g.numel() > big_cluster:
                stack.append((child, g))
            elif g.numel() > 1:
                small.append((child, g))
        ce[node] = len(cent)
        if verbose and len(cent) % 100000 < kmeans_k:
            print(f"  bkt phase A: {len(cent)} nodes, stack {len(stack)}")

    cent = np.array(cent, dtype=np.int64)
    cs = np.array(cs, dtype=np.int64)
    ce = np.array(ce, dtype=np.int64)

    # vectorized bottom: each small cluster -> groups of <= leaf_size members
    # (ordered by distance to the cluster centroid), each group an internal
    # node whose children are member leaves. Fully vectorized (no per-cluster
    # python loop) so 100M-scale builds stay fast.
    if small:
        nbase = len(cent)
        dev = xf.device
        members_all = torch.cat([g for _, g in small])
        cl_of = torch.repeat_interleave(
            torch.arange(len(small), device=dev),
            torch.tensor([g.numel() for _, g in small], device=dev))
        ncl = len(small)
        sizes_t = torch.tensor([g.numel() for _, g in small], device=dev)
        # order members within each cluster by distance to its centroid
        pts = xf[members_all]
        cen = torch.zeros((ncl, pts.shape[1]), device=dev)
        cen.index_add_(0, cl_of, pts)
        cen = cen / sizes_t[:, None].float()
        dcent = ((pts - cen[cl_of]) ** 2).sum(1)
        o1 = dcent.argsort()
        o2 = cl_of[o1].argsort(stable=True)
        order = o1[o2]
        members_all = members_all[order]          # cluster-major, dist-ordered
        mem_np = members_all.cpu().numpy()
        sizes = sizes_t.cpu().numpy()
        node_np = np.array([nd for nd, _ in small])

        gcnts = np.ceil(sizes / leaf_size).astype(np.int64)
        has_groups = sizes > leaf_size
        n_group_nodes = np.where(has_groups, gcnts, 0)
        extra = n_group_nodes + sizes              # nodes per cluster
        offs = nbase + np.concatenate([[0], np.cumsum(extra)[:-1]])
        total = int(extra.sum())
        cent2 = np.empty(total, dtype=np.int64)
        cs2 = np.full(total, -1, dtype=np.int64)
        ce2 = np.full(total, -1, dtype=np.int64)

        mstart = np.concatenate([[0], np.cumsum(sizes)[:-1]])
        # clusters WITHOUT group level: children = member leaves
        nog = ~has_groups
        cs[node_np[nog]] = offs[nog]
        ce[node_np[nog]] = offs[nog] + sizes[nog]
        # their leaf slots: positions offs..offs+size-1 (local p = offs-nbase)
        if nog.any():
            lp = offs[nog] - nbase
            idx = np.repeat(lp, sizes[nog]) + _ragged_arange(sizes[nog])
            src = mem_np[np.repeat(mstart[nog], sizes[nog]) + _ragged_arange(sizes[nog])]
            cent2[idx] = src
        # clusters WITH a group level
        wg = has_groups
        if wg.any():
            cs[node_np[wg]] = offs[wg]
            ce[node_np[wg]] = offs[wg] + gcnts[wg]
            gtot = int(gcnts[wg].sum())
            # per group: cluster idx, group idx within cluster
            gcl = np.repeat(np.where(wg)[0], gcnts[wg])
            ggi = _ragged_arange(gcnts[wg])
            gsz = np.minimum(leaf_size, sizes[gcl] - ggi * leaf_size)
            # leaf block offsets: per cluster leaves start at offs+gcnt
            leaf_base = offs[gcl] + gcnts[gcl] + ggi * leaf_size
            # careful: groups are leaf_size-packed so cumulative offset works
            gmemstart = mstart[gcl] + ggi * leaf_size
            gpos = offs[gcl] - nbase + ggi        # group node local position
            cent2[gpos] = mem_np[gmemstart]
            cs2[gpos] = leaf_base
            ce2[gpos] = leaf_base + gsz
            # leaves
            lidx = np.repeat(leaf_base - nbase, gsz) + _ragged_arange(gsz)
            lsrc = mem_np[np.repeat(gmemstart, gsz) + _ragged_arange(gsz)]
            cent2[lidx] = lsrc
            del gtot
        cent = np.concatenate([cent, cent2])
        cs = np.concatenate([cs, cs2])
        ce = np.concatenate([ce, ce2])

    # sentinel (BKTree.h:625)
    cent = np.concatenate([cent, [-1]])
    cs = np.concatenate([cs, [-1]])
    ce = np.concatenate([ce, [-1]])

    tree_nodes = np.stack([cent, cs, ce], axis=1).astype(np.int32)
    tree_start = np.array([0], dtype=np.int32)
    return tree_start, tree_nodes


# ------------------------------------------------------------------ #
# KDT tree (reference KDTree.h:84-118 BuildTrees / :286+ DivideTree:
# split on a high-variance dim at the mean; leaves encode -(vecid+1))
# ------------------------------------------------------------------ #

def build_kdt_tree(vectors, *, ntrees=1, sample_dims=16, seed=2016,
                   device=None, max_levels=64, verbose=False):
    """Level-vectorized kd-tree build. Returns (tree_start int32[ntrees],
    kdt_nodes int32[N,4]) — the 4th column holds the float split_value bit
    pattern (KDTNode layout, KDTree.h:22). Split dim = max-variance among a
    per-level random dim subset (reference: top-5 variance dims on a
    sample); split value = segment mean."""
    device = device or _dev()
    n, d = vectors.shape
    xf = torch.as_tensor(vectors, device=device)
    if xf.dtype != torch.float32:
        xf = xf.float()
    gen = torch.Generator(device=device)
    gen.manual_seed(seed + 31)

    tree_start = []
    all_nodes = []
    base = 0
    for t in range(ntrees):
        tree_start.append(base)
        # node storage (numpy, grown per level)
        left_np = np.zeros(1, dtype=np.int64)
        right_np = np.zeros(1, dtype=np.int64)
        sdim_np_all = np.zeros(1, dtype=np.int64)
        sval_np_all = np.zeros(1, dtype=np.float64)
        perm = torch.randperm(n, generator=gen, device=device)
        seg = torch.zeros(n, dtype=torch.int64, device=device)
        seg_node = np.array([0], dtype=np.int64)   # node idx per segment
        nseg = 1
        for level in range(max_levels):
            cnt = torch.bincount(seg, minlength=nseg)
            if int(cnt.max()) <= 1:
                break
            k16 = min(sample_dims, d)
            dims = torch.randperm(d, generator=gen, device=device)[:k16]
            xs = xf[perm][:, dims]                       # [n, k16]
            sums = torch.zeros(nseg, k16, device=device).index_add_(0, seg, xs)
            sqs = torch.zeros(nseg, k16, device=device).index_add_(0, seg, xs * xs)
            cf = cnt.float().clamp(min=1)[:, None]
            var = sqs / cf - (sums / cf) ** 2
            sd_local = var.argmax(1)                     # [nseg]
            split_dim = dims[sd_local]
            split_val = sums.gather(1, sd_local[:, None]).squeeze(1) / cf.squeeze(1)
            val_e = xs.gather(1, sd_local[seg][:, None]).squeeze(1)
            side = val_e >= split_val[seg]               # right side (diff >= 0)
            # guard degenerate splits (all equal): rank-split those segments
            lcnt = torch.bincount(seg[~side], minlength=nseg)
            rcnt = cnt - lcnt
            bad = ((lcnt == 0) | (rcnt == 0)) & (cnt >= 2)
            if bool(bad.any()):
                starts = torch.cumsum(cnt, 0) - cnt
                pos = torch.arange(n, device=device) - starts[seg]
                rank_side = pos >= (cnt[seg] // 2)
                side = torch.where(bad[seg], rank_side, side)
                lcnt = torch.bincount(seg[~side], minlength=nseg)
                rcnt = cnt - lcnt
            # order members: (seg, side) stable
            o1 = side.long().argsort(stable=True)
            o2 = seg[o1].argsort(stable=True)
            order = o1[o2]
            perm = perm[order]
            side = side[order]
            # record this level's nodes (only segments with cnt >= 2 are
            # internal; they were allocated a node idx in seg_node)
            cnt_np = cnt.cpu().numpy()
            l_np = lcnt.cpu().numpy()
            r_np = rcnt.cpu().numpy()
            sd_np = split_dim.cpu().numpy()
            sv_np = split_val.cpu().numpy()
            starts_np = np.concatenate([[0], np.cumsum(cnt_np)[:-1]])
            perm_np = perm.cpu().numpy()
            # children allocation
            internal = cnt_np >= 2
            child_internal_l = internal & (l_np >= 2)
            child_internal_r = internal & (r_np >= 2)
            n_new = int(child_internal_l.sum() + child_internal_r.sum())
            next_idx = len(left_np)
            new_ids_l = np.full(nseg, -1, dtype=np.int64)
            new_ids_r = np.full(nseg, -1, dtype=np.int64)
            alloc = np.cumsum(np.concatenate(
                [child_internal_l.astype(np.int64), child_internal_r.astype(np.int64)]))
            new_ids_l[child_internal_l] = next_idx + alloc[:nseg][child_internal_l] - 1
            new_ids_r[child_internal_r] = next_idx + alloc[nseg:][child_internal_r] - 1
            left_np = np.concatenate([left_np, np.zeros(n_new, dtype=np.int64)])
            right_np = np.concatenate([right_np, np.zeros(n_new, dtype=np.int64)])
            sdim_np_all = np.concatenate([sdim_np_all, np.zeros(n_new, dtype=np.int64)])
            sval_np_all = np.concatenate([sval_np_all, np.zeros(n_new)])
            # children: internal -> allocated idx, single member -> -(id+1),
            # empty side -> out-of-range leaf (ignored by search)
            lchild = np.full(nseg, -(np.int64(n) + 1), dtype=np.int64)
            rchild = np.full(nseg, -(np.int64(n) + 1), dtype=np.int64)
            lchild[child_internal_l] = new_ids_l[child_internal_l]
            rchild[child_internal_r] = new_ids_r[child_internal_r]
            leaf_l = internal & (l_np == 1)
            leaf_r = internal & (r_np == 1)
            lchild[leaf_l] = -(perm_np[starts_np[leaf_l]] + 1)
            rchild[leaf_r] = -(perm_np[starts_np[leaf_r] + l_np[leaf_r]] + 1)
            li = np.where(internal)[0]
            tgt = seg_node[li]
            left_np[tgt] = lchild[li]
            right_np[tgt] = rchild[li]
            sdim_np_all[tgt] = sd_np[li]
            sval_np_all[tgt] = sv_np[li]
            # next level: keep only members of internal children
            seg2 = seg * 2 + side.long()
            keep_seg = torch.zeros(nseg * 2, dtype=torch.bool, device=device)
            node_of = torch.full((nseg * 2,), -1, dtype=torch.int64, device=device)
            kl = torch.as_tensor(child_internal_l, device=device)
            kr = torch.as_tensor(child_internal_r, device=device)
            keep_seg[0::2] = kl
            keep_seg[1::2] = kr
            node_of[0::2] = torch.as_tensor(new_ids_l, device=device)
            node_of[1::2] = torch.as_tensor(new_ids_r, device=device)
            keep = keep_seg[seg2]
            perm = perm[keep]
            seg2 = seg2[keep]
            uniq, seg = torch.unique(seg2, return_inverse=True)
            seg_node = node_of[uniq].cpu().numpy()
            nseg = uniq.numel()
            n = n  # unchanged (total vector count for leaf encoding)
            if nseg == 0:
                break
            if verbose and level % 8 == 0:
                print(f"  kdt tree {t}: level {level}, {nseg} open segments")
        nodes = np.zeros((len(left_np), 4), dtype=np.int32)
        nodes[:, 0] = left_np + np.where(left_np >= 0, base, 0)
        nodes[:, 1] = right_np + np.where(right_np >= 0, base, 0)
        nodes[:, 2] = sdim_np_all
        nodes[:, 3] = sval_np_all.astype(np.float32).view(np.int32)
        all_nodes.append(nodes)
        base += len(left_np)
    return (np.array(tree_start, dtype=np.int32),
            np.concatenate(all_nodes).astype(np.int32))


# ------------------------------------------------------------------ #
# TP-tree KNN candidates + RNG prune
# ------------------------------------------------------------------ #

def _tpt_leaves(xf, leaf, gen):
    """Random-hyperplane tree: returns (perm, [(start, count), ...]) — leaves
    are contiguous slices of perm."""
    n, d = xf.shape
    perm = torch.arange(n, device=xf.device)
    seg = torch.zeros(n, dtype=torch.int64, device=xf.device)
    nseg = 1
    max_size = n
    while max_size > leaf:
        r = torch.randn(d, generator=gen, device=xf.device)
        proj = xf[perm] @ r
        o1 = proj.argsort()
        o2 = seg[o1].argsort(stable=True)
        order = o1[o2]
        perm = perm[order]
        seg = seg[order]                      # segments contiguous, proj-sorted
        counts = torch.bincount(seg, minlength=nseg)
        starts = torch.cumsum(counts, 0) - counts
        half = counts // 2
        pos = torch.arange(n, device=xf.device) - starts.repeat_interleave(counts)
        within_hi = pos >= half.repeat_interleave(counts)
        # only split segments still above the leaf size
        splitting = (counts > leaf).repeat_interleave(counts)
        seg = seg * 2 + (within_hi & splitting).long()
        uniq, seg = torch.unique(seg, return_inverse=True)
        nseg = uniq.numel()
        max_size = int(torch.bincount(seg, minlength=nseg).max().item())
    counts = torch.bincount(seg, minlength=nseg)
    starts = torch.cumsum(counts, 0) - counts
    return perm, list(zip(starts.tolist(), counts.tolist()))


def _leaf_knn(xf, perm, bounds, kper, chunk_rows=1_000_000):
    """Exact KNN inside each leaf. Returns (ids int32 [n,kper], dists f32) in
    original vector-id space (L2 on the f32 view; candidate generation only —
    exact metric distances are recomputed by the searcher)."""
    n = xf.shape[0]
    ids = torch.full((n, kper), -1, dtype=torch.int32, device=xf.device)
    dst = torch.full((n, kper), float("inf"), device=xf.device)
    maxleaf = max(c for _, c in bounds)
    batch = max(1, chunk_rows // maxleaf)
    starts_t = torch.tensor([s for s, _ in bounds], device=xf.device)
    counts_t = torch.tensor([c for _, c in bounds], device=xf.device)
    for i in range(0, len(bounds), batch):
        st = starts_t[i:i + batch]
        ct = counts_t[i:i + batch]
        B = st.numel()
        m = maxleaf
        pos = torch.arange(m, device=xf.device)[None, :].expand(B, m)
        mask = pos < ct[:, None]
        gidx = (st[:, None] + pos).clamp(max=n - 1)
        gather = perm[gidx]                    # [B, m]
        gather = torch.where(mask, gather, torch.zeros_like(gather))
        pts = xf[gather]                       # [B, m, d]
        sq = (pts * pts).sum(-1)
        d2 = sq[:, :, None] + sq[:, None, :] - 2.0 * torch.bmm(pts, pts.transpose(1, 2))
        d2.diagonal(dim1=1, dim2=2).fill_(float("inf"))
        d2.masked_fill_(~mask[:, None, :], float("inf"))
        k = min(kper, m - 1)
        vals, loc = torch.topk(d2, k, dim=2, largest=False)
        nbr = torch.gather(gather[:, None, :].expand(B, m, m), 2, loc)
        rows = gather[mask]
        ids[rows, :k] = nbr[mask].int()
        dst[rows, :k] = vals[mask]
    return ids, dst


def _merge_candidates(ids_a, dst_a, ids_b, dst_b, cand, self_ids,
                      row_chunk=None, out=None):
    """Merge two candidate lists per point: dedupe ids, keep `cand` nearest,
    pad to exactly `cand` columns. Row-chunked: the transient sort/gather
    tensors are ~6x the chunk size, which at 100M rows would not fit.
    `out=(ids, dst)` writes in place (may alias ids_a/dst_a: each chunk is
    fully read before it is overwritten) — at 100M rows the fresh output
    allocation alone is ~80GB."""
    n = ids_a.shape[0]
    if row_chunk is None:
        width = ids_a.shape[1] + ids_b.shape[1] + 1
        row_chunk = max(65_536, int(4e9 // (width * 24)))
    if out is not None and out[0].shape[1] == cand:
        out_i, out_d = out
    else:
        out_i = torch.empty((n, cand), dtype=ids_a.dtype, device=ids_a.device)
        out_d = torch.empty((n, cand), dtype=dst_a.dtype, device=dst_a.device)
    for s in range(0, n, row_chunk):
        e = min(n, s + row_chunk)
        cid = torch.cat([ids_a[s:e], ids_b[s:e]], dim=1)
        cdd = torch.cat([dst_a[s:e], dst_b[s:e]], dim=1)
        order = cdd.argsort(dim=1, stable=True)
        cid = torch.gather(cid, 1, order)
        cdd = torch.gather(cdd, 1, order)
        sid, sorder = cid.sort(dim=1, stable=True)
        dup_sorted = torch.zeros_like(sid, dtype=torch.bool)
        dup_sorted[:, 1:] = sid[:, 1:] == sid[:, :-1]
        dup = torch.zeros_like(dup_sorted)
        dup.scatter_(1, sorder, dup_sorted)
        invalid = dup | (cid < 0) | (cid == self_ids[s:e, None])
        cdd = cdd.masked_fill(invalid, float("inf"))
        order = cdd.argsort(dim=1, stable=True)[:, :cand]
        ci = torch.gather(cid, 1, order)
        cd = torch.gather(cdd, 1, order)
        if ci.shape[1] < cand:
            padn = cand - ci.shape[1]
            ci = torch.cat([ci, torch.full((ci.shape[0], padn), -1,
                                           dtype=ci.dtype, device=ci.device)], 1)
            cd = torch.cat([cd, torch.full((cd.shape[0], padn), float("inf"),
                                           device=cd.device)], 1)
        out_i[s:e] = ci.masked_fill(~torch.isfinite(cd), -1)
        out_d[s:e] = cd
    return out_i, out_d


def _prune_chunk(cand, d):
    """point-chunk size keeping the [B, cand, cand] pairwise tensor ~2GB"""
    return max(4096, int(2e9 // (cand * cand * 4 + cand * d * 8)))


def build_rng_graph(vectors, *, degree=32, ntrees=4, tpt_leaf=1000, cand=256,
                    rng_factor=1.0, seed=2016, device=None, point_chunk=None,
                    verbose=False):
    """Returns graph int32 [n, degree] (RNG-pruned, ascending, -1 padded)."""
    device = device or _dev()
    n = vectors.shape[0]
    xf = torch.as_tensor(vectors, device=device)
    if xf.dtype != torch.float32:
        xf = xf.float()
    gen = torch.Generator(device=device)
    gen.manual_seed(seed + 77)
    self_ids = torch.arange(n, device=device, dtype=torch.int32)

    kper = min(max(33, cand // max(ntrees, 1)), 64)
    ids = dst = None
    last_partition = None
    for t in range(ntrees):
        perm, bounds = _tpt_leaves(xf, tpt_leaf, gen)
        tids, tdst = _leaf_knn(xf, perm, bounds, kper)
        if ids is None:
            ids, dst = _merge_candidates(tids, tdst, tids[:, :0], tdst[:, :0],
                                         cand, self_ids)
        else:
            ids, dst = _merge_candidates(ids, dst, tids, tdst, cand, self_ids,
                                         out=(ids, dst))
        del tids, tdst
        last_partition = (perm, bounds)
        if verbose:
            print(f"  tpt tree {t + 1}/{ntrees} merged ({len(bounds)} leaves)")

    # Highway bridges: the reference's search-based RefineNode pool
    # (CEF=1000, NeighborhoodGraph.h:535) reaches far beyond a point's own
    # cluster, which is what creates the cross-cluster edges the RNG prune
    # then filters. Truncated in-leaf KNN pools cannot (tight clusters
    # become graph islands), so append per-point "bridge" candidates: the
    # medoids of the nearest other TP-tree leaves. They sort to the pool's
    # tail and survive (the pool is widened, not truncated), and the RNG
    # rule decides acceptance exactly as RebuildNeighbors would.
    perm, bounds = last_partition
    L = len(bounds)
    bknn = min(8, L - 1)
    bridge_ids = None
    if bknn > 0:
        # Leaf centroids + medoids WITHOUT padded [L, maxleaf, d] tensors
        # (at 100M/131k leaves those transients alone are ~90 GB and OOM
        # next to the candidate pools): index_add centroids, then a
        # packed-key (float-bits<<32 | index) segmented argmin per leaf.
        d = xf.shape[1]
        counts_t = torch.tensor([c for _, c in bounds], device=device)
        leaf_of_perm = torch.repeat_interleave(
            torch.arange(L, device=device), counts_t)
        leaf_of = torch.empty(n, dtype=torch.int64, device=device)
        leaf_of[perm] = leaf_of_perm
        centroids = torch.zeros((L, d), device=device)
        pc = 10_000_000
        for s0 in range(0, n, pc):
            e0 = min(n, s0 + pc)
            centroids.index_add_(0, leaf_of_perm[s0:e0], xf[perm[s0:e0]])
        centroids /= counts_t.float().clamp(min=1)[:, None]
        keymin = torch.full((L,), 2 ** 62, dtype=torch.int64, device=device)
        for s0 in range(0, n, pc):
            e0 = min(n, s0 + pc)
            lf = leaf_of_perm[s0:e0]
            dc = ((xf[perm[s0:e0]] - centroids[lf]) ** 2).sum(1)
            # non-negative f32 bit patterns are order-preserving as ints
            key = (dc.view(torch.int32).to(torch.int64) << 32) | \
                torch.arange(s0, e0, device=device)
            keymin.scatter_reduce_(0, lf, key, reduce="amin")
        medoid = perm[keymin & 0xffffffff]               # [L]
        # exponential-rank ladder of nearest leaves: ranks 1,2,4,...: short
        # bridges link adjacent leaves, long ones cross cluster groups —
        # scale-independent (at 100M a tight cluster spans many leaves, so
        # nearest-8 bridges alone stay inside it and islands return).
        ranks = [1, 2, 4, 8, 16, 32, 64, 128]
        ranks = [r for r in ranks if r < L][:bknn]
        kmax = max(ranks) + 1
        nleaf = torch.empty((L, len(ranks)), dtype=torch.int64, device=device)
        lchunk = max(1024, int(2e9 // (L * 4)))
        for s0 in range(0, L, lchunk):
            e0 = min(L, s0 + lchunk)
            dc = torch.cdist(centroids[s0:e0], centroids)
            idxs = dc.topk(kmax + 1, dim=1, largest=False).indices
            # drop self (rank 0), take the ladder ranks
            nleaf[s0:e0] = idxs[:, 1:][:, [r - 1 for r in ranks]]
        bridge_ids = medoid[nleaf][leaf_of].int()             # [n, len(ranks)]
        del leaf_of_perm, keymin, centroids

    if point_chunk is None:
        point_chunk = _prune_chunk(cand, xf.shape[1])
    graph = torch.full((n, degree), -1, dtype=torch.int32, device=device)
    for s in range(0, n, point_chunk):
        e = min(n, s + point_chunk)
        cid, cdd = ids[s:e], dst[s:e]
        if bridge_ids is not None:
            bi = bridge_ids[s:e]
            bd = _exact_l2(xf, torch.arange(s, e, device=device), bi.long())
            # widen the pool (cand + bknn) so bridges are never truncated away
            cid, cdd = _merge_candidates(cid, cdd, bi, bd,
                                         cid.shape[1] + bi.shape[1],
                                         self_ids[s:e])
        graph[s:e] = _rng_prune(xf, cid, cdd, degree, rng_factor, device)
        if verbose and (s // point_chunk) % 20 == 0:
            print(f"  rng prune {e}/{n}")
    return graph, ids, dst


def _rng_prune(xf, cid, cdd, degree, rng_factor, device, fill_pruned=False):
    """RNG prune one chunk of candidate lists (ascending by dist).

    fill_pruned: after the RNG rule fills what it can, pad remaining degree
    slots with the nearest REJECTED candidates (build-quality knob for
    billion-scale pools, cf. HNSW keepPrunedConnections; the reference's
    RebuildNeighbors pads -1 — its CEF=1000 pools rarely leave slots empty,
    shallow pools at 100M+ do)."""
    C = cid.shape[1]
    cvec = xf[cid.clamp(min=0).long()]
    csq = (cvec * cvec).sum(-1)
    P = csq[:, :, None] + csq[:, None, :] - 2.0 * torch.bmm(cvec, cvec.transpose(1, 2))
    valid = torch.isfinite(cdd)
    acc = torch.zeros_like(valid)
    count = torch.zeros(cid.shape[0], dtype=torch.int32, device=device)
    for j in range(C):
        viol = (rng_factor * P[:, :, j] < cdd[:, j:j + 1]) & acc
        good = valid[:, j] & ~viol.any(1) & (count < degree)
        acc[:, j] = good
        count += good.int()
    if fill_pruned:
        # fill leftover slots with the nearest valid rejected candidates
        free = (degree - count).clamp(min=0)
        rej = valid & ~acc
        rej_rank = torch.cumsum(rej.int(), dim=1)
        acc = acc | (rej & (rej_rank <= free[:, None]))
    out = torch.full((cid.shape[0], degree), -1, dtype=torch.int32, device=device)
    pos = (torch.cumsum(acc.int(), dim=1) - 1).clamp(min=0)
    rows = torch.nonzero(acc, as_tuple=True)
    out[rows[0], pos[rows]] = cid[rows]
    return out


def _rng_prune_rows(xf, cid, cdd, degree, rng_factor, device,
                    fill_pruned=False):
    """_rng_prune over row sub-chunks (the [B, C, C] pairwise tensor is the
    memory driver; C can be wider than the stored pool here)."""
    pc = _prune_chunk(cid.shape[1], xf.shape[1])
    out = torch.empty((cid.shape[0], degree), dtype=torch.int32, device=device)
    for s in range(0, cid.shape[0], pc):
        e = min(cid.shape[0], s + pc)
        out[s:e] = _rng_prune(xf, cid[s:e], cdd[s:e], degree, rng_factor,
                              device, fill_pruned=fill_pruned)
    return out


def refine_graph(vectors, graph, cand_ids, cand_dst, *, degree=32, cand=256,
                 rounds=2, hop_sample=8, rng_factor=1.0, device=None,
                 point_chunk=None, seed=2016, verbose=False):
    """Neighborhood refinement: the reference refines each node's edges from
    a CEF-sized candidate pool gathered by searching the index itself
    (NeighborhoodGraph.h:460-560 RefineGraph/RefineNode). Here the pool is
    grown NN-descent style — current candidates + two-hop neighbors +
    reverse edges — then RNG-pruned with the same RebuildNeighbors rule.
    Returns (graph int32 [n, degree], cand_ids, cand_dst)."""
    device = device or _dev()
    n = vectors.shape[0]
    xf = torch.as_tensor(vectors, device=device)
    if xf.dtype != torch.float32:
        xf = xf.float()
    g = torch.as_tensor(graph, device=device)
    self_ids = torch.arange(n, device=device, dtype=torch.int32)
    gen = torch.Generator(device=device)
    gen.manual_seed(seed + 123)
    if point_chunk is None:
        point_chunk = _prune_chunk(cand, xf.shape[1])

    for r in range(rounds):
        # reverse edges (sampled): every edge (i -> j) proposes i to j.
        # Slot by a per-round hash of the source and let collisions drop
        # (last-write-wins scatter) — O(E) with no 32n-element sort, which
        # at 100M rows would blow past cub's 2^31 sort limit and ~65 GB of
        # transients. NN-descent only needs a reverse SAMPLE per node.
        rcap = 16
        rev = torch.full((n, rcap), -1, dtype=torch.int32, device=device)
        edge_chunk = 64_000_000
        deg = g.shape[1]
        for es in range(0, n, max(1, edge_chunk // deg)):
            ee = min(n, es + max(1, edge_chunk // deg))
            srcc = self_ids[es:ee].repeat_interleave(deg)
            dstc = g[es:ee].reshape(-1).long()
            keep = dstc >= 0
            srcc, dstc = srcc[keep], dstc[keep]
            slot = ((srcc.long() * 2654435761) + r * 97) % rcap
            rev[dstc, slot] = srcc
        for s in range(0, n, point_chunk):
            e = min(n, s + point_chunk)
            B = e - s
            nbr = g[s:e]                                   # [B, deg]
            hop2 = g[nbr[:, :hop_sample].clamp(min=0).long()].reshape(B, -1)
            hop2 = torch.where((nbr[:, :hop_sample] < 0).repeat_interleave(
                g.shape[1], dim=1), torch.full_like(hop2, -1), hop2)
            newc = torch.cat([hop2, rev[s:e]], dim=1)
            newd = _exact_l2(xf, torch.arange(s, e, device=device), newc)
            cand_ids[s:e], cand_dst[s:e] = _merge_candidates(
                cand_ids[s:e], cand_dst[s:e], newc, newd, cand, self_ids[s:e])
            g[s:e] = _rng_prune(xf, cand_ids[s:e], cand_dst[s:e], degree,
                                rng_factor, device)
        if verbose:
            print(f"  refine round {r + 1}/{rounds} done")
    return g.cpu().numpy(), cand_ids, cand_dst


def _exact_l2(xf, q_rows, c_ids):
    """exact (f32 view) L2 distances between rows q_rows and candidate ids
    c_ids [B, C]; invalid ids (<0) -> inf."""
    a = xf[q_rows]                          # [B, d]
    b = xf[c_ids.clamp(min=0).long()]       # [B, C, d]
    diff = b - a[:, None, :]
    d = (diff * diff).sum(-1)
    return d.masked_fill(c_ids < 0, float("inf"))


def refine_via_search(vectors_t, tree_start, tree_nodes, graph, cand_ids,
                      cand_dst, distmethod, *, algo="BKT", degree=32,
                      cand=256, rounds=1, k=512, max_check=8192,
                      rng_factor=1.0, chunk=0, fill_pruned=False,
                      verbose=False):
    """The reference's own refinement recipe (NeighborhoodGraph.h:460-560
    RefineGraph: every node re-searches the CURRENT index and its edges are
    RNG-rebuilt from the results) — run on the PRODUCT GPU searcher via the
    C-ABI. This is what creates the cross-cluster skip edges that truncated
    candidate pools miss at 100M scale. Depth matters: the
    skip edges come from the DEEP tail of the result list (the reference
    uses CEF=1000; k=64 measured ineffective at 30M — 0.664 vs 0.649
    baseline), hence k=512 / max_check=8192 defaults. GPU-only (the
    searcher has no CPU path); vectors_t is the normalized torch tensor on
    the device; cand lists are the builder's pools (merged in)."""
    import sptag_amd
    device = vectors_t.device
    n = vectors_t.shape[0]
    xf = vectors_t if vectors_t.dtype == torch.float32 else vectors_t.float()
    self_ids = torch.arange(n, device=device, dtype=torch.int32)
    if chunk <= 0:
        # scratch-bounded chunk: visited table (scales with mc AND k — deep
        # results queues keep the traversal expanding past MaxCheck) at
        # ~8 GB, plus headroom for the overflow rerun's reference-capacity
        # global heaps (40*mc entries x 8 B per query) at ~12 GB.
        vcap = 1
        while vcap < max(4096, max_check * 4 + 64 * k):
            vcap <<= 1
        chunk = min(1_000_000,
                    int(8e9 // (vcap * 4)),
                    int(12e9 // (max_check * 40 * 8)))
        chunk = max(16_384, chunk)
    x_np = vectors_t.cpu().numpy()
    graph_t = torch.as_tensor(graph, device=device)         if not torch.is_tensor(graph) else graph
    # Search distances arrive in the index metric; the builder's pools are
    # L2 on the (normalized) float view. For cosine the EXACT conversion is
    # L2(q,v) = |q|^2 + |v|^2 - 2*dot with dot = base^2 - d_search.
    # (A pure x2 scale is exact only at |v| == base exactly; truncated int8
    # norms deviate by enough to corrupt pool ordering — the round-2 30M
    # run measured recall 0.67 with the scale shortcut.)
    nsq = (xf * xf).sum(1) if distmethod == "Cosine" else None
    base2 = (127.0 * 127.0) if vectors_t.dtype == torch.int8 else 1.0
    for r in range(rounds):
        if algo == "KDT":
            ix = sptag_amd.AnnIndex.FromArraysKDT(
                x_np, tree_start, tree_nodes, graph_t.cpu().numpy(), distmethod)
        else:
            ix = sptag_amd.AnnIndex.FromArrays(
                x_np, tree_start, tree_nodes, graph_t.cpu().numpy(), distmethod)
        d_vids = torch.empty((chunk, k), dtype=torch.int32, device=device)
        d_dists = torch.empty((chunk, k), dtype=torch.float32, device=device)
        for s0 in range(0, n, chunk):
            e0 = min(n, s0 + chunk)
            B = e0 - s0
            q = vectors_t[s0:e0].contiguous()
            ix.BatchSearchDevice(q.data_ptr(), B, k, d_vids.data_ptr(),
                                 d_dists.data_ptr(), max_check)
            sv = d_vids[:B]
            if nsq is not None:
                dot = base2 - d_dists[:B]
                sd = nsq[s0:e0, None] + nsq[sv.clamp(min=0).long()] - 2.0 * dot
            else:
                sd = d_dists[:B].clone()
            sd = sd.masked_fill(sv < 0, float("inf"))
            # Spectrum subsample of the result list: head (nearest) + a
            # strided sample of the DEEP tail. The deep candidates are what
            # the RNG prune turns into navigable mid-range edges (the
            # reference prunes from the full CEF=1000 list,
            # NeighborhoodGraph.h:535); subsampling bounds the O(C^2)
            # prune cost.
            head = min(64, k)
            sv_s = torch.cat([sv[:, :head], sv[:, head::7]], dim=1)
            sd_s = torch.cat([sd[:, :head], sd[:, head::7]], dim=1)
            # Prune from the WIDE merged list; truncating to the stored
            # pool width FIRST discards exactly those mid-range candidates
            # (measured at 30M: recall 0.86 -> 0.42).
            wide = cand + sv_s.shape[1]
            wi, wd = _merge_candidates(cand_ids[s0:e0], cand_dst[s0:e0],
                                       sv_s, sd_s, wide, self_ids[s0:e0])
            graph_t[s0:e0] = _rng_prune_rows(xf, wi, wd, degree, rng_factor,
                                             device, fill_pruned=fill_pruned)
            cand_ids[s0:e0] = wi[:, :cand]
            cand_dst[s0:e0] = wd[:, :cand]
            del wi, wd
            if verbose and (s0 // chunk) % 10 == 0:
                print(f"  search-refine round {r + 1}: {e0}/{n}")
        del ix, d_vids, d_dists
        torch.cuda.empty_cache()
        if verbose:
            print(f"  search-refine round {r + 1}/{rounds} done")
    return graph_t, cand_ids, cand_dst


def build_index_arrays(vectors, distmethod, *, algo="BKT", degree=32, ntrees=4,
                       tpt_leaf=1000, cand=256, kmeans_k=32, leaf_size=32,
                       refine_rounds=0, search_refine_rounds=0, kdt_trees=1,
                       srefine_k=512, srefine_mc=8192, fill_pruned=False,
                       seed=2016, device=None, normalized=False,
                       verbose=False):
    """Full build: returns dict(vectors, tree_start, tree_nodes, graph) ready
    for AnnIndex.FromArrays (vectors already cosine-normalized when needed,
    as the reference stores them on disk)."""
    if not normalized:
        vectors = normalize_base(vectors, distmethod)
    if algo == "KDT":
        tree_start, tree_nodes = build_kdt_tree(
            vectors, ntrees=kdt_trees, seed=seed, device=device, verbose=verbose)
    else:
        tree_start, tree_nodes = build_bkt_tree(
            vectors, kmeans_k=kmeans_k, leaf_size=leaf_size, seed=seed,
            device=device, verbose=verbose)
    if verbose:
        print(f"  {algo} tree: {len(tree_nodes)} nodes")
    graph, cids, cdst = build_rng_graph(
        vectors, degree=degree, ntrees=ntrees, tpt_leaf=tpt_leaf, cand=cand,
        seed=seed, device=device, verbose=verbose)
    if refine_rounds > 0:
        graph, cids, cdst = refine_graph(
            vectors, graph, cids, cdst, degree=degree, cand=cand,
            rounds=refine_rounds, device=device, seed=seed, verbose=verbose)
        graph = torch.as_tensor(graph, device=device or _dev())
    if search_refine_rounds > 0 and torch.cuda.is_available():
        xt = torch.as_tensor(vectors, device=device or _dev())
        graph, cids, cdst = refine_via_search(
            xt, tree_start, tree_nodes, graph, cids, cdst, distmethod,
            algo=algo, degree=degree, cand=cand, k=srefine_k,
            max_check=srefine_mc, fill_pruned=fill_pruned,
            rounds=search_refine_rounds, verbose=verbose)
        del xt
    graph = graph.cpu().numpy() if torch.is_tensor(graph) else graph
    return {"vectors": vectors, "tree_start": tree_start,
            "tree_nodes": tree_nodes, "graph": graph,
            "distmethod": distmethod, "algo": algo}
