/* sptag_amd GPU kernels — gfx950 (MI355X, CDNA4) native.
 *
 * Batched BKT best-first graph search: one query per 64-thread workgroup
 * (one wave64). The traversal replicates the reference CPU algorithm
 * decision-for-decision (citations to /root/reference/AnnService):
 *   - seed phase: BKTree::InitSearchTrees / SearchTrees (BKTree.h:697,772)
 *   - traversal:  BKT::Index<T>::Search<> loop (BKTIndex.cpp:272-352)
 *   - frontier:   Heap<NodeDistPair> semantics (Heap.h:14-110)
 *   - pruning:    DistPriorityQueue (WorkSpace.h:167-225)
 *   - top-k:      QueryResultSet AddPoint/SortResult (QueryResultSet.h:31)
 * Distances reproduce the reference binary's float summation order
 * (AVX512 16-lane chunking with fused mul+add; exact integer math for int8),
 * so int8 results are bit-exact and float results are bit-exact against
 * the oracle/_ref build (see oracle/sptag_oracle.c).
 *
 * Division of labor inside the wave:
 *   - heap/result-set mutation is inherently serial (pop order on equal
 *     keys is part of the parity contract) -> lane 0, state in LDS;
 *   - adjacency row load: lanes 0..deg-1, one coalesced 128B line;
 *   - visited-set: per-query open-addressing table in GLOBAL memory
 *     (512KB-class tables do not fit LDS at MaxCheck 8192 — SURVEY.md §7),
 *     probed in parallel with atomicCAS (set semantics only, order-free);
 *   - distances: 16-lane groups, 4 candidates in flight per wave.
 *
 * HEAPS_IN_LDS=true keeps the NG/SPT frontier heaps in LDS with reduced
 * capacities; if a heap would exceed its reduced capacity while the
 * reference's (30*maxCheck / 10*maxCheck, WorkSpace.h:265) would not, the
 * query sets an overflow flag and the host reruns it with the global-memory
 * variant at full reference capacities — never a silent semantic change.
 */
#include <hip/hip_runtime.h>

#include "common.h"

namespace sptag_amd {

#define DEV __device__ __forceinline__

static constexpr float MAXDIST = __FLT_MAX__ / 10.0f; /* Common.h:122 */

struct NodeDist { int32_t node; float distance; };
struct QRes { int32_t vid; float dist; };

/* ------------------------------------------------------------------ *
 * serial structures (lane 0 only), exact reference semantics
 * ------------------------------------------------------------------ */

/* 2^floor(log2(size)) — start of the heap's last level (Heap.h:25) */
DEV int heap_lastlevel(int size) { return 1 << (31 - __clz(size)); }

/* Binary min-heap, EXACT Heap.h semantics (pop order on equal keys is part
 * of the parity contract), with TIERED storage: entries with index <=
 * lsplit live in an LDS cache, the tail in `a` (global scratch or LDS).
 * Storage placement never changes semantics — only access latency. The
 * round-1 all-global SPT heap made the seed phase 41.5% of kernel cycles
 * (profiles/r2 phase breakdown): every percolate level was an ~HBM-latency
 * round trip. The hot top levels now stay in LDS. */
enum { SPT_LDS_TIER = SPT_LDS_TIER_H };  /* 8 levels (1..255) + sentinel [0] */

struct HeapRef {
    NodeDist* a;    /* 1-based backing array; [0] = sentinel {-1, MAXDIST} */
    NodeDist* lds;  /* optional LDS cache of [0..lsplit]; null = none */
    int lsplit;
    int cap;        /* this buffer's capacity */
    int ref_cap;    /* the reference's capacity for this heap */
};

DEV NodeDist hget(const HeapRef& h, int i)
{
    return (h.lds && i <= h.lsplit) ? h.lds[i] : h.a[i];
}

DEV void hset(const HeapRef& h, int i, NodeDist v)
{
    if (h.lds && i <= h.lsplit) h.lds[i] = v;
    else h.a[i] = v;
}

/* Heap.h:38-62 insert (incl. the full-heap last-level replace path). */
DEV void ndheap_insert(HeapRef h, int* count, NodeDist v, int* oflow)
{
    int loc;
    if (*count == h.cap) {
        if (h.cap < h.ref_cap) { *oflow = 1; return; }
        int lastlevel = heap_lastlevel(h.cap);
        int maxi = lastlevel;
        float maxd = hget(h, maxi).distance;
        for (int i = lastlevel + 1; i <= h.cap; i++) {
            float di = hget(h, i).distance;
            if (maxd < di) { maxi = i; maxd = di; }
        }
        if (v.distance > maxd) return;
        loc = maxi;
    } else {
        loc = ++(*count);
    }
    int par = loc >> 1;
    while (par > 0) {
        NodeDist p = hget(h, par);
        if (!(v.distance < p.distance)) break;
        hset(h, loc, p);
        loc = par;
        par >>= 1;
    }
    hset(h, loc, v);
}

/* Heap.h:90-105 heapify + :74-82 pop. The sinking element is the original
 * root value throughout (the reference swaps it level by level), so it is
 * carried in a register instead of re-read. */
DEV void ndheap_heapify(HeapRef h, int count)
{
    int parent = 1, next = 2;
    if (next > count) return;
    NodeDist pv = hget(h, parent);
    while (next < count) {
        NodeDist nv = hget(h, next);
        NodeDist nv2 = hget(h, next + 1);
        if (nv.distance > nv2.distance) { next++; nv = nv2; }
        if (nv.distance < pv.distance) {
            hset(h, parent, nv);
            hset(h, next, pv);
            parent = next;
            next <<= 1;
        } else {
            return;
        }
    }
    if (next == count) {
        NodeDist nv = hget(h, next);
        if (nv.distance < pv.distance) {
            hset(h, parent, nv);
            hset(h, next, pv);
        }
    }
}

DEV NodeDist ndheap_pop(HeapRef h, int* count)
{
    if (*count == 0) return hget(h, 0);
    NodeDist top = hget(h, 1);
    hset(h, 1, hget(h, *count));
    /* the vacated slot beyond count is dead until the next insert writes
     * it (Heap.h swaps the old top there; nothing ever reads it) */
    (*count)--;
    ndheap_heapify(h, *count);
    return top;
}

DEV NodeDist ndheap_top(HeapRef h, int count) { return count == 0 ? hget(h, 0) : hget(h, 1); }

/* DistPriorityQueue (WorkSpace.h:167-225): 1-based bounded float max-heap
 * seeded with MaxDist; len starts at 1. */
DEV int dpq_insert(float* a, int* len, int cap, float dist)
{
    if (dist > a[1]) return 0;
    if (*len == cap) {
        a[1] = dist;
        int parent = 1, next = 2;
        while (next < *len) {
            if (a[next] < a[next + 1]) next++;
            if (a[next] > a[parent]) {
                float t = a[parent]; a[parent] = a[next]; a[next] = t;
                parent = next;
                next <<= 1;
            } else break;
        }
        if (next == *len && a[next] > a[parent]) {
            float t = a[parent]; a[parent] = a[next]; a[next] = t;
        }
    } else {
        int next = ++(*len), parent = next >> 1;
        while (parent > 0 && dist > a[parent]) {
            a[next] = a[parent];
            next = parent;
            parent >>= 1;
        }
        a[next] = dist;
    }
    return 1;
}

/* QueryResultSet (QueryResultSet.h:17-120): 0-based k-entry max-heap on
 * (dist, vid). */
DEV int qres_lt(QRes x, QRes y)
{
    return (x.dist < y.dist) || (x.dist == y.dist && x.vid < y.vid);
}

DEV void qrs_heapify(QRes* r, int count)
{
    int parent = 0, next = 1, maxidx = count - 1;
    while (next < maxidx) {
        if (qres_lt(r[next], r[next + 1])) next++;
        if (qres_lt(r[parent], r[next])) {
            QRes t = r[next]; r[next] = r[parent]; r[parent] = t;
            parent = next;
            next = (parent << 1) + 1;
        } else break;
    }
    if (next == maxidx && qres_lt(r[parent], r[next])) {
        QRes t = r[next]; r[next] = r[parent]; r[parent] = t;
    }
}

DEV int qrs_add(QRes* r, int k, int32_t vid, float dist)
{
    if (dist < r[0].dist || (dist == r[0].dist && vid < r[0].vid)) {
        r[0].vid = vid; r[0].dist = dist;
        qrs_heapify(r, k);
        return 1;
    }
    return 0;
}

DEV void qrs_sort(QRes* r, int k)
{
    for (int i = k - 1; i >= 0; i--) {
        QRes t = r[0]; r[0] = r[i]; r[i] = t;
        qrs_heapify(r, i);
    }
}

/* ------------------------------------------------------------------ *
 * visited set — per-query global open-addressing table, atomicCAS
 * claim (set membership only; probe order is not part of the parity
 * contract — OptHashPosVector is a pure set, WorkSpace.h:113).
 * ------------------------------------------------------------------ */

/* returns 1 if idx was already present; inserts otherwise */
DEV int visited_test_insert(int32_t* tab, uint32_t mask, int32_t idx, int* oflow)
{
    uint32_t key = (uint32_t)(idx + 1);
    uint32_t h = (key * 2654435761u) & mask;
    for (int probe = 0; probe < 4096; probe++) {
        int32_t cur = atomicCAS(&tab[h], 0, (int32_t)key);
        if (cur == 0) return 0;
        if (cur == (int32_t)key) return 1;
        h = (h + 1) & mask;
    }
    *oflow = 1;
    return 1;
}

/* ------------------------------------------------------------------ *
 * distances — one candidate per 16-lane group, reference order
 * ------------------------------------------------------------------ */

/* float: the AVX512 chunk/fold order of DistanceUtils.cpp:650 (L2) /
 * the cosine analog, with fused lane accumulate (fmaf) as compiled in
 * the reference build (see oracle/sptag_oracle.c header). Result valid
 * on every lane of the group.
 *
 * The fixed-dim form fully unrolls the 16-element chunk loop so the
 * per-candidate global loads all issue before the fma chain waits on
 * them (the rolled loop costs one memory latency PER CHUNK — 8x the
 * stalls at d=128). Dims must be multiples of 16. */
template <int DM, int DFIX>
DEV float dist_f32_fix(const float* __restrict__ q, const float* __restrict__ v)
{
    const int g = threadIdx.x & 15;
    constexpr int C = DFIX / 16;
    float x[C], y[C];
#pragma unroll
    for (int c = 0; c < C; c++) { x[c] = q[c * 16 + g]; y[c] = v[c * 16 + g]; }
    float a = 0.0f;
#pragma unroll
    for (int c = 0; c < C; c++)
        a = (DM == DM_L2) ? fmaf(x[c] - y[c], x[c] - y[c], a) : fmaf(x[c], y[c], a);
    a = a + __shfl_down(a, 8, 16);
    a = a + __shfl_down(a, 4, 16);
    float sv = ((__shfl(a, 0, 16) + __shfl(a, 1, 16)) + __shfl(a, 2, 16)) + __shfl(a, 3, 16);
    return (DM == DM_L2) ? sv : 1.0f - sv;
}

template <int DM>
DEV float ref_dist_grp_f32(const float* __restrict__ q,
                           const float* __restrict__ v, int d)
{
    switch (d) {
    case 32:  return dist_f32_fix<DM, 32>(q, v);
    case 48:  return dist_f32_fix<DM, 48>(q, v);
    case 64:  return dist_f32_fix<DM, 64>(q, v);
    case 96:  return dist_f32_fix<DM, 96>(q, v);
    case 128: return dist_f32_fix<DM, 128>(q, v);
    case 256: return dist_f32_fix<DM, 256>(q, v);
    case 512: return dist_f32_fix<DM, 512>(q, v);
    case 768: return dist_f32_fix<DM, 768>(q, v);
    default: break;
    }
    const int g = threadIdx.x & 15;
    float a = 0.0f;
    const int nd16 = (d >> 4) << 4;
    for (int i = 0; i < nd16; i += 16) {
        float x = q[i + g], y = v[i + g];
        a = (DM == DM_L2) ? fmaf(x - y, x - y, a) : fmaf(x, y, a);
    }
    a = a + __shfl_down(a, 8, 16);           /* a8[g] on lanes g<8 */
    const int nd8 = (d >> 3) << 3;
    for (int i = nd16; i < nd8; i += 8) {
        if (g < 8) {
            float x = q[i + g], y = v[i + g];
            a = (DM == DM_L2) ? fmaf(x - y, x - y, a) : fmaf(x, y, a);
        }
    }
    a = a + __shfl_down(a, 4, 16);           /* a4[g] on lanes g<4 */
    const int nd4 = (d >> 2) << 2;
    for (int i = nd8; i < nd4; i += 4) {
        if (g < 4) {
            float x = q[i + g], y = v[i + g];
            a = (DM == DM_L2) ? fmaf(x - y, x - y, a) : fmaf(x, y, a);
        }
    }
    float s = ((__shfl(a, 0, 16) + __shfl(a, 1, 16)) + __shfl(a, 2, 16)) + __shfl(a, 3, 16);
    for (int i = nd4; i < d; i++) {
        float x = q[i], y = v[i];
        s = (DM == DM_L2) ? fmaf(x - y, x - y, s) : fmaf(x, y, s);
    }
    return (DM == DM_L2) ? s : 1.0f - s;
}

/* int8: exact integer accumulation (order-free; partials exact in the
 * reference's float lanes for dim*254^2 < 2^24 — guarded host-side).
 * The dword path runs on the CDNA4 int8 dot units (v_dot4c_i32_i8 via
 * __builtin_amdgcn_sdot4) — the gfx950 analog of the reference's own GPU
 * __dp4a int8 path (cuda/Distance.hxx:84-101). Integer dots are exact and
 * order-free, so results stay bit-identical to the byte-wise form:
 * L2 uses sum (xa-xb)^2 = dot(a,a) - 2 dot(a,b) + dot(b,b), every term
 * < 2^25 at the guarded dims. */
template <int DM>
DEV float ref_dist_grp_i8(const int8_t* __restrict__ q,
                          const int8_t* __restrict__ v, int d)
{
    const int g = threadIdx.x & 15;
    int s = 0;
    if ((d & 3) == 0) {
        const int W = d >> 2;
        const int32_t* qw = (const int32_t*)q;
        const int32_t* vw = (const int32_t*)v;
        if (DM == DM_COSINE) {
            for (int w = g; w < W; w += 16)
                s = __builtin_amdgcn_sdot4(qw[w], vw[w], s, false);
        } else {
            int aa = 0, ab = 0, bb = 0;
            for (int w = g; w < W; w += 16) {
                int32_t a = qw[w], b = vw[w];
                aa = __builtin_amdgcn_sdot4(a, a, aa, false);
                ab = __builtin_amdgcn_sdot4(a, b, ab, false);
                bb = __builtin_amdgcn_sdot4(b, b, bb, false);
            }
            s = aa - 2 * ab + bb;
        }
    } else {
        for (int i = g; i < d; i += 16) {
            int xa = q[i], xb = v[i];
            s += (DM == DM_L2) ? (xa - xb) * (xa - xb) : xa * xb;
        }
    }
    s += __shfl_xor(s, 8, 16);
    s += __shfl_xor(s, 4, 16);
    s += __shfl_xor(s, 2, 16);
    s += __shfl_xor(s, 1, 16);
    return (DM == DM_L2) ? (float)s : (float)(16129 - s);
}

/* load/compute split of the fixed-dim distance, for software-pipelining
 * candidate rounds (loads of round r+1 issue while round r reduces).
 * Only the candidate's elements are register-resident: the query side is
 * re-read from LDS in the reduce (cheap ds_reads), keeping the pipeline's
 * register cost at C floats per fragment. */
template <int DM, int DFIX>
struct FragF32 {
    static constexpr int C = DFIX / 16;
    float y[C];
    DEV void load(const float* __restrict__ v)
    {
        const int g = threadIdx.x & 15;
#pragma unroll
        for (int c = 0; c < C; c++) y[c] = v[c * 16 + g];
    }
    DEV float reduce(const float* __restrict__ q) const
    {
        const int g = threadIdx.x & 15;
        float a = 0.0f;
#pragma unroll
        for (int c = 0; c < C; c++) {
            float x = q[c * 16 + g];
            a = (DM == DM_L2) ? fmaf(x - y[c], x - y[c], a) : fmaf(x, y[c], a);
        }
        a = a + __shfl_down(a, 8, 16);
        a = a + __shfl_down(a, 4, 16);
        float sv = ((__shfl(a, 0, 16) + __shfl(a, 1, 16)) + __shfl(a, 2, 16)) +
                   __shfl(a, 3, 16);
        return (DM == DM_L2) ? sv : 1.0f - sv;
    }
};

template <typename T, int DM>
DEV float ref_dist_grp(const T* q, const T* v, int d);
template <> DEV float ref_dist_grp<float, DM_L2>(const float* q, const float* v, int d)
{ return ref_dist_grp_f32<DM_L2>(q, v, d); }
template <> DEV float ref_dist_grp<float, DM_COSINE>(const float* q, const float* v, int d)
{ return ref_dist_grp_f32<DM_COSINE>(q, v, d); }
template <> DEV float ref_dist_grp<int8_t, DM_L2>(const int8_t* q, const int8_t* v, int d)
{ return ref_dist_grp_i8<DM_L2>(q, v, d); }
template <> DEV float ref_dist_grp<int8_t, DM_COSINE>(const int8_t* q, const int8_t* v, int d)
{ return ref_dist_grp_i8<DM_COSINE>(q, v, d); }

/* ------------------------------------------------------------------ *
 * per-query context
 * ------------------------------------------------------------------ */

struct SerialState {          /* in LDS; lane 0 writes, wave reads after sync */
    int ng_count, spt_count, dpq_len, checked;
    int oflow, terminate, break_flag, want_tree;
    int tree_checked, no_better;   /* KDT counters */
    float fbcast;                  /* KDT upper-bound broadcast */
    NodeDist popped;
};

template <typename T>
struct QCtx {
    const DevIndex* di;
    const SearchCfg* cfg;
    const T* qlds;            /* query vector staged in LDS */
    float* dstage;            /* MAX_DEG staged distances */
    int32_t* istage;          /* MAX_DEG staged ids / compacted lanes */
    QRes* qrs;                /* k results */
    float* dpq;               /* dpq_cap+1 */
    HeapRef ng, spt;
    int32_t* vtab;            /* this query's visited table */
    uint32_t vmask;
    SerialState* ss;
    int lane;
};

template <typename T>
DEV const T* vec_at(const DevIndex& di, int32_t v)
{
    return (const T*)di.vectors + (size_t)v * di.dim;
}

DEV int not_deleted(const DevIndex& di, int32_t v)
{
    /* StaticDispatch::CheckIfNotDeleted vs AlwaysTrue (BKTIndex.cpp:437,
     * dispatch :471-507): deletes ignored when the index has none. */
    if (!di.has_deleted) return 1;
    return di.deleted[v] == 0;
}

/* pipelined fixed-dim staging: group g handles candidates g, g+4, ...,
 * double-buffering fragments so the next round's global loads are in
 * flight while the current round reduces. */
template <int DM, int DFIX>
DEV void stage_dists_pipe_f32(QCtx<float>& c, int cnt)
{
    const DevIndex& di = *c.di;
    const int grp = c.lane >> 4;
    FragF32<DM, DFIX> A, B;
    int j = grp;
    if (j < cnt) A.load(vec_at<float>(di, c.istage[j]));
    while (j < cnt) {
        int jn = j + 4;
        if (jn < cnt) B.load(vec_at<float>(di, c.istage[jn]));
        float dv = A.reduce(c.qlds);
        if ((c.lane & 15) == 0) c.dstage[j] = dv;
        A = B;
        j = jn;
    }
    __syncthreads();
}

/* stage distances of `cnt` (<=64) data vectors whose ids are in istage[0..cnt)
 * into dstage[0..cnt); whole-wave cooperative, 4 candidates in flight. */
template <typename T, int DM>
DEV void stage_dists(QCtx<T>& c, int cnt)
{
    if constexpr (sizeof(T) == 4) {
        QCtx<float>& cf = reinterpret_cast<QCtx<float>&>(c);
        switch (c.di->dim) {
        /* pipelining holds 2 fragments (2*C VGPRs) live; C <= 8 only */
        case 32:  stage_dists_pipe_f32<DM, 32>(cf, cnt); return;
        case 48:  stage_dists_pipe_f32<DM, 48>(cf, cnt); return;
        case 64:  stage_dists_pipe_f32<DM, 64>(cf, cnt); return;
        case 96:  stage_dists_pipe_f32<DM, 96>(cf, cnt); return;
        case 128: stage_dists_pipe_f32<DM, 128>(cf, cnt); return;
        default: break;
        }
    }
    for (int r = 0; r < cnt; r += 4) {
        int j = r + (c.lane >> 4);
        if (j < cnt) {
            float dv = ref_dist_grp<T, DM>(c.qlds, vec_at<T>(*c.di, c.istage[j]), c.di->dim);
            if ((c.lane & 15) == 0) c.dstage[j] = dv;
        }
    }
    __syncthreads();
}

/* Warm the GLOBAL parent-chain levels of the next `cnt` SPT insertion
 * slots (slot for the i-th insert is spt_count+1+i while the heap is not
 * full — seed-phase heaps never reach ref_cap in practice, and the warm is
 * advisory: a wrong slot guess only wastes a touch). Lane-0's serial
 * percolate-up then reads L1/L2-warm lines instead of paying an HBM
 * round trip per level — the seed phase was 31% of kernel cycles with
 * cold parent reads. */
template <typename T>
DEV void spt_warm_parents(QCtx<T>& c, int cnt)
{
    int loc = c.ss->spt_count + 1 + c.lane;
    uint32_t wsink = 0;
    if (c.lane < cnt) {
        int par = loc >> 1;
        while (par > c.spt.lsplit) {
            wsink += ((const uint8_t*)&c.spt.a[par])[0];
            par >>= 1;
        }
    }
    asm volatile("" :: "v"(wsink));
}

/* BKTree.h:772 SearchTrees — wave-cooperative, serial decisions on lane 0. */
template <typename T, int DM>
DEV void search_trees_dev(QCtx<T>& c, int limit)
{
    const DevIndex& di = *c.di;
    for (;;) {
        __syncthreads();
        if (c.ss->spt_count <= 0 || c.ss->oflow) break;
        if (c.lane == 0) c.ss->popped = ndheap_pop(c.spt, &c.ss->spt_count);
        __syncthreads();
        NodeDist bcell = c.ss->popped;
        int32_t center = di.tree_nodes[(size_t)bcell.node * 3 + 0];
        int32_t cs = di.tree_nodes[(size_t)bcell.node * 3 + 1];
        int32_t ce = di.tree_nodes[(size_t)bcell.node * 3 + 2];
        if (cs < 0) {
            if (c.lane == 0) {
                if (!visited_test_insert(c.vtab, c.vmask, center, &c.ss->oflow)) {
                    c.ss->checked++;
                    ndheap_insert(c.ng, &c.ss->ng_count, NodeDist{center, bcell.distance}, &c.ss->oflow);
                }
                c.ss->break_flag = (c.ss->checked >= limit);
            }
            __syncthreads();
            if (c.ss->break_flag) break;
        } else {
            if (c.lane == 0) {
                if (!visited_test_insert(c.vtab, c.vmask, center, &c.ss->oflow)) {
                    ndheap_insert(c.ng, &c.ss->ng_count, NodeDist{center, bcell.distance}, &c.ss->oflow);
                }
            }
            for (int base = cs; base < ce; base += 64) {
                int cnt = min(64, ce - base);
                if (c.lane < cnt) c.istage[c.lane] = di.tree_nodes[(size_t)(base + c.lane) * 3];
                __syncthreads();
                /* Issue ALL children's vector lines before the staged
                 * distance rounds start: every one of them is needed (no
                 * speculation waste), so the 16 round-pipeline latencies
                 * collapse into one. Seed-phase tree-center gathers were
                 * 31% of kernel cycles. */
                uint32_t ts = 0;
                if (c.lane < cnt) {
                    const char* vp =
                        (const char*)vec_at<T>(di, c.istage[c.lane]);
                    const int nl =
                        ((int)((unsigned)di.dim * sizeof(T)) + 127) >> 7;
                    for (int l = 0; l < nl; l++)
                        ts += (uint32_t)(uint8_t)vp[(size_t)l << 7];
                }
                stage_dists<T, DM>(c, cnt);
                asm volatile("" :: "v"(ts));   /* sink after the staging so
                                                  the touches and the first
                                                  rounds overlap */
                spt_warm_parents(c, cnt);
                if (c.lane == 0)
                    for (int i = 0; i < cnt; i++)
                        ndheap_insert(c.spt, &c.ss->spt_count,
                                      NodeDist{base + i, c.dstage[i]}, &c.ss->oflow);
                __syncthreads();
            }
        }
    }
    __syncthreads();
}

/* BKTree.h:697 InitSearchTrees (m_bfs = 0). */
template <typename T, int DM>
DEV void init_search_trees_dev(QCtx<T>& c)
{
    const DevIndex& di = *c.di;
    for (int t = 0; t < di.ntrees; t++) {
        int32_t root = di.tree_start[t];
        int32_t center = di.tree_nodes[(size_t)root * 3 + 0];
        int32_t cs = di.tree_nodes[(size_t)root * 3 + 1];
        int32_t ce = di.tree_nodes[(size_t)root * 3 + 2];
        if (cs < 0) {
            if (c.lane < 16) {
                float dv = ref_dist_grp<T, DM>(c.qlds, vec_at<T>(di, center), di.dim);
                if (c.lane == 0)
                    ndheap_insert(c.spt, &c.ss->spt_count, NodeDist{root, dv}, &c.ss->oflow);
            }
            __syncthreads();
        } else {
            for (int base = cs; base < ce; base += 64) {
                int cnt = min(64, ce - base);
                if (c.lane < cnt) c.istage[c.lane] = di.tree_nodes[(size_t)(base + c.lane) * 3];
                __syncthreads();
                /* Issue ALL children's vector lines before the staged
                 * distance rounds start: every one of them is needed (no
                 * speculation waste), so the 16 round-pipeline latencies
                 * collapse into one. Seed-phase tree-center gathers were
                 * 31% of kernel cycles. */
                uint32_t ts = 0;
                if (c.lane < cnt) {
                    const char* vp =
                        (const char*)vec_at<T>(di, c.istage[c.lane]);
                    const int nl =
                        ((int)((unsigned)di.dim * sizeof(T)) + 127) >> 7;
                    for (int l = 0; l < nl; l++)
                        ts += (uint32_t)(uint8_t)vp[(size_t)l << 7];
                }
                stage_dists<T, DM>(c, cnt);
                asm volatile("" :: "v"(ts));   /* sink after the staging so
                                                  the touches and the first
                                                  rounds overlap */
                spt_warm_parents(c, cnt);
                if (c.lane == 0)
                    for (int i = 0; i < cnt; i++)
                        ndheap_insert(c.spt, &c.ss->spt_count,
                                      NodeDist{base + i, c.dstage[i]}, &c.ss->oflow);
                __syncthreads();
            }
        }
    }
}

/* ------------------------------------------------------------------ *
 * main search kernel
 * ------------------------------------------------------------------ */

#ifndef SPTAG_LB_WAVES
#define SPTAG_LB_WAVES 1   /* min waves/SIMD hint; raise to cap VGPRs */
#endif

/* Per-phase cycle marks, diagnostic builds only (PROF template arg).
 * One wave per workgroup, so lane-0 clock64 spans are wave-accurate. */
#define PMARK() do { if (PROF && lane == 0) pmark = clock64(); } while (0)
#define PACC(i) do { if (PROF && lane == 0) { \
        uint64_t now_ = clock64(); pc[i] += now_ - pmark; pmark = now_; } \
    } while (0)

template <typename T, int DM, bool LDSHEAP, bool PROF = false>
__global__ __launch_bounds__(64, SPTAG_LB_WAVES)
void bkt_search_kernel(DevIndex di, SearchCfg cfg, SearchBufs bufs)
{
    const int q = blockIdx.x;
    if (q >= cfg.nq) return;
    const int lane = threadIdx.x;
    uint64_t pc[10] = {};
    uint64_t pmark = 0, pc_ng = 0;

    extern __shared__ char smem[];
    size_t off = 0;
    T* qlds = (T*)(smem + off);
    off += ((size_t)di.dim * sizeof(T) + 15) & ~15ul;
    float* dstage = (float*)(smem + off); off += MAX_DEG * 4;
    int32_t* istage = (int32_t*)(smem + off); off += MAX_DEG * 4;
    QRes* qrs = (QRes*)(smem + off); off += (size_t)cfg.k * 8;
    float* dpq = (float*)(smem + off); off += ((size_t)cfg.dpq_cap + 1) * 4;
    SerialState* ss = (SerialState*)(smem + off); off += 64;

    HeapRef ng, spt;
    ng.cap = cfg.ng_cap;  ng.ref_cap = cfg.max_check * 30;   /* WorkSpace.h:265 */
    ng.lds = nullptr; ng.lsplit = 0;
    spt.cap = cfg.spt_cap; spt.ref_cap = cfg.max_check * 10;
    /* SPT (tree) heap: hot top levels in an LDS tier, tail in global
     * scratch — full reference capacity without the LDS cost of the whole
     * heap (its all-global round-1 form made the seed phase 41.5% of
     * kernel cycles). */
    spt.a = (NodeDist*)bufs.gheap_spt + (size_t)q * (cfg.spt_cap + 1);
    spt.lds = (NodeDist*)(smem + off); off += ((size_t)SPT_LDS_TIER + 1) * 8;
    spt.lsplit = SPT_LDS_TIER;
    if (LDSHEAP) {
        ng.a = (NodeDist*)(smem + off); off += ((size_t)cfg.ng_cap + 1) * 8;
    } else {
        ng.a = (NodeDist*)bufs.gheap_ng + (size_t)q * (cfg.ng_cap + 1);
    }

    /* stage query */
    const T* gq = (const T*)bufs.queries + (size_t)q * di.dim;
    for (int i = lane; i < di.dim; i += 64) qlds[i] = gq[i];

    if (lane == 0) {
        ss->ng_count = 0; ss->spt_count = 0; ss->dpq_len = 1; ss->checked = 0;
        ss->oflow = 0; ss->terminate = 0; ss->break_flag = 0; ss->want_tree = 0;
        ng.a[0] = NodeDist{-1, MAXDIST};   /* Heap empty-top sentinel */
        hset(spt, 0, NodeDist{-1, MAXDIST});
        dpq[0] = MAXDIST;                  /* DistPriorityQueue seed element */
        for (int i = 0; i < cfg.k; i++) qrs[i] = QRes{-1, MAXDIST};
    }
    __syncthreads();

    QCtx<T> c{&di, &cfg, qlds, dstage, istage, qrs, dpq, ng, spt,
              bufs.visited + (size_t)q * cfg.vcap, (uint32_t)(cfg.vcap - 1),
              ss, lane};

    /* DistPriorityQueue (WorkSpace.h:167-225) as a FLAT bounded max-set:
     * its observable behavior (the kept multiset and its current maximum)
     * is independent of heap layout, so the m_Results state here is an
     * unordered LDS array + uniform registers replicated wave-wide. */
    float dmax = MAXDIST;
    int dlen = 1, dargmax = 0;

    PMARK();
    init_search_trees_dev<T, DM>(c);
    __syncthreads();
    search_trees_dev<T, DM>(c, cfg.init_pivots);
    PACC(0);

    const int deg = di.deg;
    const int checkPos = deg - 1;
    const int nline = ((int)((unsigned)di.dim * sizeof(T)) + 127) >> 7;
    int popped = 0;
    uint32_t spec_sink = 0;   /* keeps lookahead loads alive across barriers */

    for (;;) {
        __syncthreads();
        if (ss->ng_count <= 0 || ss->terminate || ss->oflow) break;
        PMARK();
        if (lane == 0) ss->popped = ndheap_pop(c.ng, &ss->ng_count);
        __syncthreads();
        PACC(1);
        asm volatile("" :: "v"(spec_sink));   /* lookahead loads land here */
        popped++;
        NodeDist gnode = ss->popped;
        const int32_t* row = di.graph + (size_t)gnode.node * deg;
        int32_t nn = lane < deg ? row[lane] : -1;
        uint64_t negm = __ballot(lane >= deg || nn < 0);
        int firstneg = negm ? (int)__builtin_ctzll(negm) : 64;
        int32_t checkNode = __shfl(nn, checkPos);
        PACC(2);

        if (lane == 0) {
            /* BKTIndex.cpp:290-331: result/termination block. The dispatch
             * flags (BKTIndex.cpp:471-507): searchDeleted makes the delete
             * filter AlwaysTrue; searchDuplicated selects CheckDup (walk
             * the duplicate chain until AddPoint rejects) vs NeverDup (add
             * the center only). */
            if (gnode.distance <= qrs[0].dist) {
                if (checkNode < -1) {
                    const int32_t* tn = &di.tree_nodes[(size_t)(-2 - checkNode) * 3];
                    int32_t i = -tn[1];
                    int32_t tmpNode = gnode.node;
                    do {
                        if (cfg.search_deleted || not_deleted(di, tmpNode)) {
                            if (cfg.search_dup) {
                                if (!qrs_add(qrs, cfg.k, tmpNode, gnode.distance)) break;
                            } else {
                                qrs_add(qrs, cfg.k, tmpNode, gnode.distance);
                                break;
                            }
                        }
                        if (i <= 0) break;
                        tmpNode = di.tree_nodes[(size_t)i * 3];
                    } while (i++ < tn[2]);
                } else {
                    if (cfg.search_deleted || not_deleted(di, gnode.node))
                        qrs_add(qrs, cfg.k, gnode.node, gnode.distance);
                }
            } else {
                if (cfg.search_deleted || not_deleted(di, gnode.node)) {
                    if (gnode.distance > dmax || ss->checked > cfg.max_check)
                        ss->terminate = 1;
                }
            }
        }
        __syncthreads();
        PACC(3);
        if (ss->terminate) break;

        /* neighbor expansion (BKTIndex.cpp:333-345): compact the unvisited
         * neighbors (row order preserved) and stage their distances.
         * spec bit0: while the visited-CAS round is in flight, warm every
         * neighbor's vector lines (one byte per 128B line) — the unvisited
         * ones are exactly what stage_dists reads next, so the two
         * dependent HBM round trips (CAS, vector gather) overlap; the
         * re-visited rows were read recently and mostly hit L2/L3. */
        int already = 1;
        uint32_t tsink = 0;
        if (lane < firstneg) {
            if (cfg.spec_flags & 1) {
                const char* vp = (const char*)vec_at<T>(di, nn);
                for (int l = 0; l < nline; l++)
                    tsink += (uint32_t)(uint8_t)vp[(size_t)l << 7];
            }
            already = visited_test_insert(c.vtab, c.vmask, nn, &ss->oflow);
        }
        uint64_t candm = __ballot(lane < firstneg && !already);
        asm volatile("" :: "v"(tsink));
        int ncand = __popcll(candm);
        if ((candm >> lane) & 1) {
            int pos = __popcll(candm & ((1ull << lane) - 1));
            istage[pos] = nn;
        }
        __syncthreads();
        PACC(4);
        stage_dists<T, DM>(c, ncand);
        PACC(5);
        /* insert phase (BKTIndex.cpp:337-344). m_Results (the flat max-set)
         * appends in O(1); a replace costs one wave-parallel max rescan
         * (any max-valued slot may be replaced — same multiset). The NG
         * frontier heap stays an exact Heap.h emulation on lane 0: its pop
         * order on equal keys is part of the parity contract. */
        for (int r = 0; r < ncand; r++) {
            float dv = dstage[r];
            if (lane == 0) ss->checked++;
            if (!(dv > dmax)) {
                if (dlen < cfg.dpq_cap) {
                    if (lane == 0) dpq[dlen] = dv;
                    dlen++;
                } else {
                    if (lane == 0) dpq[dargmax] = dv;
                    __syncthreads();
                    float m = -MAXDIST;
                    int mi = 0;
                    for (int i = lane; i < cfg.dpq_cap; i += 64) {
                        float x = dpq[i];
                        if (x > m) { m = x; mi = i; }
                    }
                    for (int o2 = 32; o2 > 0; o2 >>= 1) {
                        float om = __shfl_down(m, o2);
                        int omi = __shfl_down(mi, o2);
                        if (om > m) { m = om; mi = omi; }
                    }
                    dmax = __shfl(m, 0);
                    dargmax = __shfl(mi, 0);
                }
                if (lane == 0) {
                    if (PROF) {
                        uint64_t t0_ = clock64();
                        ndheap_insert(c.ng, &ss->ng_count,
                                      NodeDist{istage[r], dv}, &ss->oflow);
                        pc_ng += clock64() - t0_;
                    } else {
                        ndheap_insert(c.ng, &ss->ng_count,
                                      NodeDist{istage[r], dv}, &ss->oflow);
                    }
                }
            }
        }
        if (lane == 0) {
            /* dynamic pivots (BKTIndex.cpp:346-349) */
            ss->want_tree = (ndheap_top(c.ng, ss->ng_count).distance >
                             ndheap_top(c.spt, ss->spt_count).distance);
        }
        __syncthreads();
        PACC(6);
        if (ss->want_tree)
            search_trees_dev<T, DM>(c, cfg.other_pivots + ss->checked);
        PACC(7);
        /* spec bit1: lookahead on the next pop (perf-only — the frontier
         * top rarely changes between here and the next iteration's pop).
         * Read its adjacency row, probe the visited table READ-ONLY (first
         * slot; stale/partial answers only cost an extra touch), and start
         * the unvisited neighbors' vector loads. The loads stay in flight
         * across the barrier — spec_sink is consumed after the next pop. */
        if (cfg.spec_flags & 2) {
            if (lane == 0) ss->popped = ndheap_top(c.ng, ss->ng_count);
            __syncthreads();
            int32_t nxt = ss->popped.node;
            if (nxt >= 0 && lane < deg) {
                int32_t t = di.graph[(size_t)nxt * deg + lane];
                if (t >= 0) {
                    uint32_t key = (uint32_t)(t + 1);
                    uint32_t h = (key * 2654435761u) & c.vmask;
                    if (c.vtab[h] != (int32_t)key) {
                        const char* vp = (const char*)vec_at<T>(di, t);
                        for (int l = 0; l < nline; l++)
                            spec_sink += (uint32_t)(uint8_t)vp[(size_t)l << 7];
                    }
                }
            }
        }
        PACC(8);
    }
    __syncthreads();

    if (lane == 0) {
        bufs.oflow[q] = ss->oflow;
        PMARK();
        qrs_sort(qrs, cfg.k);
        PACC(9);
        if (bufs.stats) {
            if (PROF) {
                int32_t* s = bufs.stats + (size_t)q * PROF_STATS;
                s[0] = ss->checked;
                s[1] = popped;
#pragma unroll
                for (int i = 0; i < 10; i++) s[2 + i] = (int32_t)pc[i];
                s[12] = (int32_t)pc_ng;   /* NG-insert share of `insert` */
                s[13] = s[14] = s[15] = 0;
            } else {
                bufs.stats[(size_t)q * 2 + 0] = ss->checked;
                bufs.stats[(size_t)q * 2 + 1] = popped;
            }
        }
    }
    __syncthreads();
    for (int i = lane; i < cfg.k; i += 64) {
        bufs.out_vids[(size_t)q * cfg.k + i] = qrs[i].vid;
        bufs.out_dists[(size_t)q * cfg.k + i] = qrs[i].dist;
    }
}

/* ------------------------------------------------------------------ *
 * Iterative (streaming) BKT search — reference SearchIterative
 * (src/Core/BKT/BKTIndex.cpp:354-427) driven per ResultIterator::Next
 * (src/Core/ResultIterator.cpp) with per-call ResetResult
 * (BKTIndex.cpp:663). One wave per iterator; ALL traversal state
 * (frontier/tree heaps at the reference's own capacities, visited table,
 * results queue, counters) persists in global scratch between calls.
 * ------------------------------------------------------------------ */

template <typename T, int DM>
__global__ __launch_bounds__(64, SPTAG_LB_WAVES)
void bkt_iter_kernel(DevIndex di, SearchCfg cfg, IterBufs ib, int batch)
{
    const int q = blockIdx.x;
    if (q >= cfg.nq) return;
    const int lane = threadIdx.x;

    extern __shared__ char smem[];
    size_t off = 0;
    T* qlds = (T*)(smem + off);
    off += ((size_t)di.dim * sizeof(T) + 15) & ~15ul;
    float* dstage = (float*)(smem + off); off += MAX_DEG * 4;
    int32_t* istage = (int32_t*)(smem + off); off += MAX_DEG * 4;
    QRes* qrs = (QRes*)(smem + off); off += (size_t)batch * 8;
    SerialState* ss = (SerialState*)(smem + off); off += 64;

    float* dpq = ib.dpq + (size_t)q * (cfg.dpq_cap + 1);
    HeapRef ng, spt;
    ng.cap = cfg.ng_cap;  ng.ref_cap = cfg.ng_cap;   /* reference capacities */
    spt.cap = cfg.spt_cap; spt.ref_cap = cfg.spt_cap;
    /* untiered: heap state persists in global scratch across Next() calls */
    ng.lds = nullptr; ng.lsplit = 0;
    spt.lds = nullptr; spt.lsplit = 0;
    ng.a = (NodeDist*)ib.gheap_ng + (size_t)q * (cfg.ng_cap + 1);
    spt.a = (NodeDist*)ib.gheap_spt + (size_t)q * (cfg.spt_cap + 1);

    const T* gq = (const T*)ib.queries + (size_t)q * di.dim;
    for (int i = lane; i < di.dim; i += 64) qlds[i] = gq[i];

    IterState st = ib.state[q];
    if (lane == 0) {
        ss->ng_count = st.ng_count;
        ss->spt_count = st.spt_count;
        ss->oflow = st.oflow;
        /* ResetResult (WorkSpace.h:276): fresh m_Results + zeroed counters;
         * queues, visited set and the sticky relaxed flag persist. */
        ss->dpq_len = 1;
        dpq[1] = MAXDIST;
        ss->checked = 0;
        ss->tree_checked = 0;
        ss->terminate = 0; ss->break_flag = 0; ss->want_tree = 0;
        ss->no_better = st.relaxed;      /* reuse slot: sticky relaxedMono */
        for (int i = 0; i < batch; i++) qrs[i] = QRes{-1, MAXDIST};
        if (st.first) {
            ng.a[0] = NodeDist{-1, MAXDIST};
            spt.a[0] = NodeDist{-1, MAXDIST};
        }
    }
    __syncthreads();

    QCtx<T> c{&di, &cfg, qlds, dstage, istage, qrs, dpq, ng, spt,
              ib.visited + (size_t)q * cfg.vcap, (uint32_t)(cfg.vcap - 1),
              ss, lane};

    if (st.first) {
        init_search_trees_dev<T, DM>(c);
        __syncthreads();
        search_trees_dev<T, DM>(c, cfg.init_pivots);
    }

    const int deg = di.deg;
    const int checkPos = deg - 1;
    int count = 0;

    for (;;) {
        __syncthreads();
        if (ss->ng_count <= 0 || ss->oflow || count >= batch) break;
        if (lane == 0) ss->popped = ndheap_pop(c.ng, &ss->ng_count);
        __syncthreads();
        NodeDist gnode = ss->popped;
        const int32_t* row = di.graph + (size_t)gnode.node * deg;
        int32_t nn = lane < deg ? row[lane] : -1;
        uint64_t negm = __ballot(lane >= deg || nn < 0);
        int firstneg = negm ? (int)__builtin_ctzll(negm) : 64;
        int32_t checkNode = __shfl(nn, checkPos);

        if (lane == 0) {
            int cnt_new = count;
            if (not_deleted(di, gnode.node)) {
                qrs_add(qrs, batch, gnode.node, gnode.distance);
                cnt_new = count + 1;
                if (gnode.distance > dpq[1] || ss->checked > cfg.max_check)
                    ss->no_better = 1;                   /* relaxedMono */
            }
            ss->fbcast = __int_as_float(cnt_new);
        }
        __syncthreads();
        count = __float_as_int(ss->fbcast);

        if (checkNode < -1) {
            /* iterative duplicate chain (BKTIndex.cpp:387-405): duplicates
             * enter the frontier; stage chunk distances, lane 0 inserts */
            const int32_t* tn = &di.tree_nodes[(size_t)(-2 - checkNode) * 3];
            int32_t cs0 = -tn[1], ce0 = tn[2];
            for (int base = cs0; base < ce0; base += 64) {
                int cnt = min(64, ce0 - base);
                if (lane < cnt)
                    istage[lane] = di.tree_nodes[(size_t)(base + lane) * 3];
                __syncthreads();
                stage_dists<T, DM>(c, cnt);
                if (lane == 0) {
                    for (int i = 0; i < cnt; i++) {
                        int32_t tmpNode = istage[i];
                        if (!not_deleted(di, tmpNode)) continue;
                        if (!visited_test_insert(c.vtab, c.vmask, tmpNode, &ss->oflow))
                            ndheap_insert(c.ng, &ss->ng_count,
                                          NodeDist{tmpNode, dstage[i]}, &ss->oflow);
                    }
                }
                __syncthreads();
            }
        }

        /* spec bit0 (see bkt kernel): warm neighbor vector lines during
         * the visited-CAS round. */
        int already = 1;
        uint32_t tsink = 0;
        if (lane < firstneg) {
            if (cfg.spec_flags & 1) {
                const char* vp = (const char*)vec_at<T>(di, nn);
                const int nline = ((int)((unsigned)di.dim * sizeof(T)) + 127) >> 7;
                for (int l = 0; l < nline; l++)
                    tsink += (uint32_t)(uint8_t)vp[(size_t)l << 7];
            }
            already = visited_test_insert(c.vtab, c.vmask, nn, &ss->oflow);
        }
        uint64_t candm = __ballot(lane < firstneg && !already);
        asm volatile("" :: "v"(tsink));
        int ncand = __popcll(candm);
        if ((candm >> lane) & 1) {
            int pos = __popcll(candm & ((1ull << lane) - 1));
            istage[pos] = nn;
        }
        __syncthreads();
        stage_dists<T, DM>(c, ncand);
        if (lane == 0) {
            for (int r = 0; r < ncand; r++) {
                float dv = dstage[r];
                ss->checked++;
                ndheap_insert(c.ng, &ss->ng_count, NodeDist{istage[r], dv}, &ss->oflow);
                dpq_insert(dpq, &ss->dpq_len, cfg.dpq_cap, dv);
            }
            ss->want_tree = (ndheap_top(c.ng, ss->ng_count).distance >
                             ndheap_top(c.spt, ss->spt_count).distance);
        }
        __syncthreads();
        if (ss->want_tree)
            search_trees_dev<T, DM>(c, cfg.other_pivots + ss->checked);
    }
    __syncthreads();

    if (lane == 0) {
        qrs_sort(qrs, batch);
        st.ng_count = ss->ng_count;
        st.spt_count = ss->spt_count;
        st.checked = ss->checked;
        st.tree_checked = ss->tree_checked;
        st.relaxed = ss->no_better;
        st.first = 0;
        st.oflow = ss->oflow;
        ib.state[q] = st;
        ib.out_counts[q] = count;
        ib.out_relaxed[q] = st.relaxed;
    }
    __syncthreads();
    for (int i = lane; i < batch; i += 64) {
        ib.out_vids[(size_t)q * batch + i] = qrs[i].vid;
        ib.out_dists[(size_t)q * batch + i] = qrs[i].dist;
    }
}

/* ------------------------------------------------------------------ *
 * KDT search (reference src/Core/KDT/KDTIndex.cpp:184-241 +
 * inc/Core/Common/KDTree.h:213-273). Shares the heaps/visited/top-k/
 * distance machinery; differs in the seed descent (kd-tree with
 * squared-split-plane lower bounds in the SPT queue), in the frontier
 * update (every computed neighbor enters the queue — no results-queue
 * gate) and in termination (continuous-no-better-propagation counter).
 * ------------------------------------------------------------------ */

struct KdtNode { int32_t left, right, split_dim; float split_value; };

/* KDTree.h:234-271 KDTSearch, iterative; lane 0 descends (serial pointer
 * chase), the wave's first 16-lane group computes the leaf distance. */
template <typename T, int DM>
DEV void kdt_search_node_dev(QCtx<T>& c, int32_t node, float dist_bound)
{
    const DevIndex& di = *c.di;
    const KdtNode* kn = (const KdtNode*)di.tree_nodes;
    SerialState* ss = c.ss;
    /* descent: lane 0 walks, broadcasting the final leaf via ss->popped */
    if (c.lane == 0) {
        while (node >= 0) {
            KdtNode tn = kn[node];
            float qv = (float)c.qlds[tn.split_dim];
            float diff = qv - tn.split_value;
            float other_bound = dist_bound + diff * diff;
            int32_t best = diff < 0.0f ? tn.left : tn.right;
            int32_t other = diff < 0.0f ? tn.right : tn.left;
            ndheap_insert(c.spt, &ss->spt_count, NodeDist{other, other_bound},
                          &ss->oflow);
            node = best;
        }
        int32_t index = -node - 1;
        if (index >= di.n) index = -1;
        else if (visited_test_insert(c.vtab, c.vmask, index, &ss->oflow)) index = -1;
        ss->popped.node = index;
    }
    __syncthreads();
    int32_t index = ss->popped.node;
    if (index >= 0) {
        float dv = 0.0f;
        if (c.lane < 16)
            dv = ref_dist_grp<T, DM>(c.qlds, vec_at<T>(di, index), di.dim);
        if (c.lane == 0) {
            ss->tree_checked++;
            ss->checked++;
            ndheap_insert(c.ng, &ss->ng_count, NodeDist{index, dv}, &ss->oflow);
        }
    }
    __syncthreads();
}

/* KDTree.h:219-231 SearchTrees: pop bounds until the leaf budget. */
template <typename T, int DM>
DEV void kdt_search_trees_dev(QCtx<T>& c, int limits)
{
    SerialState* ss = c.ss;
    for (;;) {
        __syncthreads();
        if (ss->spt_count <= 0 || ss->checked >= limits || ss->oflow) break;
        if (c.lane == 0) ss->popped = ndheap_pop(c.spt, &ss->spt_count);
        __syncthreads();
        NodeDist tcell = ss->popped;
        kdt_search_node_dev<T, DM>(c, tcell.node, tcell.distance);
    }
    __syncthreads();
}

template <typename T, int DM, bool LDSHEAP>
__global__ __launch_bounds__(64, SPTAG_LB_WAVES)
void kdt_search_kernel(DevIndex di, SearchCfg cfg, SearchBufs bufs)
{
    const int q = blockIdx.x;
    if (q >= cfg.nq) return;
    const int lane = threadIdx.x;

    extern __shared__ char smem[];
    size_t off = 0;
    T* qlds = (T*)(smem + off);
    off += ((size_t)di.dim * sizeof(T) + 15) & ~15ul;
    float* dstage = (float*)(smem + off); off += MAX_DEG * 4;
    int32_t* istage = (int32_t*)(smem + off); off += MAX_DEG * 4;
    QRes* qrs = (QRes*)(smem + off); off += (size_t)cfg.k * 8;
    float* dpq = (float*)(smem + off); off += ((size_t)cfg.dpq_cap + 1) * 4;
    SerialState* ss = (SerialState*)(smem + off); off += 64;

    HeapRef ng, spt;
    ng.cap = cfg.ng_cap;  ng.ref_cap = cfg.max_check * 30;
    ng.lds = nullptr; ng.lsplit = 0;
    spt.cap = cfg.spt_cap; spt.ref_cap = cfg.max_check * 10;
    spt.a = (NodeDist*)bufs.gheap_spt + (size_t)q * (cfg.spt_cap + 1);
    spt.lds = (NodeDist*)(smem + off); off += ((size_t)SPT_LDS_TIER + 1) * 8;
    spt.lsplit = SPT_LDS_TIER;
    if (LDSHEAP) {
        ng.a = (NodeDist*)(smem + off); off += ((size_t)cfg.ng_cap + 1) * 8;
    } else {
        ng.a = (NodeDist*)bufs.gheap_ng + (size_t)q * (cfg.ng_cap + 1);
    }

    const T* gq = (const T*)bufs.queries + (size_t)q * di.dim;
    for (int i = lane; i < di.dim; i += 64) qlds[i] = gq[i];

    if (lane == 0) {
        ss->ng_count = 0; ss->spt_count = 0; ss->dpq_len = 1; ss->checked = 0;
        ss->oflow = 0; ss->terminate = 0; ss->break_flag = 0; ss->want_tree = 0;
        ss->tree_checked = 0; ss->no_better = 0;
        ng.a[0] = NodeDist{-1, MAXDIST};
        hset(spt, 0, NodeDist{-1, MAXDIST});
        dpq[1] = MAXDIST;
        for (int i = 0; i < cfg.k; i++) qrs[i] = QRes{-1, MAXDIST};
    }
    __syncthreads();

    QCtx<T> c{&di, &cfg, qlds, dstage, istage, qrs, dpq, ng, spt,
              bufs.visited + (size_t)q * cfg.vcap, (uint32_t)(cfg.vcap - 1),
              ss, lane};

    /* InitSearchTrees (KDTree.h:213): every tree root at bound 0 */
    for (int t = 0; t < di.ntrees; t++)
        kdt_search_node_dev<T, DM>(c, di.tree_start[t], 0.0f);
    kdt_search_trees_dev<T, DM>(c, cfg.init_pivots);

    const int deg = di.deg;
    const int nline = ((int)((unsigned)di.dim * sizeof(T)) + 127) >> 7;
    int popped = 0;
    uint32_t spec_sink = 0;

    for (;;) {
        __syncthreads();
        if (ss->ng_count <= 0 || ss->terminate || ss->oflow) break;
        if (lane == 0) ss->popped = ndheap_pop(c.ng, &ss->ng_count);
        __syncthreads();
        asm volatile("" :: "v"(spec_sink));
        popped++;
        NodeDist gnode = ss->popped;
        const int32_t* row = di.graph + (size_t)gnode.node * deg;
        int32_t nn = lane < deg ? row[lane] : -1;
        uint64_t negm = __ballot(lane >= deg || nn < 0);
        int firstneg = negm ? (int)__builtin_ctzll(negm) : 64;

        if (lane == 0) {
            /* KDTIndex.cpp:201-209: result + budget termination */
            if (not_deleted(di, gnode.node)) {
                if (!qrs_add(qrs, cfg.k, gnode.node, gnode.distance) &&
                    ss->checked > cfg.max_check)
                    ss->terminate = 1;
            }
            float worst = qrs[0].dist;
            ss->fbcast = worst > gnode.distance ? worst : gnode.distance;
        }
        __syncthreads();
        if (ss->terminate) break;
        float upper_bound = ss->fbcast;

        /* spec bit0 (see bkt kernel): warm neighbor vector lines during
         * the visited-CAS round. */
        int already = 1;
        uint32_t tsink = 0;
        if (lane < firstneg) {
            if (cfg.spec_flags & 1) {
                const char* vp = (const char*)vec_at<T>(di, nn);
                for (int l = 0; l < nline; l++)
                    tsink += (uint32_t)(uint8_t)vp[(size_t)l << 7];
            }
            already = visited_test_insert(c.vtab, c.vmask, nn, &ss->oflow);
        }
        uint64_t candm = __ballot(lane < firstneg && !already);
        asm volatile("" :: "v"(tsink));
        int ncand = __popcll(candm);
        if ((candm >> lane) & 1) {
            int pos = __popcll(candm & ((1ull << lane) - 1));
            istage[pos] = nn;
        }
        __syncthreads();
        stage_dists<T, DM>(c, ncand);
        if (lane == 0) {
            /* KDTIndex.cpp:211-233: every computed neighbor joins the
             * frontier; track whether any beat the upper bound */
            int local_opt = 1;
            for (int r = 0; r < ncand; r++) {
                float dv = dstage[r];
                if (dv <= upper_bound) local_opt = 0;
                ss->checked++;
                ndheap_insert(c.ng, &ss->ng_count, NodeDist{istage[r], dv}, &ss->oflow);
            }
            if (local_opt) ss->no_better++;
            else ss->no_better = 0;
            ss->want_tree = 0;
            ss->break_flag = 0;
            if (ss->no_better > cfg.nobetter_threshold) {
                if (ss->tree_checked <= ss->checked / 10)
                    ss->want_tree = 1;
                else if (gnode.distance > qrs[0].dist)
                    ss->break_flag = 1;
            }
        }
        __syncthreads();
        if (ss->break_flag) break;
        if (ss->want_tree)
            kdt_search_trees_dev<T, DM>(c, cfg.other_pivots + ss->checked);
        /* spec bit1 (see bkt kernel): next-pop lookahead. */
        if (cfg.spec_flags & 2) {
            if (lane == 0) ss->popped = ndheap_top(c.ng, ss->ng_count);
            __syncthreads();
            int32_t nxt = ss->popped.node;
            if (nxt >= 0 && lane < deg) {
                int32_t t = di.graph[(size_t)nxt * deg + lane];
                if (t >= 0) {
                    uint32_t key = (uint32_t)(t + 1);
                    uint32_t h = (key * 2654435761u) & c.vmask;
                    if (c.vtab[h] != (int32_t)key) {
                        const char* vp = (const char*)vec_at<T>(di, t);
                        for (int l = 0; l < nline; l++)
                            spec_sink += (uint32_t)(uint8_t)vp[(size_t)l << 7];
                    }
                }
            }
        }
    }
    __syncthreads();

    if (lane == 0) {
        bufs.oflow[q] = ss->oflow;
        if (bufs.stats) {
            bufs.stats[(size_t)q * 2 + 0] = ss->checked;
            bufs.stats[(size_t)q * 2 + 1] = popped;
        }
        qrs_sort(qrs, cfg.k);
    }
    __syncthreads();
    for (int i = lane; i < cfg.k; i += 64) {
        bufs.out_vids[(size_t)q * cfg.k + i] = qrs[i].vid;
        bufs.out_dists[(size_t)q * cfg.k + i] = qrs[i].dist;
    }
}

/* ------------------------------------------------------------------ *
 * brute-force truth kernel (TruthSet::GenerateTruth semantics,
 * TruthSet.h:163): exact top-k by (dist, vid) over all non-deleted rows.
 * One query per wave; test-scale only.
 * ------------------------------------------------------------------ */

template <typename T, int DM>
__global__ __launch_bounds__(64)
void truth_kernel(DevIndex di, const void* queries, int32_t nq, int32_t k,
                  int32_t* out_vids, float* out_dists)
{
    const int q = blockIdx.x;
    if (q >= nq) return;
    const int lane = threadIdx.x;

    extern __shared__ char smem[];
    size_t off = 0;
    T* qlds = (T*)(smem + off);
    off += ((size_t)di.dim * sizeof(T) + 15) & ~15ul;
    QRes* qrs = (QRes*)(smem + off); off += (size_t)k * 8;
    float* dstage = (float*)(smem + off);

    const T* gq = (const T*)queries + (size_t)q * di.dim;
    for (int i = lane; i < di.dim; i += 64) qlds[i] = gq[i];
    if (lane == 0)
        for (int i = 0; i < k; i++) qrs[i] = QRes{-1, MAXDIST};
    __syncthreads();

    for (int32_t base = 0; base < di.n; base += 4) {
        int cnt = min(4, di.n - base);
        int j = lane >> 4;
        if (j < cnt) {
            float dv = ref_dist_grp<T, DM>(qlds, vec_at<T>(di, base + j), di.dim);
            if ((lane & 15) == 0) dstage[j] = dv;
        }
        __syncthreads();
        if (lane == 0) {
            for (int i = 0; i < cnt; i++) {
                if (di.has_deleted && di.deleted[base + i]) continue;
                qrs_add(qrs, k, base + i, dstage[i]);
            }
        }
        __syncthreads();
    }
    if (lane == 0) qrs_sort(qrs, k);
    __syncthreads();
    for (int i = lane; i < k; i += 64) {
        out_vids[(size_t)q * k + i] = qrs[i].vid;
        out_dists[(size_t)q * k + i] = qrs[i].dist;
    }
}

/* ------------------------------------------------------------------ *
 * row gather/scatter (flagged-query reruns)
 * ------------------------------------------------------------------ */

__global__ void gather_rows_kernel(char* dst, const char* src, int row_bytes,
                                   const int32_t* idx, int nrows)
{
    int r = blockIdx.x;
    if (r >= nrows) return;
    const char* s = src + (size_t)idx[r] * row_bytes;
    char* d = dst + (size_t)r * row_bytes;
    if ((row_bytes & 3) == 0) {
        for (int i = threadIdx.x; i < row_bytes / 4; i += blockDim.x)
            ((uint32_t*)d)[i] = ((const uint32_t*)s)[i];
    } else {
        for (int i = threadIdx.x; i < row_bytes; i += blockDim.x)
            d[i] = s[i];
    }
}

__global__ void scatter_rows_kernel(char* dst, const char* src, int row_bytes,
                                    const int32_t* idx, int nrows)
{
    int r = blockIdx.x;
    if (r >= nrows) return;
    const char* s = src + (size_t)r * row_bytes;
    char* d = dst + (size_t)idx[r] * row_bytes;
    if ((row_bytes & 3) == 0) {
        for (int i = threadIdx.x; i < row_bytes / 4; i += blockDim.x)
            ((uint32_t*)d)[i] = ((const uint32_t*)s)[i];
    } else {
        for (int i = threadIdx.x; i < row_bytes; i += blockDim.x)
            d[i] = s[i];
    }
}

int launch_gather_rows(void* dst, const void* src, int row_bytes,
                       const int32_t* d_idx, int nrows, void* stream)
{
    hipLaunchKernelGGL(gather_rows_kernel, dim3(nrows), dim3(64), 0,
                       (hipStream_t)stream, (char*)dst, (const char*)src,
                       row_bytes, d_idx, nrows);
    return (int)hipGetLastError();
}

int launch_scatter_rows(void* dst, const void* src, int row_bytes,
                        const int32_t* d_idx, int nrows, void* stream)
{
    hipLaunchKernelGGL(scatter_rows_kernel, dim3(nrows), dim3(64), 0,
                       (hipStream_t)stream, (char*)dst, (const char*)src,
                       row_bytes, d_idx, nrows);
    return (int)hipGetLastError();
}

/* ------------------------------------------------------------------ *
 * launchers
 * ------------------------------------------------------------------ */

template <typename T, int DM, bool LDSHEAP>
static int launch_one(const DevIndex& di, const SearchCfg& cfg,
                      const SearchBufs& bufs, hipStream_t stream)
{
    size_t lds = lds_bytes(di.dim, sizeof(T), cfg, LDSHEAP);
    dim3 grid(cfg.nq), block(64);
    if (di.algo == ALGO_KDT)
        hipLaunchKernelGGL((kdt_search_kernel<T, DM, LDSHEAP>), grid, block, lds,
                           stream, di, cfg, bufs);
    else if (cfg.prof)
        hipLaunchKernelGGL((bkt_search_kernel<T, DM, LDSHEAP, true>), grid,
                           block, lds, stream, di, cfg, bufs);
    else
        hipLaunchKernelGGL((bkt_search_kernel<T, DM, LDSHEAP, false>), grid,
                           block, lds, stream, di, cfg, bufs);
    return (int)hipGetLastError();
}

int launch_bkt_search(int vt, int dm, bool heaps_in_lds, const DevIndex& di,
                      const SearchCfg& cfg, const SearchBufs& bufs, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    if (vt == VT_FLOAT && dm == DM_L2)
        return heaps_in_lds ? launch_one<float, DM_L2, true>(di, cfg, bufs, s)
                            : launch_one<float, DM_L2, false>(di, cfg, bufs, s);
    if (vt == VT_FLOAT && dm == DM_COSINE)
        return heaps_in_lds ? launch_one<float, DM_COSINE, true>(di, cfg, bufs, s)
                            : launch_one<float, DM_COSINE, false>(di, cfg, bufs, s);
    if (vt == VT_INT8 && dm == DM_L2)
        return heaps_in_lds ? launch_one<int8_t, DM_L2, true>(di, cfg, bufs, s)
                            : launch_one<int8_t, DM_L2, false>(di, cfg, bufs, s);
    if (vt == VT_INT8 && dm == DM_COSINE)
        return heaps_in_lds ? launch_one<int8_t, DM_COSINE, true>(di, cfg, bufs, s)
                            : launch_one<int8_t, DM_COSINE, false>(di, cfg, bufs, s);
    return (int)hipErrorInvalidValue;
}

template <typename T, int DM>
static int launch_truth_one(const DevIndex& di, const void* queries, int32_t nq,
                            int32_t k, int32_t* ov, float* od, hipStream_t s)
{
    size_t lds = (((size_t)di.dim * sizeof(T) + 15) & ~15ul) + (size_t)k * 8 + 64;
    hipLaunchKernelGGL((truth_kernel<T, DM>), dim3(nq), dim3(64), lds, s,
                       di, queries, nq, k, ov, od);
    return (int)hipGetLastError();
}

template <typename T, int DM>
static int launch_iter_one(const DevIndex& di, const SearchCfg& cfg,
                           const IterBufs& ib, int batch, hipStream_t s)
{
    size_t lds = (((size_t)di.dim * sizeof(T) + 15) & ~15ul) + MAX_DEG * 8 +
                 (size_t)batch * 8 + 64;
    hipLaunchKernelGGL((bkt_iter_kernel<T, DM>), dim3(cfg.nq), dim3(64), lds, s,
                       di, cfg, ib, batch);
    return (int)hipGetLastError();
}

int launch_bkt_iter(int vt, int dm, const DevIndex& di, const SearchCfg& cfg,
                    const IterBufs& ib, int batch, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    if (vt == VT_FLOAT && dm == DM_L2) return launch_iter_one<float, DM_L2>(di, cfg, ib, batch, s);
    if (vt == VT_FLOAT && dm == DM_COSINE) return launch_iter_one<float, DM_COSINE>(di, cfg, ib, batch, s);
    if (vt == VT_INT8 && dm == DM_L2) return launch_iter_one<int8_t, DM_L2>(di, cfg, ib, batch, s);
    if (vt == VT_INT8 && dm == DM_COSINE) return launch_iter_one<int8_t, DM_COSINE>(di, cfg, ib, batch, s);
    return (int)hipErrorInvalidValue;
}

int launch_truth(int vt, int dm, const DevIndex& di, const void* queries,
                 int32_t nq, int32_t k, int32_t* ov, float* od, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    if (vt == VT_FLOAT && dm == DM_L2) return launch_truth_one<float, DM_L2>(di, queries, nq, k, ov, od, s);
    if (vt == VT_FLOAT && dm == DM_COSINE) return launch_truth_one<float, DM_COSINE>(di, queries, nq, k, ov, od, s);
    if (vt == VT_INT8 && dm == DM_L2) return launch_truth_one<int8_t, DM_L2>(di, queries, nq, k, ov, od, s);
    if (vt == VT_INT8 && dm == DM_COSINE) return launch_truth_one<int8_t, DM_COSINE>(di, queries, nq, k, ov, od, s);
    return (int)hipErrorInvalidValue;
}

}  /* namespace sptag_amd */
