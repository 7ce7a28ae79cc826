/* ORACLE — CPU restatement of the SPTAG BKT in-memory search path.
 * See sptag_oracle.h for the role/usage contract. Reference citations are to
 * /root/reference/AnnService (microsoft/SPTAG); this file restates behavior,
 * it copies no code.
 *
 * Float rounding contract: the float-path accumulations use explicit fmaf()
 * because the _ref reference binary fuses its mul+add intrinsics (gcc -O3
 * default -ffp-contract=fast; vfmadd231ps visible in the _ref objdump).
 * Everything else (lane folds, scalar sums) must stay un-fused, hence
 * -ffp-contract=off in the Makefile.
 */
#include "sptag_oracle.h"

#include <float.h>
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

/* MaxDist = numeric_limits<float>::max() / 10 (reference Common.h:122) */
static const float ORC_MAXDIST = FLT_MAX / 10.0f;

/* ------------------------------------------------------------------ *
 * Distances — reference DistanceUtils.cpp, AVX512 runtime-dispatch
 * path (the one both build/search use on AVX512-capable hosts;
 * DistanceUtils.h:119 DistanceCalcSelector).
 * ------------------------------------------------------------------ */

/* float L2: Σ(x-y)² accumulated in 16 f32 lanes over 16-element chunks,
 * folded 16->8->8-chunks->4->4-chunks->scalar, then scalar tail — the
 * exact order of ComputeL2Distance_AVX512(float) (DistanceUtils.cpp:650)
 * as COMPILED in oracle/_ref: gcc -O3 (default -ffp-contract=fast) fuses
 * each lane's mul+add into one fma (vfmadd231ps in the _ref objdump), so
 * the accumulate is fmaf(t, t, acc). Bit-exact vs the _ref binary. */
static float dist_l2_f32(const float* x, const float* y, int d)
{
    float a16[16], a8[8], a4[4];
    int i = 0, j;
    for (j = 0; j < 16; j++) a16[j] = 0.0f;
    int nd16 = (d >> 4) << 4;
    for (; i < nd16; i += 16)
        for (j = 0; j < 16; j++) { float t = x[i + j] - y[i + j]; a16[j] = fmaf(t, t, a16[j]); }
    for (j = 0; j < 8; j++) a8[j] = a16[j] + a16[j + 8];
    int nd8 = (d >> 3) << 3;
    for (; i < nd8; i += 8)
        for (j = 0; j < 8; j++) { float t = x[i + j] - y[i + j]; a8[j] = fmaf(t, t, a8[j]); }
    for (j = 0; j < 4; j++) a4[j] = a8[j] + a8[j + 4];
    int nd4 = (d >> 2) << 2;
    for (; i < nd4; i += 4)
        for (j = 0; j < 4; j++) { float t = x[i + j] - y[i + j]; a4[j] = fmaf(t, t, a4[j]); }
    float diff = ((a4[0] + a4[1]) + a4[2]) + a4[3];
    for (; i < d; i++) { float t = x[i] - y[i]; diff = fmaf(t, t, diff); }
    return diff;
}

/* float cosine-as-distance: 1 - Σ x*y, same chunk/fold order
 * (ComputeCosineDistance_AVX512(float); base=1 for float,
 * CommonUtils.h:54 GetBase). */
static float dist_cos_f32(const float* x, const float* y, int d)
{
    float a16[16], a8[8], a4[4];
    int i = 0, j;
    for (j = 0; j < 16; j++) a16[j] = 0.0f;
    int nd16 = (d >> 4) << 4;
    for (; i < nd16; i += 16)
        for (j = 0; j < 16; j++) a16[j] = fmaf(x[i + j], y[i + j], a16[j]);
    for (j = 0; j < 8; j++) a8[j] = a16[j] + a16[j + 8];
    int nd8 = (d >> 3) << 3;
    for (; i < nd8; i += 8)
        for (j = 0; j < 8; j++) a8[j] = fmaf(x[i + j], y[i + j], a8[j]);
    for (j = 0; j < 4; j++) a4[j] = a8[j] + a8[j + 4];
    int nd4 = (d >> 2) << 2;
    for (; i < nd4; i += 4)
        for (j = 0; j < 4; j++) a4[j] = fmaf(x[i + j], y[i + j], a4[j]);
    float diff = ((a4[0] + a4[1]) + a4[2]) + a4[3];
    for (; i < d; i++) diff = fmaf(x[i], y[i], diff);
    return 1.0f - diff;
}

/* int8 L2: the reference accumulates exact int32 partial sums (madd of
 * 16-bit diffs, DistanceUtils.cpp:32-48 _mm_sqdf_epi8) converted to float
 * and float-summed; every partial is an exact integer while the total
 * < 2^24, so integer accumulation reproduces it bit-exactly
 * (SURVEY.md §8a row a4). dim guard enforced at index load. */
static float dist_l2_i8(const int8_t* x, const int8_t* y, int d)
{
    int32_t s = 0;
    for (int i = 0; i < d; i++) { int32_t t = (int32_t)x[i] - (int32_t)y[i]; s += t * t; }
    return (float)s;
}

/* int8 cosine: 127*127 - Σ x*y (ComputeCosineDistance_*: returns
 * 16129 - diff; DistanceUtils.h:76-79). Exact integers for our dims. */
static float dist_cos_i8(const int8_t* x, const int8_t* y, int d)
{
    int32_t s = 0;
    for (int i = 0; i < d; i++) s += (int32_t)x[i] * (int32_t)y[i];
    return (float)(16129 - s);
}

float orc_distance(int vt, int dm, const void* x, const void* y, int32_t dim)
{
    if (vt == ORC_VT_FLOAT)
        return dm == ORC_DM_L2 ? dist_l2_f32((const float*)x, (const float*)y, dim)
                               : dist_cos_f32((const float*)x, (const float*)y, dim);
    return dm == ORC_DM_L2 ? dist_l2_i8((const int8_t*)x, (const int8_t*)y, dim)
                           : dist_cos_i8((const int8_t*)x, (const int8_t*)y, dim);
}

/* ------------------------------------------------------------------ *
 * Index object
 * ------------------------------------------------------------------ */

typedef struct { int32_t centerid, childStart, childEnd; } BktNode; /* BKTree.h:25 */
typedef struct { int32_t left, right, split_dim; float split_value; } KdtNode; /* KDTree.h:22 */

struct OrcIndex {
    int algo;                  /* 0 = BKT, 1 = KDT */
    int32_t n, dim;
    int valuetype, distmethod;
    size_t esz;                /* element size in bytes */
    void* vectors;             /* n*dim row-major */
    int32_t ntrees;
    int32_t* tree_start;
    int32_t n_tree_nodes;
    BktNode* tree;             /* BKT */
    KdtNode* kdt;              /* KDT */
    int32_t deg;
    int32_t* graph;            /* n*deg row-major */
    uint8_t* deleted;          /* may be NULL -> none */
    int64_t deleted_count;
};

static const void* vec_at(const OrcIndex* ix, int32_t v)
{
    return (const char*)ix->vectors + (size_t)v * ix->dim * ix->esz;
}

static float idx_dist(const OrcIndex* ix, const void* q, int32_t v)
{
    return orc_distance(ix->valuetype, ix->distmethod, q, vec_at(ix, v), ix->dim);
}

/* ------------------------------------------------------------------ *
 * Heap<NodeDistPair> — exact restatement of reference Heap.h:14-110.
 * 1-based array; min-heap on distance (NodeDistPair::operator<,
 * SearchResult.h:19 compares distance only). heap[0] is the "empty top"
 * sentinel: default NodeDistPair{node=-1, distance=MaxDist}.
 * ------------------------------------------------------------------ */

typedef struct { int32_t node; float distance; } NodeDist;

typedef struct {
    NodeDist* a;    /* [0..length], 1-based */
    int length;     /* capacity */
    int count;
    int lastlevel;  /* 2^floor(log2(length)): start of the last level */
} NDHeap;

static void ndheap_init(NDHeap* h, int size)
{
    h->length = size;
    h->a = (NodeDist*)malloc(sizeof(NodeDist) * (size_t)(size + 1));
    for (int i = 0; i <= size; i++) { h->a[i].node = -1; h->a[i].distance = ORC_MAXDIST; }
    h->count = 0;
    h->lastlevel = (int)pow(2.0, floor(log2((float)size))); /* Heap.h:25 */
}

static void ndheap_destroy(NDHeap* h) { free(h->a); h->a = NULL; }

static NodeDist ndheap_top(const NDHeap* h) { return h->count == 0 ? h->a[0] : h->a[1]; }

/* Heap.h:38-62 insert: when full, scan the last level for the max and
 * replace it unless the new value is larger; else append; sift up with
 * strict '<' on distance. */
static void ndheap_insert(NDHeap* h, NodeDist v)
{
    int loc;
    if (h->count == h->length) {
        int maxi = h->lastlevel;
        for (int i = h->lastlevel + 1; i <= h->length; i++)
            if (h->a[maxi].distance < h->a[i].distance) maxi = i;
        if (v.distance > h->a[maxi].distance) return;
        loc = maxi;
    } else {
        loc = ++(h->count);
    }
    int par = loc >> 1;
    while (par > 0 && v.distance < h->a[par].distance) {
        h->a[loc] = h->a[par];
        loc = par;
        par >>= 1;
    }
    h->a[loc] = v;
}

/* Heap.h:90-105 heapify: sift-down from the root; child choice
 * 'if (heap[next] > heap[next+1]) next++' (ties keep the left child),
 * swap on strict '<' vs parent. */
static void ndheap_heapify(NDHeap* h)
{
    int parent = 1, next = 2;
    while (next < h->count) {
        if (h->a[next].distance > h->a[next + 1].distance) next++;
        if (h->a[next].distance < h->a[parent].distance) {
            NodeDist t = h->a[parent]; h->a[parent] = h->a[next]; h->a[next] = t;
            parent = next;
            next <<= 1;
        } else break;
    }
    if (next == h->count && h->a[next].distance < h->a[parent].distance) {
        NodeDist t = h->a[parent]; h->a[parent] = h->a[next]; h->a[next] = t;
    }
}

/* Heap.h:74-82 pop (reference style: swap root/last, shrink, heapify). */
static NodeDist ndheap_pop(NDHeap* h)
{
    if (h->count == 0) return h->a[0];
    NodeDist t = h->a[1]; h->a[1] = h->a[h->count]; h->a[h->count] = t;
    h->count--;
    ndheap_heapify(h);
    return h->a[h->count + 1];
}

/* ------------------------------------------------------------------ *
 * DistPriorityQueue (m_Results) — WorkSpace.h:167-225. Bounded float
 * max-heap, 1-based, pre-seeded with one MaxDist entry; insert returns
 * false iff dist > current worst; at capacity replaces the root and
 * sifts down (tie rule: 'if (m_data[next] < m_data[next+1]) next++' —
 * go right when left < right). worst() = root.
 * ------------------------------------------------------------------ */

typedef struct {
    float* a;
    int length;   /* current element count (m_length) */
    int count;    /* capacity (m_count) */
    int cap_alloc;
} DistPQ;

static void distpq_init(DistPQ* q, int capacity)
{
    q->cap_alloc = capacity;
    q->a = (float*)malloc(sizeof(float) * (size_t)(capacity + 1));
    q->a[1] = ORC_MAXDIST;
    q->length = 1;
    q->count = capacity;
}

static void distpq_destroy(DistPQ* q) { free(q->a); q->a = NULL; }

static int distpq_insert(DistPQ* q, float dist)
{
    if (dist > q->a[1]) return 0;
    if (q->length == q->count) {
        q->a[1] = dist;
        int parent = 1, next = 2;
        while (next < q->length) {
            if (q->a[next] < q->a[next + 1]) next++;
            if (q->a[next] > q->a[parent]) {
                float t = q->a[parent]; q->a[parent] = q->a[next]; q->a[next] = t;
                parent = next;
                next <<= 1;
            } else break;
        }
        if (next == q->length && q->a[next] > q->a[parent]) {
            float t = q->a[parent]; q->a[parent] = q->a[next]; q->a[next] = t;
        }
    } else {
        int next = ++(q->length), parent = next >> 1;
        while (parent > 0 && dist > q->a[parent]) {
            q->a[next] = q->a[parent];
            next = parent;
            parent >>= 1;
        }
        q->a[next] = dist;
    }
    return 1;
}

static float distpq_worst(const DistPQ* q) { return q->a[1]; }

/* ------------------------------------------------------------------ *
 * QueryResultSet — QueryResultSet.h:31-120. K-entry 0-based max-heap of
 * (Dist, VID) under BasicResult operator< (Dist, then VID; :17-20),
 * initialized to {VID=-1, Dist=MaxDist}. AddPoint replaces the root when
 * (dist,vid) < root; SortResult = in-place heapsort -> ascending.
 * ------------------------------------------------------------------ */

typedef struct { int32_t vid; float dist; } QRes;

static int qres_lt(QRes a, QRes b)
{
    return (a.dist < b.dist) || (a.dist == b.dist && a.vid < b.vid);
}

typedef struct { QRes* r; int k; } QResultSet;

static void qrs_init(QResultSet* s, QRes* storage, int k)
{
    s->r = storage; s->k = k;
    for (int i = 0; i < k; i++) { s->r[i].vid = -1; s->r[i].dist = ORC_MAXDIST; }
}

static float qrs_worst(const QResultSet* s) { return s->r[0].dist; }

static void qrs_heapify(QResultSet* s, int count)
{
    int parent = 0, next = 1, maxidx = count - 1;
    while (next < maxidx) {
        if (qres_lt(s->r[next], s->r[next + 1])) next++;
        if (qres_lt(s->r[parent], s->r[next])) {
            QRes t = s->r[next]; s->r[next] = s->r[parent]; s->r[parent] = t;
            parent = next;
            next = (parent << 1) + 1;
        } else break;
    }
    if (next == maxidx && qres_lt(s->r[parent], s->r[next])) {
        QRes t = s->r[next]; s->r[next] = s->r[parent]; s->r[parent] = t;
    }
}

static int qrs_add(QResultSet* s, int32_t vid, float dist)
{
    if (dist < s->r[0].dist || (dist == s->r[0].dist && vid < s->r[0].vid)) {
        s->r[0].vid = vid; s->r[0].dist = dist;
        qrs_heapify(s, s->k);
        return 1;
    }
    return 0;
}

static void qrs_sort(QResultSet* s)
{
    for (int i = s->k - 1; i >= 0; i--) {
        QRes t = s->r[0]; s->r[0] = s->r[i]; s->r[i] = t;
        qrs_heapify(s, i);
    }
}

/* ------------------------------------------------------------------ *
 * Visited set — semantics of OptHashPosVector (WorkSpace.h:43-165) are
 * pure set membership (CheckAndSet returns "already present"); the probe
 * sequence only affects speed, and the reference grows on overflow, so
 * implementation details are not part of the contract. We use open
 * addressing sized 8x the insert bound with linear probing.
 * ------------------------------------------------------------------ */

typedef struct { int32_t* tab; uint32_t mask; } VisitSet;

static void vset_init(VisitSet* v, int expected)
{
    uint32_t cap = 64;
    while (cap < (uint32_t)expected * 8u) cap <<= 1;
    v->mask = cap - 1;
    v->tab = (int32_t*)calloc(cap, sizeof(int32_t));
}

static void vset_destroy(VisitSet* v) { free(v->tab); v->tab = NULL; }

/* returns 1 if idx was already present (mirrors CheckAndSet's meaning at
 * the call sites: 'if (CheckAndSet(nn)) continue;'). Stores idx+1 so 0
 * means empty, as the reference does (WorkSpace.h:113). */
static int vset_check_and_set(VisitSet* v, int32_t idx)
{
    uint32_t key = (uint32_t)(idx + 1);
    uint32_t h = (key * 2654435761u) & v->mask;
    for (;;) {
        int32_t cur = v->tab[h];
        if (cur == 0) { v->tab[h] = (int32_t)key; return 0; }
        if (cur == (int32_t)key) return 1;
        h = (h + 1) & v->mask;
    }
}

/* ------------------------------------------------------------------ *
 * Search — exact restatement of the default dispatch
 * Search<AlwaysTrue|CheckIfNotDeleted, CheckDup, AlwaysTrue>
 * (BKTIndex.cpp:272-352 + :464-509 dispatch; searchDeleted=false,
 * searchDuplicated=true as in SearchIndex(QueryResult&,bool),
 * BKTIndex.cpp:596-619).
 * ------------------------------------------------------------------ */

typedef struct {
    NDHeap ng;      /* m_NGQueue,  capacity 30*maxCheck (WorkSpace.h:265) */
    NDHeap spt;     /* m_SPTQueue, capacity 10*maxCheck */
    DistPQ results; /* m_Results,  capacity max(maxCheck/16, k) */
    VisitSet visited;
    int checked;       /* m_iNumberOfCheckedLeaves */
    int tree_checked;  /* m_iNumberOfTreeCheckedLeaves (KDT) */
    int max_check;
} SearchSpace;

static int not_deleted(const OrcIndex* ix, int32_t v)
{
    /* StaticDispatch::CheckIfNotDeleted (BKTIndex.cpp:437) when the index
     * has deletes; AlwaysTrue otherwise (flag bit 2 of the dispatch). */
    if (!ix->deleted || ix->deleted_count == 0) return 1;
    return ix->deleted[v] == 0;
}

/* BKTree.h:697 InitSearchTrees (m_bfs=0 default: EnableBfs,
 * BKT/ParameterDefinitionList.h:12). */
static void init_search_trees(const OrcIndex* ix, const void* q, SearchSpace* sp)
{
    for (int t = 0; t < ix->ntrees; t++) {
        const BktNode* root = &ix->tree[ix->tree_start[t]];
        if (root->childStart < 0) {
            NodeDist nd = { ix->tree_start[t], idx_dist(ix, q, root->centerid) };
            ndheap_insert(&sp->spt, nd);
        } else {
            for (int32_t begin = root->childStart; begin < root->childEnd; begin++) {
                NodeDist nd = { begin, idx_dist(ix, q, ix->tree[begin].centerid) };
                ndheap_insert(&sp->spt, nd);
            }
        }
    }
}

/* BKTree.h:772 SearchTrees. */
static void search_trees(const OrcIndex* ix, const void* q, SearchSpace* sp, int limits)
{
    while (sp->spt.count > 0) {
        NodeDist bcell = ndheap_pop(&sp->spt);
        const BktNode* tnode = &ix->tree[bcell.node];
        if (tnode->childStart < 0) {
            if (!vset_check_and_set(&sp->visited, tnode->centerid)) {
                sp->checked++;
                NodeDist nd = { tnode->centerid, bcell.distance };
                ndheap_insert(&sp->ng, nd);
            }
            if (sp->checked >= limits) break;
        } else {
            if (!vset_check_and_set(&sp->visited, tnode->centerid)) {
                NodeDist nd = { tnode->centerid, bcell.distance };
                ndheap_insert(&sp->ng, nd);
            }
            for (int32_t begin = tnode->childStart; begin < tnode->childEnd; begin++) {
                NodeDist nd = { begin, idx_dist(ix, q, ix->tree[begin].centerid) };
                ndheap_insert(&sp->spt, nd);
            }
        }
    }
}

/* default parameters (BKT/ParameterDefinitionList.h:47-49) */
enum { ORC_INIT_PIVOTS = 50, ORC_OTHER_PIVOTS = 4 };

static int32_t orc_search_kdt(const OrcIndex* ix, const void* q, int32_t k,
                              int32_t max_check, int32_t* out_vids, float* out_dists);

static int32_t orc_search_bkt(const OrcIndex* ix, const void* q, int32_t k,
                              int32_t max_check, int search_dup,
                              int search_deleted, int32_t* out_vids,
                              float* out_dists);

int32_t orc_search(const OrcIndex* ix, const void* q, int32_t k,
                   int32_t max_check, int32_t* out_vids, float* out_dists)
{
    if (ix->algo == 1)
        return orc_search_kdt(ix, q, k, max_check, out_vids, out_dists);
    /* default entry: searchDeleted=false, searchDuplicated=true
     * (BKTIndex.cpp:615) */
    return orc_search_bkt(ix, q, k, max_check, 1, 0, out_vids, out_dists);
}

static int32_t orc_search_bkt(const OrcIndex* ix, const void* q, int32_t k,
                              int32_t max_check, int search_dup,
                              int search_deleted, int32_t* out_vids,
                              float* out_dists)
{
    SearchSpace sp;
    ndheap_init(&sp.ng, max_check * 30);
    ndheap_init(&sp.spt, max_check * 10);
    int res_cap = max_check / 16 > k ? max_check / 16 : k;
    distpq_init(&sp.results, res_cap);
    vset_init(&sp.visited, max_check * 2 + 64 * 32);
    sp.checked = 0;
    sp.tree_checked = 0;
    sp.max_check = max_check;

    QRes* storage = (QRes*)malloc(sizeof(QRes) * (size_t)k);
    QResultSet query;
    qrs_init(&query, storage, k);

    init_search_trees(ix, q, &sp);
    search_trees(ix, q, &sp, ORC_INIT_PIVOTS);

    const int checkPos = ix->deg - 1;
    int finished = 0;

    while (sp.ng.count > 0 && !finished) {
        NodeDist gnode = ndheap_pop(&sp.ng);
        int32_t tmpNode = gnode.node;
        const int32_t* row = ix->graph + (size_t)tmpNode * ix->deg;

        if (gnode.distance <= qrs_worst(&query)) {
            int32_t checkNode = row[checkPos];
            if (checkNode < -1) {
                /* duplicate-center chain: the row's last slot points at the
                 * collapsed BKT node (-2 - treeIdx; NeighborhoodGraph.h:399,
                 * BKTree.h:601-608). searchDuplicated (CheckDup): walk
                 * center + duplicates, AddPoint each until it rejects
                 * (BKTIndex.cpp:443); NeverDup (refine path): AddPoint the
                 * center only (StaticDispatch::NeverDup returns true). */
                const BktNode* tnode = &ix->tree[-2 - checkNode];
                int32_t i = -tnode->childStart;
                do {
                    if ((search_deleted || not_deleted(ix, tmpNode))) {
                        if (search_dup) {
                            if (!qrs_add(&query, tmpNode, gnode.distance)) break;
                        } else {
                            qrs_add(&query, tmpNode, gnode.distance);
                            break;
                        }
                    }
                    if (i <= 0) break;
                    tmpNode = ix->tree[i].centerid;
                } while (i++ < tnode->childEnd);
            } else {
                if ((search_deleted || not_deleted(ix, tmpNode))) qrs_add(&query, tmpNode, gnode.distance);
            }
        } else {
            if ((search_deleted || not_deleted(ix, tmpNode))) {
                if (gnode.distance > distpq_worst(&sp.results) || sp.checked > sp.max_check) {
                    finished = 1;
                    break;
                }
            }
        }
        for (int i = 0; i <= checkPos; i++) {
            int32_t nn = row[i];
            if (nn < 0) break;
            if (vset_check_and_set(&sp.visited, nn)) continue;
            float d = idx_dist(ix, q, nn);
            sp.checked++;
            if (distpq_insert(&sp.results, d)) {
                NodeDist nd = { nn, d };
                ndheap_insert(&sp.ng, nd);
            }
        }
        if (ndheap_top(&sp.ng).distance > ndheap_top(&sp.spt).distance) {
            search_trees(ix, q, &sp, ORC_OTHER_PIVOTS + sp.checked);
        }
    }

    qrs_sort(&query);
    for (int i = 0; i < k; i++) { out_vids[i] = query.r[i].vid; out_dists[i] = query.r[i].dist; }

    int32_t checked = sp.checked;
    free(storage);
    vset_destroy(&sp.visited);
    distpq_destroy(&sp.results);
    ndheap_destroy(&sp.spt);
    ndheap_destroy(&sp.ng);
    return checked;
}

/* ------------------------------------------------------------------ *
 * KDT search — exact restatement of KDT::Index<T>::Search<Q,...>
 * (src/Core/KDT/KDTIndex.cpp:184-241) and KDTree::InitSearchTrees /
 * SearchTrees / KDTSearch (inc/Core/Common/KDTree.h:213-273).
 * ------------------------------------------------------------------ */

enum { ORC_KDT_NOBETTER_THRESHOLD = 3 };  /* KDT/ParameterDefinitionList.h */

/* KDTSearch (KDTree.h:234-271): iterative form of the tail recursion.
 * distBound accumulates squared split-plane distances (lower bound);
 * the off-path child goes to the SPT queue with the tightened bound. */
static void kdt_search_node(const OrcIndex* ix, const void* q, SearchSpace* sp,
                            int32_t node, float dist_bound)
{
    while (node >= 0) {
        const KdtNode* tn = &ix->kdt[node];
        float qv;
        if (ix->valuetype == ORC_VT_FLOAT)
            qv = ((const float*)q)[tn->split_dim];
        else
            qv = (float)((const int8_t*)q)[tn->split_dim];
        float diff = qv - tn->split_value;
        float other_bound = dist_bound + diff * diff;
        int32_t best = diff < 0 ? tn->left : tn->right;
        int32_t other = diff < 0 ? tn->right : tn->left;
        NodeDist nd = { other, other_bound };
        ndheap_insert(&sp->spt, nd);
        node = best;
    }
    /* leaf: node < 0 encodes -(vector id + 1) */
    int32_t index = -node - 1;
    if (index >= ix->n) return;
    if (vset_check_and_set(&sp->visited, index)) return;
    sp->tree_checked++;
    sp->checked++;
    NodeDist nd = { index, idx_dist(ix, q, index) };
    ndheap_insert(&sp->ng, nd);
}

static void kdt_search_trees(const OrcIndex* ix, const void* q, SearchSpace* sp,
                             int limits)
{
    while (sp->spt.count > 0 && sp->checked < limits) {
        NodeDist tcell = ndheap_pop(&sp->spt);
        kdt_search_node(ix, q, sp, tcell.node, tcell.distance);
    }
}

static int32_t orc_search_kdt(const OrcIndex* ix, const void* q, int32_t k,
                              int32_t max_check, int32_t* out_vids, float* out_dists)
{
    SearchSpace sp;
    ndheap_init(&sp.ng, max_check * 30);
    ndheap_init(&sp.spt, max_check * 10);
    distpq_init(&sp.results, max_check / 16 > k ? max_check / 16 : k);
    vset_init(&sp.visited, max_check * 2 + 64 * 32);
    sp.checked = 0;
    sp.tree_checked = 0;
    sp.max_check = max_check;

    QRes* storage = (QRes*)malloc(sizeof(QRes) * (size_t)k);
    QResultSet query;
    qrs_init(&query, storage, k);

    for (int t = 0; t < ix->ntrees; t++)
        kdt_search_node(ix, q, &sp, ix->tree_start[t], 0.0f);
    kdt_search_trees(ix, q, &sp, ORC_INIT_PIVOTS);

    int no_better = 0;   /* m_iNumOfContinuousNoBetterPropagation */
    while (sp.ng.count > 0) {
        NodeDist gnode = ndheap_pop(&sp.ng);
        const int32_t* row = ix->graph + (size_t)gnode.node * ix->deg;
        if (not_deleted(ix, gnode.node)) {
            if (!qrs_add(&query, gnode.node, gnode.distance) &&
                sp.checked > sp.max_check) {
                break;
            }
        }
        float worst = qrs_worst(&query);
        float upper_bound = worst > gnode.distance ? worst : gnode.distance;
        int local_opt = 1;
        for (int i = 0; i < ix->deg; i++) {
            int32_t nn = row[i];
            if (nn < 0) break;
            if (vset_check_and_set(&sp.visited, nn)) continue;
            float d = idx_dist(ix, q, nn);
            if (d <= upper_bound) local_opt = 0;
            sp.checked++;
            NodeDist nd = { nn, d };
            ndheap_insert(&sp.ng, nd);
        }
        if (local_opt) no_better++;
        else no_better = 0;
        if (no_better > ORC_KDT_NOBETTER_THRESHOLD) {
            if (sp.tree_checked <= sp.checked / 10) {
                kdt_search_trees(ix, q, &sp, ORC_OTHER_PIVOTS + sp.checked);
            } else if (gnode.distance > qrs_worst(&query)) {
                break;
            }
        }
    }

    qrs_sort(&query);
    for (int i = 0; i < k; i++) { out_vids[i] = query.r[i].vid; out_dists[i] = query.r[i].dist; }
    int32_t checked = sp.checked;
    free(storage);
    vset_destroy(&sp.visited);
    distpq_destroy(&sp.results);
    ndheap_destroy(&sp.spt);
    ndheap_destroy(&sp.ng);
    return checked;
}

/* ------------------------------------------------------------------ *
 * Iterative (streaming) search — exact restatement of
 * BKT::Index<T>::SearchIterative (src/Core/BKT/BKTIndex.cpp:354-427) and
 * the ResultIterator driving protocol (src/Core/ResultIterator.cpp:30-54 +
 * BKTIndex.cpp:660-676 SearchIndexIterativeNext: ResetResult each call).
 * ------------------------------------------------------------------ */

struct OrcIter {
    const OrcIndex* ix;
    void* query;               /* copied */
    SearchSpace sp;
    int relaxed_mono;          /* m_relaxedMono (sticky; WorkSpace.h) */
    int first;
    int max_check;
};

OrcIter* orc_iter_create(const OrcIndex* ix, const void* query, int32_t max_check)
{
    if (ix->algo != 0) return NULL;   /* BKT only (as in the reference BKT path) */
    OrcIter* it = (OrcIter*)calloc(1, sizeof(OrcIter));
    it->ix = ix;
    size_t qb = (size_t)ix->dim * ix->esz;
    it->query = malloc(qb);
    memcpy(it->query, query, qb);
    it->max_check = max_check;
    ndheap_init(&it->sp.ng, max_check * 30);
    ndheap_init(&it->sp.spt, max_check * 10);
    distpq_init(&it->sp.results, max_check / 16 > 64 ? max_check / 16 : 64);
    vset_init(&it->sp.visited, max_check * 2 + 64 * 32);
    it->sp.checked = 0;
    it->sp.tree_checked = 0;
    it->sp.max_check = max_check;
    it->first = 1;
    return it;
}

void orc_iter_free(OrcIter* it)
{
    if (!it) return;
    free(it->query);
    vset_destroy(&it->sp.visited);
    distpq_destroy(&it->sp.results);
    ndheap_destroy(&it->sp.spt);
    ndheap_destroy(&it->sp.ng);
    free(it);
}

/* One Next(batch) call. Returns resultCount; out arrays sized batch.
 * relaxed_mono receives the sticky flag. */
int32_t orc_iter_next(OrcIter* it, int32_t batch, int32_t* out_vids,
                      float* out_dists, int32_t* relaxed_mono)
{
    const OrcIndex* ix = it->ix;
    SearchSpace* sp = &it->sp;
    const void* q = it->query;

    /* SearchIndexIterativeNext: workSpace->ResetResult(m_iMaxCheck, batch)
     * (WorkSpace.h:276-283): fresh m_Results, counters zeroed; queues,
     * visited set and m_relaxedMono persist. */
    int res_cap = it->max_check / 16 > batch ? it->max_check / 16 : batch;
    if (res_cap > sp->results.cap_alloc) {
        /* DistPriorityQueue::clear(count) reallocates when the requested
         * count exceeds m_size (WorkSpace.h:176-183); mirror that grow. */
        sp->results.a = (float*)realloc(sp->results.a,
                                        sizeof(float) * (size_t)(res_cap + 1));
        sp->results.cap_alloc = res_cap;
    }
    sp->results.a[1] = ORC_MAXDIST;
    sp->results.length = 1;
    sp->results.count = res_cap;
    sp->checked = 0;
    sp->tree_checked = 0;

    QRes* storage = (QRes*)malloc(sizeof(QRes) * (size_t)batch);
    QResultSet query;
    qrs_init(&query, storage, batch);

    if (it->first) {
        init_search_trees(ix, q, sp);
        search_trees(ix, q, sp, ORC_INIT_PIVOTS);
        it->first = 0;
    }

    int count = 0;
    const int checkPos = ix->deg - 1;
    while (sp->ng.count > 0) {
        NodeDist gnode = ndheap_pop(&sp->ng);
        int32_t tmpNode = gnode.node;
        const int32_t* row = ix->graph + (size_t)tmpNode * ix->deg;
        if (not_deleted(ix, tmpNode)) {
            qrs_add(&query, tmpNode, gnode.distance);
            count++;
            if (gnode.distance > distpq_worst(&sp->results) ||
                sp->checked > it->max_check)
                it->relaxed_mono = 1;
        }
        int32_t checkNode = row[checkPos];
        if (checkNode < -1) {
            /* iterative duplicate chain (BKTIndex.cpp:387-405): duplicates
             * enter the FRONTIER so they stream out as separate results */
            const BktNode* tnode = &ix->tree[-2 - checkNode];
            int32_t i = -tnode->childStart;
            while (i < tnode->childEnd) {
                tmpNode = ix->tree[i].centerid;
                if (not_deleted(ix, tmpNode)) {
                    float d = idx_dist(ix, q, tmpNode);
                    if (!vset_check_and_set(&sp->visited, tmpNode)) {
                        NodeDist nd = { tmpNode, d };
                        ndheap_insert(&sp->ng, nd);
                    }
                }
                i++;
            }
        }
        for (int i = 0; i <= checkPos; i++) {
            int32_t nn = row[i];
            if (nn < 0) break;
            if (vset_check_and_set(&sp->visited, nn)) continue;
            float d = idx_dist(ix, q, nn);
            sp->checked++;
            NodeDist nd = { nn, d };
            ndheap_insert(&sp->ng, nd);
            distpq_insert(&sp->results, d);
        }
        if (ndheap_top(&sp->ng).distance > ndheap_top(&sp->spt).distance)
            search_trees(ix, q, sp, ORC_OTHER_PIVOTS + sp->checked);
        if (count >= batch) break;
    }

    qrs_sort(&query);
    for (int i = 0; i < batch; i++) {
        out_vids[i] = query.r[i].vid;
        out_dists[i] = query.r[i].dist;
    }
    if (relaxed_mono) *relaxed_mono = it->relaxed_mono;
    free(storage);
    return count;
}

void orc_search_batch(const OrcIndex* ix, const void* queries, int32_t nq,
                      int32_t k, int32_t max_check, int nthreads,
                      int32_t* out_vids, float* out_dists)
{
#ifdef _OPENMP
    if (nthreads > 0) omp_set_num_threads(nthreads);
#pragma omp parallel for schedule(dynamic, 8)
#endif
    for (int32_t i = 0; i < nq; i++) {
        const char* q = (const char*)queries + (size_t)i * ix->dim * ix->esz;
        orc_search(ix, q, k, max_check, out_vids + (size_t)i * k, out_dists + (size_t)i * k);
    }
}

void orc_truth(const OrcIndex* ix, const void* queries, int32_t nq, int32_t k,
               int nthreads, int32_t* out_vids, float* out_dists)
{
#ifdef _OPENMP
    if (nthreads > 0) omp_set_num_threads(nthreads);
#pragma omp parallel for schedule(dynamic, 1)
#endif
    for (int32_t i = 0; i < nq; i++) {
        const char* q = (const char*)queries + (size_t)i * ix->dim * ix->esz;
        QRes* storage = (QRes*)malloc(sizeof(QRes) * (size_t)k);
        QResultSet rs;
        qrs_init(&rs, storage, k);
        for (int32_t v = 0; v < ix->n; v++) {
            if (ix->deleted && ix->deleted[v]) continue;
            qrs_add(&rs, v, idx_dist(ix, q, v));
        }
        qrs_sort(&rs);
        for (int j = 0; j < k; j++) {
            out_vids[(size_t)i * k + j] = rs.r[j].vid;
            out_dists[(size_t)i * k + j] = rs.r[j].dist;
        }
        free(storage);
    }
}

/* ------------------------------------------------------------------ *
 * Online add — exact restatement of BKT::Index<T>::AddIndex
 * (src/Core/BKT/BKTIndex.cpp:902-970) for the sub-rebuild-threshold case
 * (< AddCountForRebuild adds: the reference would only schedule its
 * background tree rebuild beyond that), with
 * NeighborhoodGraph::RefineNode (NeighborhoodGraph.h:535: RefineSearchIndex
 * = search at MaxCheckForRefineGraph with searchDuplicated=false, k=CEF+1;
 * then RebuildNeighbors) and RelativeNeighborhoodGraph::InsertNeighbors
 * (RelativeNeighborhoodGraph.h:37-80) two-way edge updates.
 * ------------------------------------------------------------------ */

enum { ORC_REFINE_MAXCHECK = 8192 };  /* MaxCheckForRefineGraph default */

/* RebuildNeighbors (RelativeNeighborhoodGraph.h:18-35) */
static void orc_rebuild_neighbors(OrcIndex* ix, int32_t node,
                                  const int32_t* res_vids, const float* res_dists,
                                  int num_results)
{
    int32_t* nodes = ix->graph + (size_t)node * ix->deg;
    int count = 0;
    for (int j = 0; j < num_results && count < ix->deg; j++) {
        if (res_vids[j] < 0) break;
        if (res_vids[j] == node) continue;
        int good = 1;
        for (int kk = 0; kk < count; kk++) {
            float d = orc_distance(ix->valuetype, ix->distmethod,
                                   vec_at(ix, nodes[kk]), vec_at(ix, res_vids[j]),
                                   ix->dim);
            if (1.0f * d < res_dists[j]) { good = 0; break; }  /* RNGFactor=1 */
        }
        if (good) nodes[count++] = res_vids[j];
    }
    for (int j = count; j < ix->deg; j++) nodes[j] = -1;
}

/* InsertNeighbors (RelativeNeighborhoodGraph.h:37-80) */
static void orc_insert_neighbors(OrcIndex* ix, int32_t node, int32_t insertNode,
                                 float insertDist)
{
    int32_t* nodes = ix->graph + (size_t)node * ix->deg;
    const void* nodeVec = vec_at(ix, node);
    const void* insertVec = vec_at(ix, insertNode);
    int checkSize = (nodes[ix->deg - 1] < -1) ? ix->deg - 1 : ix->deg;
    for (int k = 0; k < checkSize; k++) {
        int32_t tmpNode = nodes[k];
        if (tmpNode < 0) { nodes[k] = insertNode; break; }
        const void* tmpVec = vec_at(ix, tmpNode);
        float tmpDist = orc_distance(ix->valuetype, ix->distmethod, tmpVec,
                                     nodeVec, ix->dim);
        if (tmpDist > insertDist ||
            (insertDist == tmpDist && insertNode < tmpNode)) {
            nodes[k] = insertNode;
            while (++k < checkSize &&
                   orc_distance(ix->valuetype, ix->distmethod, tmpVec, nodeVec,
                                ix->dim) <=
                   orc_distance(ix->valuetype, ix->distmethod, tmpVec, insertVec,
                                ix->dim)) {
                int32_t t = tmpNode; tmpNode = nodes[k]; nodes[k] = t;
                if (tmpNode < 0) return;
                tmpVec = vec_at(ix, tmpNode);
            }
            break;
        } else if (orc_distance(ix->valuetype, ix->distmethod, tmpVec, insertVec,
                                ix->dim) < insertDist) {
            break;
        }
    }
}

int orc_add(OrcIndex* ix, const void* vectors, int32_t nadd, int32_t add_cef,
            int normalized)
{
    if (ix->algo != 0 || nadd <= 0) return -1;
    int32_t begin = ix->n, end = ix->n + nadd;
    size_t esz = ix->esz;
    ix->vectors = realloc(ix->vectors, (size_t)end * ix->dim * esz);
    memcpy((char*)ix->vectors + (size_t)begin * ix->dim * esz, vectors,
           (size_t)nadd * ix->dim * esz);
    ix->graph = (int32_t*)realloc(ix->graph, (size_t)end * ix->deg * 4);
    for (size_t i = (size_t)begin * ix->deg; i < (size_t)end * ix->deg; i++)
        ix->graph[i] = -1;
    if (ix->deleted) {
        ix->deleted = (uint8_t*)realloc(ix->deleted, (size_t)end);
        memset(ix->deleted + begin, 0, (size_t)nadd);
    }
    ix->n = end;

    if (ix->distmethod == ORC_DM_COSINE && !normalized) {
        /* Utils::Normalize (CommonUtils.h:62): scale to norm=base with
         * C-cast truncation for integral types */
        for (int32_t i = begin; i < end; i++) {
            if (ix->valuetype == ORC_VT_FLOAT) {
                float* v = (float*)vec_at(ix, i);
                double s = 0;
                for (int d = 0; d < ix->dim; d++) s += (double)v[d] * v[d];
                s = sqrt(s);
                if (s > 0)
                    for (int d = 0; d < ix->dim; d++) v[d] = (float)(v[d] / s);
            } else {
                int8_t* v = (int8_t*)vec_at(ix, i);
                double s = 0;
                for (int d = 0; d < ix->dim; d++) s += (double)v[d] * v[d];
                s = sqrt(s);
                if (s > 0)
                    for (int d = 0; d < ix->dim; d++)
                        v[d] = (int8_t)(v[d] * 127.0 / s);
            }
        }
    }

    int k = add_cef + 1;
    int32_t* rv = (int32_t*)malloc((size_t)k * 4);
    float* rd = (float*)malloc((size_t)k * 4);
    for (int32_t node = begin; node < end; node++) {
        /* RefineNode(this, node, updateNeighbors=true, searchDeleted=true,
         * AddCEF) — BKTIndex.cpp:968 */
        orc_search_bkt(ix, vec_at(ix, node), k, ORC_REFINE_MAXCHECK,
                       /*search_dup=*/0, /*search_deleted=*/1, rv, rd);
        orc_rebuild_neighbors(ix, node, rv, rd, k);
        for (int j = 0; j < k; j++) {
            if (rv[j] < 0) break;
            if (rv[j] == node) continue;
            orc_insert_neighbors(ix, rv[j], node, rd[j]);
        }
    }
    free(rv);
    free(rd);
    return 0;
}

/* ------------------------------------------------------------------ *
 * Construction / loading
 * ------------------------------------------------------------------ */

OrcIndex* orc_create_index(int32_t n, int32_t dim, int valuetype, int distmethod,
                           const void* vectors,
                           int32_t ntrees, const int32_t* tree_start,
                           int32_t n_tree_nodes, const int32_t* tree_nodes,
                           int32_t degree, const int32_t* graph,
                           const uint8_t* deleted)
{
    OrcIndex* ix = (OrcIndex*)calloc(1, sizeof(OrcIndex));
    ix->n = n; ix->dim = dim;
    ix->valuetype = valuetype; ix->distmethod = distmethod;
    ix->esz = valuetype == ORC_VT_FLOAT ? 4 : 1;
    size_t vbytes = (size_t)n * dim * ix->esz;
    ix->vectors = malloc(vbytes);
    memcpy(ix->vectors, vectors, vbytes);
    ix->ntrees = ntrees;
    ix->tree_start = (int32_t*)malloc(sizeof(int32_t) * (size_t)ntrees);
    memcpy(ix->tree_start, tree_start, sizeof(int32_t) * (size_t)ntrees);
    /* BKTree.h:683: append a sentinel node if the stored array does not end
     * with centerid=-1. */
    int need_sentinel = (n_tree_nodes > 0 && tree_nodes[(size_t)(n_tree_nodes - 1) * 3] != -1);
    ix->n_tree_nodes = n_tree_nodes + (need_sentinel ? 1 : 0);
    ix->tree = (BktNode*)malloc(sizeof(BktNode) * (size_t)ix->n_tree_nodes);
    memcpy(ix->tree, tree_nodes, sizeof(BktNode) * (size_t)n_tree_nodes);
    if (need_sentinel) {
        ix->tree[n_tree_nodes].centerid = -1;
        ix->tree[n_tree_nodes].childStart = -1;
        ix->tree[n_tree_nodes].childEnd = -1;
    }
    ix->deg = degree;
    ix->graph = (int32_t*)malloc(sizeof(int32_t) * (size_t)n * degree);
    memcpy(ix->graph, graph, sizeof(int32_t) * (size_t)n * degree);
    if (deleted) {
        ix->deleted = (uint8_t*)malloc((size_t)n);
        memcpy(ix->deleted, deleted, (size_t)n);
        ix->deleted_count = 0;
        for (int32_t i = 0; i < n; i++) ix->deleted_count += deleted[i] ? 1 : 0;
    }
    return ix;
}

OrcIndex* orc_create_kdt_index(int32_t n, int32_t dim, int valuetype, int distmethod,
                               const void* vectors,
                               int32_t ntrees, const int32_t* tree_start,
                               int32_t n_tree_nodes, const void* kdt_nodes,
                               int32_t degree, const int32_t* graph,
                               const uint8_t* deleted)
{
    OrcIndex* ix = (OrcIndex*)calloc(1, sizeof(OrcIndex));
    ix->algo = 1;
    ix->n = n; ix->dim = dim;
    ix->valuetype = valuetype; ix->distmethod = distmethod;
    ix->esz = valuetype == ORC_VT_FLOAT ? 4 : 1;
    size_t vbytes = (size_t)n * dim * ix->esz;
    ix->vectors = malloc(vbytes);
    memcpy(ix->vectors, vectors, vbytes);
    ix->ntrees = ntrees;
    ix->tree_start = (int32_t*)malloc(sizeof(int32_t) * (size_t)ntrees);
    memcpy(ix->tree_start, tree_start, sizeof(int32_t) * (size_t)ntrees);
    ix->n_tree_nodes = n_tree_nodes;
    ix->kdt = (KdtNode*)malloc(sizeof(KdtNode) * (size_t)n_tree_nodes);
    memcpy(ix->kdt, kdt_nodes, sizeof(KdtNode) * (size_t)n_tree_nodes);
    ix->deg = degree;
    ix->graph = (int32_t*)malloc(sizeof(int32_t) * (size_t)n * degree);
    memcpy(ix->graph, graph, sizeof(int32_t) * (size_t)n * degree);
    if (deleted) {
        ix->deleted = (uint8_t*)malloc((size_t)n);
        memcpy(ix->deleted, deleted, (size_t)n);
        for (int32_t i = 0; i < n; i++) ix->deleted_count += deleted[i] ? 1 : 0;
    }
    return ix;
}

void orc_free_index(OrcIndex* ix)
{
    if (!ix) return;
    free(ix->vectors); free(ix->tree_start); free(ix->tree); free(ix->kdt);
    free(ix->graph); free(ix->deleted); free(ix);
}

const int32_t* orc_graph_ptr(const OrcIndex* ix) { return ix->graph; }
const void* orc_vectors_ptr(const OrcIndex* ix) { return ix->vectors; }

int32_t orc_num_vectors(const OrcIndex* ix) { return ix->n; }
int32_t orc_dim(const OrcIndex* ix) { return ix->dim; }
int orc_valuetype(const OrcIndex* ix) { return ix->valuetype; }
int orc_distmethod(const OrcIndex* ix) { return ix->distmethod; }
int32_t orc_degree(const OrcIndex* ix) { return ix->deg; }

/* --- file loading (formats: SURVEY.md §3c) --- */

static void* read_all(const char* path, size_t* out_size)
{
    FILE* f = fopen(path, "rb");
    if (!f) return NULL;
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    void* buf = malloc((size_t)sz);
    if (fread(buf, 1, (size_t)sz, f) != (size_t)sz) { fclose(f); free(buf); return NULL; }
    fclose(f);
    if (out_size) *out_size = (size_t)sz;
    return buf;
}

/* minimal ini: find "key=value" in a section (SimpleIniReader semantics for
 * the keys we need; reference src/Helper/SimpleIniReader.cpp). */
static int ini_get(const char* text, const char* section, const char* key,
                   char* out, size_t outsz)
{
    char sect[128];
    snprintf(sect, sizeof sect, "[%s]", section);
    const char* p = strstr(text, sect);
    if (!p) return 0;
    p += strlen(sect);
    const char* end = strchr(p, '[');  /* next section (keys contain no '[') */
    size_t klen = strlen(key);
    while (p && (!end || p < end)) {
        while (*p == '\n' || *p == '\r' || *p == ' ') p++;
        if (!*p || *p == '[') break;
        const char* eol = strchr(p, '\n');
        if (!eol) eol = p + strlen(p);
        if (strncmp(p, key, klen) == 0 && p[klen] == '=') {
            const char* v = p + klen + 1;
            size_t vlen = (size_t)(eol - v);
            while (vlen && (v[vlen - 1] == '\r' || v[vlen - 1] == ' ')) vlen--;
            if (vlen >= outsz) vlen = outsz - 1;
            memcpy(out, v, vlen);
            out[vlen] = 0;
            return 1;
        }
        p = *eol ? eol + 1 : eol;
    }
    return 0;
}

OrcIndex* orc_load_index(const char* folder)
{
    char path[1024], val[256];
    snprintf(path, sizeof path, "%s/indexloader.ini", folder);
    size_t tsz;
    char* ini = (char*)read_all(path, &tsz);
    if (!ini) return NULL;
    ini = (char*)realloc(ini, tsz + 1);
    ini[tsz] = 0;

    int vt = ORC_VT_FLOAT, dm = ORC_DM_L2, algo = 0;
    if (ini_get(ini, "Index", "IndexAlgoType", val, sizeof val)) {
        if (strcmp(val, "BKT") == 0) algo = 0;
        else if (strcmp(val, "KDT") == 0) algo = 1;
        else {
            fprintf(stderr, "oracle: unsupported IndexAlgoType %s\n", val);
            free(ini);
            return NULL;
        }
    }
    if (ini_get(ini, "Index", "ValueType", val, sizeof val)) {
        if (strcmp(val, "Float") == 0) vt = ORC_VT_FLOAT;
        else if (strcmp(val, "Int8") == 0) vt = ORC_VT_INT8;
        else { fprintf(stderr, "oracle: unsupported ValueType %s\n", val); free(ini); return NULL; }
    }
    if (ini_get(ini, "Index", "DistCalcMethod", val, sizeof val)) {
        if (strcmp(val, "L2") == 0) dm = ORC_DM_L2;
        else if (strcmp(val, "Cosine") == 0) dm = ORC_DM_COSINE;
        else { fprintf(stderr, "oracle: unsupported DistCalcMethod %s\n", val); free(ini); return NULL; }
    }
    free(ini);

    /* vectors.bin: [int32 R][int32 C][row-major data] (Dataset.h:146) */
    size_t sz;
    snprintf(path, sizeof path, "%s/vectors.bin", folder);
    char* vb = (char*)read_all(path, &sz);
    if (!vb) return NULL;
    int32_t n = ((int32_t*)vb)[0], dim = ((int32_t*)vb)[1];

    /* tree.bin: [int32 #trees][int32 roots x #][int32 count][BKTNode x count]
     * (BKTree.h:640-686 SaveTrees/LoadTrees) */
    snprintf(path, sizeof path, "%s/tree.bin", folder);
    char* tb = (char*)read_all(path, &sz);
    if (!tb) { free(vb); return NULL; }
    int32_t ntrees = ((int32_t*)tb)[0];
    const int32_t* tstart = (int32_t*)(tb + 4);
    int32_t nnodes = *(int32_t*)(tb + 4 + 4 * (size_t)ntrees);
    const int32_t* tnodes = (int32_t*)(tb + 8 + 4 * (size_t)ntrees);

    /* graph.bin: [int32 R][int32 deg][int32 adj RxD] (NeighborhoodGraph.h:607) */
    snprintf(path, sizeof path, "%s/graph.bin", folder);
    char* gb = (char*)read_all(path, &sz);
    if (!gb) { free(vb); free(tb); return NULL; }
    int32_t gn = ((int32_t*)gb)[0], deg = ((int32_t*)gb)[1];
    if (gn != n) { fprintf(stderr, "oracle: graph R %d != vectors R %d\n", gn, n); }

    /* deletes.bin: [int32 count][int32 R][int32 1][int8 x R] (Labelset.h:78) */
    snprintf(path, sizeof path, "%s/deletes.bin", folder);
    uint8_t* del = NULL;
    char* db = (char*)read_all(path, &sz);
    int32_t delcount = 0;
    if (db) {
        delcount = ((int32_t*)db)[0];
        if (delcount > 0) del = (uint8_t*)(db + 12);
    }

    /* int8 exactness guard (see dist_l2_i8) */
    if (vt == ORC_VT_INT8 && (int64_t)dim * 254 * 254 >= (1ll << 24))
        fprintf(stderr, "oracle: warning: int8 dim %d may lose float exactness\n", dim);

    OrcIndex* ix;
    if (algo == 0)
        ix = orc_create_index(n, dim, vt, dm, vb + 8, ntrees, tstart, nnodes,
                              tnodes, deg, (int32_t*)(gb + 8), del);
    else {
        /* KDT tree.bin: [int32 #trees][int32 starts x #][int32 count]
         * [KDTNode{left,right,split_dim,split_value} x count]
         * (KDTree.h:123-135 SaveTrees / :190-201 LoadTrees) */
        const KdtNode* knodes = (const KdtNode*)(tb + 8 + 4 * (size_t)ntrees);
        ix = orc_create_kdt_index(n, dim, vt, dm, vb + 8, ntrees, tstart,
                                  nnodes, knodes, deg, (int32_t*)(gb + 8), del);
    }
    ix->deleted_count = delcount;
    free(vb); free(tb); free(gb); free(db);
    return ix;
}
