/* Internal host<->device shared definitions for the sptag_amd backend.
 * Not part of the public ABI (that is include/sptag_amd.h). */
#pragma once
#include <stdint.h>

namespace sptag_amd {

/* mirror of the public constants */
enum { VT_FLOAT = 0, VT_INT8 = 1 };
enum { DM_L2 = 0, DM_COSINE = 1 };

/* default search parameters (reference BKT/ParameterDefinitionList.h:47-49) */
enum { DEFAULT_MAXCHECK = 8192, INIT_PIVOTS = 50, OTHER_PIVOTS = 4 };

/* hard caps of the v1 kernel (host validates before launch) */
enum { MAX_DEG = 64, MAX_K = 1024, MAX_DIM = 4096 };

enum { ALGO_BKT = 0, ALGO_KDT = 1 };

/* device-resident index description (POD, passed by value to kernels) */
struct DevIndex {
    const void* vectors;       /* n*dim row-major, element size by vt */
    const int32_t* graph;      /* n*deg row-major adjacency */
    const int32_t* tree_nodes; /* BKT: {centerid,childStart,childEnd} int32x3;
                                  KDT: {left,right,split_dim,split_value} 16B */
    const int32_t* tree_start; /* ntrees roots */
    const uint8_t* deleted;    /* may be null */
    int32_t n, dim, deg, ntrees, n_tree_nodes;
    int32_t has_deleted;
    int32_t algo;
};

/* per-launch search configuration */
struct SearchCfg {
    int32_t nq, k, max_check;
    int32_t init_pivots, other_pivots;
    int32_t nobetter_threshold;  /* KDT: ThresholdOfNumberOfContinuousNoBetterPropagation */
    int32_t search_dup;          /* searchDuplicated (BKT dispatch bit 1) */
    int32_t search_deleted;      /* searchDeleted (dispatch bit 2) */
    int32_t ng_cap, spt_cap, dpq_cap;  /* heap capacities (entries) */
    int32_t vcap;                      /* visited table slots (pow2) */
    int32_t spec_flags;  /* perf-only speculation (identical results):
                            bit0 = warm neighbor vector lines during the
                                   visited-CAS round;
                            bit1 = next-pop lookahead (read next frontier
                                   top's adjacency, read-only visited probe,
                                   start unvisited vector loads) */
    int32_t prof;        /* 1 = per-phase cycle breakdown into stats
                            (stride PROF_STATS int32 per query; lane-0
                            clock64 marks — one wave per WG makes lane-0
                            spans wave-accurate). Diagnostic only. */
};

enum { PROF_STATS = 16 };  /* stats stride (int32) in prof mode; slots:
    0 checked, 1 popped, 2 seed, 3 pop, 4 row, 5 serial, 6 cas, 7 dist,
    8 insert, 9 tree, 10 spec, 11 epilogue (cycles) */

/* per-launch buffers */
struct SearchBufs {
    const void* queries;   /* nq*dim */
    int32_t* out_vids;     /* nq*k */
    float* out_dists;      /* nq*k */
    int32_t* visited;      /* nq*vcap, zeroed before launch */
    int32_t* oflow;        /* nq flags: nonzero => rerun with bigger caps */
    int32_t* stats;        /* nq*2: {checked, popped} per query (may be null) */
    /* only used by the global-heap variant: per-query strided scratch */
    void* gheap_ng;        /* nq * (ng_cap+1) * 8B */
    void* gheap_spt;       /* nq * (spt_cap+1) * 8B */
};

enum { SPT_LDS_TIER_H = 255 };  /* SPT heap LDS tier entries (see kernels) */

/* LDS bytes needed per workgroup for the LDS-heap variant (and the
 * non-heap part of the global variant). Keep in sync with kernel. */
inline size_t lds_bytes(int dim, size_t esz, const SearchCfg& c, bool heaps_in_lds)
{
    size_t b = 0;
    b += ((size_t)dim * esz + 15) & ~15ul;            /* query vector */
    b += (MAX_DEG) * 4;                               /* staged dists */
    b += (MAX_DEG) * 4;                               /* staged child centers */
    b += ((size_t)c.k) * 8;                           /* result set */
    b += ((size_t)c.dpq_cap + 1) * 4;                 /* DistPriorityQueue */
    b += 64;                                          /* scalar slots, padding */
    b += ((size_t)SPT_LDS_TIER_H + 1) * 8;            /* SPT heap LDS tier */
    if (heaps_in_lds) {
        b += ((size_t)c.ng_cap + 1) * 8;   /* SPT heap tail is global */
    }
    return b;
}

/* persistent per-iterator device state (iterative search) */
struct IterState {
    int32_t ng_count, spt_count, checked, tree_checked;
    int32_t relaxed, first, oflow, pad;
};

/* iterative-search launch buffers: persistent state + per-call outputs */
struct IterBufs {
    const void* queries;   /* nq*dim (device, owned by the iterator batch) */
    void* gheap_ng;        /* nq * (ng_cap+1) * 8 */
    void* gheap_spt;       /* nq * (spt_cap+1) * 8 */
    float* dpq;            /* nq * (dpq_cap+1) */
    int32_t* visited;      /* nq * vcap (zeroed at create) */
    IterState* state;      /* nq */
    int32_t* out_vids;     /* nq * batch */
    float* out_dists;      /* nq * batch */
    int32_t* out_counts;   /* nq */
    int32_t* out_relaxed;  /* nq */
};

int launch_bkt_iter(int valuetype, int distmethod, const DevIndex& di,
                    const SearchCfg& cfg, const IterBufs& bufs, int batch,
                    void* stream);

/* launchers implemented in kernels.hip; return hipError_t as int */
int launch_bkt_search(int valuetype, int distmethod, bool heaps_in_lds,
                      const DevIndex& di, const SearchCfg& cfg,
                      const SearchBufs& bufs, void* stream);
int launch_gather_rows(void* dst, const void* src, int row_bytes,
                       const int32_t* d_idx, int nrows, void* stream);
int launch_scatter_rows(void* dst, const void* src, int row_bytes,
                        const int32_t* d_idx, int nrows, void* stream);
int launch_truth(int valuetype, int distmethod, const DevIndex& di,
                 const void* queries, int32_t nq, int32_t k,
                 int32_t* out_vids, float* out_dists, void* stream);

}  /* namespace sptag_amd */
