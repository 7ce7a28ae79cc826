/* sptag_amd — C-ABI boundary of the MI355X-native SPTAG search backend.
 *
 * This is the drop-in seam: a host (C++, Python, or any FFI) that today
 * drives the reference SPTAG through VectorIndex can route the in-memory
 * BKT hot path through these entry points instead. Each entry point cites
 * the reference interface it replaces (paths relative to
 * /root/reference/AnnService):
 *
 *   sptag_amd_load_index    <- VectorIndex::LoadIndex(folder)
 *                              (inc/Core/VectorIndex.h:157, src/Core/
 *                              VectorIndex.cpp:618): same folder layout,
 *                              indexloader.ini + vectors/tree/graph/deletes.
 *   sptag_amd_create_index  <- BKT::Index<T>::LoadIndexDataFromMemory /
 *                              the blob form of LoadIndexData
 *                              (src/Core/BKT/BKTIndex.cpp:86).
 *   sptag_amd_search_batch  <- the batch overload
 *                              VectorIndex::SearchIndex(const void* p_vectors,
 *                              int p_vectorNum, int p_neighborCount, bool
 *                              p_withMeta, BasicResult* p_results)
 *                              (inc/Core/VectorIndex.h:103, src/Core/
 *                              VectorIndex.cpp:455): contiguous queries in,
 *                              per-query top-k (vid,dist) out, ascending by
 *                              (dist, vid), vid=-1 padding — identical result
 *                              contract, with (vid,dist) in two flat arrays
 *                              instead of BasicResult structs so no C++ types
 *                              cross the ABI.
 *   sptag_amd_save_index    <- VectorIndex::SaveIndex(folder)
 *                              (inc/Core/VectorIndex.h:87): writes the same
 *                              byte format the reference loads.
 *
 * All device state lives behind the opaque handle. Errors are negative int
 * codes; no exceptions cross this boundary. Thread-safety: search is
 * internally batched; concurrent callers on one handle are serialized by an
 * internal lock (the reference allows concurrent SearchIndex callers —
 * SURVEY.md §8b).
 *
 * THE GPU IS THE PRODUCT PATH: every search entry fails loudly (returns
 * SPTAG_AMD_ERR_NOGPU) when no HIP device is present. There is no CPU
 * fallback anywhere behind this ABI.
 */
#ifndef SPTAG_AMD_H
#define SPTAG_AMD_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* value types — subset of reference VectorValueType the hot path covers */
#define SPTAG_AMD_VT_FLOAT 0
#define SPTAG_AMD_VT_INT8  1
/* distance methods — reference DistCalcMethod */
#define SPTAG_AMD_DM_L2     0
#define SPTAG_AMD_DM_COSINE 1

#define SPTAG_AMD_OK            0
#define SPTAG_AMD_ERR_IO       -1   /* missing/corrupt index files */
#define SPTAG_AMD_ERR_PARAM    -2   /* bad argument */
#define SPTAG_AMD_ERR_NOGPU    -3   /* no HIP device / HIP runtime failure */
#define SPTAG_AMD_ERR_UNSUPP   -4   /* index feature outside the hot path */
#define SPTAG_AMD_ERR_OOM      -5
#define SPTAG_AMD_ERR_INTERNAL -6

typedef struct SptagAmdIndex SptagAmdIndex;

/* Load a reference-format index folder onto GPU `device`. */
SptagAmdIndex* sptag_amd_load_index(const char* folder, int device);

/* Assemble an index from raw blobs (layouts as in the files, no headers);
 * copies to device. tree_nodes: int32 triples {centerid,childStart,childEnd}. */
SptagAmdIndex* sptag_amd_create_index(int32_t n, int32_t dim, int valuetype,
                                      int distmethod, const void* vectors,
                                      int32_t ntrees, const int32_t* tree_start,
                                      int32_t n_tree_nodes, const int32_t* tree_nodes,
                                      int32_t degree, const int32_t* graph,
                                      const uint8_t* deleted /* NULL = none */,
                                      int device);

/* KDT variant of create (reference inc/Core/Common/KDTree.h:22 KDTNode
 * {left,right,split_dim,split_value}; kdt_nodes = 16-byte records). */
SptagAmdIndex* sptag_amd_create_index_kdt(int32_t n, int32_t dim, int valuetype,
                                          int distmethod, const void* vectors,
                                          int32_t ntrees, const int32_t* tree_start,
                                          int32_t n_tree_nodes, const void* kdt_nodes,
                                          int32_t degree, const int32_t* graph,
                                          const uint8_t* deleted, int device);

void sptag_amd_free_index(SptagAmdIndex* idx);

/* Batched search. queries: nq contiguous vectors of the index's dtype/dim.
 * out_vids/out_dists: nq*k, ascending (dist, vid), vid=-1 padding.
 * max_check <= 0 uses the index's MaxCheck (ini value or 8192 default).
 * Returns SPTAG_AMD_OK or a negative error. */
int sptag_amd_search_batch(SptagAmdIndex* idx, const void* queries, int32_t nq,
                           int32_t k, int32_t max_check,
                           int32_t* out_vids, float* out_dists);

/* Device-pointer variant: queries/outputs are HIP device buffers already
 * resident in HBM (e.g. torch tensors' data_ptr); no host<->device copies
 * inside the timed path. Same contract otherwise. */
int sptag_amd_search_batch_device(SptagAmdIndex* idx, const void* d_queries,
                                  int32_t nq, int32_t k, int32_t max_check,
                                  int32_t* d_out_vids, float* d_out_dists);

/* Instrumentation of the LAST search call on this handle: accumulated HIP
 * kernel time (events around each search launch, on the launch stream) and
 * the traversal totals across all queries (checked = distance evaluations,
 * popped = frontier pops) — the inputs to the roofline's algorithmic-bytes
 * numerator. */
void sptag_amd_last_stats(SptagAmdIndex* idx, double* kernel_ms,
                          long long* checked, long long* popped);

/* Exact brute-force top-k on the GPU (truth generation / recall gates;
 * reference TruthSet::GenerateTruth semantics, inc/Core/Common/TruthSet.h:163). */
int sptag_amd_truth(SptagAmdIndex* idx, const void* queries, int32_t nq,
                    int32_t k, int32_t* out_vids, float* out_dists);

/* Iterative (streaming) search — mirrors the reference's ResultIterator /
 * SearchIndexIterativeNext protocol (inc/Core/VectorIndex.h:43-49,
 * src/Core/ResultIterator.cpp) over a BATCH of nq per-query iterators whose
 * traversal state persists on the device between calls. Each next() call
 * returns up to `batch` further results per query, sorted, with the
 * per-query result count and the sticky relaxed-monotonicity flag. */
typedef struct SptagAmdIterBatch SptagAmdIterBatch;
SptagAmdIterBatch* sptag_amd_iter_create(SptagAmdIndex* idx, const void* queries,
                                         int32_t nq, int32_t max_check);
int sptag_amd_iter_next(SptagAmdIterBatch* it, int32_t batch,
                        int32_t* out_vids, float* out_dists,
                        int32_t* out_counts, int32_t* out_relaxed);
void sptag_amd_iter_free(SptagAmdIterBatch* it);

/* Online add — mirrors BKT AddIndex below the tree-rebuild threshold
 * (src/Core/BKT/BKTIndex.cpp:902-970): appends the vectors, then per new
 * node a GPU refine search (AddCEF=500, MaxCheckForRefineGraph) followed by
 * the reference's RNG rebuild + two-way neighbor inserts. Requires a HIP
 * device (the refine searches run on it). */
int sptag_amd_add(SptagAmdIndex* idx, const void* vectors, int32_t nadd,
                  int normalized);

/* Incremental delete — mirrors VectorIndex::DeleteIndex(SizeType)
 * (inc/Core/VectorIndex.h, BKT DeleteIndex -> Labelset::Insert,
 * src/Core/BKT/BKTIndex.cpp:896): flags the ids; subsequent searches
 * filter them with the reference's CheckIfNotDeleted dispatch (already
 * covered by the parity suite). Persisted by save_index in deletes.bin. */
int sptag_amd_delete(SptagAmdIndex* idx, const int32_t* vids, int32_t n);
/* the by-vector overload (BKTIndex.cpp:876: search CEF results, delete the
 * exact duplicates at distance < 1e-6) */
int sptag_amd_delete_by_vector(SptagAmdIndex* idx, const void* vectors,
                               int32_t n);
int64_t sptag_amd_deleted_count(const SptagAmdIndex* idx);

/* Write the index back out in the reference's byte format
 * (vectors/tree/graph/deletes + indexloader.ini). */
int sptag_amd_save_index(SptagAmdIndex* idx, const char* folder);

/* Search-parameter setter — the C-ABI form of the reference's
 * SetParameter("NumberOfInitialDynamicPivots"/"NumberOfOtherDynamicPivots"/
 * "ThresholdOfNumberOfContinuousNoBetterPropagation"/"MaxCheck", ...)
 * (BKTIndex.cpp:980, ParameterDefinitionList.h). Pass <=0 to keep a
 * value unchanged. */
void sptag_amd_set_search_params(SptagAmdIndex* idx, int32_t init_pivots,
                                 int32_t other_pivots,
                                 int32_t nobetter_threshold,
                                 int32_t default_maxcheck);

/* metadata */
int32_t sptag_amd_num_vectors(const SptagAmdIndex* idx);
int32_t sptag_amd_dim(const SptagAmdIndex* idx);
int     sptag_amd_valuetype(const SptagAmdIndex* idx);
int     sptag_amd_distmethod(const SptagAmdIndex* idx);
int32_t sptag_amd_degree(const SptagAmdIndex* idx);
int32_t sptag_amd_default_maxcheck(const SptagAmdIndex* idx);
/* 0 = BKT, 1 = KDT (the loaded indexloader.ini IndexAlgoType). */
int     sptag_amd_algo(const SptagAmdIndex* idx);

/* Runtime probe: 1 if a usable HIP device exists (search will run), else 0. */
int sptag_amd_gpu_available(void);

/* Version/build tag of the loaded library (for the loud-failure message). */
const char* sptag_amd_build_info(void);

#ifdef __cplusplus
}
#endif
#endif
