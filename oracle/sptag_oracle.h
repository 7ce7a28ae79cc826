/* ORACLE — CPU restatement of the SPTAG BKT in-memory search path.
 *
 * TEST INFRASTRUCTURE ONLY: this library is the parity checker for the HIP
 * backend. Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
 * leg may link or call it. It is never the product path: sptag_amd's search
 * fails loudly if the HIP extension is missing; it does not fall back here.
 *
 * Every function cites the reference file:line it restates (paths relative
 * to /root/reference/AnnService). Parity pinning: oracle results are checked
 * bit-exactly against the reference binaries in oracle/_ref (compiled from
 * the reference's own sources) on committed golden fixtures under
 * tests/golden/.
 */
#ifndef SPTAG_ORACLE_H
#define SPTAG_ORACLE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* value types (reference inc/Core/CommonDataStructure.h VectorValueType) */
#define ORC_VT_FLOAT 0
#define ORC_VT_INT8  1
/* distance methods (reference inc/Core/Common.h DistCalcMethod) */
#define ORC_DM_L2     0
#define ORC_DM_COSINE 1

typedef struct OrcIndex OrcIndex;

/* Load an index folder written by the reference SaveIndex
 * (vectors.bin/tree.bin/graph.bin/deletes.bin + indexloader.ini,
 * reference src/Core/VectorIndex.cpp:618 LoadIndex). Returns NULL on error. */
OrcIndex* orc_load_index(const char* folder);

/* Assemble an index from raw in-memory blobs (same layouts as the files,
 * without headers). Copies the data. tree_nodes is int32 triples
 * {centerid,childStart,childEnd} (reference BKTree.h:25). */
OrcIndex* orc_create_index(int32_t n, int32_t dim, int valuetype, int distmethod,
                           const void* vectors,
                           int32_t ntrees, const int32_t* tree_start,
                           int32_t n_tree_nodes, const int32_t* tree_nodes,
                           int32_t degree, const int32_t* graph,
                           const uint8_t* deleted /* may be NULL */);

/* KDT variant (reference inc/Core/Common/KDTree.h:22 KDTNode
 * {left,right,split_dim,split_value}); kdt_nodes is 16-byte records. */
OrcIndex* orc_create_kdt_index(int32_t n, int32_t dim, int valuetype, int distmethod,
                               const void* vectors,
                               int32_t ntrees, const int32_t* tree_start,
                               int32_t n_tree_nodes, const void* kdt_nodes,
                               int32_t degree, const int32_t* graph,
                               const uint8_t* deleted);

void orc_free_index(OrcIndex* idx);

int32_t orc_num_vectors(const OrcIndex* idx);
int32_t orc_dim(const OrcIndex* idx);
int orc_valuetype(const OrcIndex* idx);
int orc_distmethod(const OrcIndex* idx);
int32_t orc_degree(const OrcIndex* idx);
/* read-only views of the live index (tests: post-add graph comparison) */
const int32_t* orc_graph_ptr(const OrcIndex* idx);
const void* orc_vectors_ptr(const OrcIndex* idx);

/* Exact restatement of BKT::Index<T>::SearchIndex for one query
 * (reference src/Core/BKT/BKTIndex.cpp:272-352 Search<> +
 *  BKTree.h:697 InitSearchTrees, BKTree.h:772 SearchTrees).
 * searchDeleted=0, searchDuplicated=1 semantics (the default
 * SearchIndex(QueryResult&,bool) entry, BKTIndex.cpp:596-619).
 * Writes k (vid,dist) pairs ascending by (dist, vid); vid=-1 padding.
 * Returns the number of checked leaves (distance evaluations). */
int32_t orc_search(const OrcIndex* idx, const void* query, int32_t k,
                   int32_t max_check, int32_t* out_vids, float* out_dists);

/* Batch search, OpenMP over queries with nthreads (<=0: all cores). */
void orc_search_batch(const OrcIndex* idx, const void* queries, int32_t nq,
                      int32_t k, int32_t max_check, int nthreads,
                      int32_t* out_vids, float* out_dists);

/* Exact brute-force top-k (truth generation; reference
 * inc/Core/Common/TruthSet.h:163 GenerateTruth semantics: full scan with the
 * same distance function, top-k by (dist, vid)). */
void orc_truth(const OrcIndex* idx, const void* queries, int32_t nq, int32_t k,
               int nthreads, int32_t* out_vids, float* out_dists);

/* Online add — reference BKT AddIndex (BKTIndex.cpp:902-970), valid below
 * the tree-rebuild threshold (AddCountForRebuild). add_cef: reference
 * AddCEF default 500. */
int orc_add(OrcIndex* idx, const void* vectors, int32_t nadd, int32_t add_cef,
            int normalized);

/* Iterative (streaming) search — reference SearchIterative /
 * ResultIterator (BKTIndex.cpp:354-427, ResultIterator.cpp). BKT only. */
typedef struct OrcIter OrcIter;
OrcIter* orc_iter_create(const OrcIndex* idx, const void* query,
                         int32_t max_check);
int32_t orc_iter_next(OrcIter* it, int32_t batch, int32_t* out_vids,
                      float* out_dists, int32_t* relaxed_mono);
void orc_iter_free(OrcIter* it);

/* Distance between two raw vectors with the reference's summation order
 * (DistanceUtils.cpp AVX512 chunk/fold order for float; exact integer math
 * for int8). Exposed for unit tests. */
float orc_distance(int valuetype, int distmethod, const void* x, const void* y,
                   int32_t dim);

#ifdef __cplusplus
}
#endif
#endif
