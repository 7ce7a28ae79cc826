/* sptag_amd host library — implements the C-ABI in include/sptag_amd.h.
 *
 * Host code is C++ (as the reference's is); all compute runs in the HIP
 * kernels of kernels.hip. There is NO CPU search path here: if the HIP
 * runtime or device is unavailable, every search call fails loudly with
 * SPTAG_AMD_ERR_NOGPU.
 *
 * File formats are byte-compatible with the reference
 * (/root/reference/AnnService):
 *   vectors.bin  [int32 R][int32 C][row-major T]      Dataset.h:146
 *   tree.bin     [int32 #trees][int32 roots x #]
 *                [int32 count][{centerid,childStart,childEnd} x count]
 *                                                     BKTree.h:640-686
 *   graph.bin    [int32 R][int32 deg][int32 adj RxD]  NeighborhoodGraph.h:607
 *   deletes.bin  [int32 count][int32 R][int32 1][int8 x R]  Labelset.h:78
 *   indexloader.ini                                   VectorIndex.cpp:198-222
 */
#include "../../include/sptag_amd.h"

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <algorithm>
#include <cctype>
#include <cstring>
#include <mutex>
#include <string>
#include <vector>

#include "common.h"

using namespace sptag_amd;

namespace {

struct HostIndex {
    int device = 0;
    int algo = ALGO_BKT;
    int vt = VT_FLOAT, dm = DM_L2;
    int32_t n = 0, dim = 0, deg = 0, ntrees = 0, n_tree_nodes = 0;
    int32_t default_maxcheck = DEFAULT_MAXCHECK;
    int32_t init_pivots = INIT_PIVOTS, other_pivots = OTHER_PIVOTS;
    int32_t nobetter_threshold = 3;  /* KDT ParameterDefinitionList */
    bool refine_mode = false;  /* internal: RefineSearchIndex dispatch flags */
    bool has_deleted = false;
    /* device allocations */
    void* d_vectors = nullptr;
    int32_t* d_graph = nullptr;
    int32_t* d_tree = nullptr;
    int32_t* d_tree_start = nullptr;
    uint8_t* d_deleted = nullptr;
    /* host copies kept for save_index and shard assembly */
    std::vector<char> h_vectors;
    std::vector<int32_t> h_graph, h_tree, h_tree_start;
    std::vector<uint8_t> h_deleted;
    int64_t deleted_count = 0;
    std::mutex lock;
    /* cached per-handle search scratch (grown on demand) */
    int32_t* d_visited = nullptr; size_t visited_cap = 0;
    int32_t* d_oflow = nullptr;   size_t oflow_cap = 0;
    int32_t* d_stats = nullptr;   size_t stats_cap = 0;
    void* d_gng = nullptr;        size_t gng_cap = 0;
    void* d_gspt = nullptr;       size_t gspt_cap = 0;
    void* d_redo = nullptr;       size_t redo_cap = 0;
    void* d_gspt0 = nullptr;      size_t gspt0_cap = 0;
    hipEvent_t ev0 = nullptr, ev1 = nullptr;
    double last_kernel_ms = 0;
    long long last_checked = 0, last_popped = 0;

    size_t esz() const { return vt == VT_FLOAT ? 4 : 1; }
    DevIndex dev() const {
        DevIndex di;
        di.vectors = d_vectors;
        di.graph = d_graph;
        di.tree_nodes = d_tree;
        di.tree_start = d_tree_start;
        di.deleted = d_deleted;
        di.n = n; di.dim = dim; di.deg = deg; di.ntrees = ntrees;
        di.n_tree_nodes = n_tree_nodes;
        di.has_deleted = has_deleted ? 1 : 0;
        di.algo = algo;
        return di;
    }
};

#define HIP_OR_FAIL(expr, ret)                                              \
    do {                                                                    \
        hipError_t _e = (expr);                                             \
        if (_e != hipSuccess) {                                             \
            fprintf(stderr, "sptag_amd: HIP error %s at %s:%d\n",           \
                    hipGetErrorString(_e), __FILE__, __LINE__);             \
            return ret;                                                     \
        }                                                                   \
    } while (0)

bool read_file(const std::string& path, std::vector<char>& out)
{
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) return false;
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    out.resize((size_t)sz);
    bool ok = fread(out.data(), 1, (size_t)sz, f) == (size_t)sz;
    fclose(f);
    return ok;
}

/* SimpleIniReader-compatible lookup (reference src/Helper/
 * SimpleIniReader.cpp): case-insensitive section and key names, leading/
 * trailing whitespace around the key and value tolerated, ';' comment
 * lines skipped. This is the drop-in seam, so it accepts what the
 * reference's own parser would. */
static bool ieq(const std::string& a, const std::string& b)
{
    if (a.size() != b.size()) return false;
    for (size_t i = 0; i < a.size(); i++)
        if (tolower((unsigned char)a[i]) != tolower((unsigned char)b[i]))
            return false;
    return true;
}

static std::string trim(const std::string& s)
{
    size_t b = s.find_first_not_of(" \t\r");
    if (b == std::string::npos) return "";
    size_t e = s.find_last_not_of(" \t\r");
    return s.substr(b, e - b + 1);
}

bool ini_get(const std::string& text, const char* section, const char* key,
             std::string& out)
{
    bool in_section = false;
    size_t p = 0;
    while (p < text.size()) {
        size_t eol = text.find('\n', p);
        if (eol == std::string::npos) eol = text.size();
        std::string line = trim(text.substr(p, eol - p));
        p = eol + 1;
        if (line.empty() || line[0] == ';') continue;
        if (line.front() == '[') {
            size_t close = line.find(']');
            if (close == std::string::npos) continue;
            in_section = ieq(trim(line.substr(1, close - 1)), section);
            continue;
        }
        if (!in_section) continue;
        size_t eq = line.find('=');
        if (eq == std::string::npos) continue;
        if (ieq(trim(line.substr(0, eq)), key)) {
            out = trim(line.substr(eq + 1));
            return true;
        }
    }
    return false;
}

int upload(HostIndex* ix)
{
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMalloc(&ix->d_vectors, ix->h_vectors.size()), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(ix->d_vectors, ix->h_vectors.data(), ix->h_vectors.size(),
                          hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMalloc(&ix->d_graph, ix->h_graph.size() * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(ix->d_graph, ix->h_graph.data(), ix->h_graph.size() * 4,
                          hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMalloc(&ix->d_tree, ix->h_tree.size() * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(ix->d_tree, ix->h_tree.data(), ix->h_tree.size() * 4,
                          hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMalloc(&ix->d_tree_start, ix->h_tree_start.size() * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(ix->d_tree_start, ix->h_tree_start.data(),
                          ix->h_tree_start.size() * 4, hipMemcpyHostToDevice),
                SPTAG_AMD_ERR_NOGPU);
    if (ix->has_deleted) {
        HIP_OR_FAIL(hipMalloc(&ix->d_deleted, ix->h_deleted.size()), SPTAG_AMD_ERR_OOM);
        HIP_OR_FAIL(hipMemcpy(ix->d_deleted, ix->h_deleted.data(), ix->h_deleted.size(),
                              hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    }
    return SPTAG_AMD_OK;
}

uint32_t next_pow2(uint32_t x)
{
    uint32_t p = 1;
    while (p < x) p <<= 1;
    return p;
}

}  // namespace

struct SptagAmdIndex : HostIndex {};

extern "C" {

int sptag_amd_gpu_available(void)
{
    int n = 0;
    return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

const char* sptag_amd_build_info(void)
{
    return "sptag_amd gfx950 v1 (BKT search; HIP " __DATE__ ")";
}

SptagAmdIndex* sptag_amd_create_index(int32_t n, int32_t dim, int valuetype,
                                      int distmethod, const void* vectors,
                                      int32_t ntrees, const int32_t* tree_start,
                                      int32_t n_tree_nodes, const int32_t* tree_nodes,
                                      int32_t degree, const int32_t* graph,
                                      const uint8_t* deleted, int device)
{
    if (n <= 0 || dim <= 0 || dim > MAX_DIM || degree <= 0 || degree > MAX_DEG ||
        ntrees <= 0 || n_tree_nodes <= 0)
        return nullptr;
    if (valuetype == SPTAG_AMD_VT_INT8 && (int64_t)dim * 254 * 254 >= (1ll << 24)) {
        fprintf(stderr, "sptag_amd: int8 dim %d exceeds exact-float range\n", dim);
        return nullptr;
    }
    auto* ix = new SptagAmdIndex();
    ix->device = device;
    ix->vt = valuetype;
    ix->dm = distmethod;
    ix->n = n; ix->dim = dim; ix->deg = degree;
    ix->ntrees = ntrees;
    ix->h_vectors.assign((const char*)vectors,
                         (const char*)vectors + (size_t)n * dim * ix->esz());
    ix->h_graph.assign(graph, graph + (size_t)n * degree);
    ix->h_tree_start.assign(tree_start, tree_start + ntrees);
    ix->h_tree.assign(tree_nodes, tree_nodes + (size_t)n_tree_nodes * 3);
    /* BKTree.h:683: append sentinel if the stored array does not end with
     * centerid == -1 */
    if (ix->h_tree[(size_t)(n_tree_nodes - 1) * 3] != -1) {
        ix->h_tree.push_back(-1); ix->h_tree.push_back(-1); ix->h_tree.push_back(-1);
        n_tree_nodes += 1;
    }
    ix->n_tree_nodes = n_tree_nodes;
    if (deleted) {
        ix->h_deleted.assign(deleted, deleted + n);
        for (int32_t i = 0; i < n; i++) ix->deleted_count += deleted[i] ? 1 : 0;
        ix->has_deleted = ix->deleted_count > 0;
    }
    if (sptag_amd_gpu_available()) {
        if (upload(ix) != SPTAG_AMD_OK) {
            sptag_amd_free_index(ix);
            return nullptr;
        }
    }
    /* without a GPU the handle still supports metadata queries + save_index;
     * search fails with ERR_NOGPU. */
    return ix;
}

SptagAmdIndex* sptag_amd_create_index_kdt(int32_t n, int32_t dim, int valuetype,
                                          int distmethod, const void* vectors,
                                          int32_t ntrees, const int32_t* tree_start,
                                          int32_t n_tree_nodes, const void* kdt_nodes,
                                          int32_t degree, const int32_t* graph,
                                          const uint8_t* deleted, int device)
{
    if (n <= 0 || dim <= 0 || dim > MAX_DIM || degree <= 0 || degree > MAX_DEG ||
        ntrees <= 0 || n_tree_nodes <= 0)
        return nullptr;
    if (valuetype == SPTAG_AMD_VT_INT8 && (int64_t)dim * 254 * 254 >= (1ll << 24)) {
        /* same exact-float-range guard as the BKT create path: beyond it the
         * int8 distances stop being exact integers in float. */
        fprintf(stderr, "sptag_amd: int8 dim %d exceeds exact-float range\n", dim);
        return nullptr;
    }
    auto* ix = new SptagAmdIndex();
    ix->device = device;
    ix->algo = ALGO_KDT;
    ix->vt = valuetype;
    ix->dm = distmethod;
    ix->n = n; ix->dim = dim; ix->deg = degree;
    ix->ntrees = ntrees;
    ix->n_tree_nodes = n_tree_nodes;
    ix->h_vectors.assign((const char*)vectors,
                         (const char*)vectors + (size_t)n * dim * ix->esz());
    ix->h_graph.assign(graph, graph + (size_t)n * degree);
    ix->h_tree_start.assign(tree_start, tree_start + ntrees);
    const int32_t* kw = (const int32_t*)kdt_nodes;
    ix->h_tree.assign(kw, kw + (size_t)n_tree_nodes * 4);
    if (deleted) {
        ix->h_deleted.assign(deleted, deleted + n);
        for (int32_t i = 0; i < n; i++) ix->deleted_count += deleted[i] ? 1 : 0;
        ix->has_deleted = ix->deleted_count > 0;
    }
    if (sptag_amd_gpu_available()) {
        if (upload(ix) != SPTAG_AMD_OK) {
            sptag_amd_free_index(ix);
            return nullptr;
        }
    }
    return ix;
}

SptagAmdIndex* sptag_amd_load_index(const char* folder, int device)
{
    std::string dir(folder);
    if (!dir.empty() && dir.back() != '/') dir += '/';

    std::vector<char> raw;
    if (!read_file(dir + "indexloader.ini", raw)) return nullptr;
    std::string ini(raw.begin(), raw.end());

    std::string val;
    int algo = ALGO_BKT;
    if (ini_get(ini, "Index", "IndexAlgoType", val)) {
        if (val == "BKT") algo = ALGO_BKT;
        else if (val == "KDT") algo = ALGO_KDT;
        else {
            fprintf(stderr, "sptag_amd: IndexAlgoType %s not supported\n",
                    val.c_str());
            return nullptr;
        }
    }
    int vt = VT_FLOAT, dm = DM_L2;
    if (ini_get(ini, "Index", "ValueType", val)) {
        if (val == "Float") vt = VT_FLOAT;
        else if (val == "Int8") vt = VT_INT8;
        else {
            fprintf(stderr, "sptag_amd: ValueType %s not supported\n", val.c_str());
            return nullptr;
        }
    }
    if (ini_get(ini, "Index", "DistCalcMethod", val)) {
        if (val == "L2") dm = DM_L2;
        else if (val == "Cosine" || val == "InnerProduct") dm = DM_COSINE;
        else {
            fprintf(stderr, "sptag_amd: DistCalcMethod %s not supported\n", val.c_str());
            return nullptr;
        }
    }

    std::vector<char> vb, tb, gb, db;
    if (!read_file(dir + "vectors.bin", vb) || !read_file(dir + "tree.bin", tb) ||
        !read_file(dir + "graph.bin", gb))
        return nullptr;
    bool have_del = read_file(dir + "deletes.bin", db);

    const int32_t* vh = (const int32_t*)vb.data();
    int32_t n = vh[0], dim = vh[1];
    const int32_t* th = (const int32_t*)tb.data();
    int32_t ntrees = th[0];
    const int32_t* tstart = th + 1;
    int32_t nnodes = th[1 + ntrees];
    const int32_t* tnodes = th + 2 + ntrees;
    const int32_t* gh = (const int32_t*)gb.data();
    int32_t gn = gh[0], deg = gh[1];
    if (gn != n) {
        fprintf(stderr, "sptag_amd: graph rows %d != vectors %d\n", gn, n);
        return nullptr;
    }
    const uint8_t* del = nullptr;
    if (have_del && db.size() >= 12 + (size_t)n) {
        int32_t delcount = ((const int32_t*)db.data())[0];
        if (delcount > 0) del = (const uint8_t*)db.data() + 12;
    }

    SptagAmdIndex* ix;
    if (algo == ALGO_BKT)
        ix = sptag_amd_create_index(n, dim, vt, dm, vb.data() + 8, ntrees, tstart,
                                    nnodes, tnodes, deg, gh + 2, del, device);
    else
        ix = sptag_amd_create_index_kdt(n, dim, vt, dm, vb.data() + 8, ntrees,
                                        tstart, nnodes, tnodes, deg, gh + 2, del,
                                        device);
    if (!ix) return nullptr;
    if (ini_get(ini, "Index", "ThresholdOfNumberOfContinuousNoBetterPropagation", val))
        ix->nobetter_threshold = atoi(val.c_str());
    if (ini_get(ini, "Index", "MaxCheck", val)) ix->default_maxcheck = atoi(val.c_str());
    if (ini_get(ini, "Index", "NumberOfInitialDynamicPivots", val))
        ix->init_pivots = atoi(val.c_str());
    if (ini_get(ini, "Index", "NumberOfOtherDynamicPivots", val))
        ix->other_pivots = atoi(val.c_str());
    if (ini_get(ini, "Index", "EnableBfs", val) && atoi(val.c_str()) != 0) {
        fprintf(stderr, "sptag_amd: EnableBfs not supported\n");
        sptag_amd_free_index(ix);
        return nullptr;
    }
    return ix;
}

void sptag_amd_free_index(SptagAmdIndex* ix)
{
    if (!ix) return;
    if (ix->d_vectors) (void)hipFree(ix->d_vectors);
    if (ix->d_graph) (void)hipFree(ix->d_graph);
    if (ix->d_tree) (void)hipFree(ix->d_tree);
    if (ix->d_tree_start) (void)hipFree(ix->d_tree_start);
    if (ix->d_deleted) (void)hipFree(ix->d_deleted);
    if (ix->d_visited) (void)hipFree(ix->d_visited);
    if (ix->d_oflow) (void)hipFree(ix->d_oflow);
    if (ix->d_stats) (void)hipFree(ix->d_stats);
    if (ix->d_gng) (void)hipFree(ix->d_gng);
    if (ix->d_gspt) (void)hipFree(ix->d_gspt);
    if (ix->d_redo) (void)hipFree(ix->d_redo);
    if (ix->d_gspt0) (void)hipFree(ix->d_gspt0);
    if (ix->ev0) (void)hipEventDestroy(ix->ev0);
    if (ix->ev1) (void)hipEventDestroy(ix->ev1);
    delete ix;
}

static int ensure_cap(void** p, size_t* cap, size_t need)
{
    if (*cap >= need) return SPTAG_AMD_OK;
    if (*p) (void)hipFree(*p);
    *p = nullptr;
    *cap = 0;
    if (hipMalloc(p, need) != hipSuccess) return SPTAG_AMD_ERR_OOM;
    *cap = need;
    return SPTAG_AMD_OK;
}

/* core search on DEVICE buffers; scratch cached on the handle. */
static int search_device_core(SptagAmdIndex* ix, const void* d_q, int32_t nq,
                              int32_t k, int32_t max_check,
                              int32_t* d_vids, float* d_dists)
{
    if (max_check <= 0) max_check = ix->default_maxcheck;

    SearchCfg cfg;
    cfg.nq = nq;
    cfg.k = k;
    cfg.max_check = max_check;
    cfg.init_pivots = ix->init_pivots;
    cfg.other_pivots = ix->other_pivots;
    cfg.nobetter_threshold = ix->nobetter_threshold;
    cfg.search_dup = 1;        /* SearchIndex(QueryResult&,bool) defaults */
    cfg.search_deleted = 0;
    if (ix->refine_mode) { cfg.search_dup = 0; cfg.search_deleted = 1; }
    cfg.dpq_cap = std::max(max_check / 16, k);   /* WorkSpace.h:268 */
    /* k-deep (refine-class) searches keep expanding far past MaxCheck —
     * the budget is only consulted when a pop misses the k-deep results
     * queue (BKTIndex.cpp:326) — so the visited table must scale with k
     * as well (the reference's OptHashPosVector grows on demand,
     * WorkSpace.h:119; a fixed GPU table errors loudly on saturation). */
    cfg.vcap = (int32_t)next_pow2(
        (uint32_t)std::max(4096, max_check * 4 + 64 * k));
    /* LDS-variant capacities: sized to the TYPICAL traversal occupancy so
     * several workgroups fit per CU; the rare query that outgrows them is
     * rerun on the global-heap variant at the reference's own capacities
     * (WorkSpace.h:265) — a speed tradeoff, never a semantic one. */
    /* KDT inserts every computed neighbor into the frontier (no results-
     * queue gate), so its occupancy tracks `checked` itself. */
    cfg.ng_cap = ix->algo == ALGO_KDT ? max_check + 2048 : max_check / 2 + 512;
    if (k > 64) cfg.ng_cap += 4 * k;   /* k-deep frontiers run larger */
    if (const char* e = getenv("SPTAG_AMD_NG_CAP"))   /* perf experiments */
        cfg.ng_cap = std::max(256, atoi(e));
    cfg.spt_cap = 4096;
    /* measured on MI355X (gpurun_out/ab_v5.txt): both speculation forms
     * LOSE throughput at 10M/mc2048 (745k QPS off vs 622k on) — the extra
     * touch loads consume request bandwidth the resident waves already
     * saturate. Off by default; kept for experiments. */
    cfg.spec_flags = 0;
    if (const char* e = getenv("SPTAG_AMD_SPEC")) cfg.spec_flags = atoi(e);
    /* phase-cycle diagnostic (BKT kernels only — the KDT kernel writes
     * plain stride-2 stats) */
    cfg.prof = (getenv("SPTAG_AMD_PROF") && ix->algo == ALGO_BKT) ? 1 : 0;
    const int sstride = cfg.prof ? PROF_STATS : 2;

    int lds_limit = 64 * 1024;
    (void)hipDeviceGetAttribute(&lds_limit, hipDeviceAttributeMaxSharedMemoryPerBlock,
                                ix->device);
    bool lds_ok = lds_bytes(ix->dim, ix->esz(), cfg, true) <= (size_t)lds_limit;

    if (ensure_cap((void**)&ix->d_visited, &ix->visited_cap,
                   (size_t)nq * cfg.vcap * 4) != SPTAG_AMD_OK) return SPTAG_AMD_ERR_OOM;
    if (ensure_cap((void**)&ix->d_oflow, &ix->oflow_cap, (size_t)nq * 4) != SPTAG_AMD_OK)
        return SPTAG_AMD_ERR_OOM;
    if (ensure_cap((void**)&ix->d_stats, &ix->stats_cap,
                   (size_t)nq * sstride * 4) != SPTAG_AMD_OK)
        return SPTAG_AMD_ERR_OOM;
    if (!ix->ev0) { (void)hipEventCreate(&ix->ev0); (void)hipEventCreate(&ix->ev1); }

    if (ensure_cap(&ix->d_gspt0, &ix->gspt0_cap,
                   (size_t)nq * ((size_t)cfg.spt_cap + 1) * 8) != SPTAG_AMD_OK)
        return SPTAG_AMD_ERR_OOM;

    SearchBufs bufs;
    bufs.queries = d_q;
    bufs.out_vids = d_vids;
    bufs.out_dists = d_dists;
    bufs.visited = ix->d_visited;
    bufs.oflow = ix->d_oflow;
    bufs.stats = ix->d_stats;
    bufs.gheap_ng = nullptr;
    bufs.gheap_spt = ix->d_gspt0;

    ix->last_kernel_ms = 0;
    ix->last_checked = 0;
    ix->last_popped = 0;
    std::vector<int32_t> stats((size_t)sstride * nq);
    std::vector<int32_t> redo;
    bool all_global = !lds_ok;
    if (lds_ok) {
        HIP_OR_FAIL(hipMemsetAsync(ix->d_visited, 0, (size_t)nq * cfg.vcap * 4),
                    SPTAG_AMD_ERR_NOGPU);
        (void)hipEventRecord(ix->ev0);
        int err = launch_bkt_search(ix->vt, ix->dm, true, ix->dev(), cfg, bufs, nullptr);
        if (err != 0) {
            fprintf(stderr, "sptag_amd: launch failed %d\n", err);
            return SPTAG_AMD_ERR_INTERNAL;
        }
        (void)hipEventRecord(ix->ev1);
        HIP_OR_FAIL(hipDeviceSynchronize(), SPTAG_AMD_ERR_NOGPU);
        float ms = 0;
        (void)hipEventElapsedTime(&ms, ix->ev0, ix->ev1);
        ix->last_kernel_ms += ms;
        std::vector<int32_t> oflow(nq);
        HIP_OR_FAIL(hipMemcpy(oflow.data(), ix->d_oflow, (size_t)nq * 4,
                              hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
        HIP_OR_FAIL(hipMemcpy(stats.data(), ix->d_stats,
                              (size_t)nq * sstride * 4,
                              hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
        for (int32_t i = 0; i < nq; i++) {
            if (oflow[i]) {
                redo.push_back(i);
            } else {
                ix->last_checked += stats[(size_t)sstride * i];
                ix->last_popped += stats[(size_t)sstride * i + 1];
            }
        }
        if (cfg.prof && ix->algo == ALGO_BKT) {
            static const char* pname[10] = {
                "seed", "pop", "row", "serial", "cas",
                "dist", "insert", "tree", "spec", "sort"};
            double tot[10] = {};
            for (int32_t i = 0; i < nq; i++)
                for (int p = 0; p < 10; p++)
                    tot[p] += (double)stats[(size_t)sstride * i + 2 + p];
            double all = 0;
            for (int p = 0; p < 10; p++) all += tot[p];
            fprintf(stderr, "sptag_amd PROF (sum of per-query wave cycles, "
                            "%d queries, kernel %.2f ms):\n", nq, ms);
            for (int p = 0; p < 10; p++)
                fprintf(stderr, "  %-7s %14.0f  (%5.1f%%)\n", pname[p], tot[p],
                        all > 0 ? 100.0 * tot[p] / all : 0.0);
            double ngi = 0;
            for (int32_t i = 0; i < nq; i++)
                ngi += (double)stats[(size_t)sstride * i + 12];
            fprintf(stderr, "  (insert = ng-heap %14.0f + results-set rest)\n",
                    ngi);
        }
        if (!redo.empty() && getenv("SPTAG_AMD_DEBUG"))
            fprintf(stderr, "sptag_amd: %zu/%d queries overflowed (rerun)\n",
                    redo.size(), nq);
        if (redo.size() * 2 > (size_t)nq) {
            all_global = true;   /* cheaper to redo the whole batch */
            redo.clear();
            ix->last_checked = 0;
            ix->last_popped = 0;
        }
    }

    int32_t gq_n = all_global ? nq : (int32_t)redo.size();
    if (gq_n > 0) {
        SearchCfg c2 = cfg;
        c2.nq = gq_n;
        c2.ng_cap = max_check * 30;     /* reference capacities */
        c2.spt_cap = max_check * 10;
        if (ensure_cap(&ix->d_gng, &ix->gng_cap,
                       (size_t)gq_n * ((size_t)c2.ng_cap + 1) * 8) != SPTAG_AMD_OK)
            return SPTAG_AMD_ERR_OOM;
        if (ensure_cap(&ix->d_gspt, &ix->gspt_cap,
                       (size_t)gq_n * ((size_t)c2.spt_cap + 1) * 8) != SPTAG_AMD_OK)
            return SPTAG_AMD_ERR_OOM;
        SearchBufs b2 = bufs;
        b2.gheap_ng = ix->d_gng;
        b2.gheap_spt = ix->d_gspt;
        size_t row = (size_t)ix->dim * ix->esz();
        if (!all_global) {
            /* compact the flagged queries on-device */
            size_t qseg = ((size_t)gq_n * row + 15) & ~15ul;  /* align */
            size_t need = (size_t)gq_n * 4 + 16 + qseg + (size_t)gq_n * k * 8;
            if (ensure_cap(&ix->d_redo, &ix->redo_cap, need) != SPTAG_AMD_OK)
                return SPTAG_AMD_ERR_OOM;
            int32_t* d_idx = (int32_t*)ix->d_redo;
            char* d_q2 = (char*)ix->d_redo + (((size_t)gq_n * 4 + 15) & ~15ul);
            int32_t* d_v2 = (int32_t*)(d_q2 + qseg);
            float* d_d2 = (float*)(d_v2 + (size_t)gq_n * k);
            HIP_OR_FAIL(hipMemcpy(d_idx, redo.data(), (size_t)gq_n * 4,
                                  hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
            launch_gather_rows(d_q2, d_q, (int)row, d_idx, gq_n, nullptr);
            b2.queries = d_q2;
            b2.out_vids = d_v2;
            b2.out_dists = d_d2;
        }
        HIP_OR_FAIL(hipMemsetAsync(ix->d_visited, 0, (size_t)gq_n * cfg.vcap * 4),
                    SPTAG_AMD_ERR_NOGPU);
        (void)hipEventRecord(ix->ev0);
        int err = launch_bkt_search(ix->vt, ix->dm, false, ix->dev(), c2, b2, nullptr);
        if (err != 0) {
            fprintf(stderr, "sptag_amd: global-variant launch failed %d\n", err);
            return SPTAG_AMD_ERR_INTERNAL;
        }
        (void)hipEventRecord(ix->ev1);
        if (!all_global) {
            int32_t* d_idx = (int32_t*)ix->d_redo;
            launch_scatter_rows(d_vids, b2.out_vids, k * 4, d_idx, gq_n, nullptr);
            launch_scatter_rows(d_dists, b2.out_dists, k * 4, d_idx, gq_n, nullptr);
        }
        HIP_OR_FAIL(hipDeviceSynchronize(), SPTAG_AMD_ERR_NOGPU);
        float ms = 0;
        (void)hipEventElapsedTime(&ms, ix->ev0, ix->ev1);
        ix->last_kernel_ms += ms;
        HIP_OR_FAIL(hipMemcpy(stats.data(), ix->d_stats,
                              (size_t)gq_n * sstride * 4,
                              hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
        for (int32_t i = 0; i < gq_n; i++) {
            ix->last_checked += stats[(size_t)sstride * i];
            ix->last_popped += stats[(size_t)sstride * i + 1];
        }
        /* a query can still overflow at reference capacities only through
         * visited-table saturation (probe give-up) — that would be a silent
         * result truncation, so it is an error, not a degradation. */
        std::vector<int32_t> oflow2(gq_n);
        HIP_OR_FAIL(hipMemcpy(oflow2.data(), ix->d_oflow, (size_t)gq_n * 4,
                              hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
        for (int32_t i = 0; i < gq_n; i++) {
            if (oflow2[i]) {
                fprintf(stderr,
                        "sptag_amd: query overflowed at reference capacities "
                        "(visited-table saturation)\n");
                return SPTAG_AMD_ERR_INTERNAL;
            }
        }
    }
    return SPTAG_AMD_OK;
}

static int check_search_args(SptagAmdIndex* ix, const void* q, int32_t nq, int32_t k)
{
    if (!ix || !q || nq <= 0 || k <= 0 || k > MAX_K) return SPTAG_AMD_ERR_PARAM;
    if (!sptag_amd_gpu_available() || !ix->d_vectors) {
        fprintf(stderr,
                "sptag_amd: search requires a HIP device (%s); no CPU fallback\n",
                sptag_amd_build_info());
        return SPTAG_AMD_ERR_NOGPU;
    }
    return SPTAG_AMD_OK;
}

int sptag_amd_search_batch_device(SptagAmdIndex* ix, const void* d_queries,
                                  int32_t nq, int32_t k, int32_t max_check,
                                  int32_t* d_out_vids, float* d_out_dists)
{
    int rc = check_search_args(ix, d_queries, nq, k);
    if (rc != SPTAG_AMD_OK) return rc;
    std::lock_guard<std::mutex> g(ix->lock);
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);
    return search_device_core(ix, d_queries, nq, k, max_check, d_out_vids,
                              d_out_dists);
}

int sptag_amd_search_batch(SptagAmdIndex* ix, const void* queries, int32_t nq,
                           int32_t k, int32_t max_check,
                           int32_t* out_vids, float* out_dists)
{
    int rc = check_search_args(ix, queries, nq, k);
    if (rc != SPTAG_AMD_OK) return rc;
    std::lock_guard<std::mutex> g(ix->lock);
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);

    size_t qbytes = (size_t)nq * ix->dim * ix->esz();
    void* d_q = nullptr;
    int32_t* d_vids = nullptr;
    float* d_dists = nullptr;
    rc = SPTAG_AMD_ERR_NOGPU;
    do {
        if (hipMalloc(&d_q, qbytes) != hipSuccess) { rc = SPTAG_AMD_ERR_OOM; break; }
        if (hipMalloc(&d_vids, (size_t)nq * k * 4) != hipSuccess) { rc = SPTAG_AMD_ERR_OOM; break; }
        if (hipMalloc(&d_dists, (size_t)nq * k * 4) != hipSuccess) { rc = SPTAG_AMD_ERR_OOM; break; }
        if (hipMemcpy(d_q, queries, qbytes, hipMemcpyHostToDevice) != hipSuccess) break;
        rc = search_device_core(ix, d_q, nq, k, max_check, d_vids, d_dists);
        if (rc != SPTAG_AMD_OK) break;
        if (hipMemcpy(out_vids, d_vids, (size_t)nq * k * 4, hipMemcpyDeviceToHost)
            != hipSuccess) { rc = SPTAG_AMD_ERR_NOGPU; break; }
        if (hipMemcpy(out_dists, d_dists, (size_t)nq * k * 4, hipMemcpyDeviceToHost)
            != hipSuccess) { rc = SPTAG_AMD_ERR_NOGPU; break; }
        rc = SPTAG_AMD_OK;
    } while (0);
    if (d_q) (void)hipFree(d_q);
    if (d_vids) (void)hipFree(d_vids);
    if (d_dists) (void)hipFree(d_dists);
    return rc;
}

void sptag_amd_last_stats(SptagAmdIndex* ix, double* kernel_ms,
                          long long* checked, long long* popped)
{
    if (!ix) return;
    if (kernel_ms) *kernel_ms = ix->last_kernel_ms;
    if (checked) *checked = ix->last_checked;
    if (popped) *popped = ix->last_popped;
}

struct SptagAmdIterBatch {
    SptagAmdIndex* ix;
    int32_t nq, max_check, vcap, ng_cap, spt_cap, dpq_alloc;
    void* d_q = nullptr;
    void* d_ng = nullptr;
    void* d_spt = nullptr;
    float* d_dpq = nullptr;
    int32_t* d_visited = nullptr;
    IterState* d_state = nullptr;
    int32_t* d_ov = nullptr;
    float* d_od = nullptr;
    int32_t* d_oc = nullptr;
    int32_t* d_or = nullptr;
    int32_t out_cap = 0;
};

SptagAmdIterBatch* sptag_amd_iter_create(SptagAmdIndex* ix, const void* queries,
                                         int32_t nq, int32_t max_check)
{
    if (!ix || !queries || nq <= 0) return nullptr;
    if (ix->algo != ALGO_BKT) {
        /* reference parity: KDT::Index<T>::GetIterator logs "ITERATIVE NOT
         * SUPPORT FOR KDT" and returns null (KDTIndex.cpp:322-346) */
        fprintf(stderr, "sptag_amd: iterative search is BKT-only "
                        "(as in the reference: KDTIndex.cpp:322)\n");
        return nullptr;
    }
    if (!sptag_amd_gpu_available() || !ix->d_vectors) {
        fprintf(stderr, "sptag_amd: iterator requires a HIP device\n");
        return nullptr;
    }
    if (max_check <= 0) max_check = ix->default_maxcheck;
    auto* it = new SptagAmdIterBatch();
    it->ix = ix;
    it->nq = nq;
    it->max_check = max_check;
    it->ng_cap = max_check * 30;    /* reference WorkSpace.h:265 capacities */
    it->spt_cap = max_check * 10;
    it->dpq_alloc = std::max(max_check / 16, (int)MAX_K);
    it->vcap = (int32_t)next_pow2((uint32_t)std::max(4096, max_check * 4));
    size_t qb = (size_t)nq * ix->dim * ix->esz();
    bool ok = hipMalloc(&it->d_q, qb) == hipSuccess &&
        hipMemcpy(it->d_q, queries, qb, hipMemcpyHostToDevice) == hipSuccess &&
        hipMalloc(&it->d_ng, (size_t)nq * (it->ng_cap + 1) * 8) == hipSuccess &&
        hipMalloc(&it->d_spt, (size_t)nq * (it->spt_cap + 1) * 8) == hipSuccess &&
        hipMalloc(&it->d_dpq, (size_t)nq * (it->dpq_alloc + 1) * 4) == hipSuccess &&
        hipMalloc(&it->d_visited, (size_t)nq * it->vcap * 4) == hipSuccess &&
        hipMemset(it->d_visited, 0, (size_t)nq * it->vcap * 4) == hipSuccess &&
        hipMalloc(&it->d_state, (size_t)nq * sizeof(IterState)) == hipSuccess &&
        hipMalloc(&it->d_oc, (size_t)nq * 4) == hipSuccess &&
        hipMalloc(&it->d_or, (size_t)nq * 4) == hipSuccess;
    if (ok) {
        std::vector<IterState> init(nq);
        for (auto& st : init) { st = IterState{}; st.first = 1; }
        ok = hipMemcpy(it->d_state, init.data(), (size_t)nq * sizeof(IterState),
                       hipMemcpyHostToDevice) == hipSuccess;
    }
    if (!ok) {
        sptag_amd_iter_free(it);
        return nullptr;
    }
    return it;
}

void sptag_amd_iter_free(SptagAmdIterBatch* it)
{
    if (!it) return;
    if (it->d_q) (void)hipFree(it->d_q);
    if (it->d_ng) (void)hipFree(it->d_ng);
    if (it->d_spt) (void)hipFree(it->d_spt);
    if (it->d_dpq) (void)hipFree(it->d_dpq);
    if (it->d_visited) (void)hipFree(it->d_visited);
    if (it->d_state) (void)hipFree(it->d_state);
    if (it->d_ov) (void)hipFree(it->d_ov);
    if (it->d_od) (void)hipFree(it->d_od);
    if (it->d_oc) (void)hipFree(it->d_oc);
    if (it->d_or) (void)hipFree(it->d_or);
    delete it;
}

int sptag_amd_iter_next(SptagAmdIterBatch* it, int32_t batch,
                        int32_t* out_vids, float* out_dists,
                        int32_t* out_counts, int32_t* out_relaxed)
{
    if (!it || batch <= 0 || batch > MAX_K) return SPTAG_AMD_ERR_PARAM;
    SptagAmdIndex* ix = it->ix;
    std::lock_guard<std::mutex> g(ix->lock);
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);
    if (batch > it->out_cap) {
        if (it->d_ov) (void)hipFree(it->d_ov);
        if (it->d_od) (void)hipFree(it->d_od);
        it->d_ov = nullptr; it->d_od = nullptr; it->out_cap = 0;
        if (hipMalloc(&it->d_ov, (size_t)it->nq * batch * 4) != hipSuccess ||
            hipMalloc(&it->d_od, (size_t)it->nq * batch * 4) != hipSuccess)
            return SPTAG_AMD_ERR_OOM;
        it->out_cap = batch;
    }
    SearchCfg cfg;
    cfg.nq = it->nq;
    cfg.k = batch;
    cfg.max_check = it->max_check;
    cfg.init_pivots = ix->init_pivots;
    cfg.other_pivots = ix->other_pivots;
    cfg.nobetter_threshold = ix->nobetter_threshold;
    cfg.search_dup = 1;
    cfg.search_deleted = 0;
    cfg.dpq_cap = std::max(it->max_check / 16, batch);  /* ResetResult cap */
    cfg.vcap = it->vcap;
    cfg.ng_cap = it->ng_cap;
    cfg.spt_cap = it->spt_cap;
    cfg.spec_flags = 0;
    if (const char* e = getenv("SPTAG_AMD_SPEC")) cfg.spec_flags = atoi(e);
    cfg.prof = 0;

    IterBufs ib;
    ib.queries = it->d_q;
    ib.gheap_ng = it->d_ng;
    ib.gheap_spt = it->d_spt;
    ib.dpq = it->d_dpq;
    ib.visited = it->d_visited;
    ib.state = it->d_state;
    ib.out_vids = it->d_ov;
    ib.out_dists = it->d_od;
    ib.out_counts = it->d_oc;
    ib.out_relaxed = it->d_or;
    int err = launch_bkt_iter(ix->vt, ix->dm, ix->dev(), cfg, ib, batch, nullptr);
    if (err != 0) {
        fprintf(stderr, "sptag_amd: iter launch failed %d\n", err);
        return SPTAG_AMD_ERR_INTERNAL;
    }
    HIP_OR_FAIL(hipDeviceSynchronize(), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMemcpy(out_vids, it->d_ov, (size_t)it->nq * batch * 4,
                          hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMemcpy(out_dists, it->d_od, (size_t)it->nq * batch * 4,
                          hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMemcpy(out_counts, it->d_oc, (size_t)it->nq * 4,
                          hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
    if (out_relaxed)
        HIP_OR_FAIL(hipMemcpy(out_relaxed, it->d_or, (size_t)it->nq * 4,
                              hipMemcpyDeviceToHost), SPTAG_AMD_ERR_NOGPU);
    return SPTAG_AMD_OK;
}

int sptag_amd_truth(SptagAmdIndex* ix, const void* queries, int32_t nq,
                    int32_t k, int32_t* out_vids, float* out_dists)
{
    if (!ix || !queries || nq <= 0 || k <= 0 || k > MAX_K) return SPTAG_AMD_ERR_PARAM;
    if (!sptag_amd_gpu_available() || !ix->d_vectors) return SPTAG_AMD_ERR_NOGPU;
    std::lock_guard<std::mutex> g(ix->lock);
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);

    size_t qbytes = (size_t)nq * ix->dim * ix->esz();
    void* d_q = nullptr;
    int32_t* d_v = nullptr;
    float* d_d = nullptr;
    HIP_OR_FAIL(hipMalloc(&d_q, qbytes), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_v, (size_t)nq * k * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_d, (size_t)nq * k * 4), SPTAG_AMD_ERR_OOM);
    int rc = SPTAG_AMD_ERR_NOGPU;
    if (hipMemcpy(d_q, queries, qbytes, hipMemcpyHostToDevice) == hipSuccess) {
        int err = launch_truth(ix->vt, ix->dm, ix->dev(), d_q, nq, k, d_v, d_d,
                               nullptr);
        if (err == 0 && hipDeviceSynchronize() == hipSuccess &&
            hipMemcpy(out_vids, d_v, (size_t)nq * k * 4,
                      hipMemcpyDeviceToHost) == hipSuccess &&
            hipMemcpy(out_dists, d_d, (size_t)nq * k * 4,
                      hipMemcpyDeviceToHost) == hipSuccess)
            rc = SPTAG_AMD_OK;
    }
    (void)hipFree(d_q);
    (void)hipFree(d_v);
    (void)hipFree(d_d);
    return rc;
}

/* Host-side distance restatements for the add path's edge arithmetic
 * (RebuildNeighbors / InsertNeighbors recompute pair distances on the
 * host, as the reference does on CPU): same rounding contracts as the
 * kernels/oracle (fused 16-lane order for f32, exact ints for int8). */
static float host_dist(const SptagAmdIndex* ix, const void* a, const void* b)
{
    int d = ix->dim;
    if (ix->vt == VT_INT8) {
        const int8_t* x = (const int8_t*)a;
        const int8_t* y = (const int8_t*)b;
        int32_t s = 0;
        if (ix->dm == DM_L2) {
            for (int i = 0; i < d; i++) { int32_t t = (int32_t)x[i] - y[i]; s += t * t; }
            return (float)s;
        }
        for (int i = 0; i < d; i++) s += (int32_t)x[i] * y[i];
        return (float)(16129 - s);
    }
    const float* x = (const float*)a;
    const float* y = (const float*)b;
    float a16[16], a8[8], a4[4];
    int i = 0, j;
    for (j = 0; j < 16; j++) a16[j] = 0.0f;
    int nd16 = (d >> 4) << 4;
    for (; i < nd16; i += 16)
        for (j = 0; j < 16; j++) {
            float xx = x[i + j], yy = y[i + j];
            a16[j] = ix->dm == DM_L2 ? fmaf(xx - yy, xx - yy, a16[j])
                                     : fmaf(xx, yy, a16[j]);
        }
    for (j = 0; j < 8; j++) a8[j] = a16[j] + a16[j + 8];
    int nd8 = (d >> 3) << 3;
    for (; i < nd8; i += 8)
        for (j = 0; j < 8; j++) {
            float xx = x[i + j], yy = y[i + j];
            a8[j] = ix->dm == DM_L2 ? fmaf(xx - yy, xx - yy, a8[j])
                                    : fmaf(xx, yy, a8[j]);
        }
    for (j = 0; j < 4; j++) a4[j] = a8[j] + a8[j + 4];
    int nd4 = (d >> 2) << 2;
    for (; i < nd4; i += 4)
        for (j = 0; j < 4; j++) {
            float xx = x[i + j], yy = y[i + j];
            a4[j] = ix->dm == DM_L2 ? fmaf(xx - yy, xx - yy, a4[j])
                                    : fmaf(xx, yy, a4[j]);
        }
    float diff = ((a4[0] + a4[1]) + a4[2]) + a4[3];
    for (; i < d; i++) {
        float xx = x[i], yy = y[i];
        diff = ix->dm == DM_L2 ? fmaf(xx - yy, xx - yy, diff) : fmaf(xx, yy, diff);
    }
    return ix->dm == DM_L2 ? diff : 1.0f - diff;
}

static const void* hvec(const SptagAmdIndex* ix, int32_t v)
{
    return ix->h_vectors.data() + (size_t)v * ix->dim * ix->esz();
}

/* RelativeNeighborhoodGraph::RebuildNeighbors (RelativeNeighborhoodGraph.h:18) */
static void host_rebuild_neighbors(SptagAmdIndex* ix, int32_t node,
                                   const int32_t* rv, const float* rd, int nres)
{
    int32_t* nodes = ix->h_graph.data() + (size_t)node * ix->deg;
    int count = 0;
    for (int j = 0; j < nres && count < ix->deg; j++) {
        if (rv[j] < 0) break;
        if (rv[j] == node) continue;
        bool good = true;
        for (int kk = 0; kk < count; kk++) {
            if (host_dist(ix, hvec(ix, nodes[kk]), hvec(ix, rv[j])) < rd[j]) {
                good = false;
                break;
            }
        }
        if (good) nodes[count++] = rv[j];
    }
    for (int j = count; j < ix->deg; j++) nodes[j] = -1;
}

/* RelativeNeighborhoodGraph::InsertNeighbors (RelativeNeighborhoodGraph.h:37) */
static void host_insert_neighbors(SptagAmdIndex* ix, int32_t node,
                                  int32_t insertNode, float insertDist)
{
    int32_t* nodes = ix->h_graph.data() + (size_t)node * ix->deg;
    const void* nodeVec = hvec(ix, node);
    const void* insertVec = hvec(ix, insertNode);
    int checkSize = (nodes[ix->deg - 1] < -1) ? ix->deg - 1 : ix->deg;
    for (int k = 0; k < checkSize; k++) {
        int32_t tmpNode = nodes[k];
        if (tmpNode < 0) { nodes[k] = insertNode; break; }
        const void* tmpVec = hvec(ix, tmpNode);
        float tmpDist = host_dist(ix, tmpVec, nodeVec);
        if (tmpDist > insertDist ||
            (insertDist == tmpDist && insertNode < tmpNode)) {
            nodes[k] = insertNode;
            while (++k < checkSize &&
                   host_dist(ix, tmpVec, nodeVec) <= host_dist(ix, tmpVec, insertVec)) {
                std::swap(tmpNode, nodes[k]);
                if (tmpNode < 0) return;
                tmpVec = hvec(ix, tmpNode);
            }
            break;
        } else if (host_dist(ix, tmpVec, insertVec) < insertDist) {
            break;
        }
    }
}

int sptag_amd_add(SptagAmdIndex* ix, const void* vectors, int32_t nadd,
                  int normalized)
{
    if (!ix || !vectors || nadd <= 0) return SPTAG_AMD_ERR_PARAM;
    if (ix->algo != ALGO_BKT) return SPTAG_AMD_ERR_UNSUPP;
    if (!sptag_amd_gpu_available() || !ix->d_vectors) {
        fprintf(stderr, "sptag_amd: add requires a HIP device (refine searches)\n");
        return SPTAG_AMD_ERR_NOGPU;
    }
    std::lock_guard<std::mutex> g(ix->lock);
    HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);

    int32_t begin = ix->n, end = ix->n + nadd;
    size_t esz = ix->esz();
    /* host copies (AddIndex: m_pSamples/m_pGraph/m_deletedID AddBatch) */
    ix->h_vectors.resize((size_t)end * ix->dim * esz);
    memcpy(ix->h_vectors.data() + (size_t)begin * ix->dim * esz, vectors,
           (size_t)nadd * ix->dim * esz);
    ix->h_graph.resize((size_t)end * ix->deg, -1);
    if (!ix->h_deleted.empty()) ix->h_deleted.resize((size_t)end, 0);
    ix->n = end;

    if (ix->dm == DM_COSINE && !normalized) {
        /* Utils::Normalize to norm=base, C-cast truncation for int8
         * (CommonUtils.h:62) */
        for (int32_t i = begin; i < end; i++) {
            char* vp = ix->h_vectors.data() + (size_t)i * ix->dim * esz;
            double s = 0;
            if (ix->vt == VT_FLOAT) {
                float* v = (float*)vp;
                for (int d = 0; d < ix->dim; d++) s += (double)v[d] * v[d];
                s = sqrt(s);
                if (s > 0) for (int d = 0; d < ix->dim; d++) v[d] = (float)(v[d] / s);
            } else {
                int8_t* v = (int8_t*)vp;
                for (int d = 0; d < ix->dim; d++) s += (double)v[d] * v[d];
                s = sqrt(s);
                if (s > 0) for (int d = 0; d < ix->dim; d++)
                    v[d] = (int8_t)(v[d] * 127.0 / s);
            }
        }
    }

    /* device: re-allocate vectors+graph at the new size */
    void* nv = nullptr;
    int32_t* ng = nullptr;
    HIP_OR_FAIL(hipMalloc(&nv, ix->h_vectors.size()), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(nv, ix->h_vectors.data(), ix->h_vectors.size(),
                          hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    HIP_OR_FAIL(hipMalloc(&ng, ix->h_graph.size() * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMemcpy(ng, ix->h_graph.data(), ix->h_graph.size() * 4,
                          hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
    (void)hipFree(ix->d_vectors);
    (void)hipFree(ix->d_graph);
    ix->d_vectors = nv;
    ix->d_graph = ng;
    if (ix->has_deleted) {
        (void)hipFree(ix->d_deleted);
        HIP_OR_FAIL(hipMalloc(&ix->d_deleted, ix->h_deleted.size()), SPTAG_AMD_ERR_OOM);
        HIP_OR_FAIL(hipMemcpy(ix->d_deleted, ix->h_deleted.data(),
                              ix->h_deleted.size(), hipMemcpyHostToDevice),
                    SPTAG_AMD_ERR_NOGPU);
    }

    /* per new node: RefineNode (search CEF+1 at MaxCheckForRefineGraph with
     * the refine dispatch flags, then RNG rebuild + two-way inserts).
     * Sequential, as the reference's add loop is (BKTIndex.cpp:966-969);
     * each refine sees the edges of previously added nodes. */
    const int k = 500 + 1;   /* AddCEF default (ParameterDefinitionList) */
    std::vector<int32_t> rv(k);
    std::vector<float> rd(k);
    int32_t* d_v = nullptr;
    float* d_d = nullptr;
    void* d_q = nullptr;
    /* staging for the batched changed-row upload: one H2D + one scatter
     * per added node instead of one 128 B hipMemcpy per touched edge (the
     * round-1 form did up to ~500 tiny copies per add). The SEARCH stays
     * sequential per node — reference AddIndex semantics (BKTIndex.cpp:
     * 966-969: each refine sees every previously added node's edges). */
    int32_t* d_rows = nullptr;     /* staged row payloads */
    int32_t* d_rowidx = nullptr;   /* their row ids */
    HIP_OR_FAIL(hipMalloc(&d_v, (size_t)k * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_d, (size_t)k * 4), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_q, (size_t)ix->dim * esz), SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_rows, (size_t)(k + 1) * ix->deg * 4),
                SPTAG_AMD_ERR_OOM);
    HIP_OR_FAIL(hipMalloc(&d_rowidx, (size_t)(k + 1) * 4), SPTAG_AMD_ERR_OOM);
    std::vector<int32_t> hrows((size_t)(k + 1) * ix->deg);
    std::vector<int32_t> hidx(k + 1);
    int rc = SPTAG_AMD_OK;
    for (int32_t node = begin; node < end && rc == SPTAG_AMD_OK; node++) {
        HIP_OR_FAIL(hipMemcpy(d_q, hvec(ix, node), (size_t)ix->dim * esz,
                              hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
        ix->refine_mode = true;
        rc = search_device_core(ix, d_q, 1, k, 8192 /*MaxCheckForRefineGraph*/,
                                d_v, d_d);
        ix->refine_mode = false;
        if (rc != SPTAG_AMD_OK) break;
        HIP_OR_FAIL(hipMemcpy(rv.data(), d_v, (size_t)k * 4, hipMemcpyDeviceToHost),
                    SPTAG_AMD_ERR_NOGPU);
        HIP_OR_FAIL(hipMemcpy(rd.data(), d_d, (size_t)k * 4, hipMemcpyDeviceToHost),
                    SPTAG_AMD_ERR_NOGPU);
        host_rebuild_neighbors(ix, node, rv.data(), rd.data(), k);
        int nrows = 0;
        hidx[nrows++] = node;
        for (int j = 0; j < k; j++) {
            if (rv[j] < 0) break;
            if (rv[j] == node) continue;
            host_insert_neighbors(ix, rv[j], node, rd[j]);
            hidx[nrows++] = rv[j];
        }
        /* dedup row ids, stage payloads, one H2D + one scatter */
        std::sort(hidx.begin(), hidx.begin() + nrows);
        nrows = (int)(std::unique(hidx.begin(), hidx.begin() + nrows) -
                      hidx.begin());
        for (int j = 0; j < nrows; j++)
            memcpy(&hrows[(size_t)j * ix->deg],
                   ix->h_graph.data() + (size_t)hidx[j] * ix->deg,
                   (size_t)ix->deg * 4);
        HIP_OR_FAIL(hipMemcpy(d_rows, hrows.data(),
                              (size_t)nrows * ix->deg * 4,
                              hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
        HIP_OR_FAIL(hipMemcpy(d_rowidx, hidx.data(), (size_t)nrows * 4,
                              hipMemcpyHostToDevice), SPTAG_AMD_ERR_NOGPU);
        launch_scatter_rows(ix->d_graph, d_rows, ix->deg * 4, d_rowidx, nrows,
                            nullptr);
        HIP_OR_FAIL(hipDeviceSynchronize(), SPTAG_AMD_ERR_NOGPU);
    }
    (void)hipFree(d_v);
    (void)hipFree(d_d);
    (void)hipFree(d_q);
    (void)hipFree(d_rows);
    (void)hipFree(d_rowidx);
    return rc;
}

int sptag_amd_delete_by_vector(SptagAmdIndex* ix, const void* vectors,
                               int32_t n)
{
    /* reference DeleteIndex(const void*, SizeType) (BKTIndex.cpp:876-891):
     * search CEF=1000 results per vector, delete every id at distance
     * < 1e-6 (exact duplicates of the given vector). */
    if (!ix || !vectors || n <= 0) return SPTAG_AMD_ERR_PARAM;
    if (ix->algo != ALGO_BKT) return SPTAG_AMD_ERR_UNSUPP;
    if (!sptag_amd_gpu_available() || !ix->d_vectors) return SPTAG_AMD_ERR_NOGPU;
    const int k = 1000;   /* m_iCEF default */
    std::vector<int32_t> rv((size_t)n * k);
    std::vector<float> rd((size_t)n * k);
    int rc = sptag_amd_search_batch(ix, vectors, n, k, 0, rv.data(), rd.data());
    if (rc != SPTAG_AMD_OK) return rc;
    std::vector<int32_t> dels;
    for (int32_t i = 0; i < n; i++)
        for (int j = 0; j < k; j++) {
            if (rv[(size_t)i * k + j] < 0) break;
            if (rd[(size_t)i * k + j] < 1e-6f) dels.push_back(rv[(size_t)i * k + j]);
        }
    if (dels.empty()) return SPTAG_AMD_OK;
    return sptag_amd_delete(ix, dels.data(), (int32_t)dels.size());
}

int sptag_amd_delete(SptagAmdIndex* ix, const int32_t* vids, int32_t n)
{
    if (!ix || !vids || n <= 0) return SPTAG_AMD_ERR_PARAM;
    std::lock_guard<std::mutex> g(ix->lock);
    if (ix->h_deleted.empty()) ix->h_deleted.assign((size_t)ix->n, 0);
    for (int32_t i = 0; i < n; i++) {
        int32_t v = vids[i];
        if (v < 0 || v >= ix->n) return SPTAG_AMD_ERR_PARAM;
        if (!ix->h_deleted[v]) {
            ix->h_deleted[v] = 1;
            ix->deleted_count++;
        }
    }
    ix->has_deleted = ix->deleted_count > 0;
    if (sptag_amd_gpu_available() && ix->d_vectors) {
        HIP_OR_FAIL(hipSetDevice(ix->device), SPTAG_AMD_ERR_NOGPU);
        if (!ix->d_deleted)
            HIP_OR_FAIL(hipMalloc(&ix->d_deleted, ix->h_deleted.size()),
                        SPTAG_AMD_ERR_OOM);
        HIP_OR_FAIL(hipMemcpy(ix->d_deleted, ix->h_deleted.data(),
                              ix->h_deleted.size(), hipMemcpyHostToDevice),
                    SPTAG_AMD_ERR_NOGPU);
    }
    return SPTAG_AMD_OK;
}

int64_t sptag_amd_deleted_count(const SptagAmdIndex* ix)
{
    return ix ? ix->deleted_count : -1;
}

int sptag_amd_save_index(SptagAmdIndex* ix, const char* folder)
{
    if (!ix || !folder) return SPTAG_AMD_ERR_PARAM;
    std::string dir(folder);
    if (!dir.empty() && dir.back() != '/') dir += '/';

    auto wfile = [&](const std::string& name, const void* hdr, size_t hdrsz,
                     const void* body, size_t bodysz) -> bool {
        FILE* f = fopen((dir + name).c_str(), "wb");
        if (!f) return false;
        bool ok = fwrite(hdr, 1, hdrsz, f) == hdrsz &&
                  (bodysz == 0 || fwrite(body, 1, bodysz, f) == bodysz);
        fclose(f);
        return ok;
    };

    int32_t vh[2] = {ix->n, ix->dim};
    if (!wfile("vectors.bin", vh, 8, ix->h_vectors.data(), ix->h_vectors.size()))
        return SPTAG_AMD_ERR_IO;
    {
        FILE* f = fopen((dir + "tree.bin").c_str(), "wb");
        if (!f) return SPTAG_AMD_ERR_IO;
        fwrite(&ix->ntrees, 4, 1, f);
        fwrite(ix->h_tree_start.data(), 4, ix->ntrees, f);
        fwrite(&ix->n_tree_nodes, 4, 1, f);
        fwrite(ix->h_tree.data(), ix->algo == ALGO_KDT ? 16 : 12,
               ix->n_tree_nodes, f);
        fclose(f);
    }
    int32_t gh[2] = {ix->n, ix->deg};
    if (!wfile("graph.bin", gh, 8, ix->h_graph.data(), ix->h_graph.size() * 4))
        return SPTAG_AMD_ERR_IO;
    {
        std::vector<uint8_t> del = ix->h_deleted;
        if (del.empty()) del.assign((size_t)ix->n, 0);
        int32_t dh[3] = {(int32_t)ix->deleted_count, ix->n, 1};
        if (!wfile("deletes.bin", dh, 12, del.data(), del.size()))
            return SPTAG_AMD_ERR_IO;
    }
    {
        FILE* f = fopen((dir + "indexloader.ini").c_str(), "wb");
        if (!f) return SPTAG_AMD_ERR_IO;
        if (ix->algo == ALGO_KDT) {
            fprintf(f,
                    "[Index]\nIndexAlgoType=KDT\nValueType=%s\n\n"
                    "TreeFilePath=tree.bin\nGraphFilePath=graph.bin\n"
                    "VectorFilePath=vectors.bin\nDeleteVectorFilePath=deletes.bin\n"
                    "KDTNumber=%d\nNumTopDimensionKDTSplit=5\nSamples=100\n"
                    "TPTNumber=32\nTPTLeafSize=2000\nNumTopDimensionTPTSplit=5\n"
                    "NeighborhoodSize=%d\nGraphNeighborhoodScale=2.000000\n"
                    "GraphCEFScale=2.000000\nRefineIterations=2\nCEF=1000\n"
                    "MaxCheckForRefineGraph=8192\nNumberOfThreads=4\n"
                    "DistCalcMethod=%s\nMaxCheck=%d\n"
                    "ThresholdOfNumberOfContinuousNoBetterPropagation=%d\n"
                    "NumberOfInitialDynamicPivots=%d\n"
                    "NumberOfOtherDynamicPivots=%d\nHashTableExponent=2\n"
                    "DataBlockSize=1048576\nDataCapacity=2147483647\n"
                    "MetaRecordSize=10\n",
                    ix->vt == VT_FLOAT ? "Float" : "Int8", ix->ntrees, ix->deg,
                    ix->dm == DM_L2 ? "L2" : "Cosine", ix->default_maxcheck,
                    ix->nobetter_threshold, ix->init_pivots, ix->other_pivots);
            fclose(f);
            return SPTAG_AMD_OK;
        }
        fprintf(f,
                "[Index]\n"
                "IndexAlgoType=BKT\n"
                "ValueType=%s\n\n"
                "TreeFilePath=tree.bin\nGraphFilePath=graph.bin\n"
                "VectorFilePath=vectors.bin\nDeleteVectorFilePath=deletes.bin\n"
                "EnableBfs=0\nBKTNumber=%d\nBKTKmeansK=32\nBKTLeafSize=8\n"
                "Samples=1000\nBKTLambdaFactor=100.000000\nTPTNumber=32\n"
                "TPTLeafSize=2000\nNumTopDimensionTpTreeSplit=5\n"
                "NeighborhoodSize=%d\nGraphNeighborhoodScale=2.000000\n"
                "GraphCEFScale=2.000000\nRefineIterations=2\nEnableRebuild=0\n"
                "CEF=1000\nAddCEF=500\nMaxCheckForRefineGraph=8192\n"
                "RNGFactor=1.000000\nGPUGraphType=2\nGPURefineSteps=0\n"
                "GPURefineDepth=30\nGPULeafSize=500\nHeadNumGPUs=1\n"
                "TPTBalanceFactor=2\nNumberOfThreads=4\nDistCalcMethod=%s\n"
                "DeletePercentageForRefine=0.400000\nAddCountForRebuild=1000\n"
                "MaxCheck=%d\n"
                "ThresholdOfNumberOfContinuousNoBetterPropagation=3\n"
                "NumberOfInitialDynamicPivots=%d\nNumberOfOtherDynamicPivots=%d\n"
                "HashTableExponent=2\nDataBlockSize=1048576\n"
                "DataCapacity=2147483647\nMetaRecordSize=10\n",
                ix->vt == VT_FLOAT ? "Float" : "Int8", ix->ntrees, ix->deg,
                ix->dm == DM_L2 ? "L2" : "Cosine", ix->default_maxcheck,
                ix->init_pivots, ix->other_pivots);
        fclose(f);
    }
    return SPTAG_AMD_OK;
}

int32_t sptag_amd_num_vectors(const SptagAmdIndex* ix) { return ix ? ix->n : -1; }
int32_t sptag_amd_dim(const SptagAmdIndex* ix) { return ix ? ix->dim : -1; }
int sptag_amd_valuetype(const SptagAmdIndex* ix) { return ix ? ix->vt : -1; }
int sptag_amd_distmethod(const SptagAmdIndex* ix) { return ix ? ix->dm : -1; }
int32_t sptag_amd_degree(const SptagAmdIndex* ix) { return ix ? ix->deg : -1; }
int32_t sptag_amd_default_maxcheck(const SptagAmdIndex* ix)
{ return ix ? ix->default_maxcheck : -1; }
int sptag_amd_algo(const SptagAmdIndex* ix) { return ix ? ix->algo : -1; }

void sptag_amd_set_search_params(SptagAmdIndex* ix, int32_t init_pivots,
                                 int32_t other_pivots,
                                 int32_t nobetter_threshold,
                                 int32_t default_maxcheck)
{
    if (!ix) return;
    if (init_pivots > 0) ix->init_pivots = init_pivots;
    if (other_pivots > 0) ix->other_pivots = other_pivots;
    if (nobetter_threshold > 0) ix->nobetter_threshold = nobetter_threshold;
    if (default_maxcheck > 0) ix->default_maxcheck = default_maxcheck;
}

}  /* extern "C" */
