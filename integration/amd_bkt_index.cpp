/* AmdBktIndex implementation — see amd_bkt_index.h. Routes the reference's
 * VectorIndex search virtuals (inc/Core/VectorIndex.h:41,103) to the
 * MI355X backend C-ABI (include/sptag_amd.h / libsptag_amd.so). */
#include "amd_bkt_index.h"

#include "sptag_amd.h"

#include <cstdio>
#include <cstring>

namespace SPTAG {
namespace AMD {

AmdBktIndex::AmdBktIndex(SptagAmdIndex* h) : m_h(h) {}

AmdBktIndex::~AmdBktIndex()
{
    if (m_h) sptag_amd_free_index(m_h);
}

std::shared_ptr<VectorIndex> AmdBktIndex::Load(const std::string& folder,
                                               int device)
{
    SptagAmdIndex* h = sptag_amd_load_index(folder.c_str(), device);
    if (!h) return nullptr;
    auto ix = std::make_shared<AmdBktIndex>(h);
    ix->SetReady(true);
    return ix;
}

/* VectorIndex.h:41 — the per-query virtual the reference CLI calls in its
 * thread loop (IndexSearcher/main.cpp:206). One query = a 1-row GPU batch;
 * concurrent callers are legal (the backend serializes on the handle). */
ErrorCode AmdBktIndex::SearchIndex(QueryResult& p_query,
                                   bool p_searchDeleted) const
{
    if (p_searchDeleted && sptag_amd_deleted_count(m_h) > 0)
        return ErrorCode::Undefined;   /* searchDeleted not exposed in ABI */
    int k = p_query.GetResultNum();
    std::vector<int32_t> vids(k);
    std::vector<float> dists(k);
    int rc = sptag_amd_search_batch(m_h, p_query.GetTarget(), 1, k, m_maxCheck,
                                    vids.data(), dists.data());
    if (rc != SPTAG_AMD_OK) return ErrorCode::Fail;
    for (int i = 0; i < k; i++) {
        BasicResult* r = p_query.GetResult(i);
        r->VID = vids[i];
        r->Dist = dists[i];
    }
    return ErrorCode::Success;
}

/* VectorIndex.h:103 — the batch overload (the seam the Python wrapper's
 * BatchSearch calls, Wrappers/src/CoreInterface.cpp:230). The base class
 * runs an omp-for of per-query searches; here it is ONE GPU launch. */
ErrorCode AmdBktIndex::SearchIndex(const void* p_vectors, int p_vectorCount,
                                   int p_neighborCount, bool p_withMeta,
                                   BasicResult* p_results) const
{
    (void)p_withMeta;   /* metadata sets are out of the hot-path scope */
    std::vector<int32_t> vids((size_t)p_vectorCount * p_neighborCount);
    std::vector<float> dists((size_t)p_vectorCount * p_neighborCount);
    int rc = sptag_amd_search_batch(m_h, p_vectors, p_vectorCount,
                                    p_neighborCount, m_maxCheck, vids.data(),
                                    dists.data());
    if (rc != SPTAG_AMD_OK) return ErrorCode::Fail;
    for (size_t i = 0; i < vids.size(); i++) {
        p_results[i].VID = vids[i];
        p_results[i].Dist = dists[i];
    }
    return ErrorCode::Success;
}

ErrorCode AmdBktIndex::AddIndex(const void* p_data, SizeType p_vectorNum,
                                DimensionType p_dimension,
                                std::shared_ptr<MetadataSet> p_metadataSet,
                                bool p_withMetaIndex, bool p_normalized)
{
    (void)p_metadataSet;
    (void)p_withMetaIndex;
    if (p_dimension != GetFeatureDim()) return ErrorCode::DimensionSizeMismatch;
    return sptag_amd_add(m_h, p_data, p_vectorNum, p_normalized ? 1 : 0) ==
                   SPTAG_AMD_OK
               ? ErrorCode::Success
               : ErrorCode::Fail;
}

ErrorCode AmdBktIndex::DeleteIndex(const void* p_vectors, SizeType p_vectorNum)
{
    return sptag_amd_delete_by_vector(m_h, p_vectors, p_vectorNum) ==
                   SPTAG_AMD_OK
               ? ErrorCode::Success
               : ErrorCode::Fail;
}

ErrorCode AmdBktIndex::DeleteIndex(const SizeType& p_id)
{
    int32_t id = (int32_t)p_id;
    return sptag_amd_delete(m_h, &id, 1) == SPTAG_AMD_OK ? ErrorCode::Success
                                                         : ErrorCode::Fail;
}

ErrorCode AmdBktIndex::SaveIndex(const std::string& p_folderPath)
{
    return sptag_amd_save_index(m_h, p_folderPath.c_str()) == SPTAG_AMD_OK
               ? ErrorCode::Success
               : ErrorCode::Fail;
}

DimensionType AmdBktIndex::GetFeatureDim() const { return sptag_amd_dim(m_h); }
SizeType AmdBktIndex::GetNumSamples() const { return sptag_amd_num_vectors(m_h); }
SizeType AmdBktIndex::GetNumDeleted() const
{
    return (SizeType)sptag_amd_deleted_count(m_h);
}
DistCalcMethod AmdBktIndex::GetDistCalcMethod() const
{
    return sptag_amd_distmethod(m_h) == SPTAG_AMD_DM_L2 ? DistCalcMethod::L2
                                                        : DistCalcMethod::Cosine;
}
IndexAlgoType AmdBktIndex::GetIndexAlgoType() const
{
    return sptag_amd_algo(m_h) == 1 ? IndexAlgoType::KDT : IndexAlgoType::BKT;
}
VectorValueType AmdBktIndex::GetVectorValueType() const
{
    return sptag_amd_valuetype(m_h) == SPTAG_AMD_VT_INT8
               ? VectorValueType::Int8
               : VectorValueType::Float;
}

std::string AmdBktIndex::GetParameter(const char* p_param,
                                      const char* p_section) const
{
    (void)p_section;
    if (std::strcmp(p_param, "MaxCheck") == 0)
        return std::to_string(m_maxCheck > 0 ? m_maxCheck
                                             : sptag_amd_default_maxcheck(m_h));
    return "Undefined!";
}

ErrorCode AmdBktIndex::SetParameter(const char* p_param, const char* p_value,
                                    const char* p_section)
{
    (void)p_section;
    /* the parameter the searcher CLI sweeps (main.cpp:185) */
    if (std::strcmp(p_param, "MaxCheck") == 0) {
        m_maxCheck = std::atoi(p_value);
        return ErrorCode::Success;
    }
    return ErrorCode::ParamNotFound;
}

/* ---- out-of-scope virtuals (SURVEY.md §8b scope note): loud stubs ---- */
#define AMD_STUB(ret)                                                        \
    do {                                                                     \
        fprintf(stderr, "AmdBktIndex: %s is outside the hot-path scope\n",   \
                __func__);                                                   \
        return ret;                                                          \
    } while (0)

ErrorCode AmdBktIndex::BuildIndex(const void*, SizeType, DimensionType, bool,
                                  bool)
{
    AMD_STUB(ErrorCode::Undefined);
}
std::shared_ptr<ResultIterator> AmdBktIndex::GetIterator(const void*,
                                                         bool) const
{
    /* the backend's iterator API (sptag_amd_iter_*) covers this surface;
     * the reference-class shim does not bridge COMMON::WorkSpace. */
    AMD_STUB(nullptr);
}
ErrorCode AmdBktIndex::SearchIndexIterativeNext(QueryResult&,
                                                COMMON::WorkSpace*, int, int&,
                                                bool, bool) const
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::SearchIndexIterativeEnd(
    std::unique_ptr<COMMON::WorkSpace>) const
{
    AMD_STUB(ErrorCode::Undefined);
}
bool AmdBktIndex::SearchIndexIterativeFromNeareast(QueryResult&,
                                                   COMMON::WorkSpace*, bool,
                                                   bool) const
{
    AMD_STUB(false);
}
std::unique_ptr<COMMON::WorkSpace> AmdBktIndex::RentWorkSpace(int) const
{
    AMD_STUB(nullptr);
}
ErrorCode AmdBktIndex::RefineSearchIndex(QueryResult&, bool) const
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::SearchIndexWithFilter(
    QueryResult&, std::function<bool(const ByteArray&)>, int, bool) const
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::SearchTree(QueryResult&) const
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::RefineIndex(std::shared_ptr<VectorIndex>&)
{
    AMD_STUB(ErrorCode::Undefined);
}
float AmdBktIndex::AccurateDistance(const void*, const void*) const
{
    AMD_STUB(0.0f);
}
float AmdBktIndex::ComputeDistance(const void*, const void*) const
{
    AMD_STUB(0.0f);
}
float AmdBktIndex::GetDistance(const void*, const SizeType) const
{
    AMD_STUB(0.0f);
}
const void* AmdBktIndex::GetSample(const SizeType) const { AMD_STUB(nullptr); }
bool AmdBktIndex::ContainSample(const SizeType idx) const
{
    return idx >= 0 && idx < GetNumSamples();
}
bool AmdBktIndex::NeedRefine() const { return false; }
ErrorCode AmdBktIndex::UpdateIndex() { return ErrorCode::Success; }
void AmdBktIndex::SetQuantizer(std::shared_ptr<COMMON::IQuantizer>)
{
    fprintf(stderr, "AmdBktIndex: quantizers are outside the hot-path scope\n");
}

std::shared_ptr<std::vector<std::uint64_t>> AmdBktIndex::BufferSize() const
{
    AMD_STUB(nullptr);
}
std::shared_ptr<std::vector<std::string>> AmdBktIndex::GetIndexFiles() const
{
    AMD_STUB(nullptr);
}
ErrorCode AmdBktIndex::SaveConfig(std::shared_ptr<Helper::DiskIO>)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::SaveIndexData(
    const std::vector<std::shared_ptr<Helper::DiskIO>>&)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::LoadConfig(Helper::IniReader&)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::LoadIndexData(
    const std::vector<std::shared_ptr<Helper::DiskIO>>&)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::LoadIndexDataFromMemory(const std::vector<ByteArray>&)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::RefineIndex(
    const std::vector<std::shared_ptr<Helper::DiskIO>>&, IAbortOperation*)
{
    AMD_STUB(ErrorCode::Undefined);
}
ErrorCode AmdBktIndex::SetWorkSpaceFactory(
    std::unique_ptr<COMMON::IWorkSpaceFactory<COMMON::IWorkSpace>>)
{
    AMD_STUB(ErrorCode::Undefined);
}

}  // namespace AMD
}  // namespace SPTAG
