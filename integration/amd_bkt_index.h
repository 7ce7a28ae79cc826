/* AmdBktIndex — a SPTAG::VectorIndex subclass (compiled against the
 * reference's own headers, /root/reference/AnnService/inc/Core/
 * VectorIndex.h:28) whose search virtuals route to the MI355X backend
 * through the C-ABI in include/sptag_amd.h. This is the literal C++
 * drop-in for the hot path: a caller holding a `VectorIndex*` (the
 * reference indexsearcher loop, main.cpp:206, or the SWIG wrappers,
 * Wrappers/src/CoreInterface.cpp:230) calls the same virtuals and gets
 * bit-identical results computed on the GPU.
 *
 * Scope: live methods are the search path —
 * SearchIndex(QueryResult&,bool) (VectorIndex.h:41), the batch overload
 * SearchIndex(const void*,int,int,bool,BasicResult*) (VectorIndex.h:103,
 * overridden to ONE GPU launch instead of the base's omp-for),
 * AddIndex/DeleteIndex, SaveIndex(folder), the metadata getters and
 * Set/GetParameter("MaxCheck" etc.). Out-of-scope virtuals (quantizers,
 * SPANN-era refine plumbing, streaming iterators via COMMON::WorkSpace —
 * the backend's own iterator API covers that surface) fail loudly with
 * ErrorCode::Undefined rather than silently degrading.
 *
 * This header is compiled ONLY where /root/reference exists (dev
 * container); the resulting binary lands in oracle/_ref/ and travels to
 * the GPU box like the other reference-built artifacts.
 */
#pragma once

#include "inc/Core/VectorIndex.h"

#include <string>
#include <vector>

struct SptagAmdIndex;

namespace SPTAG {
namespace AMD {

class AmdBktIndex : public VectorIndex {
public:
    explicit AmdBktIndex(SptagAmdIndex* h);
    ~AmdBktIndex() override;

    /* folder load through the C-ABI (byte-compatible with the reference's
     * LoadIndex folder format, VectorIndex.cpp:618). */
    static std::shared_ptr<VectorIndex> Load(const std::string& folder,
                                             int device = 0);

    /* ---- live search surface ---- */
    ErrorCode SearchIndex(QueryResult& p_query,
                          bool p_searchDeleted = false) const override;
    ErrorCode SearchIndex(const void* p_vectors, int p_vectorCount,
                          int p_neighborCount, bool p_withMeta,
                          BasicResult* p_results) const override;
    ErrorCode AddIndex(const void* p_data, SizeType p_vectorNum,
                       DimensionType p_dimension,
                       std::shared_ptr<MetadataSet> p_metadataSet,
                       bool p_withMetaIndex = false,
                       bool p_normalized = false) override;
    ErrorCode DeleteIndex(const void* p_vectors, SizeType p_vectorNum) override;
    ErrorCode DeleteIndex(const SizeType& p_id) override;
    ErrorCode SaveIndex(const std::string& p_folderPath) override;

    DimensionType GetFeatureDim() const override;
    SizeType GetNumSamples() const override;
    SizeType GetNumDeleted() const override;
    DistCalcMethod GetDistCalcMethod() const override;
    IndexAlgoType GetIndexAlgoType() const override;
    VectorValueType GetVectorValueType() const override;

    std::string GetParameter(const char* p_param,
                             const char* p_section = nullptr) const override;
    ErrorCode SetParameter(const char* p_param, const char* p_value,
                           const char* p_section = nullptr) override;

    /* ---- out-of-scope virtuals: loud stubs ---- */
    ErrorCode BuildIndex(const void*, SizeType, DimensionType, bool,
                         bool) override;
    std::shared_ptr<ResultIterator> GetIterator(const void*,
                                                bool) const override;
    ErrorCode SearchIndexIterativeNext(QueryResult&, COMMON::WorkSpace*, int,
                                       int&, bool, bool) const override;
    ErrorCode SearchIndexIterativeEnd(
        std::unique_ptr<COMMON::WorkSpace>) const override;
    bool SearchIndexIterativeFromNeareast(QueryResult&, COMMON::WorkSpace*,
                                          bool, bool) const override;
    std::unique_ptr<COMMON::WorkSpace> RentWorkSpace(int) const override;
    ErrorCode RefineSearchIndex(QueryResult&, bool) const override;
    ErrorCode SearchIndexWithFilter(QueryResult&,
                                    std::function<bool(const ByteArray&)>,
                                    int, bool) const override;
    ErrorCode SearchTree(QueryResult&) const override;
    ErrorCode RefineIndex(std::shared_ptr<VectorIndex>&) override;
    float AccurateDistance(const void*, const void*) const override;
    float ComputeDistance(const void*, const void*) const override;
    float GetDistance(const void*, const SizeType) const override;
    const void* GetSample(const SizeType) const override;
    bool ContainSample(const SizeType) const override;
    bool NeedRefine() const override;
    ErrorCode UpdateIndex() override;
    void SetQuantizer(std::shared_ptr<COMMON::IQuantizer>) override;

protected:
    std::shared_ptr<std::vector<std::uint64_t>> BufferSize() const override;
    std::shared_ptr<std::vector<std::string>> GetIndexFiles() const override;
    ErrorCode SaveConfig(std::shared_ptr<Helper::DiskIO>) override;
    ErrorCode SaveIndexData(
        const std::vector<std::shared_ptr<Helper::DiskIO>>&) override;
    ErrorCode LoadConfig(Helper::IniReader&) override;
    ErrorCode LoadIndexData(
        const std::vector<std::shared_ptr<Helper::DiskIO>>&) override;
    ErrorCode LoadIndexDataFromMemory(const std::vector<ByteArray>&) override;
    ErrorCode RefineIndex(const std::vector<std::shared_ptr<Helper::DiskIO>>&,
                          IAbortOperation*) override;
    ErrorCode SetWorkSpaceFactory(
        std::unique_ptr<COMMON::IWorkSpaceFactory<COMMON::IWorkSpace>>)
        override;

private:
    SptagAmdIndex* m_h;
    int m_maxCheck = 0;   /* 0 = the index's stored default */
};

}  // namespace AMD
}  // namespace SPTAG
