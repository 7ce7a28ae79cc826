// Golden-answer harness for the ONLINE ADD path, linked against the
// reference objects in oracle/_ref.
// Usage: addprobe <index_folder> <add_vectors.bin> <out_folder>
// Loads the index, AddIndex()es the vectors, saves the post-add index.
#include "inc/Core/VectorIndex.h"
#include <cstdio>
#include <vector>

int main(int argc, char** argv)
{
    if (argc < 4) { fprintf(stderr, "args\n"); return 2; }
    std::shared_ptr<SPTAG::VectorIndex> index;
    if (SPTAG::VectorIndex::LoadIndex(argv[1], index) != SPTAG::ErrorCode::Success)
        return 1;
    FILE* f = fopen(argv[2], "rb");
    int32_t n = 0, dim = 0;
    if (fread(&n, 4, 1, f) != 1 || fread(&dim, 4, 1, f) != 1) return 1;
    size_t esz = index->GetVectorValueType() == SPTAG::VectorValueType::Float ? 4 : 1;
    std::vector<char> v((size_t)n * dim * esz);
    if (fread(v.data(), 1, v.size(), f) != v.size()) return 1;
    fclose(f);
    if (index->AddIndex(v.data(), n, dim, nullptr) != SPTAG::ErrorCode::Success)
        return 1;
    if (index->SaveIndex(argv[3]) != SPTAG::ErrorCode::Success) return 1;
    return 0;
}
