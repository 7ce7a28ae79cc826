// Golden-answer harness for the ITERATIVE search API, linked against the
// reference objects in oracle/_ref (built from /root/reference sources).
// Usage: iterprobe <index_folder> <queries.bin> <batch> <ncalls> <out.bin>
// Output: [int32 nq][int32 batch][int32 ncalls] then per query per call:
// [int32 count][int32 relaxed][batch x {int32 vid, float dist}]
#include "inc/Core/VectorIndex.h"
#include "inc/Core/ResultIterator.h"
#include <cstdio>
#include <cstdlib>
#include <vector>

int main(int argc, char** argv)
{
    if (argc < 6) { fprintf(stderr, "args\n"); return 2; }
    const char* folder = argv[1];
    const char* qfile = argv[2];
    int batch = atoi(argv[3]);
    int ncalls = atoi(argv[4]);
    const char* outfile = argv[5];

    std::shared_ptr<SPTAG::VectorIndex> index;
    if (SPTAG::VectorIndex::LoadIndex(folder, index) != SPTAG::ErrorCode::Success) {
        fprintf(stderr, "load failed\n");
        return 1;
    }
    FILE* qf = fopen(qfile, "rb");
    int32_t nq = 0, dim = 0;
    if (fread(&nq, 4, 1, qf) != 1 || fread(&dim, 4, 1, qf) != 1) return 1;
    size_t esz = index->GetVectorValueType() == SPTAG::VectorValueType::Float ? 4 : 1;
    std::vector<char> q((size_t)nq * dim * esz);
    if (fread(q.data(), 1, q.size(), qf) != q.size()) return 1;
    fclose(qf);

    FILE* out = fopen(outfile, "wb");
    fwrite(&nq, 4, 1, out);
    fwrite(&batch, 4, 1, out);
    fwrite(&ncalls, 4, 1, out);
    for (int32_t i = 0; i < nq; i++) {
        auto it = index->GetIterator(q.data() + (size_t)i * dim * esz, false);
        for (int c = 0; c < ncalls; c++) {
            auto res = it->Next(batch);
            int32_t count = res->GetResultNum();
            int32_t relaxed = it->GetRelaxedMono() ? 1 : 0;
            fwrite(&count, 4, 1, out);
            fwrite(&relaxed, 4, 1, out);
            for (int j = 0; j < batch; j++) {
                int32_t vid = j < count ? res->GetResult(j)->VID : -1;
                float dist = j < count ? res->GetResult(j)->Dist : 0.0f;
                fwrite(&vid, 4, 1, out);
                fwrite(&dist, 4, 1, out);
            }
        }
        it->Close();
    }
    fclose(out);
    return 0;
}
