/* amd_searcher — drives the GPU backend through the reference's OWN C++
 * virtual interface, exactly as the reference CLI does:
 *   load -> SetParameter("MaxCheck", ...) -> VectorIndex::SearchIndex()
 * (reference flow: src/IndexSearcher/main.cpp:185,206 per-query loop and
 * the batch overload VectorIndex.h:103). All index calls go through a
 * `std::shared_ptr<SPTAG::VectorIndex>` BASE pointer, so the virtual
 * dispatch boundary is what is exercised. Results are written in the
 * searcher's binary output format ([int32 nq][int32 k] + nq*k *
 * {int32 VID, float Dist}) for byte-diffing against the reference CPU
 * searcher's recorded answers (tests/golden/). The per-query virtual and
 * the one-launch batch virtual are cross-checked for equality here.
 *
 * Usage: amd_searcher <index_folder> <queries.bin> <k> <maxcheck> <out.bin>
 */
#include "amd_bkt_index.h"

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

int main(int argc, char** argv)
{
    if (argc < 6) {
        fprintf(stderr,
                "usage: amd_searcher <index_folder> <queries.bin> <k> "
                "<maxcheck> <out.bin>\n");
        return 2;
    }
    const char* folder = argv[1];
    const char* qfile = argv[2];
    int k = atoi(argv[3]);
    int mc = atoi(argv[4]);
    const char* outfile = argv[5];

    std::shared_ptr<SPTAG::VectorIndex> index =
        SPTAG::AMD::AmdBktIndex::Load(folder);
    if (!index) {
        fprintf(stderr, "amd_searcher: load failed: %s\n", folder);
        return 1;
    }
    if (index->SetParameter("MaxCheck", std::to_string(mc), "Index") !=
        SPTAG::ErrorCode::Success) {
        fprintf(stderr, "amd_searcher: SetParameter failed\n");
        return 1;
    }

    FILE* f = fopen(qfile, "rb");
    if (!f) {
        fprintf(stderr, "amd_searcher: cannot open %s\n", qfile);
        return 1;
    }
    int32_t nq = 0, dim = 0;
    if (fread(&nq, 4, 1, f) != 1 || fread(&dim, 4, 1, f) != 1) return 1;
    if (dim != index->GetFeatureDim()) {
        fprintf(stderr, "amd_searcher: query dim %d != index dim %d\n", dim,
                index->GetFeatureDim());
        return 1;
    }
    size_t esz =
        index->GetVectorValueType() == SPTAG::VectorValueType::Float ? 4 : 1;
    std::vector<char> queries((size_t)nq * dim * esz);
    if (fread(queries.data(), 1, queries.size(), f) != queries.size()) return 1;
    fclose(f);

    /* batch overload: one GPU launch for the whole set */
    std::vector<SPTAG::BasicResult> batch((size_t)nq * k);
    if (index->SearchIndex(queries.data(), nq, k, false, batch.data()) !=
        SPTAG::ErrorCode::Success) {
        fprintf(stderr, "amd_searcher: batch SearchIndex failed\n");
        return 1;
    }

    /* per-query virtual (the CLI thread-loop form), cross-checked */
    int mismatches = 0;
    for (int32_t qi = 0; qi < nq; qi++) {
        SPTAG::QueryResult qr(queries.data() + (size_t)qi * dim * esz, k,
                              false);
        if (index->SearchIndex(qr) != SPTAG::ErrorCode::Success) {
            fprintf(stderr, "amd_searcher: SearchIndex(QueryResult) failed\n");
            return 1;
        }
        for (int i = 0; i < k; i++) {
            const SPTAG::BasicResult* r = qr.GetResult(i);
            const SPTAG::BasicResult& b = batch[(size_t)qi * k + i];
            if (r->VID != b.VID || r->Dist != b.Dist) mismatches++;
        }
    }
    if (mismatches) {
        fprintf(stderr,
                "amd_searcher: %d per-query/batch result mismatches\n",
                mismatches);
        return 1;
    }

    FILE* out = fopen(outfile, "wb");
    if (!out) return 1;
    fwrite(&nq, 4, 1, out);
    fwrite(&k, 4, 1, out);
    for (size_t i = 0; i < batch.size(); i++) {
        int32_t vid = batch[i].VID;
        float dist = batch[i].Dist;
        fwrite(&vid, 4, 1, out);
        fwrite(&dist, 4, 1, out);
    }
    fclose(out);
    printf("amd_searcher: %d queries, k=%d, mc=%d -> %s (per-query == batch)\n",
           nq, k, mc, outfile);
    return 0;
}
