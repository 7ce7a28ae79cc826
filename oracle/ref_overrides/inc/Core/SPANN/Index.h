// Build-time stub for the reference compile in oracle/_ref (see oracle/Makefile).
// The SPANN disk index is out of scope (zstd submodule not vendored)
// and is never exercised; VectorIndex.cpp only needs the symbol SPANN::Index<T>
// to instantiate in its CreateInstance switch (reference VectorIndex.cpp:600-605).
// We alias it to the in-memory BKT index so the reference core links without zstd.
#ifndef _SPTAG_SPANN_INDEX_H_
#define _SPTAG_SPANN_INDEX_H_

#include "inc/Core/BKT/Index.h"

namespace SPTAG
{
    namespace SPANN
    {
        template <typename T>
        class Index : public SPTAG::BKT::Index<T>
        {
        };
    }
}

#endif
