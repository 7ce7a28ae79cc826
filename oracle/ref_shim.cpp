// Shim TU for the oracle/_ref reference build (see oracle/Makefile).
// Intentionally empty: all needed globals come from the compiled reference
// TUs (AsyncFileReader.cpp provides Helper::SetThreadAffinity; VectorIndex.cpp
// provides SPTAG::rg and f_createIO).
