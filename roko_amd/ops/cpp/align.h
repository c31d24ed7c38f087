#pragma once
#include <cstdint>
#include <string>

namespace rk {

struct AlignStats {
    int64_t edit_distance = 0;
    int64_t matches = 0;
    int64_t mismatches = 0;
    int64_t insertions = 0;  // extra bases in the query w.r.t. the target
    int64_t deletions = 0;   // target bases missing from the query
};

// Banded unit-cost global alignment of query vs target (band = half-width
// in diagonals beyond the length difference). Throws when the optimal path
// cannot be represented in the band. When `cigar` is non-null it receives
// the run-length encoded alignment path (query as SEQ: M/I/D ops, '='/'X'
// collapsed into M — the form BAM records use).
AlignStats align_stats(const std::string& query, const std::string& target,
                       int band, std::string* cigar = nullptr);

}  // namespace rk
