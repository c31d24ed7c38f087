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
// cannot be represented in the band.
AlignStats align_stats(const std::string& query, const std::string& target,
                       int band);

}  // namespace rk
