// Sampled-read window feature extraction (the framework's equivalent of the
// reference's generate.cpp:28-160 hot loop, re-architected as a single
// column-store sweep instead of an mpileup multi-iterator + per-column hash
// maps — see SURVEY.md §3.1 for the reference's structure).

#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace rk {

struct FeatureParams {
    int rows = 200;        // sampled read rows per window
    int cols = 90;         // (position, insertion) columns per window
    int stride = 30;       // columns between window starts
    int max_ins = 3;       // insertion slots materialised per position
    uint32_t filter_flag = 0x4 | 0x100 | 0x200 | 0x400 | 0x800;
    uint8_t min_mapq = 10;
    uint64_t seed = 0;     // mixed with (contig, start) by the caller
};

struct FeatureResult {
    int64_t n_windows = 0;
    std::vector<int32_t> positions;  // n_windows * cols * 2, (ref_pos, ins)
    std::vector<uint8_t> matrices;   // n_windows * rows * cols, base ids 0..11
};

// Build feature windows for pileup columns in [start, end) of `contig`.
// Columns are (position, insertion-slot) pairs; every column covered by the
// filtered pileup enters a sliding queue and windows of `cols` columns are
// emitted every `stride` columns. Rows are reads sampled uniformly WITH
// replacement from the reads overlapping the window (deterministic under
// `seed`). Windows whose column span no read covers are skipped (the
// reference has undefined behaviour there, generate.cpp:123).
FeatureResult extract_features(const std::string& bam_path, const std::string& contig,
                               int64_t start, int64_t end, const FeatureParams& params);

}  // namespace rk
