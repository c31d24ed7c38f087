// Column-store window builder. See pileup.h.
//
// Pipeline: (1) fetch filtered reads overlapping the region; (2) walk each
// CIGAR once, scattering (read, slot, base) events into a per-position column
// store; (3) sweep positions in order, materialising (pos, ins) column keys
// into a sliding queue and emitting a sampled-read window every `stride`
// columns. Semantics match the reference's emitted matrices:
//  * base ids 0-5 = A C G T GAP UNKNOWN, +6 when the read is reverse-strand
//    (the offset applies to GAP/UNKNOWN too, generate.cpp:145);
//  * a sampled read shows GAP at columns it doesn't cover while
//    ref_start <= pos <= ref_end and UNKNOWN outside (generate.cpp:134-139;
//    note the reference's `> ref_end` keeps pos == exclusive-end INSIDE the
//    read — replicated deliberately for feature-distribution parity with
//    published checkpoints);
//  * insertions are attached to the preceding aligned base, capped at
//    max_ins, and dropped after deletions (htslib pileup semantics as the
//    reference consumes them, generate.cpp:66-84);
//  * ambiguous read bases map to UNKNOWN (the reference throws on IUPAC codes
//    other than N, models.h:135-136 — widened here).
// Deliberate fixes over the reference (documented in SURVEY.md §7 step 2):
// deterministic seeding instead of srand(time) per call, and zero-coverage
// windows are skipped instead of undefined behaviour.

#include "pileup.h"

#include <algorithm>
#include <cstring>
#include <deque>
#include <random>
#include <stdexcept>

#include "bam.h"

namespace rk {

namespace {

constexpr uint8_t B_GAP = 4;
constexpr uint8_t B_UNKNOWN = 5;
constexpr uint8_t STRAND_OFFSET = 6;

inline uint8_t base_from_code4(uint8_t c) {
    switch (c) {
        case 1: return 0;   // A
        case 2: return 1;   // C
        case 4: return 2;   // G
        case 8: return 3;   // T
        default: return B_UNKNOWN;  // N and IUPAC ambiguity codes
    }
}

struct Event {
    uint32_t read;
    uint8_t slot;  // 0 = aligned/deleted base, 1..max_ins = insertion slots
    uint8_t base;  // 0..5
};

struct ReadMeta {
    int64_t ref_start;
    int64_t ref_end;  // exclusive
    uint8_t offset;   // 0 forward, 6 reverse — added to every emitted base
};

struct Column {
    std::vector<Event> events;
    uint8_t max_slot = 0;
    bool covered = false;  // has at least one slot-0 event
};

// splitmix64: deterministic per-region stream derivation
inline uint64_t mix64(uint64_t x) {
    x += 0x9e3779b97f4a7c15ull;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
    return x ^ (x >> 31);
}

}  // namespace

FeatureResult extract_features(const std::string& bam_path, const std::string& contig,
                               int64_t start, int64_t end, const FeatureParams& P) {
    BamReader bam(bam_path);
    int tid = bam.tid_of(contig);
    if (tid < 0) throw std::runtime_error("contig not in BAM header: " + contig);
    int64_t ref_len = bam.references()[tid].length;
    if (end > ref_len) end = ref_len;
    if (start < 0) start = 0;
    FeatureResult out;
    if (start >= end) return out;

    std::vector<Column> columns(static_cast<size_t>(end - start));
    std::vector<ReadMeta> reads;

    // ---- phase 1+2: fetch + CIGAR walk into the column store ----------------
    bam.fetch(tid, start, end, [&](const BamRecord& rec) {
        if (rec.flag & P.filter_flag) return;
        if ((rec.flag & FLAG_PAIRED) && !(rec.flag & FLAG_PROPER_PAIR)) return;
        if (rec.mapq < P.min_mapq) return;
        if (rec.l_seq <= 0 || rec.cigar.empty()) return;

        uint32_t rid = uint32_t(reads.size());
        reads.push_back({rec.pos, rec.ref_end(), rec.is_reverse() ? STRAND_OFFSET : uint8_t(0)});

        int64_t rpos = rec.pos;
        int64_t qpos = 0;
        const size_t nops = rec.cigar.size();
        for (size_t ci = 0; ci < nops; ++ci) {
            uint32_t op = rec.cigar[ci] & 0xf;
            int64_t len = rec.cigar[ci] >> 4;
            switch (op) {
                case CIG_M:
                case CIG_EQ:
                case CIG_X: {
                    for (int64_t k = 0; k < len; ++k) {
                        int64_t p = rpos + k;
                        if (p < start || p >= end) continue;
                        Column& col = columns[size_t(p - start)];
                        col.events.push_back({rid, 0, base_from_code4(rec.seqi(qpos + k))});
                        col.covered = true;
                    }
                    // insertion following the last aligned base of this chunk
                    if (ci + 1 < nops && (rec.cigar[ci + 1] & 0xf) == CIG_I) {
                        int64_t p = rpos + len - 1;
                        int64_t ins_len = rec.cigar[ci + 1] >> 4;
                        if (p >= start && p < end) {
                            Column& col = columns[size_t(p - start)];
                            int n = int(std::min<int64_t>(ins_len, P.max_ins));
                            for (int i = 1; i <= n; ++i) {
                                col.events.push_back(
                                    {rid, uint8_t(i),
                                     base_from_code4(rec.seqi(qpos + len - 1 + i))});
                                if (uint8_t(i) > col.max_slot) col.max_slot = uint8_t(i);
                            }
                        }
                    }
                    rpos += len;
                    qpos += len;
                    break;
                }
                case CIG_I:
                    // handled by the preceding aligned chunk; a leading
                    // insertion (or one after D/N) attaches nowhere — same as
                    // the reference's pileup view (is_del reads never expose
                    // insertions, generate.cpp:66-68)
                    qpos += len;
                    break;
                case CIG_D: {
                    for (int64_t k = 0; k < len; ++k) {
                        int64_t p = rpos + k;
                        if (p < start || p >= end) continue;
                        Column& col = columns[size_t(p - start)];
                        col.events.push_back({rid, 0, B_GAP});
                        col.covered = true;
                    }
                    rpos += len;
                    break;
                }
                case CIG_N:
                    rpos += len;  // refskip: no events (generate.cpp:54)
                    break;
                case CIG_S:
                    qpos += len;
                    break;
                default:
                    break;  // H, P consume nothing relevant
            }
        }
    });

    if (reads.empty()) return out;

    // ---- phase 3: sweep columns into the sliding window queue --------------
    struct Key {
        int32_t pos;
        int32_t ins;
        std::vector<std::pair<uint32_t, uint8_t>> entries;  // (read, base 0..5)
    };
    std::deque<Key> queue;

    const int W = P.cols, R = P.rows, S = P.stride;
    std::mt19937_64 rng(mix64(P.seed ^ mix64(uint64_t(tid) << 32 ^ uint64_t(start))));

    // scratch reused across windows
    std::vector<uint32_t> valid;                 // read ids with >=1 visible base
    std::vector<int64_t> valid_mark(reads.size(), -1);
    std::vector<int32_t> row_of(reads.size(), -1);
    std::vector<uint8_t> dense;                  // V x W staging matrix
    int64_t window_id = 0;

    auto emit_windows = [&]() {
        while (int64_t(queue.size()) >= W) {
            // -- collect reads visible in these W columns
            valid.clear();
            for (int s = 0; s < W; ++s) {
                const Key& k = queue[size_t(s)];
                for (const auto& e : k.entries) {
                    if (e.second != B_UNKNOWN && valid_mark[e.first] != window_id) {
                        valid_mark[e.first] = window_id;
                        valid.push_back(e.first);
                    }
                }
            }
            ++window_id;
            if (!valid.empty()) {
                std::sort(valid.begin(), valid.end());
                const int V = int(valid.size());
                for (int v = 0; v < V; ++v) row_of[valid[size_t(v)]] = v;

                // -- dense V x W staging: defaults by bounds, then events
                dense.assign(size_t(V) * W, B_UNKNOWN);
                const int32_t wposs = queue[0].pos;
                const int32_t wpose = queue[size_t(W - 1)].pos;
                for (int v = 0; v < V; ++v) {
                    const ReadMeta& m = reads[valid[size_t(v)]];
                    if (m.ref_start > wpose || m.ref_end < wposs) continue;
                    // find column range with ref_start <= pos <= ref_end
                    // (ref_end exclusive-end counted inside — see header note)
                    int lo = 0, hi = W - 1;
                    while (lo < W && queue[size_t(lo)].pos < m.ref_start) ++lo;
                    while (hi >= 0 && queue[size_t(hi)].pos > m.ref_end) --hi;
                    if (lo <= hi)
                        std::memset(&dense[size_t(v) * W + lo], B_GAP, size_t(hi - lo + 1));
                }
                for (int s = 0; s < W; ++s) {
                    const Key& k = queue[size_t(s)];
                    for (const auto& e : k.entries) {
                        int v = row_of[e.first];
                        if (v >= 0) dense[size_t(v) * W + s] = e.second;
                    }
                }

                // -- emit: positions + R sampled rows
                size_t pbase = out.positions.size();
                out.positions.resize(pbase + size_t(W) * 2);
                for (int s = 0; s < W; ++s) {
                    out.positions[pbase + 2 * size_t(s)] = queue[size_t(s)].pos;
                    out.positions[pbase + 2 * size_t(s) + 1] = queue[size_t(s)].ins;
                }
                size_t xbase = out.matrices.size();
                out.matrices.resize(xbase + size_t(R) * W);
                for (int r = 0; r < R; ++r) {
                    uint32_t pick = uint32_t(rng() % uint64_t(V));
                    uint32_t rid = valid[pick];
                    const uint8_t off = reads[rid].offset;
                    const uint8_t* src = &dense[size_t(row_of[rid]) * W];
                    uint8_t* dst = &out.matrices[xbase + size_t(r) * W];
                    for (int s = 0; s < W; ++s) dst[s] = uint8_t(src[s] + off);
                }
                out.n_windows++;
                for (int v = 0; v < V; ++v) row_of[valid[size_t(v)]] = -1;
            }
            // -- slide
            queue.erase(queue.begin(), queue.begin() + S);
        }
    };

    for (int64_t p = start; p < end; ++p) {
        Column& col = columns[size_t(p - start)];
        if (!col.covered) continue;
        // bucket events by slot into queue keys (slot order 0,1,..,max_slot)
        size_t kbase = queue.size();
        for (int slotn = 0; slotn <= col.max_slot; ++slotn)
            queue.push_back({int32_t(p), int32_t(slotn), {}});
        for (const Event& e : col.events)
            queue[kbase + e.slot].entries.emplace_back(e.read, e.base);
        col.events.clear();
        col.events.shrink_to_fit();
        emit_windows();
    }
    // Columns left in the queue (< W of them after the final emit) are
    // dropped — the region overlap covers the joins (features CLI), matching
    // the reference's tail behaviour (SURVEY.md §3.1 tail note).

    return out;
}

}  // namespace rk
