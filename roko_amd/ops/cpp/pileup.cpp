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

// one flat, append-only event stream instead of 300k per-column vectors
// (the vector-per-column version spent its time in allocator churn):
// events are counting-sorted by (column, slot) key in one O(N) pass
struct Event {
    uint32_t key;   // (pos - start) * (max_ins + 1) + slot
    uint32_t read;
    uint8_t base;   // 0..5
};

struct ReadMeta {
    int64_t ref_start;
    int64_t ref_end;  // exclusive
    uint8_t offset;   // 0 forward, 6 reverse — added to every emitted base
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

    const uint32_t SLOTS = uint32_t(P.max_ins) + 1;
    std::vector<Event> events;
    events.reserve(1 << 20);
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
                    int64_t k0 = std::max<int64_t>(0, start - rpos);
                    int64_t k1 = std::min<int64_t>(len, end - rpos);
                    for (int64_t k = k0; k < k1; ++k) {
                        events.push_back(
                            {uint32_t(rpos + k - start) * SLOTS, rid,
                             base_from_code4(rec.seqi(qpos + k))});
                    }
                    // insertion following the last aligned base of this chunk
                    if (ci + 1 < nops && (rec.cigar[ci + 1] & 0xf) == CIG_I) {
                        int64_t p = rpos + len - 1;
                        int64_t ins_len = rec.cigar[ci + 1] >> 4;
                        if (p >= start && p < end) {
                            int n = int(std::min<int64_t>(ins_len, P.max_ins));
                            for (int i = 1; i <= n; ++i) {
                                events.push_back(
                                    {uint32_t(p - start) * SLOTS + uint32_t(i), rid,
                                     base_from_code4(rec.seqi(qpos + len - 1 + i))});
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
                    int64_t k0 = std::max<int64_t>(0, start - rpos);
                    int64_t k1 = std::min<int64_t>(len, end - rpos);
                    for (int64_t k = k0; k < k1; ++k)
                        events.push_back(
                            {uint32_t(rpos + k - start) * SLOTS, rid, B_GAP});
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

    if (reads.empty() || events.empty()) return out;

    // ---- counting sort by (column, slot) key: one stable O(N) scatter -----
    const size_t nkeys = size_t(end - start) * SLOTS;
    std::vector<uint32_t> cnt(nkeys + 1, 0);
    for (const Event& e : events) cnt[size_t(e.key) + 1]++;
    for (size_t i = 0; i < nkeys; ++i) cnt[i + 1] += cnt[i];
    std::vector<Event> sorted(events.size());
    {
        std::vector<uint32_t> cur(cnt.begin(), cnt.end() - 1);
        for (const Event& e : events) sorted[cur[size_t(e.key)]++] = e;
    }
    events.clear();
    events.shrink_to_fit();

    // ---- sweep columns into the sliding window queue -----------------------
    struct Key {
        int32_t pos;
        int32_t ins;
        uint32_t beg;   // span into `sorted`
        uint32_t end_;
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
            for (int sc = 0; sc < W; ++sc) {
                const Key& k = queue[size_t(sc)];
                for (uint32_t i = k.beg; i < k.end_; ++i) {
                    const Event& e = sorted[i];
                    if (e.base != B_UNKNOWN && valid_mark[e.read] != window_id) {
                        valid_mark[e.read] = window_id;
                        valid.push_back(e.read);
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
                    // column range with ref_start <= pos <= ref_end
                    // (ref_end exclusive-end counted inside — header note)
                    int lo = 0, hi = W - 1;
                    while (lo < W && queue[size_t(lo)].pos < m.ref_start) ++lo;
                    while (hi >= 0 && queue[size_t(hi)].pos > m.ref_end) --hi;
                    if (lo <= hi)
                        std::memset(&dense[size_t(v) * W + lo], B_GAP, size_t(hi - lo + 1));
                }
                for (int sc = 0; sc < W; ++sc) {
                    const Key& k = queue[size_t(sc)];
                    for (uint32_t i = k.beg; i < k.end_; ++i) {
                        const Event& e = sorted[i];
                        int v = row_of[e.read];
                        if (v >= 0) dense[size_t(v) * W + sc] = e.base;
                    }
                }

                // -- emit: positions + R sampled rows
                size_t pbase = out.positions.size();
                out.positions.resize(pbase + size_t(W) * 2);
                for (int sc = 0; sc < W; ++sc) {
                    out.positions[pbase + 2 * size_t(sc)] = queue[size_t(sc)].pos;
                    out.positions[pbase + 2 * size_t(sc) + 1] = queue[size_t(sc)].ins;
                }
                size_t xbase = out.matrices.size();
                out.matrices.resize(xbase + size_t(R) * W);
                for (int r = 0; r < R; ++r) {
                    uint32_t pick = uint32_t(rng() % uint64_t(V));
                    uint32_t rid = valid[pick];
                    const uint8_t off = reads[rid].offset;
                    const uint8_t* src = &dense[size_t(row_of[rid]) * W];
                    uint8_t* dst = &out.matrices[xbase + size_t(r) * W];
                    for (int sc = 0; sc < W; ++sc) dst[sc] = uint8_t(src[sc] + off);
                }
                out.n_windows++;
                for (int v = 0; v < V; ++v) row_of[valid[size_t(v)]] = -1;
            }
            // -- slide
            queue.erase(queue.begin(), queue.begin() + S);
        }
    };

    for (int64_t p = start; p < end; ++p) {
        const size_t kb = size_t(p - start) * SLOTS;
        if (cnt[kb + 1] == cnt[kb]) continue;  // no slot-0 event: not covered
        int max_slot = 0;
        for (int sl = int(SLOTS) - 1; sl >= 1; --sl)
            if (cnt[kb + size_t(sl) + 1] > cnt[kb + size_t(sl)]) {
                max_slot = sl;
                break;
            }
        for (int sl = 0; sl <= max_slot; ++sl)
            queue.push_back({int32_t(p), int32_t(sl), cnt[kb + size_t(sl)],
                             cnt[kb + size_t(sl) + 1]});
        emit_windows();
    }
    // Columns left in the queue (< W of them after the final emit) are
    // dropped — the region overlap covers the joins (features CLI), matching
    // the reference's tail behaviour (SURVEY.md §3.1 tail note).

    return out;
}

}  // namespace rk
