// Banded global alignment (unit-cost Needleman-Wunsch) with traceback,
// for assembly-quality assessment: polished-vs-truth error counts broken
// down into mismatches / insertions / deletions — the quality metric the
// reference publishes (reference README.md:97-112, evaluated externally
// with pomoxis assess_assembly; this framework evaluates in-repo).
//
// Band: diagonals d = j - i within [min(0, m-n) - W, max(0, m-n) + W].
// Unit costs make the DP an edit distance; W >= the true error count
// guarantees an optimal path stays in band (callers size W from the
// expected divergence; the Python wrapper grows it on saturation).

#include "align.h"

#include <algorithm>
#include <cstring>
#include <limits>
#include <stdexcept>
#include <vector>

namespace rk {

AlignStats align_stats(const std::string& query, const std::string& target,
                       int band, std::string* cigar) {
    const int64_t n = int64_t(query.size());   // rows: query (assembly)
    const int64_t m = int64_t(target.size());  // cols: target (truth)
    if (band < 1) band = 1;
    const int64_t dlo = std::min<int64_t>(0, m - n) - band;
    const int64_t dhi = std::max<int64_t>(0, m - n) + band;
    const int64_t K = dhi - dlo + 1;  // band width in diagonals
    if ((n + 1) * K > (int64_t(1) << 33))
        throw std::runtime_error("align_stats: band*length too large");

    const int32_t INF = std::numeric_limits<int32_t>::max() / 2;
    std::vector<int32_t> prev(size_t(K), INF), cur(size_t(K), INF);
    std::vector<uint8_t> tb(size_t(n + 1) * size_t(K));  // 0 diag 1 up 2 left

    // row 0: dp(0, j) = j for j in band
    for (int64_t d = dlo; d <= dhi; ++d) {
        const int64_t j = d;  // i = 0
        if (j >= 0 && j <= m) {
            prev[size_t(d - dlo)] = int32_t(j);
            tb[size_t(d - dlo)] = 2;
        }
    }

    for (int64_t i = 1; i <= n; ++i) {
        std::fill(cur.begin(), cur.end(), INF);
        uint8_t* tbrow = tb.data() + size_t(i) * size_t(K);
        for (int64_t d = dlo; d <= dhi; ++d) {
            const int64_t j = i + d;
            if (j < 0 || j > m) continue;
            const size_t k = size_t(d - dlo);
            int32_t best = INF;
            uint8_t dir = 0;
            // up: (i-1, j) -> diagonal d+1 in the previous row
            if (d + 1 <= dhi) {
                const int32_t v = prev[k + 1];
                if (v < INF && v + 1 < best) { best = v + 1; dir = 1; }
            }
            // left: (i, j-1) -> diagonal d-1 in this row
            if (j >= 1 && d - 1 >= dlo) {
                const int32_t v = cur[k - 1];
                if (v < INF && v + 1 < best) { best = v + 1; dir = 2; }
            }
            // diag: (i-1, j-1) -> same diagonal, previous row
            if (j >= 1) {
                const int32_t v = prev[k];
                if (v < INF) {
                    const int32_t c = v + (query[size_t(i - 1)] ==
                                           target[size_t(j - 1)] ? 0 : 1);
                    if (c <= best) { best = c; dir = 0; }  // prefer diagonal
                }
            }
            cur[k] = best;
            tbrow[k] = dir;
        }
        std::swap(prev, cur);
    }

    const int64_t dend = m - n;
    AlignStats s{};
    if (dend < dlo || dend > dhi || prev[size_t(dend - dlo)] >= INF)
        throw std::runtime_error("align_stats: end cell outside band");
    s.edit_distance = prev[size_t(dend - dlo)];

    // traceback from (n, m)
    std::vector<std::pair<int64_t, char>> ops;  // reversed run-length ops
    auto push_op = [&](char op) {
        if (!ops.empty() && ops.back().second == op) ++ops.back().first;
        else ops.emplace_back(1, op);
    };
    int64_t i = n, j = m;
    while (i > 0 || j > 0) {
        const int64_t d = j - i;
        const uint8_t dir = tb[size_t(i) * size_t(K) + size_t(d - dlo)];
        if (i > 0 && j > 0 && dir == 0) {
            if (query[size_t(i - 1)] == target[size_t(j - 1)]) ++s.matches;
            else ++s.mismatches;
            if (cigar) push_op('M');
            --i; --j;
        } else if (i > 0 && (dir == 1 || j == 0)) {
            ++s.insertions;  // extra base in query w.r.t. target
            if (cigar) push_op('I');
            --i;
        } else {
            ++s.deletions;   // base of target missing from query
            if (cigar) push_op('D');
            --j;
        }
    }
    if (cigar) {
        cigar->clear();
        for (auto it = ops.rbegin(); it != ops.rend(); ++it) {
            *cigar += std::to_string(it->first);
            *cigar += it->second;
        }
    }
    return s;
}

}  // namespace rk
