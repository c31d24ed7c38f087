// Self-contained BAM/BGZF/BAI reader for the feature-extraction core.
//
// The reference links vendored htslib-1.9 for this capability (SURVEY.md §2.1
// L0: hts_open/sam_index_load/bam_mplp_*); this image ships no htslib, so the
// framework carries its own minimal, read-only implementation of the three
// on-disk formats it needs (BGZF blocks over zlib, BAM records, BAI index).
// Formats per the SAM/BAM specification (samtools/hts-specs, SAMv1.pdf).
//
// Design notes (deliberately NOT an htslib translation):
//  * no pileup engine here — the window builder (pileup.cpp) walks CIGARs
//    directly into a column store, which is both simpler and faster than a
//    per-position multi-iterator;
//  * single-threaded sequential decode per region; process-level parallelism
//    comes from the Python side fanning regions out over workers
//    (reference: features.py:141-143).

#pragma once

#include <cstdint>
#include <cstdio>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace rk {

// ---------------------------------------------------------------------------
// BGZF
// ---------------------------------------------------------------------------

class Bgzf {
public:
    explicit Bgzf(const std::string& path);
    ~Bgzf();
    Bgzf(const Bgzf&) = delete;
    Bgzf& operator=(const Bgzf&) = delete;

    // Read n bytes across block boundaries. Returns bytes read (< n only at EOF).
    size_t read(void* dst, size_t n);
    // Skip n bytes of uncompressed stream.
    void skip(size_t n);
    // Virtual offset = (compressed block offset << 16) | intra-block offset.
    void seek_virtual(uint64_t voff);
    uint64_t tell_virtual() const;
    bool eof();

private:
    bool load_block(uint64_t coffset);  // false at physical EOF / EOF marker

    FILE* f_ = nullptr;
    uint64_t block_coffset_ = ~0ull;  // file offset of the loaded block
    uint64_t next_coffset_ = 0;       // file offset of the following block
    std::vector<uint8_t> ubuf_;       // inflated payload of the loaded block
    size_t upos_ = 0;                 // cursor within ubuf_
    std::vector<uint8_t> cbuf_;       // scratch for the compressed payload
    bool phys_eof_ = false;
};

// ---------------------------------------------------------------------------
// BAM records
// ---------------------------------------------------------------------------

// CIGAR op codes per the spec: MIDNSHP=X
enum : uint8_t { CIG_M = 0, CIG_I, CIG_D, CIG_N, CIG_S, CIG_H, CIG_P, CIG_EQ, CIG_X };

constexpr uint16_t FLAG_PAIRED = 0x1;
constexpr uint16_t FLAG_PROPER_PAIR = 0x2;
constexpr uint16_t FLAG_UNMAP = 0x4;
constexpr uint16_t FLAG_REVERSE = 0x10;
constexpr uint16_t FLAG_SECONDARY = 0x100;
constexpr uint16_t FLAG_QCFAIL = 0x200;
constexpr uint16_t FLAG_DUP = 0x400;
constexpr uint16_t FLAG_SUPPLEMENTARY = 0x800;

// One decoded alignment record (owning copy of the variable-length payload).
struct BamRecord {
    int32_t tid = -1;
    int32_t pos = -1;  // 0-based leftmost ref position
    uint16_t flag = 0;
    uint8_t mapq = 0;
    std::string qname;
    std::vector<uint32_t> cigar;  // len<<4 | op
    std::vector<uint8_t> seq4;    // 4-bit packed, as stored
    int32_t l_seq = 0;

    bool is_reverse() const { return flag & FLAG_REVERSE; }
    int64_t ref_end() const;       // exclusive end on the reference
    int64_t query_length_cigar() const;

    // 4-bit code of query base i (1=A 2=C 4=G 8=T 15=N, spec "=ACMGRSVTWYHKDBN")
    uint8_t seqi(int64_t i) const {
        uint8_t b = seq4[i >> 1];
        return (i & 1) ? (b & 0xf) : (b >> 4);
    }
    char seq_char(int64_t i) const {
        static const char* tbl = "=ACMGRSVTWYHKDBN";
        return tbl[seqi(i)];
    }
};

struct RefInfo {
    std::string name;
    int64_t length;
};

// ---------------------------------------------------------------------------
// BAI index
// ---------------------------------------------------------------------------

struct Chunk {
    uint64_t beg, end;  // virtual offsets
};

class BaiIndex {
public:
    // Loads <bam>.bai or <bam minus .bam>.bai; returns nullptr if absent.
    static std::unique_ptr<BaiIndex> load(const std::string& bam_path);

    // Merged candidate chunks for records overlapping [beg, end) on ref tid.
    std::vector<Chunk> query(int tid, int64_t beg, int64_t end) const;

private:
    struct Bin {
        uint32_t id;
        std::vector<Chunk> chunks;
    };
    struct Ref {
        std::vector<Bin> bins;
        std::vector<uint64_t> ioffsets;  // 16 kb linear index
    };
    std::vector<Ref> refs_;
};

// ---------------------------------------------------------------------------
// BAM reader
// ---------------------------------------------------------------------------

class BamReader {
public:
    explicit BamReader(const std::string& path);

    const std::vector<RefInfo>& references() const { return refs_; }
    int tid_of(const std::string& name) const;
    bool has_index() const { return index_ != nullptr; }

    // Sequential record read from the current stream position.
    // Returns false at EOF.
    bool next(BamRecord& rec);

    // Iterate records overlapping [start, end) of reference `tid`, in
    // coordinate order. Uses the BAI when present, otherwise a linear scan
    // (fixtures / small files). Calls fn for every overlapping record that
    // passes no filter — filtering is the caller's business.
    template <class Fn>
    void fetch(int tid, int64_t start, int64_t end, Fn&& fn) {
        if (index_) {
            auto chunks = index_->query(tid, start, end);
            for (const auto& ck : chunks) {
                bgzf_.seek_virtual(ck.beg);
                BamRecord rec;
                while (bgzf_.tell_virtual() < ck.end && next(rec)) {
                    if (rec.tid != tid) continue;
                    if (rec.pos >= end) break;  // coordinate-sorted
                    if (rec.ref_end() <= start) continue;
                    fn(rec);
                }
            }
        } else {
            bgzf_.seek_virtual(data_voffset_);
            BamRecord rec;
            while (next(rec)) {
                if (rec.tid != tid || rec.pos >= end) {
                    if (rec.tid > tid) break;  // sorted past our ref
                    continue;
                }
                if (rec.ref_end() <= start) continue;
                fn(rec);
            }
        }
    }

private:
    Bgzf bgzf_;
    std::vector<RefInfo> refs_;
    std::unique_ptr<BaiIndex> index_;
    uint64_t data_voffset_ = 0;  // virtual offset of the first record
    std::vector<uint8_t> scratch_;
};

}  // namespace rk
