// BGZF / BAM / BAI implementation. See bam.h for scope and design notes.

#include "bam.h"

#include <zlib.h>

#include <algorithm>
#include <cstring>

namespace rk {

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------

static uint16_t rd_u16(const uint8_t* p) { return uint16_t(p[0]) | uint16_t(p[1]) << 8; }
static uint32_t rd_u32(const uint8_t* p) {
    return uint32_t(p[0]) | uint32_t(p[1]) << 8 | uint32_t(p[2]) << 16 | uint32_t(p[3]) << 24;
}
static int32_t rd_i32(const uint8_t* p) { return int32_t(rd_u32(p)); }
static uint64_t rd_u64(const uint8_t* p) {
    return uint64_t(rd_u32(p)) | uint64_t(rd_u32(p + 4)) << 32;
}

// ---------------------------------------------------------------------------
// Bgzf
// ---------------------------------------------------------------------------

Bgzf::Bgzf(const std::string& path) {
    f_ = std::fopen(path.c_str(), "rb");
    if (!f_) throw std::runtime_error("cannot open " + path);
    if (!load_block(0)) throw std::runtime_error(path + ": empty or not BGZF");
}

Bgzf::~Bgzf() {
    if (f_) std::fclose(f_);
}

bool Bgzf::load_block(uint64_t coffset) {
    if (block_coffset_ == coffset && !ubuf_.empty()) {
        upos_ = 0;
        return true;
    }
    if (std::fseek(f_, long(coffset), SEEK_SET) != 0) return false;
    uint8_t hdr[18];
    size_t got = std::fread(hdr, 1, sizeof hdr, f_);
    if (got == 0) {
        phys_eof_ = true;
        return false;
    }
    if (got < 18 || hdr[0] != 0x1f || hdr[1] != 0x8b || hdr[2] != 8 || !(hdr[3] & 4)) {
        // recognise the common wrong-format cases and say so (this reader
        // supports coordinate-sorted BAM+BAI only; CRAM/SAM are documented
        // out of scope — convert with `samtools view -b`)
        if (got >= 4 && std::memcmp(hdr, "CRAM", 4) == 0)
            throw std::runtime_error(
                "input is a CRAM file; this framework reads BAM only — "
                "convert with `samtools view -b -o out.bam in.cram`");
        if (got >= 3 && hdr[0] == '@' &&
            (hdr[1] == 'H' || hdr[1] == 'S' || hdr[1] == 'R' || hdr[1] == 'P'))
            throw std::runtime_error(
                "input looks like uncompressed SAM text; this framework "
                "reads BAM only — convert with `samtools view -b`");
        throw std::runtime_error("corrupt BGZF block header");
    }
    uint16_t xlen = rd_u16(hdr + 10);
    // Find the BC subfield carrying BSIZE. The fixed 18-byte read already
    // includes the first 6 bytes of the extra field (the common case where
    // BC is the only subfield).
    std::vector<uint8_t> extra(xlen);
    size_t have = std::min<size_t>(6, xlen);
    std::memcpy(extra.data(), hdr + 12, have);
    if (xlen > have &&
        std::fread(extra.data() + have, 1, xlen - have, f_) != size_t(xlen - have))
        throw std::runtime_error("truncated BGZF extra field");
    int bsize = -1;
    for (size_t i = 0; i + 4 <= extra.size();) {
        uint8_t si1 = extra[i], si2 = extra[i + 1];
        uint16_t slen = rd_u16(&extra[i + 2]);
        if (si1 == 66 && si2 == 67 && slen == 2) bsize = rd_u16(&extra[i + 4]) + 1;
        i += 4 + slen;
    }
    if (bsize < 0) throw std::runtime_error("BGZF block without BSIZE");
    size_t cdata_len = size_t(bsize) - 12 - xlen - 8;  // hdr(12)+extra+crc/isize(8)
    cbuf_.resize(cdata_len);
    if (std::fseek(f_, long(coffset + 12 + xlen), SEEK_SET) != 0)
        throw std::runtime_error("BGZF seek failed");
    if (std::fread(cbuf_.data(), 1, cdata_len, f_) != cdata_len)
        throw std::runtime_error("truncated BGZF block");
    uint8_t tail[8];
    if (std::fread(tail, 1, 8, f_) != 8) throw std::runtime_error("truncated BGZF tail");
    uint32_t isize = rd_u32(tail + 4);

    ubuf_.resize(isize);
    if (isize > 0) {
        z_stream zs{};
        if (inflateInit2(&zs, -15) != Z_OK) throw std::runtime_error("inflateInit2 failed");
        zs.next_in = cbuf_.data();
        zs.avail_in = uInt(cdata_len);
        zs.next_out = ubuf_.data();
        zs.avail_out = uInt(isize);
        int rc = inflate(&zs, Z_FINISH);
        inflateEnd(&zs);
        if (rc != Z_STREAM_END) throw std::runtime_error("BGZF inflate failed");
    }
    block_coffset_ = coffset;
    next_coffset_ = coffset + bsize;
    upos_ = 0;
    if (isize == 0) {
        // EOF marker block (or any empty block): try the next one.
        return load_block(next_coffset_);
    }
    return true;
}

size_t Bgzf::read(void* dst, size_t n) {
    uint8_t* out = static_cast<uint8_t*>(dst);
    size_t done = 0;
    while (done < n) {
        if (upos_ >= ubuf_.size()) {
            if (!load_block(next_coffset_)) break;
        }
        size_t take = std::min(n - done, ubuf_.size() - upos_);
        std::memcpy(out + done, ubuf_.data() + upos_, take);
        upos_ += take;
        done += take;
    }
    return done;
}

void Bgzf::skip(size_t n) {
    while (n > 0) {
        if (upos_ >= ubuf_.size()) {
            if (!load_block(next_coffset_)) throw std::runtime_error("BGZF skip past EOF");
        }
        size_t take = std::min(n, ubuf_.size() - upos_);
        upos_ += take;
        n -= take;
    }
}

void Bgzf::seek_virtual(uint64_t voff) {
    uint64_t coff = voff >> 16;
    size_t uoff = voff & 0xffff;
    phys_eof_ = false;
    if (!load_block(coff)) throw std::runtime_error("BGZF seek to bad virtual offset");
    if (uoff > ubuf_.size()) throw std::runtime_error("BGZF intra-block offset out of range");
    upos_ = uoff;
}

uint64_t Bgzf::tell_virtual() const {
    if (upos_ >= ubuf_.size()) return next_coffset_ << 16;
    return (block_coffset_ << 16) | uint64_t(upos_);
}

bool Bgzf::eof() {
    if (upos_ < ubuf_.size()) return false;
    if (!load_block(next_coffset_)) return true;
    return false;
}

// ---------------------------------------------------------------------------
// BamRecord
// ---------------------------------------------------------------------------

int64_t BamRecord::ref_end() const {
    int64_t p = pos;
    for (uint32_t c : cigar) {
        uint32_t op = c & 0xf, len = c >> 4;
        if (op == CIG_M || op == CIG_D || op == CIG_N || op == CIG_EQ || op == CIG_X)
            p += len;
    }
    return p;
}

int64_t BamRecord::query_length_cigar() const {
    int64_t q = 0;
    for (uint32_t c : cigar) {
        uint32_t op = c & 0xf, len = c >> 4;
        if (op == CIG_M || op == CIG_I || op == CIG_S || op == CIG_EQ || op == CIG_X)
            q += len;
    }
    return q;
}

// ---------------------------------------------------------------------------
// BaiIndex
// ---------------------------------------------------------------------------

std::unique_ptr<BaiIndex> BaiIndex::load(const std::string& bam_path) {
    std::string p1 = bam_path + ".bai";
    FILE* f = std::fopen(p1.c_str(), "rb");
    if (!f) {
        std::string p2 = bam_path;
        auto dot = p2.rfind(".bam");
        if (dot != std::string::npos) {
            p2 = p2.substr(0, dot) + ".bai";
            f = std::fopen(p2.c_str(), "rb");
        }
    }
    if (!f) return nullptr;
    std::vector<uint8_t> buf;
    uint8_t tmp[1 << 16];
    size_t n;
    while ((n = std::fread(tmp, 1, sizeof tmp, f)) > 0) buf.insert(buf.end(), tmp, tmp + n);
    std::fclose(f);
    if (buf.size() < 8 || std::memcmp(buf.data(), "BAI\1", 4) != 0)
        throw std::runtime_error("bad BAI magic");

    auto idx = std::unique_ptr<BaiIndex>(new BaiIndex());
    size_t off = 4;
    auto need = [&](size_t k) {
        if (off + k > buf.size()) throw std::runtime_error("truncated BAI");
    };
    need(4);
    int32_t n_ref = rd_i32(&buf[off]);
    off += 4;
    idx->refs_.resize(n_ref);
    for (int r = 0; r < n_ref; ++r) {
        need(4);
        int32_t n_bin = rd_i32(&buf[off]);
        off += 4;
        auto& ref = idx->refs_[r];
        ref.bins.reserve(n_bin);
        for (int b = 0; b < n_bin; ++b) {
            need(8);
            uint32_t bin_id = rd_u32(&buf[off]);
            int32_t n_chunk = rd_i32(&buf[off + 4]);
            off += 8;
            Bin bin;
            bin.id = bin_id;
            bin.chunks.reserve(n_chunk);
            for (int c = 0; c < n_chunk; ++c) {
                need(16);
                bin.chunks.push_back({rd_u64(&buf[off]), rd_u64(&buf[off + 8])});
                off += 16;
            }
            // bin 37450 is the samtools pseudo-bin with meta data — keep it out
            if (bin.id != 37450u) ref.bins.push_back(std::move(bin));
        }
        need(4);
        int32_t n_intv = rd_i32(&buf[off]);
        off += 4;
        ref.ioffsets.reserve(n_intv);
        for (int i = 0; i < n_intv; ++i) {
            need(8);
            ref.ioffsets.push_back(rd_u64(&buf[off]));
            off += 8;
        }
    }
    return idx;
}

// bins overlapping [beg, end), 6-level binning per the SAM spec
static void reg2bins(int64_t beg, int64_t end, std::vector<uint32_t>& bins) {
    if (beg >= end) return;
    --end;
    bins.push_back(0);
    for (int64_t k = 1 + (beg >> 26); k <= 1 + (end >> 26); ++k) bins.push_back(uint32_t(k));
    for (int64_t k = 9 + (beg >> 23); k <= 9 + (end >> 23); ++k) bins.push_back(uint32_t(k));
    for (int64_t k = 73 + (beg >> 20); k <= 73 + (end >> 20); ++k) bins.push_back(uint32_t(k));
    for (int64_t k = 585 + (beg >> 17); k <= 585 + (end >> 17); ++k) bins.push_back(uint32_t(k));
    for (int64_t k = 4681 + (beg >> 14); k <= 4681 + (end >> 14); ++k) bins.push_back(uint32_t(k));
}

std::vector<Chunk> BaiIndex::query(int tid, int64_t beg, int64_t end) const {
    std::vector<Chunk> out;
    if (tid < 0 || size_t(tid) >= refs_.size()) return out;
    const Ref& ref = refs_[tid];
    uint64_t min_off = 0;
    size_t iv = size_t(beg >> 14);
    if (!ref.ioffsets.empty()) {
        if (iv >= ref.ioffsets.size()) iv = ref.ioffsets.size() - 1;
        min_off = ref.ioffsets[iv];
    }
    std::vector<uint32_t> bins;
    reg2bins(beg, end, bins);
    std::sort(bins.begin(), bins.end());
    for (const Bin& b : ref.bins) {
        if (!std::binary_search(bins.begin(), bins.end(), b.id)) continue;
        for (const Chunk& c : b.chunks)
            if (c.end > min_off) out.push_back(c);
    }
    std::sort(out.begin(), out.end(), [](const Chunk& a, const Chunk& b) { return a.beg < b.beg; });
    // merge overlapping / adjacent chunks
    std::vector<Chunk> merged;
    for (const Chunk& c : out) {
        if (!merged.empty() && c.beg <= merged.back().end)
            merged.back().end = std::max(merged.back().end, c.end);
        else
            merged.push_back(c);
    }
    return merged;
}

// ---------------------------------------------------------------------------
// BamReader
// ---------------------------------------------------------------------------

BamReader::BamReader(const std::string& path) : bgzf_(path) {
    uint8_t magic[4];
    if (bgzf_.read(magic, 4) != 4 || std::memcmp(magic, "BAM\1", 4) != 0)
        throw std::runtime_error(path + ": not a BAM file");
    uint8_t b4[4];
    if (bgzf_.read(b4, 4) != 4) throw std::runtime_error("truncated BAM header");
    int32_t l_text = rd_i32(b4);
    bgzf_.skip(size_t(l_text));  // SAM header text: unused, contigs come from refs
    if (bgzf_.read(b4, 4) != 4) throw std::runtime_error("truncated BAM header");
    int32_t n_ref = rd_i32(b4);
    refs_.reserve(n_ref);
    for (int i = 0; i < n_ref; ++i) {
        if (bgzf_.read(b4, 4) != 4) throw std::runtime_error("truncated BAM refs");
        int32_t l_name = rd_i32(b4);
        std::string name(size_t(l_name), '\0');
        if (bgzf_.read(&name[0], size_t(l_name)) != size_t(l_name))
            throw std::runtime_error("truncated BAM refs");
        if (!name.empty() && name.back() == '\0') name.pop_back();
        if (bgzf_.read(b4, 4) != 4) throw std::runtime_error("truncated BAM refs");
        refs_.push_back({name, rd_i32(b4)});
    }
    data_voffset_ = bgzf_.tell_virtual();
    index_ = BaiIndex::load(path);
}

int BamReader::tid_of(const std::string& name) const {
    for (size_t i = 0; i < refs_.size(); ++i)
        if (refs_[i].name == name) return int(i);
    return -1;
}

bool BamReader::next(BamRecord& rec) {
    uint8_t b4[4];
    size_t got = bgzf_.read(b4, 4);
    if (got == 0) return false;
    if (got != 4) throw std::runtime_error("truncated BAM record");
    uint32_t block_size = rd_u32(b4);
    scratch_.resize(block_size);
    if (bgzf_.read(scratch_.data(), block_size) != block_size)
        throw std::runtime_error("truncated BAM record body");
    const uint8_t* p = scratch_.data();
    if (block_size < 32) throw std::runtime_error("BAM record too small");
    rec.tid = rd_i32(p);
    rec.pos = rd_i32(p + 4);
    uint8_t l_read_name = p[8];
    rec.mapq = p[9];
    uint16_t n_cigar = rd_u16(p + 12);
    rec.flag = rd_u16(p + 14);
    rec.l_seq = rd_i32(p + 16);
    size_t off = 32;
    rec.qname.assign(reinterpret_cast<const char*>(p + off), l_read_name > 0 ? l_read_name - 1 : 0);
    off += l_read_name;
    rec.cigar.resize(n_cigar);
    for (int i = 0; i < n_cigar; ++i) rec.cigar[i] = rd_u32(p + off + 4 * size_t(i));
    off += 4 * size_t(n_cigar);
    size_t seq_bytes = size_t(rec.l_seq + 1) / 2;
    if (off + seq_bytes > block_size) throw std::runtime_error("BAM record seq overflow");
    rec.seq4.assign(p + off, p + off + seq_bytes);
    // qual + aux tags are not needed by the feature path; skip.
    return true;
}

}  // namespace rk
