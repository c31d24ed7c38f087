// pybind11 bindings: roko_amd.ops._pileup
//
// The framework's Python<->C++ boundary for the data path (the reference's
// equivalent is the hand-rolled CPython module gen.cpp:10-63 exposing one
// function; here we also expose raw record fetch so label generation
// (labels.py) needs no pysam).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "align.h"
#include "bam.h"
#include "pileup.h"

namespace py = pybind11;

static py::tuple generate_features(const std::string& bam, const std::string& contig,
                                   int64_t start, int64_t end, int rows, int cols,
                                   int stride, int max_ins, uint32_t filter_flag,
                                   int min_mapq, uint64_t seed) {
    rk::FeatureParams P;
    P.rows = rows;
    P.cols = cols;
    P.stride = stride;
    P.max_ins = max_ins;
    P.filter_flag = filter_flag;
    P.min_mapq = uint8_t(min_mapq);
    P.seed = seed;

    rk::FeatureResult res;
    {
        py::gil_scoped_release release;
        res = rk::extract_features(bam, contig, start, end, P);
    }

    auto positions = py::array_t<int32_t>({res.n_windows, int64_t(cols), int64_t(2)});
    auto matrices = py::array_t<uint8_t>({res.n_windows, int64_t(rows), int64_t(cols)});
    if (res.n_windows > 0) {
        std::memcpy(positions.mutable_data(), res.positions.data(),
                    res.positions.size() * sizeof(int32_t));
        std::memcpy(matrices.mutable_data(), res.matrices.data(), res.matrices.size());
    }
    return py::make_tuple(positions, matrices);
}

static py::list bam_references(const std::string& path) {
    rk::BamReader bam(path);
    py::list out;
    for (const auto& r : bam.references()) out.append(py::make_tuple(r.name, r.length));
    return out;
}

// Records overlapping [start, end) of `contig`, unfiltered:
// (qname, flag, pos, mapq, cigar uint32 array, seq string).
static py::list fetch_records(const std::string& path, const std::string& contig,
                              int64_t start, int64_t end) {
    rk::BamReader bam(path);
    int tid = bam.tid_of(contig);
    if (tid < 0) throw std::runtime_error("contig not in BAM header: " + contig);
    py::list out;
    bam.fetch(tid, start, end, [&](const rk::BamRecord& rec) {
        auto cig = py::array_t<uint32_t>(int64_t(rec.cigar.size()));
        std::memcpy(cig.mutable_data(), rec.cigar.data(), rec.cigar.size() * 4);
        std::string seq(size_t(rec.l_seq), 'N');
        for (int64_t i = 0; i < rec.l_seq; ++i) seq[size_t(i)] = rec.seq_char(i);
        out.append(py::make_tuple(rec.qname, rec.flag, rec.pos, rec.mapq, cig, seq));
    });
    return out;
}

PYBIND11_MODULE(_pileup, m) {
    m.doc() = "roko-mi355x native data path: BGZF/BAM/BAI reader + window builder";
    m.def("generate_features", &generate_features, py::arg("bam"), py::arg("contig"),
          py::arg("start"), py::arg("end"), py::arg("rows") = 200, py::arg("cols") = 90,
          py::arg("stride") = 30, py::arg("max_ins") = 3,
          py::arg("filter_flag") = 0xf04u, py::arg("min_mapq") = 10,
          py::arg("seed") = 0);
    m.def("bam_references", &bam_references, py::arg("path"));
    m.def("fetch_records", &fetch_records, py::arg("path"), py::arg("contig"),
          py::arg("start"), py::arg("end"));
    m.def(
        "align_stats",
        [](const std::string& query, const std::string& target, int band) {
            rk::AlignStats s;
            {
                py::gil_scoped_release release;
                s = rk::align_stats(query, target, band);
            }
            py::dict d;
            d["edit_distance"] = s.edit_distance;
            d["matches"] = s.matches;
            d["mismatches"] = s.mismatches;
            d["insertions"] = s.insertions;
            d["deletions"] = s.deletions;
            return d;
        },
        py::arg("query"), py::arg("target"), py::arg("band") = 128,
        "Banded global alignment stats of query vs target (assembly QC)");
    m.def(
        "align_cigar",
        [](const std::string& query, const std::string& target, int band) {
            rk::AlignStats s;
            std::string cig;
            {
                py::gil_scoped_release release;
                s = rk::align_stats(query, target, band, &cig);
            }
            py::dict d;
            d["edit_distance"] = s.edit_distance;
            d["matches"] = s.matches;
            d["mismatches"] = s.mismatches;
            d["insertions"] = s.insertions;
            d["deletions"] = s.deletions;
            d["cigar"] = cig;
            return d;
        },
        py::arg("query"), py::arg("target"), py::arg("band") = 128,
        "align_stats plus the M/I/D CIGAR of query against target");
}
