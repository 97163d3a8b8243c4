// _racon: python bindings over the native engine, used by tests/ and bench.py.
// The compute path is pure C++/HIP; python only drives files in, FASTA out.
//
// Error semantics: fatal input errors (bad extensions, malformed records,
// missing GPU when requested) terminate the process with a message on
// stderr — the same die-on-error contract the reference pins with its
// EXPECT_DEATH tests (racon_test.cpp:55-86). Callers that need isolation
// run polish in a subprocess (as the CLI tests do).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <memory>
#include <string>
#include <utility>
#include <vector>

#include "align/pairwise.hpp"
#include "align/poa.hpp"
#include "core/polisher.hpp"
#include "core/sequence.hpp"
#include "hip/comm.hpp"

namespace rga::hip {
int runtime_device_count();  // provided by the HIP backend (or the stub)
void runtime_device_synchronize();
std::vector<std::tuple<std::string, int32_t, int32_t>> align_pairs(
    const std::vector<std::pair<std::string, std::string>>& pairs, uint32_t band_width);
std::vector<std::pair<std::string, bool>> poa_windows_gpu(
    const std::vector<std::vector<std::tuple<std::string, std::string, uint32_t, uint32_t>>>&
        window_layers,
    int8_t match, int8_t mismatch, int8_t gap, bool banded, bool trim, bool tgs);
}

namespace py = pybind11;

namespace {

std::vector<std::pair<std::string, std::string>> polish(
    const std::string& sequences_path, const std::string& overlaps_path,
    const std::string& target_path, bool fragment_correction, uint32_t window_length,
    double quality_threshold, double error_threshold, bool trim, int8_t match, int8_t mismatch,
    int8_t gap, uint32_t threads, uint32_t poa_batches, bool banded_poa, uint32_t aligner_batches,
    uint32_t aligner_band_width, bool include_unpolished) {
  rga::PolisherConfig config;
  config.type = fragment_correction ? rga::PolisherType::kF : rga::PolisherType::kC;
  config.window_length = window_length;
  config.quality_threshold = quality_threshold;
  config.error_threshold = error_threshold;
  config.trim = trim;
  config.match = match;
  config.mismatch = mismatch;
  config.gap = gap;
  config.num_threads = threads;
  config.poa_batches = poa_batches;
  config.banded_poa = banded_poa;
  config.aligner_batches = aligner_batches;
  config.aligner_band_width = aligner_band_width;

  std::vector<std::pair<std::string, std::string>> result;
  {
    py::gil_scoped_release release;
    auto polisher =
        rga::createPolisher(sequences_path, overlaps_path, target_path, config);
    polisher->initialize();
    std::vector<std::unique_ptr<rga::Sequence>> polished;
    polisher->polish(polished, !include_unpolished);
    result.reserve(polished.size());
    for (auto& s : polished) {
      result.emplace_back(s->name(), s->data());
    }
  }
  return result;
}

int64_t edit_distance_py(const std::string& a, const std::string& b) {
  py::gil_scoped_release release;
  return rga::edit_distance(a.c_str(), static_cast<uint32_t>(a.size()), b.c_str(),
                            static_cast<uint32_t>(b.size()));
}

std::string align_cigar_py(const std::string& q, const std::string& t) {
  py::gil_scoped_release release;
  return rga::align_global_cigar(q.c_str(), static_cast<uint32_t>(q.size()), t.c_str(),
                                 static_cast<uint32_t>(t.size()));
}

std::string reverse_complement(const std::string& s) {
  rga::Sequence seq(std::string("x"), s);
  seq.make_reverse_complement();
  return seq.reverse_complement();
}

// CPU POA consensus over a set of window layers (first = backbone); exposed
// for kernel-vs-CPU numerics tests.
std::string poa_consensus(const std::vector<std::string>& seqs,
                          const std::vector<std::string>& quals, int8_t match, int8_t mismatch,
                          int8_t gap) {
  py::gil_scoped_release release;
  rga::poa::Graph graph;
  rga::poa::NWEngine engine(match, mismatch, gap);
  for (size_t i = 0; i < seqs.size(); ++i) {
    rga::poa::Alignment alignment;
    if (i > 0) {
      alignment = engine.align(seqs[i].c_str(), static_cast<uint32_t>(seqs[i].size()), graph);
    }
    if (i < quals.size() && !quals[i].empty()) {
      graph.add_alignment(alignment, seqs[i].c_str(), static_cast<uint32_t>(seqs[i].size()),
                          quals[i].c_str(), static_cast<uint32_t>(quals[i].size()));
    } else {
      graph.add_alignment(alignment, seqs[i].c_str(), static_cast<uint32_t>(seqs[i].size()));
    }
  }
  return graph.generate_consensus(nullptr);
}

// CPU side of the window-level differ: the same raw-window inputs as
// hip::poa_windows_gpu, run through the pipeline's own Window::generate_consensus
// (subgraph trick, trim, tie-breaks — not the simplified poa_consensus above).
std::vector<std::pair<std::string, bool>> poa_windows_cpu(
    const std::vector<std::vector<std::tuple<std::string, std::string, uint32_t, uint32_t>>>&
        window_layers,
    int8_t match, int8_t mismatch, int8_t gap, bool trim, bool tgs) {
  py::gil_scoped_release release;
  rga::poa::NWEngine engine(match, mismatch, gap);
  std::vector<std::pair<std::string, bool>> out;
  out.reserve(window_layers.size());
  for (const auto& layers : window_layers) {
    if (layers.empty()) {
      throw std::runtime_error("poa_windows_cpu: window without a backbone");
    }
    const auto& bb = layers.front();
    auto w = rga::createWindow(0, 0, tgs ? rga::WindowType::kTGS : rga::WindowType::kNGS,
                               std::get<0>(bb).data(),
                               static_cast<uint32_t>(std::get<0>(bb).size()),
                               std::get<1>(bb).data(),
                               static_cast<uint32_t>(std::get<1>(bb).size()));
    for (size_t i = 1; i < layers.size(); ++i) {
      const auto& l = layers[i];
      const std::string& q = std::get<1>(l);
      w->add_layer(std::get<0>(l).data(), static_cast<uint32_t>(std::get<0>(l).size()),
                   q.empty() ? nullptr : q.data(), static_cast<uint32_t>(q.size()),
                   std::get<2>(l), std::get<3>(l));
    }
    bool polished = w->generate_consensus(engine, trim);
    out.emplace_back(w->consensus(), polished);
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_racon, m) {
  m.doc() = "MI355X-native consensus polishing engine (racon capability parity)";

  m.def("polish", &polish, py::arg("sequences"), py::arg("overlaps"), py::arg("targets"),
        py::arg("fragment_correction") = false, py::arg("window_length") = 500,
        py::arg("quality_threshold") = 10.0, py::arg("error_threshold") = 0.3,
        py::arg("trim") = true, py::arg("match") = 3, py::arg("mismatch") = -5,
        py::arg("gap") = -4, py::arg("threads") = 1, py::arg("poa_batches") = 0,
        py::arg("banded_poa") = false, py::arg("aligner_batches") = 0,
        py::arg("aligner_band_width") = 0, py::arg("include_unpolished") = false,
        "Polish targets with reads+overlaps; returns [(name, sequence)].");

  m.def("device_count", [] { return rga::hip::runtime_device_count(); },
        "Number of visible HIP devices.");
  m.def("device_synchronize", [] {
    py::gil_scoped_release release;
    rga::hip::runtime_device_synchronize();
  });

  // distributed communicator (hip/comm.hpp): TCP control plane + RCCL data
  // plane; one process per GPU
  m.def("comm_init", [](int rank, int world, const std::string& host, int port, bool use_gpu) {
    py::gil_scoped_release release;
    rga::comm::world_comm().init(rank, world, host, port, use_gpu);
  }, py::arg("rank"), py::arg("world"), py::arg("host"), py::arg("port"), py::arg("use_gpu"));
  m.def("comm_gather", [](const py::bytes& payload, int root) {
    std::string data = payload;
    std::vector<std::string> parts;
    {
      py::gil_scoped_release release;
      parts = rga::comm::world_comm().gather(data, root);
    }
    py::list out;
    for (auto& p : parts) {
      out.append(py::bytes(p));
    }
    return out;
  }, py::arg("payload"), py::arg("root") = 0);
  m.def("comm_allreduce_max", [](double v) {
    py::gil_scoped_release release;
    return rga::comm::world_comm().allreduce_max(v);
  });
  m.def("comm_allreduce_sum", [](double v) {
    py::gil_scoped_release release;
    return rga::comm::world_comm().allreduce_sum(v);
  });
  m.def("comm_barrier", [] {
    py::gil_scoped_release release;
    rga::comm::world_comm().barrier();
  });
  m.def("comm_finalize", [] { rga::comm::world_comm().finalize(); });
  m.def("edit_distance", &edit_distance_py, py::arg("a"), py::arg("b"));
  m.def("align_cigar", &align_cigar_py, py::arg("query"), py::arg("target"));
  m.def("gpu_align", [](const std::vector<std::pair<std::string, std::string>>& pairs,
                        uint32_t band_width) {
          py::gil_scoped_release release;
          return rga::hip::align_pairs(pairs, band_width);
        },
        py::arg("pairs"), py::arg("band_width") = 0,
        "GPU Myers aligner on raw (query, target) pairs -> (cigar, edit_distance, status)");
  m.def("reverse_complement", &reverse_complement, py::arg("sequence"));
  // window-level CPU-vs-GPU differ: identical raw-window inputs through
  // both engines (layers = [(seq, qual, begin, end)], slot 0 = backbone)
  m.def("poa_windows_cpu", &poa_windows_cpu, py::arg("windows"), py::arg("match") = 3,
        py::arg("mismatch") = -5, py::arg("gap") = -4, py::arg("trim") = true,
        py::arg("tgs") = true);
  m.def("poa_windows_gpu", [](const std::vector<std::vector<
                                  std::tuple<std::string, std::string, uint32_t, uint32_t>>>& w,
                              int8_t match, int8_t mismatch, int8_t gap, bool banded,
                              bool trim, bool tgs) {
    py::gil_scoped_release release;
    return rga::hip::poa_windows_gpu(w, match, mismatch, gap, banded, trim, tgs);
  }, py::arg("windows"), py::arg("match") = 3, py::arg("mismatch") = -5,
     py::arg("gap") = -4, py::arg("banded") = false, py::arg("trim") = true,
     py::arg("tgs") = true);
  m.def("poa_consensus", &poa_consensus, py::arg("sequences"),
        py::arg("qualities") = std::vector<std::string>(), py::arg("match") = 5,
        py::arg("mismatch") = -4, py::arg("gap") = -8);
  m.attr("__version__") = "1.0.0";
}
