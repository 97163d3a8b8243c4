#include "core/window.hpp"

#include <algorithm>
#include <cstdio>
#include <cstdlib>

#include "align/poa.hpp"

namespace rga {

namespace {

// Trim both consensus ends up to the first base whose coverage reaches the
// floor. Returns false when the trimmed range is empty or inverted (the
// window straddles a likely chimeric join and is left untrimmed).
bool trim_to_coverage(std::string* consensus, const std::vector<uint32_t>& coverages,
                      uint32_t floor) {
  const int32_t n = static_cast<int32_t>(consensus->size());
  int32_t lo = 0;
  while (lo < n && coverages[lo] < floor) {
    ++lo;
  }
  int32_t hi = n - 1;
  while (hi >= 0 && coverages[hi] < floor) {
    --hi;
  }
  if (lo >= hi) {
    return false;
  }
  *consensus = consensus->substr(lo, hi - lo + 1);
  return true;
}

}  // namespace

std::shared_ptr<Window> createWindow(uint64_t id, uint32_t rank, WindowType type,
                                     const char* backbone, uint32_t backbone_length,
                                     const char* quality, uint32_t quality_length) {
  if (backbone_length == 0 || backbone_length != quality_length) {
    fprintf(stderr,
            "[rga::createWindow] error: backbone must be non-empty with a "
            "quality string of the same length\n");
    exit(1);
  }
  return std::make_shared<Window>(id, rank, type, backbone, backbone_length, quality,
                                  quality_length);
}

Window::Window(uint64_t contig_id, uint32_t rank, WindowType type, const char* backbone,
               uint32_t backbone_len, const char* backbone_qual, uint32_t qual_len)
    : contig_id_(contig_id), rank_(rank), type_(type) {
  Layer base;
  base.seq = backbone;
  base.seq_len = backbone_len;
  base.qual = backbone_qual;
  base.qual_len = qual_len;
  layers_.push_back(base);
}

void Window::add_layer(const char* sequence, uint32_t sequence_length, const char* quality,
                       uint32_t quality_length, uint32_t begin, uint32_t end) {
  if (sequence_length == 0 || begin == end) {
    return;  // nothing routed into this window
  }
  if (quality != nullptr && sequence_length != quality_length) {
    fprintf(stderr, "[rga::Window::add_layer] error: quality length does not match the segment\n");
    exit(1);
  }
  const uint32_t backbone_len = backbone_length();
  if (begin >= end || begin > backbone_len || end > backbone_len) {
    fprintf(stderr, "[rga::Window::add_layer] error: segment span outside the backbone\n");
    exit(1);
  }
  Layer l;
  l.seq = sequence;
  l.seq_len = sequence_length;
  l.qual = quality;
  l.qual_len = quality_length;
  l.begin = begin;
  l.end = end;
  layers_.push_back(l);
}

std::vector<uint32_t> Window::layer_order() const {
  std::vector<uint32_t> order(layers_.size());
  for (uint32_t i = 0; i < order.size(); ++i) {
    order[i] = i;
  }
  // Deliberately an UNSTABLE sort over everything but the backbone: the
  // pinned CPU goldens encode libstdc++ introsort's ordering of equal start
  // positions (reference behavior), so this must stay std::sort with a
  // strict-weak begin-only comparison.
  std::sort(order.begin() + 1, order.end(), [this](uint32_t a, uint32_t b) {
    return layers_[a].begin < layers_[b].begin;
  });
  return order;
}

bool Window::generate_consensus(poa::NWEngine& engine, bool trim) {
  const Layer& base = layers_.front();
  if (layers_.size() < 3) {
    // under-covered: pass the backbone through, flagged unpolished
    consensus_.assign(base.seq, base.seq_len);
    return false;
  }

  poa::Graph graph;
  graph.add_alignment(poa::Alignment(), base.seq, base.seq_len, base.qual, base.qual_len);

  // a layer within 1% of both window edges counts as spanning the whole
  // backbone and is aligned against the full graph; anything shorter goes
  // through the subgraph of its own backbone range
  const uint32_t edge_margin = static_cast<uint32_t>(0.01 * base.seq_len);
  const std::vector<uint32_t> order = layer_order();
  for (uint32_t k = 1; k < order.size(); ++k) {
    const Layer& l = layers_[order[k]];

    poa::Alignment aln;
    const bool spans_window = l.begin < edge_margin && l.end > base.seq_len - edge_margin;
    if (spans_window) {
      aln = engine.align(l.seq, l.seq_len, graph);
    } else {
      std::vector<int32_t> node_map;
      poa::Graph ranged = graph.subgraph(l.begin, l.end, &node_map);
      aln = engine.align(l.seq, l.seq_len, ranged);
      poa::Graph::update_alignment(&aln, node_map);
    }

    if (l.qual == nullptr) {
      graph.add_alignment(aln, l.seq, l.seq_len);
    } else {
      graph.add_alignment(aln, l.seq, l.seq_len, l.qual, l.qual_len);
    }
  }

  std::vector<uint32_t> coverages;
  consensus_ = graph.generate_consensus(&coverages);

  if (type_ == WindowType::kTGS && trim) {
    const uint32_t cov_floor = static_cast<uint32_t>(layers_.size() - 1) / 2;
    if (!trim_to_coverage(&consensus_, coverages, cov_floor)) {
      fprintf(stderr,
              "[rga::Window::generate_consensus] warning: window %u of contig %lu "
              "has no half-coverage core (possible chimera); left untrimmed\n",
              rank_, static_cast<unsigned long>(contig_id_));
    }
  }

  return true;
}

}  // namespace rga
