// One POA consensus task: a slice of a target contig (the backbone) plus the
// read segments ("layers") that the breaking-point walk routed onto it.
//
// Capability parity with reference src/window.{hpp,cpp}: fewer than three
// layers copies the backbone through unpolished; layers enter the graph in
// start-position order (unstable sort — introsort tie order is part of the
// pinned CPU goldens); a layer that does not reach within ~1% of both window
// edges is aligned against a subgraph of its backbone range; long-read (kTGS)
// windows have consensus ends below half-average coverage trimmed off.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace rga {

namespace poa {
class NWEngine;
}

enum class WindowType {
  kNGS,  // short reads (mean length <= 1000)
  kTGS,  // long reads
};

class Window {
 public:
  // A sequence segment participating in this window's consensus. Views only:
  // the bytes live in the Polisher's Sequence objects, which must outlive the
  // window. Slot 0 is always the backbone (span unset).
  struct Layer {
    const char* seq = nullptr;
    uint32_t seq_len = 0;
    const char* qual = nullptr;  // nullptr when the source read had none
    uint32_t qual_len = 0;
    uint32_t begin = 0;  // backbone coordinates of the routed segment
    uint32_t end = 0;
  };

  Window(uint64_t contig_id, uint32_t rank, WindowType type, const char* backbone,
         uint32_t backbone_len, const char* backbone_qual, uint32_t qual_len);

  uint64_t id() const { return contig_id_; }
  uint32_t rank() const { return rank_; }
  WindowType type() const { return type_; }
  const std::string& consensus() const { return consensus_; }
  void set_consensus(std::string consensus) { consensus_ = std::move(consensus); }

  uint32_t num_layers() const { return static_cast<uint32_t>(layers_.size()); }
  uint32_t backbone_length() const { return layers_.front().seq_len; }

  // Pair-view accessors kept for the batch packers (index 0 = backbone).
  std::pair<const char*, uint32_t> sequence(uint32_t i) const {
    return {layers_[i].seq, layers_[i].seq_len};
  }
  std::pair<const char*, uint32_t> quality(uint32_t i) const {
    return {layers_[i].qual, layers_[i].qual_len};
  }
  // Backbone span of a routed layer (begin, inclusive end); (0, 0) = backbone.
  std::pair<uint32_t, uint32_t> span(uint32_t i) const {
    return {layers_[i].begin, layers_[i].end};
  }

  // Indices of all layers with the backbone first and the rest ordered by
  // start position. Shared by the CPU and HIP consensus paths so a window's
  // result never depends on which device polished it.
  std::vector<uint32_t> layer_order() const;

  void add_layer(const char* sequence, uint32_t sequence_length, const char* quality,
                 uint32_t quality_length, uint32_t begin, uint32_t end);

  // CPU POA consensus; returns true when a real consensus was generated
  // (false = backbone copy for under-covered windows).
  bool generate_consensus(poa::NWEngine& engine, bool trim);

 private:
  uint64_t contig_id_;
  uint32_t rank_;
  WindowType type_;
  std::string consensus_;
  std::vector<Layer> layers_;
};

std::shared_ptr<Window> createWindow(uint64_t id, uint32_t rank, WindowType type,
                                     const char* backbone, uint32_t backbone_length,
                                     const char* quality, uint32_t quality_length);

}  // namespace rga
