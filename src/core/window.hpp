// One POA consensus unit: a backbone slice of a target contig plus the read
// segments (layers) routed into it. Behavioral parity with reference
// src/window.{hpp,cpp}: <3 layers copies the backbone; layers are added in
// start-position-sorted order (std::sort, same tie behavior); layers not
// spanning ~98% of the window are aligned to a subgraph of the backbone range;
// TGS windows trim consensus ends below half-average coverage.
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
  Window(uint64_t id, uint32_t rank, WindowType type, const char* backbone,
         uint32_t backbone_length, const char* quality, uint32_t quality_length);

  uint64_t id() const { return id_; }
  uint32_t rank() const { return rank_; }
  WindowType type() const { return type_; }
  const std::string& consensus() const { return consensus_; }
  void set_consensus(std::string consensus) { consensus_ = std::move(consensus); }

  uint32_t num_layers() const { return static_cast<uint32_t>(sequences_.size()); }
  // Layer views (index 0 is the backbone).
  const std::pair<const char*, uint32_t>& sequence(uint32_t i) const { return sequences_[i]; }
  const std::pair<const char*, uint32_t>& quality(uint32_t i) const { return qualities_[i]; }
  const std::pair<uint32_t, uint32_t>& position(uint32_t i) const { return positions_[i]; }

  // Layer order sorted by start position (backbone stays first); shared by the
  // CPU and HIP consensus paths so results do not depend on the device.
  std::vector<uint32_t> layer_order() const;

  void add_layer(const char* sequence, uint32_t sequence_length, const char* quality,
                 uint32_t quality_length, uint32_t begin, uint32_t end);

  // CPU POA consensus; returns true when a real consensus was generated.
  bool generate_consensus(poa::NWEngine& engine, bool trim);

 private:
  uint64_t id_;
  uint32_t rank_;
  WindowType type_;
  std::string consensus_;
  std::vector<std::pair<const char*, uint32_t>> sequences_;
  std::vector<std::pair<const char*, uint32_t>> qualities_;
  std::vector<std::pair<uint32_t, uint32_t>> positions_;
};

std::shared_ptr<Window> createWindow(uint64_t id, uint32_t rank, WindowType type,
                                     const char* backbone, uint32_t backbone_length,
                                     const char* quality, uint32_t quality_length);

}  // namespace rga
