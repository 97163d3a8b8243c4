// Two-phase polishing pipeline: initialize() parses and routes read segments
// into per-target windows; polish() runs POA consensus per window and merges
// results in window order. Behavioral parity with reference
// src/polisher.{hpp,cpp}; the HIP-accelerated pipeline (src/hip/) subclasses
// this and overrides the two phase hooks, with per-item CPU fallback.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <unordered_map>
#include <vector>

#include "core/overlap.hpp"
#include "core/sequence.hpp"
#include "core/window.hpp"
#include "io/parsers.hpp"
#include "util/logger.hpp"
#include "util/threadpool.hpp"

namespace rga {

namespace poa {
class NWEngine;
}

enum class PolisherType {
  kC,  // contig polishing
  kF,  // fragment correction
};

struct PolisherConfig {
  PolisherType type = PolisherType::kC;
  uint32_t window_length = 500;
  double quality_threshold = 10.0;
  double error_threshold = 0.3;
  bool trim = true;
  int8_t match = 3;
  int8_t mismatch = -5;
  int8_t gap = -4;
  uint32_t num_threads = 1;
  // HIP acceleration (0 batches = CPU path), mirroring the reference's
  // cudapoa-batches / cuda-banded-alignment / cudaaligner-* flags.
  uint32_t poa_batches = 0;
  bool banded_poa = false;
  uint32_t aligner_batches = 0;
  uint32_t aligner_band_width = 0;
};

class Polisher {
 public:
  Polisher(std::unique_ptr<SequenceParser> sparser, std::unique_ptr<OverlapParser> oparser,
           std::unique_ptr<SequenceParser> tparser, PolisherConfig config);
  virtual ~Polisher();

  void initialize();
  virtual void polish(std::vector<std::unique_ptr<Sequence>>& dst, bool drop_unpolished);

 protected:
  // Computes breaking points for every overlap (CPU Myers NW over the pool);
  // the HIP subclass aligns on-device and falls back here for skipped items.
  virtual void find_overlap_breaking_points(std::vector<std::unique_ptr<Overlap>>& overlaps);

  // Runs CPU consensus for windows_[i] where mask[i] is true (mask empty =
  // all windows); returns per-window polish status.
  void generate_consensus_cpu(std::vector<bool>& polished, const std::vector<bool>* todo);

  // Serial ordered merge of window consensuses into polished contigs.
  void collect(std::vector<std::unique_ptr<Sequence>>& dst, bool drop_unpolished,
               const std::vector<bool>& polished);

  std::unique_ptr<SequenceParser> sparser_;
  std::unique_ptr<OverlapParser> oparser_;
  std::unique_ptr<SequenceParser> tparser_;
  PolisherConfig config_;

  std::vector<std::unique_ptr<poa::NWEngine>> engines_;  // one per pool thread
  std::vector<std::unique_ptr<Sequence>> sequences_;
  std::string dummy_quality_;
  std::vector<std::shared_ptr<Window>> windows_;
  std::vector<uint64_t> targets_coverages_;
  WindowType window_type_ = WindowType::kTGS;

  std::unique_ptr<ThreadPool> thread_pool_;
  std::unique_ptr<Logger> logger_;

 private:
  // initialize() phases. Targets load whole; reads and overlaps stream in
  // bounded chunks so peak RSS stays flat on genome-scale inputs.
  uint64_t load_targets(SequenceIndex* index);
  void load_reads(SequenceIndex* index, uint64_t num_targets, std::vector<bool>* keep_name,
                  std::vector<bool>* keep_fwd, std::vector<bool>* keep_rev);
  void load_overlaps(const SequenceIndex& index, std::vector<bool>* keep_fwd,
                     std::vector<bool>* keep_rev, std::vector<std::unique_ptr<Overlap>>* overlaps);
  void build_windows(uint64_t num_targets);
  void route_layers(std::vector<std::unique_ptr<Overlap>>& overlaps, uint64_t num_targets);

  std::vector<uint64_t> first_window_of_target_;
};

std::unique_ptr<Polisher> createPolisher(const std::string& sequences_path,
                                         const std::string& overlaps_path,
                                         const std::string& target_path, PolisherConfig config);

}  // namespace rga
