// Host adapter: Window -> HIP POA kernel batches.
// Capability parity: reference src/cuda/cudabatch.{hpp,cpp} (addWindow /
// generateConsensus / reset contract, CPU-parity post-processing, per-seq
// skip accounting for effective coverage) on the MI355X kernel of
// poa_kernel.hip. One batch owns one device stream and a slab arena sized
// from the memory budget (reference: 90% free / batches).
#pragma once

#include <cstdint>
#include <memory>
#include <vector>

#include "core/window.hpp"
#include "hip/poa_types.hpp"

namespace rga::hip {

class PoaBatch {
 public:
  PoaBatch(int device, size_t mem_budget, int8_t match, int8_t mismatch, int8_t gap,
           bool banded, uint32_t max_depth);
  ~PoaBatch();

  PoaBatch(const PoaBatch&) = delete;
  PoaBatch& operator=(const PoaBatch&) = delete;

  // Reserves arena space for the window's layers (sorted, CPU-identical
  // order). Bookkeeping only, callable under the shared queue lock; the
  // copies happen in pack() / generate(). Returns false when the batch is
  // full (caller retries with a fresh batch) or the window cannot fit this
  // configuration at all (never_fits set).
  bool add_window(const std::shared_ptr<Window>& window, bool* never_fits);

  // Copies every reserved window into the pinned staging buffers; called
  // outside the queue lock (generate() calls it implicitly).
  void pack();

  uint32_t size() const { return static_cast<uint32_t>(windows_.size()); }
  uint32_t capacity() const { return num_slabs_; }

  // Runs the kernel and writes consensus into the windows; returns per-window
  // polish status (false = needs CPU fallback or <3-layer backbone copy).
  std::vector<bool> generate(bool trim);

  void reset();

 private:
  // Constructor phases (see .cpp): OOM-degrading arena allocation and the
  // matching cleanup used by both the destructor and a throwing constructor.
  void allocate_arenas(bool banded);
  void release_all();

  int device_;
  void* stream_ = nullptr;

  PoaLimits limits_;
  uint32_t num_slabs_;
  size_t seq_arena_cap_;
  int8_t match_, mismatch_, gap_;
  uint32_t max_depth_;

  // pinned host staging
  uint8_t* h_seq_ = nullptr;
  uint8_t* h_weight_ = nullptr;
  uint32_t* h_layer_ends_ = nullptr;
  uint32_t* h_layer_span_ = nullptr;
  uint32_t* h_layer_index_ = nullptr;
  PoaWindowDesc* h_desc_ = nullptr;
  uint8_t* h_consensus_ = nullptr;
  uint16_t* h_coverage_ = nullptr;
  uint32_t* h_consensus_len_ = nullptr;
  int32_t* h_status_ = nullptr;

  // device memory (single allocation, carved)
  void* d_pool_ = nullptr;
  PoaDeviceArena arena_{};

  size_t seq_bytes_ = 0;
  size_t num_layer_ends_ = 0;
  std::vector<std::shared_ptr<Window>> windows_;
  std::vector<uint32_t> seqs_added_;  // layers shipped per window (coverage)
  std::vector<std::vector<uint32_t>> pending_orders_;  // layer order per reserved window
  size_t packed_upto_ = 0;
};

}  // namespace rga::hip
