// Host adapter: Overlap -> HIP banded aligner batches.
// Capability parity: reference src/cuda/cudaaligner.{hpp,cpp}
// (addOverlap / alignAll / generate_cigar_strings / reset; skip statuses
// leave the CIGAR empty so the CPU pairwise aligner picks the overlap up).
#pragma once

#include <cstdint>
#include <memory>
#include <vector>

#include "core/overlap.hpp"
#include "core/sequence.hpp"
#include "hip/aligner_types.hpp"

namespace rga::hip {

class AlignerBatch {
 public:
  AlignerBatch(int device, size_t mem_budget);
  ~AlignerBatch();

  AlignerBatch(const AlignerBatch&) = delete;
  AlignerBatch& operator=(const AlignerBatch&) = delete;

  // Reserves arena space for the overlap's query/target spans (bookkeeping
  // only — callable under the shared queue lock). Returns false when the
  // batch is full; never_fits is set when this overlap cannot run on the GPU
  // at all. The actual copies happen in pack().
  bool add_overlap(Overlap* overlap, const std::vector<std::unique_ptr<Sequence>>& sequences,
                   bool* never_fits);

  // Copies every reserved span into the pinned staging buffers. Called
  // outside the queue lock so fills from different batches proceed in
  // parallel.
  void pack();

  uint32_t size() const { return static_cast<uint32_t>(overlaps_.size()); }

  // Runs the kernel and writes CIGAR strings into the accepted overlaps;
  // returns how many fell back (band-edge failures).
  uint32_t align_and_emit();

  void reset();

 private:
  int device_;
  void* stream_ = nullptr;
  AlnLimits limits_;

  size_t seq_cap_, moves_cap_dw_, path_cap_;
  uint32_t max_alignments_;

  uint8_t* h_seqs_ = nullptr;
  AlnDesc* h_descs_ = nullptr;
  uint8_t* h_path_ = nullptr;
  uint32_t* h_path_len_ = nullptr;
  int32_t* h_status_ = nullptr;

  void* d_pool_ = nullptr;
  AlnDeviceArena arena_{};

  size_t seq_bytes_ = 0;
  size_t moves_dw_ = 0;
  size_t path_bytes_ = 0;
  std::vector<Overlap*> overlaps_;
  struct PendingSpan {
    const char* q;
    const char* t;
  };
  std::vector<PendingSpan> pending_;  // parallel to overlaps_; consumed by pack()
  size_t packed_upto_ = 0;
};

}  // namespace rga::hip
