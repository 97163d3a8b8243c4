// Host adapter: Overlap -> HIP Myers aligner batches.
// Capability parity: reference src/cuda/cudaaligner.{hpp,cpp}
// (addOverlap / alignAll / generate_cigar_strings / reset; skip statuses
// leave the CIGAR empty so the CPU pairwise aligner picks the overlap up).
// Alignments are sorted by target length into 64-lane waves (uniform loop
// bounds per wave) and dispatched in as many sub-launches as the traceback
// arena requires.
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
  // band_width: requested band in cells (rounded to K in {4,8,16} blocks);
  // 0 selects the default 512.
  AlignerBatch(int device, size_t mem_budget, uint32_t band_width);
  ~AlignerBatch();

  AlignerBatch(const AlignerBatch&) = delete;
  AlignerBatch& operator=(const AlignerBatch&) = delete;

  // Reserves arena space for the overlap's query/target spans (bookkeeping
  // only — callable under the shared queue lock). Returns false when the
  // batch is full; never_fits is set when this overlap cannot run on the GPU
  // at all. The actual copies happen in pack().
  bool add_overlap(Overlap* overlap, const std::vector<std::unique_ptr<Sequence>>& sequences,
                   bool* never_fits);

  // Raw-span variant (testing / direct use): returns the slot index, -1 when
  // the batch is full, -2 when the pair can never run here.
  int32_t reserve_span(const char* q, uint32_t q_len, const char* t, uint32_t t_len);

  // Runs the kernel over everything reserved (pack + sub-launches + D2H).
  void run();
  // Post-run accessors per slot.
  int32_t status_of(uint32_t slot) const { return h_status_[slot]; }
  int32_t edit_distance_of(uint32_t slot) const { return h_edit_[slot]; }
  std::string cigar_of(uint32_t slot) const;

  // Copies every reserved span into the pinned staging buffers. Called
  // outside the queue lock so fills from different batches proceed in
  // parallel.
  void pack();

  uint32_t size() const { return static_cast<uint32_t>(overlaps_.size()); }

  // Runs the kernel (possibly several sub-launches), writes CIGAR strings
  // into the accepted overlaps AND walks them into breaking points right on
  // the emit threads (window_length > 0) — the walk overlaps the other
  // batches' kernels instead of serializing in the post-GPU CPU pass.
  // Returns how many overlaps fell back (band-edge -> CPU aligner).
  uint32_t align_and_emit(uint32_t window_length = 0);

  void reset();

 private:
  void allocate_arenas(size_t mem_budget);
  void release_all();


 private:
  int device_;
  void* stream_ = nullptr;
  AlnLimits limits_;
  uint32_t band_k_;  // blocks per band (4, 8 or 16)

  size_t seq_cap_, path_cap_;
  uint64_t tb_cap_u64_, s_cap_i32_, peq_cap_u64_;
  uint32_t max_alignments_;

  uint8_t* h_seqs_ = nullptr;
  AlnDesc* h_descs_ = nullptr;
  uint8_t* h_path_ = nullptr;
  uint32_t* h_path_len_ = nullptr;
  int32_t* h_status_ = nullptr;
  int32_t* h_edit_ = nullptr;
  uint32_t* h_order_ = nullptr;
  AlnWaveDesc* h_waves_ = nullptr;

  void* d_pool_ = nullptr;
  AlnDeviceArena arena_{};
  uint32_t* d_order_ = nullptr;
  AlnWaveDesc* d_waves_ = nullptr;

  size_t seq_bytes_ = 0;
  size_t path_bytes_ = 0;
  uint64_t tb_reserved_ = 0;  // u64 units of per-column state reserved
  std::vector<Overlap*> overlaps_;
  struct PendingSpan {
    const char* q;
    const char* t;
  };
  std::vector<PendingSpan> pending_;  // parallel to overlaps_; consumed by pack()
  size_t packed_upto_ = 0;
};

}  // namespace rga::hip
