#include "hip/aligner_batch.hpp"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstring>
#include <string>

#include "hip/hip_common.hpp"

namespace rga::hip {

namespace {
// arena split: moves dominate; per ~30 kbp+30 kbp alignment the moves cost
// (q+t+1)*256 B ~ 15 MB, seqs q+t, path q+t.
constexpr double kMovesShare = 0.97;
}  // namespace

AlignerBatch::AlignerBatch(int device, size_t mem_budget) : device_(device) {
  RGA_HIP_CHECK(hipSetDevice(device_));
  hipStream_t s;
  RGA_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  stream_ = s;

  moves_cap_dw_ = static_cast<size_t>(mem_budget * kMovesShare) / 4;
  seq_cap_ = std::max<size_t>(16u << 20, mem_budget / 128);
  path_cap_ = seq_cap_;
  max_alignments_ = 65536;

  RGA_HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_seqs_), seq_cap_));
  RGA_HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_descs_),
                              max_alignments_ * sizeof(AlnDesc)));
  RGA_HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_path_), path_cap_));
  RGA_HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_path_len_), max_alignments_ * 4));
  RGA_HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_status_), max_alignments_ * 4));

  size_t total = 0;
  auto carve = [&total](size_t bytes) {
    size_t off = total;
    total += (bytes + 255) & ~size_t(255);
    return off;
  };
  size_t o_seqs = carve(seq_cap_);
  size_t o_descs = carve(max_alignments_ * sizeof(AlnDesc));
  size_t o_moves = carve(moves_cap_dw_ * 4);
  size_t o_path = carve(path_cap_);
  size_t o_plen = carve(max_alignments_ * 4);
  size_t o_status = carve(max_alignments_ * 4);
  size_t o_ed = carve(max_alignments_ * 4);

  RGA_HIP_CHECK(hipMalloc(&d_pool_, total));
  auto base = static_cast<uint8_t*>(d_pool_);
  arena_.seqs = base + o_seqs;
  arena_.descs = reinterpret_cast<AlnDesc*>(base + o_descs);
  arena_.moves = reinterpret_cast<uint32_t*>(base + o_moves);
  arena_.path = base + o_path;
  arena_.path_len = reinterpret_cast<uint32_t*>(base + o_plen);
  arena_.status = reinterpret_cast<int32_t*>(base + o_status);
  arena_.edit_distance = reinterpret_cast<int32_t*>(base + o_ed);
  arena_.limits = limits_;
}

AlignerBatch::~AlignerBatch() {
  (void)hipSetDevice(device_);
  if (d_pool_ != nullptr) {
    (void)hipFree(d_pool_);
  }
  for (void* p : {static_cast<void*>(h_seqs_), static_cast<void*>(h_descs_),
                  static_cast<void*>(h_path_), static_cast<void*>(h_path_len_),
                  static_cast<void*>(h_status_)}) {
    if (p != nullptr) {
      (void)hipHostFree(p);
    }
  }
  if (stream_ != nullptr) {
    (void)hipStreamDestroy(static_cast<hipStream_t>(stream_));
  }
}

bool AlignerBatch::add_overlap(Overlap* overlap,
                               const std::vector<std::unique_ptr<Sequence>>& sequences,
                               bool* never_fits) {
  *never_fits = false;
  auto q = overlap->query_span(sequences);
  auto t = overlap->target_span(sequences);
  if (q.second == 0 || t.second == 0 || q.second > limits_.max_len ||
      t.second > limits_.max_len) {
    *never_fits = true;  // reference: exceeded_max_length -> CPU fallback
    return false;
  }
  const size_t bytes = static_cast<size_t>(q.second) + t.second;
  const size_t mdw = (static_cast<size_t>(q.second) + t.second + 1) * 64;
  if (overlaps_.size() >= max_alignments_ || seq_bytes_ + bytes > seq_cap_ ||
      moves_dw_ + mdw > moves_cap_dw_ || path_bytes_ + bytes > path_cap_) {
    return false;
  }

  AlnDesc d;
  d.q_offset = static_cast<uint32_t>(seq_bytes_);
  d.q_len = q.second;
  seq_bytes_ += q.second;
  d.t_offset = static_cast<uint32_t>(seq_bytes_);
  d.t_len = t.second;
  seq_bytes_ += t.second;
  d.moves_offset = moves_dw_;
  moves_dw_ += mdw;
  d.path_offset = static_cast<uint32_t>(path_bytes_);
  path_bytes_ += bytes;

  h_descs_[overlaps_.size()] = d;
  overlaps_.emplace_back(overlap);
  pending_.push_back({q.first, t.first});
  return true;
}

void AlignerBatch::pack() {
  for (; packed_upto_ < overlaps_.size(); ++packed_upto_) {
    const AlnDesc& d = h_descs_[packed_upto_];
    const PendingSpan& p = pending_[packed_upto_];
    std::memcpy(h_seqs_ + d.q_offset, p.q, d.q_len);
    std::memcpy(h_seqs_ + d.t_offset, p.t, d.t_len);
  }
}

uint32_t AlignerBatch::align_and_emit() {
  if (overlaps_.empty()) {
    return 0;
  }
  pack();
  RGA_HIP_CHECK(hipSetDevice(device_));
  auto s = static_cast<hipStream_t>(stream_);
  RGA_HIP_CHECK(hipMemcpyAsync(const_cast<uint8_t*>(arena_.seqs), h_seqs_, seq_bytes_,
                               hipMemcpyHostToDevice, s));
  RGA_HIP_CHECK(hipMemcpyAsync(const_cast<AlnDesc*>(arena_.descs), h_descs_,
                               overlaps_.size() * sizeof(AlnDesc), hipMemcpyHostToDevice, s));

  launch_aligner_kernel(arena_, static_cast<uint32_t>(overlaps_.size()), stream_);

  RGA_HIP_CHECK(hipMemcpyAsync(h_path_, arena_.path, path_bytes_, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_path_len_, arena_.path_len, overlaps_.size() * 4,
                               hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_status_, arena_.status, overlaps_.size() * 4,
                               hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipStreamSynchronize(s));

  uint32_t failed = 0;
  std::string cigar;
  static const char kOps[3] = {'M', 'I', 'D'};
  for (size_t i = 0; i < overlaps_.size(); ++i) {
    if (h_status_[i] != kAlnOk || h_path_len_[i] == 0) {
      ++failed;  // empty CIGAR -> CPU pairwise fallback
      continue;
    }
    const uint8_t* path = h_path_ + h_descs_[i].path_offset;
    const uint32_t plen = h_path_len_[i];
    cigar.clear();
    // path is reversed (walked from (n, m)); emit forward with run-lengths
    uint32_t run = 0;
    uint8_t run_op = 255;
    char buf[16];
    for (int64_t k = static_cast<int64_t>(plen) - 1; k >= 0; --k) {
      uint8_t op = path[k];
      if (op == run_op) {
        ++run;
      } else {
        if (run > 0) {
          cigar.append(buf, snprintf(buf, sizeof(buf), "%u%c", run, kOps[run_op]));
        }
        run_op = op;
        run = 1;
      }
    }
    if (run > 0) {
      cigar.append(buf, snprintf(buf, sizeof(buf), "%u%c", run, kOps[run_op]));
    }
    overlaps_[i]->set_cigar(cigar);
  }
  return failed;
}

void AlignerBatch::reset() {
  overlaps_.clear();
  pending_.clear();
  packed_upto_ = 0;
  seq_bytes_ = 0;
  moves_dw_ = 0;
  path_bytes_ = 0;
}

}  // namespace rga::hip
