#include "hip/poa_batch.hpp"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <chrono>
#include <cstring>
#include <stdexcept>

#include "hip/hip_common.hpp"

namespace rga::hip {

namespace {

constexpr size_t kSeqArenaPerWindow = 96 * 1024;  // bases+weights staging per window (avg)

size_t slab_bytes(const PoaLimits& L) {
  size_t n = L.max_nodes;
  size_t b = 0;
  b += 4 * n;                       // letters, in_cnt, out_cnt, ring_cnt
  b += n * L.max_edges * 8;         // in_edges(2) + in_weights(4) + out_edges(2)
  b += n * L.max_ring * 2;          // ring
  b += n * 2 * 2;                   // nseq, rank
  b += n * 12;                      // hb_score(8) + hb_pred(4)
  b += (2 * L.matrix_width + n) * 8;  // aln_nodes + aln_seq
  b += (n + 1) * L.matrix_width * 2;  // matrix
  b += (n + 1) * L.matrix_width;      // moves
  b += n * 8;                         // row_desc
  b += 64;                            // timing
  return b;
}

}  // namespace

PoaBatch::PoaBatch(int device, size_t mem_budget, int8_t match, int8_t mismatch, int8_t gap,
                   bool banded, uint32_t max_depth)
    : device_(device), match_(match), mismatch_(mismatch), gap_(gap), max_depth_(max_depth) {
  // int16 score guard: worst |score| <= (max_nodes + matrix_width) * max|param|
  // (banded mode additionally reserves values below -28000 as out-of-band)
  int32_t worst = static_cast<int32_t>(limits_.max_nodes + limits_.matrix_width) *
                  std::max({std::abs(static_cast<int32_t>(match)),
                            std::abs(static_cast<int32_t>(mismatch)),
                            std::abs(static_cast<int32_t>(gap))});
  // the kernel clamps stored scores at -28000 in every mode, so the usable
  // magnitude is 28000 (not the int16 32767 limit); banded mode additionally
  // reserves values below -28000 as out-of-band sentinels
  if (worst > (banded ? 27000 : 28000)) {
    char msg[160];
    snprintf(msg, sizeof(msg),
             "[rga::hip::PoaBatch] score parameters too large for int16 GPU "
             "scores (|m|,|x|,|g| must keep (%u+%u)*max <= %d)",
             limits_.max_nodes, limits_.matrix_width, banded ? 27000 : 28000);
    throw std::runtime_error(msg);
  }

  // The kernel compiles the capacity model as constants (poa_kernel.hip
  // keeps slab addressing in immediates); this object must not deviate
  // from the PoaLimits defaults without recompiling the kernel.
  {
    PoaLimits defaults;
    if (limits_.max_seq_len != defaults.max_seq_len ||
        limits_.max_nodes != defaults.max_nodes ||
        limits_.max_edges != defaults.max_edges ||
        limits_.max_ring != defaults.max_ring ||
        limits_.matrix_width != defaults.matrix_width ||
        limits_.max_consensus != defaults.max_consensus) {
      throw std::runtime_error(
          "[rga::hip::PoaBatch] PoaLimits diverged from the compile-time "
          "kernel capacity model");
    }
  }

  RGA_HIP_TRY(hipSetDevice(device_));
  hipStream_t s;
  RGA_HIP_TRY(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  stream_ = s;

  const size_t per_window = slab_bytes(limits_) + kSeqArenaPerWindow +
                            limits_.max_consensus * 3 + sizeof(PoaWindowDesc) + 1024;
  // slab cap: 4096 default; RGA_POA_SLABS raises it for single-mega-batch
  // experiments (all windows in one launch vs several contending launches)
  static const size_t slab_cap = [] {
    const char* e = getenv("RGA_POA_SLABS");
    long v = e != nullptr ? atol(e) : 0;
    return v > 0 ? static_cast<size_t>(v) : static_cast<size_t>(8192);
  }();
  num_slabs_ = static_cast<uint32_t>(
      std::min<size_t>(slab_cap, std::max<size_t>(32, mem_budget / per_window)));

  // Exception safety: the destructor does not run when the constructor
  // throws, so free whatever was acquired before rethrowing (the polisher
  // catches and falls back to the CPU engine).
  try {
    allocate_arenas(banded);
  } catch (...) {
    release_all();
    throw;
  }
}

// ---- device pool + pinned staging. The device pool is the big allocation:
// on OOM the slab count is halved and the layout retried — shared-device
// setups (rehearsals, several ranks per GPU) must degrade to smaller
// batches, not die (the round-1 sizing assumed one exclusive device).
void PoaBatch::allocate_arenas(bool banded) {
  const PoaLimits& L = limits_;
  const size_t n = L.max_nodes;
  size_t total = 0;
  auto carve = [&total](size_t bytes) {
    size_t off = total;
    total += (bytes + 255) & ~size_t(255);
    return off;
  };
  size_t max_layers = 0;
  size_t o_spans = 0;
  size_t o_seq = 0, o_wt = 0, o_ends = 0, o_ends_idx = 0, o_desc = 0, o_letters = 0,
         o_in_cnt = 0, o_out_cnt = 0, o_ring_cnt = 0, o_in_edges = 0, o_in_w = 0,
         o_out_edges = 0, o_ring = 0, o_nseq = 0, o_rank = 0, o_hb_score = 0, o_hb_pred = 0,
         o_aln_n = 0, o_aln_s = 0, o_matrix = 0, o_moves = 0, o_rd = 0, o_timing = 0,
         o_cons = 0, o_cov = 0, o_clen = 0, o_status = 0;
  for (;; num_slabs_ /= 2) {
    seq_arena_cap_ = static_cast<size_t>(num_slabs_) * kSeqArenaPerWindow;
    max_layers = static_cast<size_t>(num_slabs_) * (max_depth_ + 1);
    total = 0;
    o_seq = carve(seq_arena_cap_);
    o_wt = carve(seq_arena_cap_);
    o_ends = carve(max_layers * 4);
    o_spans = carve(max_layers * 4);
    o_ends_idx = carve((num_slabs_ + 1) * 4);
    o_desc = carve(num_slabs_ * sizeof(PoaWindowDesc));
    o_letters = carve(num_slabs_ * n);
    o_in_cnt = carve(num_slabs_ * n);
    o_out_cnt = carve(num_slabs_ * n);
    o_ring_cnt = carve(num_slabs_ * n);
    o_in_edges = carve(num_slabs_ * n * L.max_edges * 2);
    o_in_w = carve(num_slabs_ * n * L.max_edges * 4);
    o_out_edges = carve(num_slabs_ * n * L.max_edges * 2);
    o_ring = carve(num_slabs_ * n * L.max_ring * 2);
    o_nseq = carve(num_slabs_ * n * 2);
    o_rank = carve(num_slabs_ * n * 2);
    o_hb_score = carve(num_slabs_ * n * 8);
    o_hb_pred = carve(num_slabs_ * n * 4);
    o_aln_n = carve(num_slabs_ * (2 * L.matrix_width + n) * 4);
    o_aln_s = carve(num_slabs_ * (2 * L.matrix_width + n) * 4);
    o_matrix = carve(num_slabs_ * (n + 1) * L.matrix_width * 2);
    o_moves = carve(num_slabs_ * (n + 1) * L.matrix_width);
    o_rd = carve(num_slabs_ * n * 8);
    o_timing = carve(static_cast<size_t>(num_slabs_) * 8 * 8);
    o_cons = carve(static_cast<size_t>(num_slabs_) * L.max_consensus);
    o_cov = carve(static_cast<size_t>(num_slabs_) * L.max_consensus * 2);
    o_clen = carve(num_slabs_ * 4);
    o_status = carve(num_slabs_ * 4);

    hipError_t err = hipMalloc(&d_pool_, total);
    if (err == hipSuccess) {
      break;
    }
    d_pool_ = nullptr;
    (void)hipGetLastError();  // clear the sticky OOM
    if (num_slabs_ <= 64) {
      throw std::runtime_error(
          "[rga::hip::PoaBatch] device arena allocation failed (out of memory)");
    }
  }

  // ---- pinned host staging, sized to the slab count that fit ----
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_seq_), seq_arena_cap_));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_weight_), seq_arena_cap_));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_layer_ends_), max_layers * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_layer_span_), max_layers * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_layer_index_), (num_slabs_ + 1) * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_desc_),
                            num_slabs_ * sizeof(PoaWindowDesc)));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_consensus_),
                            static_cast<size_t>(num_slabs_) * limits_.max_consensus));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_coverage_),
                            static_cast<size_t>(num_slabs_) * limits_.max_consensus * 2));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_consensus_len_), num_slabs_ * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_status_), num_slabs_ * 4));

  auto base = static_cast<uint8_t*>(d_pool_);
  arena_.seq_data = base + o_seq;
  arena_.weight_data = base + o_wt;
  arena_.layer_ends = reinterpret_cast<uint32_t*>(base + o_ends);
  arena_.layer_spans = reinterpret_cast<uint32_t*>(base + o_spans);
  arena_.layer_ends_index = reinterpret_cast<uint32_t*>(base + o_ends_idx);
  arena_.windows = reinterpret_cast<PoaWindowDesc*>(base + o_desc);
  arena_.letters = base + o_letters;
  arena_.in_cnt = base + o_in_cnt;
  arena_.out_cnt = base + o_out_cnt;
  arena_.ring_cnt = base + o_ring_cnt;
  arena_.in_edges = reinterpret_cast<uint16_t*>(base + o_in_edges);
  arena_.in_weights = reinterpret_cast<int32_t*>(base + o_in_w);
  arena_.out_edges = reinterpret_cast<uint16_t*>(base + o_out_edges);
  arena_.ring = reinterpret_cast<uint16_t*>(base + o_ring);
  arena_.nseq = reinterpret_cast<uint16_t*>(base + o_nseq);
  arena_.rank = reinterpret_cast<uint16_t*>(base + o_rank);
  arena_.hb_score = reinterpret_cast<int64_t*>(base + o_hb_score);
  arena_.hb_pred = reinterpret_cast<int32_t*>(base + o_hb_pred);
  arena_.aln_nodes = reinterpret_cast<int32_t*>(base + o_aln_n);
  arena_.aln_seq = reinterpret_cast<int32_t*>(base + o_aln_s);
  arena_.matrix = reinterpret_cast<int16_t*>(base + o_matrix);
  arena_.moves = base + o_moves;
  arena_.row_desc = reinterpret_cast<uint64_t*>(base + o_rd);
  arena_.timing = reinterpret_cast<unsigned long long*>(base + o_timing);
  arena_.consensus = base + o_cons;
  arena_.coverage = reinterpret_cast<uint16_t*>(base + o_cov);
  arena_.consensus_len = reinterpret_cast<uint32_t*>(base + o_clen);
  arena_.status = reinterpret_cast<int32_t*>(base + o_status);
  arena_.vstore_mode = [] {
    const char* e = getenv("RGA_VSTORE");
    return e != nullptr ? static_cast<uint32_t>(atoi(e)) : 1u;
  }();
  arena_.match = match_;
  arena_.mismatch = mismatch_;
  arena_.gap = gap_;
  arena_.band_width = banded ? 256 : 0;  // reference static band 256
  arena_.limits = limits_;
}

PoaBatch::~PoaBatch() { release_all(); }

void PoaBatch::release_all() {
  (void)hipSetDevice(device_);
  if (d_pool_ != nullptr) {
    (void)hipFree(d_pool_);
  }
  for (void* p : {static_cast<void*>(h_seq_), static_cast<void*>(h_weight_),
                  static_cast<void*>(h_layer_ends_), static_cast<void*>(h_layer_span_),
                  static_cast<void*>(h_layer_index_),
                  static_cast<void*>(h_desc_), static_cast<void*>(h_consensus_),
                  static_cast<void*>(h_coverage_), static_cast<void*>(h_consensus_len_),
                  static_cast<void*>(h_status_)}) {
    if (p != nullptr) {
      (void)hipHostFree(p);
    }
  }
  if (stream_ != nullptr) {
    (void)hipStreamDestroy(static_cast<hipStream_t>(stream_));
  }
}

bool PoaBatch::add_window(const std::shared_ptr<Window>& window, bool* never_fits) {
  *never_fits = false;
  const uint32_t total_layers = window->num_layers();

  // backbone must fit the DP row; otherwise this window can never run here
  if (window->sequence(0).second + 1 > limits_.matrix_width) {
    *never_fits = true;
    return false;
  }
  if (windows_.size() >= num_slabs_) {
    return false;
  }

  // CPU-identical layer order: backbone, then layers by window start position
  std::vector<uint32_t> order = window->layer_order();

  // count + measure what ships (skip too-long layers, cap depth)
  uint32_t shipped = 1;
  uint32_t max_len = window->sequence(0).second;
  size_t bytes = window->sequence(0).second;
  for (uint32_t k = 1; k < total_layers && shipped < max_depth_ + 1; ++k) {
    uint32_t i = order[k];
    uint32_t len = window->sequence(i).second;
    if (len + 1 > limits_.matrix_width) {
      continue;  // reference: exceeded_maximum_sequence_size -> skipped
    }
    bytes += len;
    max_len = std::max(max_len, len);
    ++shipped;
  }

  if (seq_bytes_ + bytes > seq_arena_cap_) {
    return false;  // arena full; try the next batch round
  }

  // reserve (bookkeeping only; copies happen in pack() outside the queue lock)
  const uint32_t win_idx = static_cast<uint32_t>(windows_.size());
  PoaWindowDesc desc;
  desc.seq_offset = static_cast<uint32_t>(seq_bytes_);
  desc.scratch_idx = win_idx;
  desc.num_seqs = shipped;
  desc.max_len = max_len;
  h_layer_index_[win_idx] = static_cast<uint32_t>(num_layer_ends_);
  seq_bytes_ += bytes;
  num_layer_ends_ += shipped;
  h_desc_[win_idx] = desc;

  windows_.emplace_back(window);
  pending_orders_.emplace_back(std::move(order));
  seqs_added_.emplace_back(shipped - 1);  // layers only (effective coverage)
  return true;
}

void PoaBatch::pack() {
  for (; packed_upto_ < windows_.size(); ++packed_upto_) {
    const auto& window = windows_[packed_upto_];
    const auto& order = pending_orders_[packed_upto_];
    const PoaWindowDesc& desc = h_desc_[packed_upto_];
    size_t off = desc.seq_offset;
    uint32_t ends_at = h_layer_index_[packed_upto_];
    uint32_t rel_end = 0;
    uint32_t packed = 0;
    const uint32_t total_layers = window->num_layers();
    const uint32_t bb_len = window->sequence(0).second;
    // CPU spanning rule (Window::generate_consensus): a layer within 1% of
    // both window edges aligns against the full graph
    const uint32_t edge_margin = static_cast<uint32_t>(0.01 * bb_len);
    for (uint32_t k = 0; k < total_layers && packed < desc.num_seqs; ++k) {
      uint32_t i = order[k];
      auto seq = window->sequence(i);
      auto qual = window->quality(i);
      if (k > 0 && seq.second + 1 > limits_.matrix_width) {
        continue;
      }
      uint32_t span_word = 0xFFFFFFFFu;  // backbone / spanning layers: full DP
      if (k > 0) {
        auto sp = window->span(i);
        const bool spans_window =
            sp.first < edge_margin && sp.second > bb_len - edge_margin;
        if (!spans_window) {
          span_word = (sp.first << 16) | (sp.second & 0xffffu);
        }
      }
      h_layer_span_[ends_at] = span_word;
      std::memcpy(h_seq_ + off, seq.first, seq.second);
      if (qual.first != nullptr) {
        for (uint32_t b = 0; b < seq.second; ++b) {
          h_weight_[off + b] = static_cast<uint8_t>(qual.first[b]) - 33;
        }
      } else {
        std::memset(h_weight_ + off, 1, seq.second);
      }
      off += seq.second;
      rel_end += seq.second;
      h_layer_ends_[ends_at++] = rel_end;
      ++packed;
    }
  }
}

std::vector<bool> PoaBatch::generate(bool trim) {
  std::vector<bool> polished(windows_.size(), false);
  if (windows_.empty()) {
    return polished;
  }
  using clk = std::chrono::steady_clock;
  const bool host_timing = getenv("RGA_POA_HOSTTIME") != nullptr;
  auto t0 = clk::now();

  pack();
  const size_t nw0 = windows_.size();
  // order: columns-per-lane bucket first (each bucket is its own kernel
  // instantiation launched over a contiguous desc range), heaviest windows
  // first within a bucket (blocks launch roughly in order, so the long
  // poles start early instead of defining the tail)
  std::vector<uint32_t> perm(nw0);
  for (uint32_t i = 0; i < nw0; ++i) perm[i] = i;
  auto bucket = [&](uint32_t w) -> uint32_t {
    // (wb, ring-width) variants — see launch_poa_kernel. The LDS ring is
    // indexed by ABSOLUTE column, so its width follows the true longest
    // row (aw); only the columns-per-pass choice (WB) follows the banded
    // clamp. 9-wide single-pass was tried and spills 44-64 B/lane of
    // scratch, which is catastrophically slow; 8-wide multi-pass covers
    // any width, and the ring width picks the smallest LDS footprint the
    // rows fit in.
    const uint32_t aw = h_desc_[w].max_len;
    uint32_t pass_w = aw;
    if (arena_.band_width != 0) {
      pass_w = std::min(pass_w, arena_.band_width + 64);
    }
    const bool narrow_pass = pass_w <= 320;
    if (aw <= 320) return 0;  // WB5, 384-wide ring
    if (aw <= 575) {
      if (narrow_pass) return 3;  // banded: WB5 passes, 576-wide ring
      return h_desc_[w].num_seqs <= 96
                 ? 1   // WB8, 576-wide ring, 1536-node Kahn (deep windows
                       // can outgrow the smaller node cap; route them to
                       // the full variant instead of bouncing via the CPU)
                 : 2;  // WB8, full-width ring, full Kahn
    }
    return narrow_pass ? 4   // banded: WB5 passes, full 1024-wide ring
                       : 2;  // WB8, full 1024-wide ring
  };
  auto cost = [&](uint32_t w) {
    const uint32_t first = h_layer_index_[w];
    return h_layer_ends_[first + h_desc_[w].num_seqs - 1];  // total layer bytes
  };
  // bucket reads h_desc_, which is overwritten with the sorted descriptors
  // below — snapshot per original window up front so both the comparator and
  // the launch-range split see the pre-sort values
  std::vector<uint32_t> bucket_of(nw0);
  for (uint32_t i = 0; i < nw0; ++i) bucket_of[i] = bucket(i);
  std::sort(perm.begin(), perm.end(), [&](uint32_t a, uint32_t b) {
    const uint32_t ba = bucket_of[a], bb = bucket_of[b];
    if (ba != bb) return ba < bb;
    const uint32_t ca = cost(a), cb = cost(b);
    if (ca != cb) return ca > cb;
    return a < b;
  });
  std::vector<PoaWindowDesc> desc_sorted(nw0);
  std::vector<uint32_t> lidx_sorted(nw0);
  for (uint32_t w = 0; w < nw0; ++w) {
    desc_sorted[w] = h_desc_[perm[w]];
    lidx_sorted[w] = h_layer_index_[perm[w]];
  }
  std::copy(desc_sorted.begin(), desc_sorted.end(), h_desc_);
  std::copy(lidx_sorted.begin(), lidx_sorted.end(), h_layer_index_);

  RGA_HIP_CHECK(hipSetDevice(device_));
  auto s = static_cast<hipStream_t>(stream_);
  auto d = [&](const void* dst, const void* src, size_t bytes) {
    RGA_HIP_CHECK(hipMemcpyAsync(const_cast<void*>(dst), src, bytes, hipMemcpyHostToDevice, s));
  };
  d(arena_.seq_data, h_seq_, seq_bytes_);
  d(arena_.weight_data, h_weight_, seq_bytes_);
  d(arena_.layer_ends, h_layer_ends_, num_layer_ends_ * 4);
  d(arena_.layer_spans, h_layer_span_, num_layer_ends_ * 4);
  d(arena_.layer_ends_index, h_layer_index_, windows_.size() * 4);
  d(arena_.windows, h_desc_, windows_.size() * sizeof(PoaWindowDesc));

  // one launch per columns-per-lane bucket over its contiguous range
  {
    uint32_t begin = 0;
    while (begin < nw0) {
      const uint32_t wb = bucket_of[perm[begin]];
      uint32_t end = begin + 1;
      while (end < nw0 && bucket_of[perm[end]] == wb) {
        ++end;
      }
      launch_poa_kernel(arena_, begin, end - begin, wb, stream_);
      begin = end;
    }
  }

  const size_t nw = windows_.size();
  RGA_HIP_CHECK(hipMemcpyAsync(h_consensus_, arena_.consensus,
                               nw * limits_.max_consensus, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_coverage_, arena_.coverage,
                               nw * limits_.max_consensus * 2, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_consensus_len_, arena_.consensus_len, nw * 4,
                               hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_status_, arena_.status, nw * 4, hipMemcpyDeviceToHost, s));
  auto t1 = clk::now();
  RGA_HIP_CHECK(hipStreamSynchronize(s));
  auto t2 = clk::now();

  if (getenv("RGA_POA_TIMING") != nullptr) {
    std::vector<unsigned long long> t(nw * 8);
    RGA_HIP_CHECK(hipMemcpy(t.data(), arena_.timing, nw * 8 * 8, hipMemcpyDeviceToHost));
    unsigned long long sum[8] = {0};
    std::vector<unsigned long long> totals(nw);
    for (size_t i = 0; i < nw; ++i) {
      for (int k = 0; k < 8; ++k) sum[k] += t[i * 8 + k];
      totals[i] = t[i * 8 + 6];
    }
    std::sort(totals.begin(), totals.end());
    fprintf(stderr,
            "[rga::hip::PoaBatch] timing (wall ticks, %zu windows): dp=%llu tb=%llu "
            "add=%llu topo=%llu rdesc=%llu cons=%llu total=%llu layers=%llu\n"
            "[rga::hip::PoaBatch] window total ticks: p50=%llu p90=%llu p99=%llu max=%llu\n",
            nw, sum[0], sum[1], sum[2], sum[3], sum[4], sum[5], sum[6], sum[7],
            totals[nw / 2], totals[nw * 9 / 10], totals[nw * 99 / 100], totals[nw - 1]);
  }

  // CPU-parity post-processing (reference cudabatch.cpp:199-261).
  // Kernel slot i handled original window perm[i].
  for (size_t i = 0; i < nw; ++i) {
    const uint32_t orig = perm[i];
    auto& window = windows_[orig];
    if (h_status_[i] != kPoaOk || h_consensus_len_[i] == 0) {
      polished[orig] = false;  // device-side failure -> CPU fallback
      continue;
    }
    std::string consensus(reinterpret_cast<char*>(h_consensus_ + i * limits_.max_consensus),
                          h_consensus_len_[i]);
    bool status = true;
    if (window->type() == WindowType::kTGS && trim) {
      const uint16_t* cov = h_coverage_ + i * limits_.max_consensus;
      uint32_t average = seqs_added_[orig] / 2;
      int32_t begin = 0, end = static_cast<int32_t>(consensus.size()) - 1;
      for (; begin < static_cast<int32_t>(consensus.size()); ++begin) {
        if (cov[begin] >= average) {
          break;
        }
      }
      for (; end >= 0; --end) {
        if (cov[end] >= average) {
          break;
        }
      }
      if (begin >= end) {
        fprintf(stderr, "[rga::hip::PoaBatch] warning: contig %lu might be chimeric in window %u!\n",
                static_cast<unsigned long>(window->id()), window->rank());
        status = false;
      } else {
        consensus = consensus.substr(begin, end - begin + 1);
      }
    }
    if (status) {
      window->set_consensus(std::move(consensus));
    }
    polished[orig] = status;
  }
  if (host_timing) {
    auto t3 = clk::now();
    auto ms = [](auto a, auto b) {
      return std::chrono::duration_cast<std::chrono::microseconds>(b - a).count() / 1000.0;
    };
    fprintf(stderr,
            "[rga::hip::PoaBatch] host timing (%zu windows): pack+sort+h2d+launch %.1f ms, "
            "gpu sync %.1f ms, post %.1f ms\n",
            nw, ms(t0, t1), ms(t1, t2), ms(t2, t3));
  }
  return polished;
}

void PoaBatch::reset() {
  windows_.clear();
  seqs_added_.clear();
  pending_orders_.clear();
  packed_upto_ = 0;
  seq_bytes_ = 0;
  num_layer_ends_ = 0;
}

}  // namespace rga::hip
