#include "hip/aligner_batch.hpp"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <atomic>
#include <cstring>
#include <numeric>
#include <string>
#include <thread>
#include <vector>

#include "hip/hip_common.hpp"

namespace rga::hip {

namespace {
constexpr uint32_t kLanes = 64;

// Band blocks for a requested width. The polisher's auto heuristic clamps
// to 256 (K=4; see hip_polisher.cpp) — explicit --cudaaligner-band-width
// values still select up to K=16.
uint32_t pick_band_k(uint32_t band_width) {
  if (band_width == 0) return 8;  // default band 512
  uint32_t blocks = (band_width + 63) / 64;
  if (blocks <= 4) return 4;
  if (blocks <= 8) return 8;
  return 16;
}


}  // namespace

AlignerBatch::AlignerBatch(int device, size_t mem_budget, uint32_t band_width)
    : device_(device), band_k_(pick_band_k(band_width)) {
  limits_.band = band_k_ * 64;
  try {
    allocate_arenas(mem_budget);
  } catch (...) {
    release_all();
    throw;
  }
}

void AlignerBatch::allocate_arenas(size_t mem_budget) {
  RGA_HIP_TRY(hipSetDevice(device_));
  hipStream_t s;
  RGA_HIP_TRY(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  stream_ = s;

  // arena split: the per-column band state dominates — per alignment
  // ~ (m+1) * K * (16 B Pv/Mv + 4 B S); seqs/path/peq are q+t-scale.
  // Cap the device pool so construction stays cheap and the POA arenas
  // (pooled at the same time) keep headroom on 288 GB parts: at K=4 a
  // 20 kbp alignment's state is ~1.3 MB, so 12 GB holds ~9k alignments.
  // On OOM (shared-device setups) the pool halves and retries.
  size_t pool = std::min<size_t>(mem_budget, 12ull << 30);
  size_t total = 0;
  auto carve = [&total](size_t bytes) {
    size_t off = total;
    total += (bytes + 255) & ~size_t(255);
    return off;
  };
  size_t o_seqs = 0, o_descs = 0, o_peq = 0, o_tb = 0, o_s = 0, o_path = 0, o_plen = 0,
         o_status = 0, o_ed = 0, o_order = 0, o_waves = 0;
  uint32_t max_waves = 0;
  for (;; pool /= 2) {
    seq_cap_ = std::max<size_t>(16u << 20, pool / 96);
    path_cap_ = seq_cap_;
    peq_cap_u64_ = seq_cap_ / 8;  // 4 codes per 64 bases = q_bytes/2 of u64s
                                  // is generous; /8 covers wave-max padding
    max_alignments_ = 65536;

    const size_t fixed = 2 * seq_cap_ + peq_cap_u64_ * 8 +
                         max_alignments_ * (sizeof(AlnDesc) + 16) + (2u << 20);
    const size_t state = pool > fixed ? pool - fixed : (16u << 20);
    tb_cap_u64_ = state / 20 * 16 / 8;  // 16/20 of state bytes as u64
    s_cap_i32_ = state / 20 * 4 / 4;    // 4/20 of state bytes as i32

    // sized for the SMALLEST lanes-per-wave the launcher may pick (16 for
    // small jobs), not kLanes — and the per-sub-launch descriptor slices
    // accumulate across the whole run
    max_waves = max_alignments_ / 16 + 2;
    total = 0;
    o_seqs = carve(seq_cap_);
    o_descs = carve(max_alignments_ * sizeof(AlnDesc));
    o_peq = carve(peq_cap_u64_ * 8);
    o_tb = carve(tb_cap_u64_ * 8);
    o_s = carve(s_cap_i32_ * 4);
    o_path = carve(path_cap_);
    o_plen = carve(max_alignments_ * 4);
    o_status = carve(max_alignments_ * 4);
    o_ed = carve(max_alignments_ * 4);
    o_order = carve(max_alignments_ * 4);
    o_waves = carve(max_waves * sizeof(AlnWaveDesc));

    hipError_t err = hipMalloc(&d_pool_, total);
    if (err == hipSuccess) {
      break;
    }
    d_pool_ = nullptr;
    (void)hipGetLastError();
    if (pool <= (256u << 20)) {
      throw std::runtime_error(
          "[rga::hip::AlignerBatch] device arena allocation failed (out of memory)");
    }
  }

  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_seqs_), seq_cap_));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_descs_),
                            max_alignments_ * sizeof(AlnDesc)));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_path_), path_cap_));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_path_len_), max_alignments_ * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_status_), max_alignments_ * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_edit_), max_alignments_ * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_order_), max_alignments_ * 4));
  RGA_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&h_waves_),
                            max_waves * sizeof(AlnWaveDesc)));
  auto base = static_cast<uint8_t*>(d_pool_);
  arena_.seqs = base + o_seqs;
  arena_.descs = reinterpret_cast<AlnDesc*>(base + o_descs);
  arena_.peq = reinterpret_cast<uint64_t*>(base + o_peq);
  arena_.tb = reinterpret_cast<uint64_t*>(base + o_tb);
  arena_.sbuf = reinterpret_cast<int32_t*>(base + o_s);
  arena_.path = base + o_path;
  arena_.path_len = reinterpret_cast<uint32_t*>(base + o_plen);
  arena_.status = reinterpret_cast<int32_t*>(base + o_status);
  arena_.edit_distance = reinterpret_cast<int32_t*>(base + o_ed);
  d_order_ = reinterpret_cast<uint32_t*>(base + o_order);
  d_waves_ = reinterpret_cast<AlnWaveDesc*>(base + o_waves);
  arena_.order = d_order_;
  arena_.waves = d_waves_;
  arena_.lanes_per_wave = kLanes;
  arena_.limits = limits_;
}

AlignerBatch::~AlignerBatch() { release_all(); }

void AlignerBatch::release_all() {
  (void)hipSetDevice(device_);
  if (d_pool_ != nullptr) {
    (void)hipFree(d_pool_);
  }
  for (void* p : {static_cast<void*>(h_seqs_), static_cast<void*>(h_descs_),
                  static_cast<void*>(h_path_), static_cast<void*>(h_path_len_),
                  static_cast<void*>(h_status_), static_cast<void*>(h_edit_),
                  static_cast<void*>(h_order_), static_cast<void*>(h_waves_)}) {
    if (p != nullptr) {
      (void)hipHostFree(p);
    }
  }
  if (stream_ != nullptr) {
    (void)hipStreamDestroy(static_cast<hipStream_t>(stream_));
  }
}

int32_t AlignerBatch::reserve_span(const char* q, uint32_t q_len, const char* t,
                                   uint32_t t_len) {
  if (q_len == 0 || t_len == 0 || q_len > limits_.max_len || t_len > limits_.max_len) {
    return -2;  // reference: exceeded_max_length -> CPU fallback
  }
  const size_t bytes = static_cast<size_t>(q_len) + t_len;
  // per-lane share of the wave's tb region (the wave total is 64 of these,
  // and the arena-capacity check is against the sum of all shares). The
  // per-fill cap is HALF a tb arena: a greedy first fill at the full cap
  // would swallow most of the overlap queue and starve the other batch
  // threads (the pull model balances only when rounds are smaller than
  // queue/threads).
  const uint64_t tb_need = static_cast<uint64_t>(t_len + 1) * band_k_ * 2;
  if (overlaps_.size() >= max_alignments_ || seq_bytes_ + bytes > seq_cap_ ||
      path_bytes_ + bytes > path_cap_ || tb_reserved_ + tb_need > tb_cap_u64_ / 2) {
    return -1;
  }

  AlnDesc d;
  d.q_offset = static_cast<uint32_t>(seq_bytes_);
  d.q_len = q_len;
  seq_bytes_ += q_len;
  d.t_offset = static_cast<uint32_t>(seq_bytes_);
  d.t_len = t_len;
  seq_bytes_ += t_len;
  d.path_offset = static_cast<uint32_t>(path_bytes_);
  path_bytes_ += bytes;
  tb_reserved_ += tb_need;

  const int32_t slot = static_cast<int32_t>(overlaps_.size());
  h_descs_[slot] = d;
  overlaps_.emplace_back(nullptr);
  pending_.push_back({q, t});
  return slot;
}

bool AlignerBatch::add_overlap(Overlap* overlap,
                               const std::vector<std::unique_ptr<Sequence>>& sequences,
                               bool* never_fits) {
  *never_fits = false;
  auto q = overlap->query_span(sequences);
  auto t = overlap->target_span(sequences);
  int32_t slot = reserve_span(q.first, q.second, t.first, t.second);
  if (slot == -2) {
    *never_fits = true;
    return false;
  }
  if (slot < 0) {
    return false;
  }
  overlaps_[slot] = overlap;
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

void AlignerBatch::run() {
  if (overlaps_.empty()) {
    return;
  }
  pack();
  RGA_HIP_CHECK(hipSetDevice(device_));
  auto s = static_cast<hipStream_t>(stream_);
  const uint32_t na = static_cast<uint32_t>(overlaps_.size());
  RGA_HIP_CHECK(hipMemcpyAsync(const_cast<uint8_t*>(arena_.seqs), h_seqs_, seq_bytes_,
                               hipMemcpyHostToDevice, s));
  RGA_HIP_CHECK(hipMemcpyAsync(const_cast<AlnDesc*>(arena_.descs), h_descs_,
                               na * sizeof(AlnDesc), hipMemcpyHostToDevice, s));

  // sort into waves by descending target length (uniform per-wave loops)
  std::vector<uint32_t> order(na);
  std::iota(order.begin(), order.end(), 0u);
  std::sort(order.begin(), order.end(), [&](uint32_t x, uint32_t y) {
    if (h_descs_[x].t_len != h_descs_[y].t_len) return h_descs_[x].t_len > h_descs_[y].t_len;
    return x < y;
  });
  std::copy(order.begin(), order.end(), h_order_);
  RGA_HIP_CHECK(hipMemcpyAsync(d_order_, h_order_, na * 4, hipMemcpyHostToDevice, s));

  // greedy sub-launches bounded by the tb/s/peq arenas. Waves are
  // deliberately under-filled for small jobs: at 64 alignments/wave a
  // typical polish run yields ~1.5 waves per CU and every gather latency
  // lands on the wall clock; 16-32/wave gives each CU interleavable waves
  // at the cost of idle lanes (which issue no extra instructions).
  const uint32_t K = band_k_;
  // Full 64-lane packing for big jobs; small jobs still under-fill waves so
  // every CU gets interleavable work. (The 32-lane middle tier predated the
  // back-to-back sub-launch pipeline; measured post-pipeline, 64 wins at
  // the flagship size: 6.00 vs 5.89 Mbp/s.)
  uint32_t lanes = kLanes;
  if (na < 16384) {
    lanes = 16;
  }
  // tuning override (RGA_ALN_LANES in {16,32,64}): alignments packed per
  // 64-lane wave — fewer per wave = more interleavable waves per CU at the
  // cost of idle lanes
  static const long lanes_env = [] {
    const char* e = getenv("RGA_ALN_LANES");
    return e != nullptr ? atol(e) : 0;
  }();
  if (lanes_env == 16 || lanes_env == 32 || lanes_env == 64) {
    lanes = static_cast<uint32_t>(lanes_env);
  }
  uint32_t wave_begin = 0;  // in wave units
  uint32_t wave_desc_off = 0;  // h_waves_/d_waves_ slice per sub-launch, so
                               // sub-launches enqueue back-to-back with no
                               // intervening stream sync
  const uint32_t num_waves_total = (na + lanes - 1) / lanes;
  std::vector<uint32_t> never_run;  // original indices of un-runnable waves
  auto wave_needs = [&](uint32_t w, uint64_t* peq_need, uint64_t* tb_need,
                        uint64_t* s_need) {
    uint32_t nb = K, mmax = 0;
    for (uint32_t l = 0; l < lanes; ++l) {
      const uint32_t slot = w * lanes + l;
      if (slot >= na) break;
      const AlnDesc& d = h_descs_[h_order_[slot]];
      nb = std::max(nb, (d.q_len + 63) / 64);
      mmax = std::max(mmax, d.t_len);
    }
    *peq_need = static_cast<uint64_t>(nb) * 4 * kLanes;
    *tb_need = static_cast<uint64_t>(mmax + 1) * K * 2 * kLanes;
    *s_need = static_cast<uint64_t>(mmax + 1) * K * kLanes;
    return std::make_pair(nb, mmax);
  };
  while (wave_begin < num_waves_total) {
    // a wave whose state alone exceeds an arena (giant target at a wide
    // explicit band: e.g. 262 kbp at K=16 needs ~4.3 GB of tb) can never
    // launch — fail its alignments to the CPU pairwise fallback instead of
    // writing out of bounds. No construction-time carve guarantees a full
    // max_len wave fits; this check is the guarantee.
    {
      uint64_t peq_need, tb_need, s_need;
      wave_needs(wave_begin, &peq_need, &tb_need, &s_need);
      if (peq_need > peq_cap_u64_ || tb_need > tb_cap_u64_ || s_need > s_cap_i32_) {
        for (uint32_t l = 0; l < lanes; ++l) {
          const uint32_t slot = wave_begin * lanes + l;
          if (slot >= na) break;
          never_run.push_back(h_order_[slot]);
        }
        ++wave_begin;
        continue;
      }
    }
    uint64_t peq_off = 0, tb_off = 0, s_off = 0;
    uint32_t w = wave_begin;
    uint32_t launch_waves = 0;
    for (; w < num_waves_total; ++w) {
      uint64_t peq_need, tb_need, s_need;
      auto nbm = wave_needs(w, &peq_need, &tb_need, &s_need);
      if (launch_waves > 0 && (peq_off + peq_need > peq_cap_u64_ ||
                               tb_off + tb_need > tb_cap_u64_ || s_off + s_need > s_cap_i32_)) {
        break;
      }
      AlnWaveDesc wd;
      wd.peq_off = peq_off;
      wd.tb_off = tb_off;
      wd.s_off = s_off;
      wd.nb = nbm.first;
      wd.mmax = nbm.second;
      h_waves_[wave_desc_off + launch_waves] = wd;
      peq_off += peq_need;
      tb_off += tb_need;
      s_off += s_need;
      ++launch_waves;
    }
    const uint32_t launch_align =
        std::min(na - wave_begin * lanes, launch_waves * lanes);
    RGA_HIP_CHECK(hipMemcpyAsync(d_waves_ + wave_desc_off, h_waves_ + wave_desc_off,
                                 launch_waves * sizeof(AlnWaveDesc),
                                 hipMemcpyHostToDevice, s));
    AlnDeviceArena launch_arena = arena_;
    launch_arena.order = d_order_ + wave_begin * lanes;
    launch_arena.waves = d_waves_ + wave_desc_off;
    launch_arena.lanes_per_wave = lanes;
    launch_aligner_kernel(launch_arena, launch_waves, launch_align, K, stream_);
    wave_begin += launch_waves;
    wave_desc_off += launch_waves;
  }

  RGA_HIP_CHECK(hipMemcpyAsync(h_path_, arena_.path, path_bytes_, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_path_len_, arena_.path_len, na * 4, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_status_, arena_.status, na * 4, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipMemcpyAsync(h_edit_, arena_.edit_distance, na * 4, hipMemcpyDeviceToHost, s));
  RGA_HIP_CHECK(hipStreamSynchronize(s));
  for (uint32_t orig : never_run) {
    h_status_[orig] = kAlnNotRun;
    h_path_len_[orig] = 0;
    h_edit_[orig] = -1;
  }
}

std::string AlignerBatch::cigar_of(uint32_t slot) const {
  std::string cigar;
  if (h_status_[slot] != kAlnOk || h_path_len_[slot] == 0) {
    return cigar;
  }
  const uint8_t* path = h_path_ + h_descs_[slot].path_offset;
  const uint32_t plen = h_path_len_[slot];
  static const char kOps[3] = {'M', 'I', 'D'};
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
  return cigar;
}

uint32_t AlignerBatch::align_and_emit(uint32_t window_length) {
  if (overlaps_.empty()) {
    return 0;
  }
  run();
  // CIGAR strings are independent per slot: build them on a few threads
  // (the run-length walk over ~30 kbp paths x thousands of alignments is
  // otherwise a serial tail after every batch round)
  const size_t n = overlaps_.size();
  const uint32_t nthreads = std::min<uint32_t>(8, std::max<uint32_t>(1, n / 512));
  std::atomic<uint32_t> failed{0};
  auto emit_range = [&](size_t begin, size_t end) {
    uint32_t local_failed = 0;
    for (size_t i = begin; i < end; ++i) {
      if (overlaps_[i] == nullptr) {
        continue;
      }
      std::string cigar = cigar_of(static_cast<uint32_t>(i));
      if (cigar.empty()) {
        ++local_failed;  // empty CIGAR -> CPU pairwise fallback
        continue;
      }
      overlaps_[i]->set_cigar(std::move(cigar));
      if (window_length > 0) {
        // walk into breaking points here (overlapped with other batches'
        // GPU work); the polisher's final CPU pass then no-ops for this
        // overlap and only realigns the skipped ones
        overlaps_[i]->find_breaking_points_from_cigar(window_length);
        overlaps_[i]->set_cigar(std::string());  // consumed; free the bytes
      }
    }
    failed += local_failed;
  };
  if (nthreads <= 1) {
    emit_range(0, n);
  } else {
    std::vector<std::thread> threads;
    const size_t step = (n + nthreads - 1) / nthreads;
    for (uint32_t t = 0; t < nthreads; ++t) {
      const size_t b = t * step, e = std::min(n, b + step);
      if (b < e) {
        threads.emplace_back(emit_range, b, e);
      }
    }
    for (auto& t : threads) {
      t.join();
    }
  }
  return failed.load();
}

void AlignerBatch::reset() {
  overlaps_.clear();
  pending_.clear();
  packed_upto_ = 0;
  seq_bytes_ = 0;
  path_bytes_ = 0;
  tb_reserved_ = 0;
}

}  // namespace rga::hip
