// MI355X pipeline orchestration: devices, batch workers, pull-model window
// queue, per-item CPU fallback.
// Capability parity: reference src/cuda/cudapolisher.cpp — device
// enumeration + warm-up, 90%-free-memory batch sizing, one host thread per
// batch pulling from a global mutex-guarded queue, failed windows re-polished
// on CPU, serial ordered merge identical to the CPU path.
#include <hip/hip_runtime.h>

#include <atomic>
#include <chrono>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <tuple>
#include <utility>
#include <vector>

#include "core/polisher.hpp"
#include "hip/aligner_batch.hpp"
#include "hip/hip_common.hpp"
#include "hip/poa_batch.hpp"

namespace rga {

namespace {

// Process-wide batch pools: arena construction (tens of GB of device memory
// + pinned staging) is expensive, and repeated polish() calls in one process
// (bench steps, services) would otherwise pay it every time. Batches are
// keyed by their construction parameters; budget differences are ignored
// (arenas are internally capped).
struct BatchPools {
  std::mutex mutex;
  std::vector<std::pair<uint64_t, std::unique_ptr<hip::AlignerBatch>>> aligner;
  std::vector<std::pair<uint64_t, std::unique_ptr<hip::PoaBatch>>> poa;
};
BatchPools& pools() {
  static BatchPools p;
  return p;
}

std::unique_ptr<hip::AlignerBatch> acquire_aligner(int device, size_t budget, uint32_t band) {
  const uint64_t key = (static_cast<uint64_t>(device) << 32) | band;
  auto& p = pools();
  {
    std::lock_guard<std::mutex> lock(p.mutex);
    for (auto it = p.aligner.begin(); it != p.aligner.end(); ++it) {
      if (it->first == key) {
        auto b = std::move(it->second);
        p.aligner.erase(it);
        return b;
      }
    }
  }
  return std::make_unique<hip::AlignerBatch>(device, budget, band);
}

void release_aligner(int device, uint32_t band, std::unique_ptr<hip::AlignerBatch> b) {
  const uint64_t key = (static_cast<uint64_t>(device) << 32) | band;
  b->reset();
  auto& p = pools();
  std::lock_guard<std::mutex> lock(p.mutex);
  p.aligner.emplace_back(key, std::move(b));
}

uint64_t poa_key(int device, int8_t m, int8_t x, int8_t g, bool banded, uint32_t depth) {
  return (static_cast<uint64_t>(device) << 40) |
         (static_cast<uint64_t>(static_cast<uint8_t>(m)) << 32) |
         (static_cast<uint64_t>(static_cast<uint8_t>(x)) << 24) |
         (static_cast<uint64_t>(static_cast<uint8_t>(g)) << 16) |
         (static_cast<uint64_t>(banded) << 15) | depth;
}

std::unique_ptr<hip::PoaBatch> acquire_poa(int device, size_t budget, int8_t m, int8_t x,
                                           int8_t g, bool banded, uint32_t depth) {
  const uint64_t key = poa_key(device, m, x, g, banded, depth);
  auto& p = pools();
  {
    std::lock_guard<std::mutex> lock(p.mutex);
    for (auto it = p.poa.begin(); it != p.poa.end(); ++it) {
      if (it->first == key) {
        auto b = std::move(it->second);
        p.poa.erase(it);
        return b;
      }
    }
  }
  return std::make_unique<hip::PoaBatch>(device, budget, m, x, g, banded, depth);
}

void release_poa(uint64_t key, std::unique_ptr<hip::PoaBatch> b) {
  b->reset();
  auto& p = pools();
  std::lock_guard<std::mutex> lock(p.mutex);
  p.poa.emplace_back(key, std::move(b));
}

// Fraction of free HBM a polish phase may claim (per process). One process
// per exclusive GPU wants the reference's 0.9; several ranks sharing one
// device (rehearsals) export RGA_MEM_FRACTION to split it explicitly.
double mem_fraction() {
  static const double frac = [] {
    const char* e = getenv("RGA_MEM_FRACTION");
    double f = e != nullptr ? atof(e) : 0.9;
    return f > 0.0 && f <= 0.95 ? f : 0.9;
  }();
  return frac;
}

}  // namespace

class HipPolisher : public Polisher {
 public:
  HipPolisher(std::unique_ptr<SequenceParser> sparser, std::unique_ptr<OverlapParser> oparser,
              std::unique_ptr<SequenceParser> tparser, PolisherConfig config)
      : Polisher(std::move(sparser), std::move(oparser), std::move(tparser), config) {
    int n = hip::device_count();
    if (n < 1) {
      fprintf(stderr, "[rga::HipPolisher] error: no HIP devices available!\n");
      exit(1);
    }
    for (int d = 0; d < n; ++d) {
      RGA_HIP_CHECK(hipSetDevice(d));
      RGA_HIP_CHECK(hipFree(nullptr));  // create the context up front
      devices_.emplace_back(d);
    }
    fprintf(stderr, "[rga::HipPolisher] using %d GPU(s)\n", n);
  }

  // GPU overlap alignment; overlaps the GPU skips or fails keep an empty
  // CIGAR and are aligned by the CPU pairwise path inside the base-class
  // call at the end (reference cudapolisher.cpp:74-213).
  void find_overlap_breaking_points(std::vector<std::unique_ptr<Overlap>>& overlaps) override {
    if (config_.aligner_batches < 1) {
      Polisher::find_overlap_breaking_points(overlaps);
      return;
    }

    // auto band: 10% of the mean overlap span, forced even
    // (reference cudapolisher.cpp:150-163)
    uint32_t band = config_.aligner_band_width;
    if (band == 0) {
      uint64_t total_len = 0, count = 0;
      for (const auto& o : overlaps) {
        if (o && !o->has_cigar()) {
          total_len += o->t_end() - o->t_begin();
          ++count;
        }
      }
      if (count > 0) {
        // 10% of mean span like the reference, clamped scale-aware: the
        // Myers band is exact inside the band and escapes fall back per
        // item to the exact CPU aligner, so the clamp is a throughput
        // knob, not a quality one (measured, docs/DESIGN.md "Band clamp"):
        // at <=20 kbp spans / 6% error a 256 band completes 24/24 overlaps
        // exactly; at 100 kbp it escapes 5/24 (6% err) and 13/24 (12%),
        // all of which the 512 band completes exactly. Wide explicit bands
        // stay available via --cudaaligner-band-width.
        const uint64_t mean_span = total_len / count;
        band = static_cast<uint32_t>(mean_span / 10) & ~1u;
        band = band < 64 ? 64 : (band > 256 ? 256 : band);
        if (mean_span >= 40000) {
          band = 512;  // long-read drift outruns the 256 band (see above)
        }
      }
    }

    std::vector<std::unique_ptr<hip::AlignerBatch>> batches;
    std::vector<int> batch_devices;
    try {
      for (int d : devices_) {
        RGA_HIP_CHECK(hipSetDevice(d));
        size_t free_mem = 0, total_mem = 0;
        RGA_HIP_CHECK(hipMemGetInfo(&free_mem, &total_mem));
        size_t budget =
            static_cast<size_t>(free_mem * mem_fraction()) / config_.aligner_batches;
        for (uint32_t b = 0; b < config_.aligner_batches; ++b) {
          batches.emplace_back(acquire_aligner(d, budget, band));
          batch_devices.emplace_back(d);
        }
      }
    } catch (const std::exception& e) {
      fprintf(stderr,
              "[rga::HipPolisher] warning: %s; aligning overlaps on the CPU instead\n",
              e.what());
      for (size_t b = 0; b < batches.size(); ++b) {
        release_aligner(batch_devices[b], band, std::move(batches[b]));
      }
      Polisher::find_overlap_breaking_points(overlaps);
      return;
    }

    std::mutex queue_mutex;
    uint64_t next_overlap = 0;
    std::atomic<uint64_t> skipped{0};
    std::atomic<int64_t> t_fill_ns{0}, t_gpu_ns{0};

    auto worker = [&](hip::AlignerBatch* batch) {
      using clk = std::chrono::steady_clock;
      while (true) {
        batch->reset();
        uint32_t pulled = 0;
        auto t0 = clk::now();
        {
          std::lock_guard<std::mutex> lock(queue_mutex);
          while (next_overlap < overlaps.size()) {
            auto* o = overlaps[next_overlap].get();
            if (o->has_cigar()) {  // SAM input already has an alignment
              ++next_overlap;
              continue;
            }
            bool never_fits = false;
            if (batch->add_overlap(o, sequences_, &never_fits)) {
              ++pulled;
              ++next_overlap;
            } else if (never_fits) {
              ++skipped;
              ++next_overlap;
            } else {
              break;
            }
          }
        }
        auto t1 = clk::now();
        t_fill_ns += (t1 - t0).count();
        if (pulled == 0) {
          return;
        }
        skipped += batch->align_and_emit(config_.window_length);
        t_gpu_ns += (clk::now() - t1).count();
      }
    };

    std::vector<std::thread> threads;
    threads.reserve(batches.size());
    for (auto& b : batches) {
      threads.emplace_back(worker, b.get());
    }
    for (auto& t : threads) {
      t.join();
    }
    for (size_t b = 0; b < batches.size(); ++b) {
      release_aligner(batch_devices[b], band, std::move(batches[b]));
    }
    batches.clear();

    fprintf(stderr,
            "[rga::HipPolisher] align timings: queue %.3f s, pack+gpu+cigar %.3f s "
            "(sum over %zu batch threads)\n",
            t_fill_ns.load() / 1e9, t_gpu_ns.load() / 1e9, threads.size());
    if (skipped.load() > 0) {
      fprintf(stderr, "[rga::HipPolisher] %lu overlap(s) aligned on CPU\n",
              static_cast<unsigned long>(skipped.load()));
    }
    // CPU pass: walks CIGARs into breaking points; aligns leftovers with the
    // CPU pairwise engine
    Polisher::find_overlap_breaking_points(overlaps);
  }

  void polish(std::vector<std::unique_ptr<Sequence>>& dst, bool drop_unpolished) override {
    if (config_.poa_batches < 1) {
      Polisher::polish(dst, drop_unpolished);
      return;
    }
    // int16 device scores bound the usable score magnitudes
    // (poa_batch.cpp guard); prefer the CPU engine over dying
    {
      const int32_t worst_param =
          std::max({std::abs(static_cast<int32_t>(config_.match)),
                    std::abs(static_cast<int32_t>(config_.mismatch)),
                    std::abs(static_cast<int32_t>(config_.gap))});
      // thresholds must match the PoaBatch constructor guard exactly: the
      // kernel clamps stored scores at -28000 in every mode, and banded mode
      // reserves values below that as out-of-band sentinels
      if (static_cast<int32_t>(2047 + 1024) * worst_param >
          (config_.banded_poa ? 27000 : 28000)) {
        fprintf(stderr,
                "[rga::HipPolisher] warning: score parameters too large for the "
                "int16 GPU POA scores; polishing windows on the CPU instead\n");
        Polisher::polish(dst, drop_unpolished);
        return;
      }
    }

    logger_->log();

    // one batch object per (device, batch) pair, reference cudapolisher.cpp:228-240
    constexpr uint32_t kMaxDepth = 200;  // MAX_DEPTH_PER_WINDOW, cudapolisher.cpp:226
    std::vector<std::unique_ptr<hip::PoaBatch>> batches;
    std::vector<uint64_t> batch_keys;
    try {
      for (int d : devices_) {
        RGA_HIP_CHECK(hipSetDevice(d));
        size_t free_mem = 0, total_mem = 0;
        RGA_HIP_CHECK(hipMemGetInfo(&free_mem, &total_mem));
        size_t budget = static_cast<size_t>(free_mem * mem_fraction()) / config_.poa_batches;
        for (uint32_t b = 0; b < config_.poa_batches; ++b) {
          batches.emplace_back(acquire_poa(d, budget, config_.match, config_.mismatch,
                                           config_.gap, config_.banded_poa, kMaxDepth));
          batch_keys.emplace_back(poa_key(d, config_.match, config_.mismatch, config_.gap,
                                          config_.banded_poa, kMaxDepth));
        }
      }
    } catch (const std::exception& e) {
      fprintf(stderr,
              "[rga::HipPolisher] warning: %s; polishing windows on the CPU instead\n",
              e.what());
      for (size_t b = 0; b < batches.size(); ++b) {
        release_poa(batch_keys[b], std::move(batches[b]));
      }
      Polisher::polish(dst, drop_unpolished);
      return;
    }

    std::vector<bool> polished(windows_.size(), false);
    std::vector<bool> gpu_handled(windows_.size(), false);
    std::mutex queue_mutex;
    uint64_t next_window = 0;
    std::mutex status_mutex;
    std::atomic<int64_t> t_fill_ns{0}, t_gpu_ns{0};

    auto worker = [&](hip::PoaBatch* batch) {
      using clk = std::chrono::steady_clock;
      std::vector<uint64_t> batch_indices;
      // a window claimed from the queue but refused by a full batch is
      // carried into the worker's next round (claim under the lock is just
      // an index increment; the per-window bookkeeping — layer-order sort,
      // measurements — runs outside so the workers do not serialize on it)
      uint64_t carried = UINT64_MAX;
      while (true) {
        batch_indices.clear();
        batch->reset();
        auto t0 = clk::now();
        while (true) {
          uint64_t i;
          if (carried != UINT64_MAX) {
            i = carried;
            carried = UINT64_MAX;
          } else {
            std::lock_guard<std::mutex> lock(queue_mutex);
            if (next_window >= windows_.size()) {
              break;
            }
            i = next_window++;
          }
          auto& w = windows_[i];
          if (w->num_layers() < 3) {
            // backbone copy, never worth a GPU trip (reference semantics:
            // consensus = backbone, unpolished)
            w->set_consensus(std::string(w->sequence(0).first, w->sequence(0).second));
            continue;
          }
          bool never_fits = false;
          if (batch->add_window(w, &never_fits)) {
            batch_indices.emplace_back(i);
          } else if (never_fits) {
            continue;  // leave for the CPU fallback pass
          } else {
            carried = i;  // batch full; keep the claim for the next round
            break;
          }
        }
        auto t1 = clk::now();
        t_fill_ns += (t1 - t0).count();
        if (batch_indices.empty()) {
          return;
        }
        auto status = batch->generate(config_.trim);
        t_gpu_ns += (clk::now() - t1).count();
        {
          std::lock_guard<std::mutex> lock(status_mutex);
          for (size_t i = 0; i < batch_indices.size(); ++i) {
            polished[batch_indices[i]] = status[i];
            gpu_handled[batch_indices[i]] = status[i];
          }
        }
      }
    };

    std::vector<std::thread> threads;
    threads.reserve(batches.size());
    for (auto& b : batches) {
      threads.emplace_back(worker, b.get());
    }
    for (auto& t : threads) {
      t.join();
    }
    for (size_t b = 0; b < batches.size(); ++b) {
      release_poa(batch_keys[b], std::move(batches[b]));
    }
    batches.clear();
    fprintf(stderr,
            "[rga::HipPolisher] poa timings: queue %.3f s, pack+gpu+post %.3f s "
            "(sum over %zu batch threads)\n",
            t_fill_ns.load() / 1e9, t_gpu_ns.load() / 1e9, threads.size());

    // CPU fallback for every window the GPU did not polish
    // (reference cudapolisher.cpp:354-383)
    uint64_t num_fallback = 0;
    std::vector<bool> todo(windows_.size(), false);
    for (uint64_t i = 0; i < windows_.size(); ++i) {
      if (!gpu_handled[i] && windows_[i]->num_layers() >= 3) {
        todo[i] = true;
        ++num_fallback;
      }
    }
    if (num_fallback > 0) {
      fprintf(stderr, "[rga::HipPolisher] %lu window(s) re-polished on CPU\n",
              static_cast<unsigned long>(num_fallback));
      generate_consensus_cpu(polished, &todo);
    } else {
      logger_->log("[rga::Polisher] generated consensus");
    }

    collect(dst, drop_unpolished, polished);
  }

 private:
  std::vector<int> devices_;
};

namespace hip {
int runtime_device_count() { return device_count(); }

void runtime_device_synchronize() { RGA_HIP_CHECK(hipDeviceSynchronize()); }

// Window-level CPU-vs-GPU differ entry: runs the HIP POA kernel directly on
// raw windows (backbone first; layer spans in backbone coordinates), so a
// divergent window found end-to-end can be isolated and replayed. Returns
// (consensus, polished) per window — exactly what the pipeline would store.
// Windows the GPU cannot run (< 3 layers, overflow) return the same CPU
// fallback semantics as HipPolisher::polish.
std::vector<std::pair<std::string, bool>> poa_windows_gpu(
    const std::vector<std::vector<std::tuple<std::string, std::string, uint32_t, uint32_t>>>&
        window_layers,
    int8_t match, int8_t mismatch, int8_t gap, bool banded, bool trim, bool tgs) {
  std::vector<std::pair<std::string, bool>> out(window_layers.size());
  std::vector<std::shared_ptr<Window>> windows;
  windows.reserve(window_layers.size());
  for (const auto& layers : window_layers) {
    if (layers.empty()) {
      throw std::runtime_error("poa_windows_gpu: window without a backbone");
    }
    const auto& bb = layers.front();
    auto w = createWindow(0, 0, tgs ? WindowType::kTGS : WindowType::kNGS,
                          std::get<0>(bb).data(),
                          static_cast<uint32_t>(std::get<0>(bb).size()),
                          std::get<1>(bb).data(),
                          static_cast<uint32_t>(std::get<1>(bb).size()));
    for (size_t i = 1; i < layers.size(); ++i) {
      const auto& l = layers[i];
      const std::string& q = std::get<1>(l);
      w->add_layer(std::get<0>(l).data(), static_cast<uint32_t>(std::get<0>(l).size()),
                   q.empty() ? nullptr : q.data(), static_cast<uint32_t>(q.size()),
                   std::get<2>(l), std::get<3>(l));
    }
    windows.push_back(std::move(w));
  }

  hip::PoaBatch batch(0, 4ull << 30, match, mismatch, gap, banded, 200);
  size_t begin = 0;
  std::vector<int64_t> slot_of(windows.size(), -1);
  auto flush = [&](size_t end) {
    std::vector<bool> status = batch.generate(trim);
    size_t k = 0;
    for (size_t i = begin; i < end; ++i) {
      if (slot_of[i] >= 0) {
        out[i] = {windows[i]->consensus(), status[k++]};
      }
    }
    batch.reset();
    begin = end;
  };
  size_t next_slot = 0;
  for (size_t i = 0; i < windows.size(); ++i) {
    if (windows[i]->num_layers() < 3) {
      auto bb = windows[i]->sequence(0);
      out[i] = {std::string(bb.first, bb.second), false};
      continue;
    }
    bool never_fits = false;
    if (batch.add_window(windows[i], &never_fits)) {
      slot_of[i] = static_cast<int64_t>(next_slot++);
      continue;
    }
    if (never_fits) {
      out[i] = {std::string(), false};
      continue;
    }
    flush(i);
    next_slot = 0;
    if (batch.add_window(windows[i], &never_fits)) {
      slot_of[i] = static_cast<int64_t>(next_slot++);
    } else {
      out[i] = {std::string(), false};
    }
  }
  flush(windows.size());
  return out;
}

// Direct GPU alignment of raw (query, target) pairs — numerics testing
// entry (GPU edit distance must equal the CPU optimum; CIGARs must be
// consistent). Returns (cigar, edit_distance, status) per pair.
std::vector<std::tuple<std::string, int32_t, int32_t>> align_pairs(
    const std::vector<std::pair<std::string, std::string>>& pairs, uint32_t band_width) {
  std::vector<std::tuple<std::string, int32_t, int32_t>> out(pairs.size());
  if (pairs.empty()) {
    return out;
  }
  AlignerBatch batch(0, 4ull << 30, band_width);
  size_t begin = 0;
  std::vector<int32_t> slots(pairs.size(), -1);
  auto flush = [&](size_t end) {
    batch.run();
    for (size_t i = begin; i < end; ++i) {
      if (slots[i] >= 0) {
        out[i] = {batch.cigar_of(slots[i]), batch.edit_distance_of(slots[i]),
                  batch.status_of(slots[i])};
      } else {
        out[i] = {std::string(), -1, kAlnNotRun};
      }
    }
    batch.reset();
    begin = end;
  };
  for (size_t i = 0; i < pairs.size(); ++i) {
    slots[i] = batch.reserve_span(pairs[i].first.data(),
                                  static_cast<uint32_t>(pairs[i].first.size()),
                                  pairs[i].second.data(),
                                  static_cast<uint32_t>(pairs[i].second.size()));
    if (slots[i] == -1) {
      flush(i);
      slots[i] = batch.reserve_span(pairs[i].first.data(),
                                    static_cast<uint32_t>(pairs[i].first.size()),
                                    pairs[i].second.data(),
                                    static_cast<uint32_t>(pairs[i].second.size()));
    }
  }
  flush(pairs.size());
  return out;
}
}  // namespace hip

std::unique_ptr<Polisher> createHipPolisher(std::unique_ptr<SequenceParser> sparser,
                                            std::unique_ptr<OverlapParser> oparser,
                                            std::unique_ptr<SequenceParser> tparser,
                                            PolisherConfig config) {
  return std::make_unique<HipPolisher>(std::move(sparser), std::move(oparser), std::move(tparser),
                                       config);
}

}  // namespace rga
