// Device-side batch layout for the HIP pairwise (overlap) aligner.
//
// Capability parity target: GenomeWorks cudaaligner as driven by reference
// src/cuda/cudaaligner.cpp (add_alignment / align_all / get CIGARs, skip
// statuses feeding the CPU pairwise fallback) — re-designed for CDNA4:
// ONE LANE PER ALIGNMENT banded Myers bit-vector DP. Each lane advances a
// sliding band of K 64-row blocks (band = 64*K cells) column by column with
// the blocked Myers recurrence — 64 DP cells per VALU instruction, no
// cross-lane traffic in the hot loop. Per-column Pv/Mv words and block-bottom
// scores are stored wave-coalesced for the on-device O(1)-per-step traceback
// (popcount score reconstruction). Alignments whose optimal path leaves the
// band fail with kAlnBandEdge and fall back to the CPU aligner (reference
// contract: cudaaligner skip statuses -> edlib, src/cuda/cudaaligner.cpp:63-72).
#pragma once

#include <cstdint>

namespace rga::hip {

struct AlnLimits {
  uint32_t band = 512;        // band cells (64 * K); K in {4, 8, 16}
  uint32_t max_len = 262144;  // per-side length cap
};

enum AlnStatus : int32_t {
  kAlnOk = 0,
  kAlnBandEdge = 1,  // traceback hit an invalid cell: band too narrow
  kAlnNotRun = 2,
};

struct AlnDesc {
  uint32_t q_offset;  // into packed seq arena
  uint32_t q_len;     // n (DP rows)
  uint32_t t_offset;
  uint32_t t_len;      // m (DP columns)
  uint32_t path_offset;  // bytes into the path arena (capacity q_len+t_len)
};

// Per-wave offsets into the shared arenas (element units). Lanes of a wave
// share one region so per-column stores coalesce across the 64 alignments.
struct AlnWaveDesc {
  uint64_t peq_off;  // u64 units: [(block*4 + code)*64 + lane]
  uint64_t tb_off;   // u64 units: [((col*K + block)*2 + {Pv,Mv})*64 + lane]
  uint64_t s_off;    // i32 units: [(col*K + block)*64 + lane]
  uint32_t nb;       // query blocks allocated in peq for this wave (>= K)
  uint32_t mmax;     // max t_len in this wave (uniform column loop bound)
};

struct AlnDeviceArena {
  const uint8_t* seqs;       // packed query/target bytes
  const AlnDesc* descs;      // original order
  const uint32_t* order;     // launch: alignment index per sorted lane slot
  const AlnWaveDesc* waves;  // launch: per-wave arena offsets
  uint64_t* peq;             // match bit-vectors per query block/code
  uint64_t* tb;              // per-column Pv/Mv band state (traceback)
  int32_t* sbuf;             // per-column block-bottom scores
  uint8_t* path;             // per alignment: ops from (n,m); 0=M,1=I,2=D
  uint32_t* path_len;        // per alignment (original index)
  int32_t* status;           // per alignment
  int32_t* edit_distance;    // per alignment (diagnostic)
  uint32_t lanes_per_wave;   // alignments packed per 64-lane wave; fewer
                             // than 64 when the job is too small to give
                             // every CU multiple waves (latency hiding)
  AlnLimits limits;
};

// Launches the K-block Myers kernel over `num_align` sorted alignments
// packed arena.lanes_per_wave to a wave. band_k must be 4, 8 or 16.
void launch_aligner_kernel(const AlnDeviceArena& arena, uint32_t num_waves, uint32_t num_align,
                           uint32_t band_k, void* stream);

}  // namespace rga::hip
