// Device-side batch layout for the HIP pairwise (overlap) aligner.
//
// Capability parity target: GenomeWorks cudaaligner as driven by reference
// src/cuda/cudaaligner.cpp (add_alignment / align_all / get CIGARs, skip
// statuses feeding the CPU edlib fallback) — re-designed for CDNA4: one
// 64-lane wavefront per alignment, banded anti-diagonal edit-distance DP
// with the band center following the rectangle diagonal, 2-bit moves packed
// 16-per-dword in HBM, and an LDS-tiled on-device traceback.
#pragma once

#include <cstdint>

namespace rga::hip {

struct AlnLimits {
  uint32_t band = 1024;          // band cells per anti-diagonal (16 regs/lane)
  uint32_t max_len = 262144;     // per-side length cap
};

enum AlnStatus : int32_t {
  kAlnOk = 0,
  kAlnBandEdge = 1,   // traceback hit an invalid cell: band too narrow
  kAlnNotRun = 2,
};

struct AlnDesc {
  uint32_t q_offset;     // into packed seq arena
  uint32_t q_len;
  uint32_t t_offset;
  uint32_t t_len;
  uint64_t moves_offset;  // dwords into the moves arena
  uint32_t path_offset;   // bytes into the path arena (capacity q_len+t_len)
};

struct AlnDeviceArena {
  const uint8_t* seqs;      // packed query/target bytes
  const AlnDesc* descs;
  uint32_t* moves;          // 2-bit moves, 16 per dword
  uint8_t* path;            // per alignment: ops walked back from (n,m); 0=M,1=I,2=D
  uint32_t* path_len;       // per alignment
  int32_t* status;          // per alignment
  int32_t* edit_distance;   // per alignment (diagnostic)
  AlnLimits limits;
};

void launch_aligner_kernel(const AlnDeviceArena& arena, uint32_t num_alignments, void* stream);

}  // namespace rga::hip
