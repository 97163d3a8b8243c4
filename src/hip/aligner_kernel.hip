// HIP/CDNA4 banded pairwise aligner: one 64-lane wavefront per alignment.
//
// Anti-diagonal edit-distance DP over a band of B=1024 cells whose center
// follows the rectangle diagonal (i ~ d*n/(n+m)). Band cells live in
// register arrays (16 int32 per lane, strided k = r*64 + lane); the per-
// diagonal "shift" of the band center is a single cross-lane rotate pass.
// 2-bit moves are packed 16-per-dword (one dword per lane per diagonal,
// coalesced 256 B stores) and walked back on-device through an LDS-staged
// tile, emitting the reversed op string. Alignments whose optimal path
// leaves the band fail with kAlnBandEdge and fall back to the CPU aligner
// (reference contract: cudaaligner skip statuses -> edlib,
// src/cuda/cudaaligner.cpp:63-72).
#include <hip/hip_runtime.h>

#include "hip/aligner_types.hpp"

namespace rga::hip {

namespace {

constexpr int kLanes = 64;
constexpr int kRB = 16;  // band regs per lane: band = kRB * 64 = 1024
constexpr int32_t kInf = 1 << 28;
constexpr uint32_t kTraceTile = 16;  // diagonals staged in LDS per refill

// in-place rotate: arr[k] <- arr[k+1] (band slides down by one cell).
// Processing r ascending keeps arr[r+1] original when lane 63 borrows it.
template <int R>
__device__ inline void rotate_plus1(int32_t (&arr)[R], int lane) {
#pragma unroll
  for (int r = 0; r < R; ++r) {
    int32_t borrow = (r == R - 1) ? kInf : __shfl(arr[r + 1], 0, kLanes);
    int32_t dn = __shfl_down(arr[r], 1, kLanes);  // lane l gets lane l+1
    arr[r] = (lane == kLanes - 1) ? borrow : dn;
  }
}

__launch_bounds__(kLanes, 2)
__global__ void aligner_kernel(AlnDeviceArena a, uint32_t num_alignments) {
  const uint32_t idx = blockIdx.x;
  if (idx >= num_alignments) {
    return;
  }
  const int lane = threadIdx.x;
  const AlnDesc desc = a.descs[idx];
  const uint8_t* q = a.seqs + desc.q_offset;
  const uint8_t* t = a.seqs + desc.t_offset;
  const int32_t n = static_cast<int32_t>(desc.q_len);  // query rows (i)
  const int32_t m = static_cast<int32_t>(desc.t_len);  // target cols (j)
  const int32_t total = n + m;
  constexpr int32_t kBand = kRB * kLanes;
  constexpr int32_t kHalf = kBand / 2;

  uint32_t* moves = a.moves + desc.moves_offset;

  __shared__ uint32_t lds_tile[kTraceTile * kLanes];

  // band offset for diagonal d: first band cell's i-index
  auto off_of = [&](int32_t d) -> int32_t {
    int32_t center = total == 0 ? 0 : static_cast<int32_t>(
        (static_cast<int64_t>(d) * n) / total);
    return center - kHalf;
  };

  int32_t A1[kRB], A2[kRB];  // diagonals d-1 (aligned to off(d)) and d-2 (off(d)-1)
#pragma unroll
  for (int r = 0; r < kRB; ++r) {
    A1[r] = kInf;
    A2[r] = kInf;
  }

  int32_t off_prev = off_of(0);
  // d = 0 seed: cell (0,0) = 0 sits at k = -off(0) = kHalf
  {
    int32_t k0 = -off_prev;
#pragma unroll
    for (int r = 0; r < kRB; ++r) {
      int32_t k = r * kLanes + lane;
      if (k == k0) {
        A1[r] = 0;
      }
    }
    if (lane == 0) {
      moves[0 * kLanes] = 0xffffffffu;  // no moves on d=0
    }
  }

  int32_t final_score = kInf;

  // One diagonal step. Aprev = D(d-1, off+k), Aprev2 = D(d-2, off+k) on
  // entry; on exit Aprev2 holds D(d, off+k) (A0 overwrites the dead array;
  // the caller alternates the argument roles instead of copying 32 regs).
  auto step = [&](int32_t d, int32_t (&Aprev)[kRB], int32_t (&Aprev2)[kRB]) {
    const int32_t off = off_of(d);
    if (off != off_prev) {  // band center moved down by one
      rotate_plus1(Aprev, lane);
      rotate_plus1(Aprev2, lane);
    }
    uint32_t mv_word = 0;
    int32_t a1_hi_prev = kInf;  // lane-63 value of Aprev[r-1] (original)
    int32_t a2_hi_prev = kInf;
#pragma unroll
    for (int r = 0; r < kRB; ++r) {
      const int32_t a1 = Aprev[r];
      const int32_t a2 = Aprev2[r];
      const int32_t a1_hi = __shfl(a1, kLanes - 1, kLanes);
      const int32_t a2_hi = __shfl(a2, kLanes - 1, kLanes);
      int32_t a1m1 = __shfl_up(a1, 1, kLanes);  // D(d-1, off+k-1)
      int32_t a2m1 = __shfl_up(a2, 1, kLanes);  // D(d-2, off+k-1)
      if (lane == 0) {
        a1m1 = a1_hi_prev;
        a2m1 = a2_hi_prev;
      }
      a1_hi_prev = a1_hi;
      a2_hi_prev = a2_hi;

      const int32_t k = r * kLanes + lane;
      const int32_t i = off + k;
      const int32_t j = d - i;
      int32_t best = kInf;
      uint32_t mv = 3;
      if (i >= 0 && i <= n && j >= 0 && j <= m) {
        if (i == 0) {
          best = j;
          mv = 2;  // 'D' chain along the top boundary
        } else if (j == 0) {
          best = i;
          mv = 1;  // 'I' chain along the left boundary
        } else {
          const int32_t sub = (q[i - 1] != t[j - 1]) ? 1 : 0;
          best = a2m1 + sub;  // diagonal
          mv = 0;
          const int32_t ci = a1m1 + 1;  // consume query
          if (ci < best) {
            best = ci;
            mv = 1;
          }
          const int32_t cd = a1 + 1;  // consume target
          if (cd < best) {
            best = cd;
            mv = 2;
          }
          if (best >= kInf) {
            best = kInf;
            mv = 3;
          }
        }
      }
      Aprev2[r] = best;  // becomes D(d, off+k)
      mv_word |= mv << (2 * r);
      if (i == n && j == m) {
        final_score = best;
      }
    }
    moves[static_cast<size_t>(d) * kLanes + lane] = mv_word;
    off_prev = off;
  };

  for (int32_t d = 1; d + 1 <= total; d += 2) {
    step(d, A1, A2);      // A2 <- D(d)
    step(d + 1, A2, A1);  // A1 <- D(d+1)
  }
  if ((total & 1) == 1) {
    step(total, A1, A2);
  }

  // broadcast the final score (exactly one lane saw (n, m))
  {
    int32_t v = final_score;
#pragma unroll
    for (int s = 32; s > 0; s >>= 1) {
      v = min(v, __shfl_xor(v, s, kLanes));
    }
    final_score = v;
    if (lane == 0) {
      a.edit_distance[idx] = final_score;
    }
  }

  // ---- traceback: LDS-staged tile, lane 0 walks ----
  uint8_t* path = a.path + desc.path_offset;
  int32_t i = n, j = m;
  uint32_t plen = 0;
  int32_t status = (final_score >= kInf) ? kAlnBandEdge : kAlnOk;

  while (status == kAlnOk && (i != 0 || j != 0)) {
    const int64_t tile_hi = i + j;  // stage diagonals (tile_hi - kTraceTile, tile_hi]
    for (uint32_t dd = 0; dd < kTraceTile; ++dd) {
      const int64_t ds = tile_hi - dd;
      if (ds >= 0) {
        lds_tile[dd * kLanes + lane] = moves[static_cast<size_t>(ds) * kLanes + lane];
      }
    }
    __threadfence_block();

    if (lane == 0) {
      while (i != 0 || j != 0) {
        const int32_t dcur = i + j;
        const int32_t dd = tile_hi - dcur;
        if (dd >= static_cast<int32_t>(kTraceTile) || dd < 0) {
          break;  // refill
        }
        if (i == 0) {  // top boundary: all D
          path[plen++] = 2;
          --j;
          continue;
        }
        if (j == 0) {  // left boundary: all I
          path[plen++] = 1;
          --i;
          continue;
        }
        const int32_t k = i - off_of(dcur);
        if (k < 0 || k >= kBand) {
          status = kAlnBandEdge;
          break;
        }
        const uint32_t word = lds_tile[dd * kLanes + (k & (kLanes - 1))];
        const uint32_t mv = (word >> (2 * (k >> 6))) & 3u;
        if (mv == 3u) {
          status = kAlnBandEdge;
          break;
        }
        path[plen++] = static_cast<uint8_t>(mv);
        if (mv == 0) {
          --i;
          --j;
        } else if (mv == 1) {
          --i;
        } else {
          --j;
        }
      }
    }
    i = __shfl(i, 0, kLanes);
    j = __shfl(j, 0, kLanes);
    status = __shfl(status, 0, kLanes);
    plen = __shfl(plen, 0, kLanes);
  }

  if (lane == 0) {
    a.path_len[idx] = (status == kAlnOk) ? plen : 0;
    a.status[idx] = status;
  }
}

}  // namespace

void launch_aligner_kernel(const AlnDeviceArena& arena, uint32_t num_alignments, void* stream) {
  hipLaunchKernelGGL(aligner_kernel, dim3(num_alignments), dim3(kLanes), 0,
                     static_cast<hipStream_t>(stream), arena, num_alignments);
}

}  // namespace rga::hip
