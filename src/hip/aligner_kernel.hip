// HIP/CDNA4 banded Myers bit-vector aligner: one LANE per alignment.
//
// Each lane runs the blocked Myers edit-distance recurrence (Myers 1999,
// Hyyro 2003 block form) over a band of K 64-row blocks that slides along
// the rectangle diagonal — 64 DP cells per 64-bit VALU op, no cross-lane
// communication in the hot loop (the previous anti-diagonal design spent
// ~85 cycles per cell on ds_bpermute shuffles; this one spends ~0.5).
// Column state (Pv/Mv per block + block-bottom scores) is stored
// wave-coalesced; the traceback walks (n,m)->(0,0) reconstructing the three
// predecessor scores per step in O(1) via popcount over the stored vertical
// deltas. Band-edge escapes are detected during traceback and reported as
// kAlnBandEdge -> CPU pairwise fallback (reference contract:
// src/cuda/cudaaligner.cpp:63-72).
#include <hip/hip_runtime.h>

#include "hip/aligner_types.hpp"

namespace rga::hip {

namespace {

constexpr int kLanes = 64;

__device__ inline uint32_t base_code(uint8_t b) {
  // A=0 C=1 G=2 T=3, anything else 4 (matches nothing)
  switch (b) {
    case 'A': return 0;
    case 'C': return 1;
    case 'G': return 2;
    case 'T': return 3;
    default: return 4;
  }
}

// band top block for column j: clamp(center/64 - K/2, [0, nbt-K]).
// center = (j * n) / m computed as a fixed-point multiply with a
// precomputed 32.32 step — the integer division was microcoded and ran
// once per column in the forward pass and twice per traceback step.
template <int K>
__device__ inline int32_t btop_of(uint64_t j, uint64_t step, int32_t nbt) {
  int32_t center_blk = static_cast<int32_t>((j * step) >> 32) >> 6;
  int32_t top = center_blk - K / 2;
  int32_t hi = nbt > K ? nbt - K : 0;
  return top < 0 ? 0 : (top > hi ? hi : top);
}

// D(row, col) from stored column state: S = bottom score of block rb,
// k = (row-1) & 63 its delta bit. Subtracts the deltas of rows below it.
__device__ inline int32_t score_at(int32_t S, uint64_t Pv, uint64_t Mv, uint32_t k) {
  const uint64_t mask = (k == 63) ? 0ull : (~0ull << (k + 1));
  return S - (__popcll(Pv & mask) - __popcll(Mv & mask));
}

template <int K>
__launch_bounds__(kLanes)
__global__ void myers_kernel(AlnDeviceArena a, uint32_t num_align) {
  const uint32_t wave = blockIdx.x;
  const int lane = threadIdx.x;
  // lanes_per_wave < 64 under-fills waves on purpose: small jobs otherwise
  // give each CU ~1 wave and every load latency lands on the wall clock
  const uint32_t slot = wave * a.lanes_per_wave + lane;
  const bool active = lane < static_cast<int>(a.lanes_per_wave) && slot < num_align;

  const AlnWaveDesc wd = a.waves[wave];
  const uint32_t idx = active ? a.order[slot] : 0u;
  const AlnDesc desc = a.descs[idx];
  const int32_t n = active ? static_cast<int32_t>(desc.q_len) : 0;
  const int32_t m = active ? static_cast<int32_t>(desc.t_len) : 0;
  const uint8_t* q = a.seqs + desc.q_offset;
  const uint8_t* t = a.seqs + desc.t_offset;
  const int32_t nbt = (n + 63) >> 6;  // query blocks

  uint64_t* peq = a.peq + wd.peq_off;
  uint64_t* tb = a.tb + wd.tb_off;
  int32_t* sb = a.sbuf + wd.s_off;
  // 32.32 fixed-point slope for the band center (one division per lane)
  const uint64_t step = (m > 0) ? ((static_cast<uint64_t>(n) << 32) / static_cast<uint32_t>(m))
                                : 0;

  // ---- fill Peq (wave-coalesced layout [(b*4+c)*64+lane]) ----
  for (uint32_t b = 0; b < wd.nb; ++b) {
    uint64_t mask[4] = {0, 0, 0, 0};
    const int32_t base = static_cast<int32_t>(b) << 6;
    if (base < n) {
      const int32_t lim = min(64, n - base);
      for (int32_t k = 0; k < lim; ++k) {
        const uint32_t c = base_code(q[base + k]);
        if (c < 4) {
          mask[c] |= 1ull << k;
        }
      }
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      peq[(static_cast<uint64_t>(b) * 4 + c) * kLanes + lane] = mask[c];
    }
  }

  // ---- column 0 state: D(i,0) = i ----
  uint64_t Pv[K], Mv[K];
  int32_t S[K];
  int32_t btop = 0;
#pragma unroll
  for (int b = 0; b < K; ++b) {
    Pv[b] = ~0ull;
    Mv[b] = 0ull;
    S[b] = (b + 1) * 64;
  }
  {
#pragma unroll
    for (int b = 0; b < K; ++b) {
      const uint64_t o = (static_cast<uint64_t>(0) * K + b) * 2;
      tb[(o + 0) * kLanes + lane] = Pv[b];
      tb[(o + 1) * kLanes + lane] = Mv[b];
      sb[(static_cast<uint64_t>(0) * K + b) * kLanes + lane] = S[b];
    }
  }

  // ---- column loop (uniform bound; finished lanes coast) ----
  // Eq words for the next column are prefetched during the current column's
  // recurrence: the gather (64 lanes x scattered 8 B) is the only global
  // read in the loop and would otherwise sit on the critical path.
  uint64_t EqN[K];
  int32_t btopN = 0;
  if (m > 0) {
    const uint32_t c1 = base_code(t[0]);
    btopN = btop_of<K>(1, step, nbt);
#pragma unroll
    for (int b = 0; b < K; ++b) {
      EqN[b] =
          (c1 < 4) ? peq[(static_cast<uint64_t>(btopN + b) * 4 + c1) * kLanes + lane] : 0ull;
    }
  }
  for (int32_t j = 1; j <= static_cast<int32_t>(wd.mmax); ++j) {
    if (j <= m) {
      uint64_t EqC[K];
#pragma unroll
      for (int b = 0; b < K; ++b) {
        EqC[b] = EqN[b];
      }
      const int32_t btop_new = btopN;
      if (j < m) {  // issue next column's gather before the compute chain
        const uint32_t cn = base_code(t[j]);
        btopN = btop_of<K>(j + 1, step, nbt);
#pragma unroll
        for (int b = 0; b < K; ++b) {
          EqN[b] = (cn < 4)
                       ? peq[(static_cast<uint64_t>(btopN + b) * 4 + cn) * kLanes + lane]
                       : 0ull;
        }
      }
      while (btop < btop_new) {
        // band slides down one block: drop top, append pessimistic bottom
#pragma unroll
        for (int b = 0; b < K - 1; ++b) {
          Pv[b] = Pv[b + 1];
          Mv[b] = Mv[b + 1];
          S[b] = S[b + 1];
        }
        Pv[K - 1] = ~0ull;
        Mv[K - 1] = 0ull;
        S[K - 1] = S[K - 2] + 64;
        ++btop;
      }

      int32_t hin = 1;  // top boundary: D(top-1, j) - D(top-1, j-1) = +1
#pragma unroll
      for (int b = 0; b < K; ++b) {
        uint64_t Eq = EqC[b];
        const uint64_t hin_neg = (hin < 0) ? 1ull : 0ull;
        const uint64_t hin_pos = (hin > 0) ? 1ull : 0ull;
        const uint64_t Xv = Eq | Mv[b];
        Eq |= hin_neg;
        const uint64_t Xh = (((Eq & Pv[b]) + Pv[b]) ^ Pv[b]) | Eq;
        uint64_t Ph = Mv[b] | ~(Xh | Pv[b]);
        uint64_t Mh = Pv[b] & Xh;
        const int32_t hout =
            static_cast<int32_t>((Ph >> 63) & 1) - static_cast<int32_t>((Mh >> 63) & 1);
        Ph = (Ph << 1) | hin_pos;
        Mh = (Mh << 1) | hin_neg;
        Pv[b] = Mh | ~(Xv | Ph);
        Mv[b] = Ph & Xv;
        S[b] += hout;
        hin = hout;
      }

#pragma unroll
      for (int b = 0; b < K; ++b) {
        const uint64_t o = (static_cast<uint64_t>(j) * K + b) * 2;
        tb[(o + 0) * kLanes + lane] = Pv[b];
        tb[(o + 1) * kLanes + lane] = Mv[b];
        sb[(static_cast<uint64_t>(j) * K + b) * kLanes + lane] = S[b];
      }
    }
  }

  if (!active) {
    return;
  }

  // ---- final score at (n, m) ----
  int32_t st = kAlnOk;
  int32_t D = 0;
  {
    const int32_t rb = ((n - 1) >> 6) - btop;
    if (rb < 0 || rb >= K) {
      st = kAlnBandEdge;
    } else {
      D = score_at(S[rb], Pv[rb], Mv[rb], (n - 1) & 63);
    }
  }
  a.edit_distance[idx] = (st == kAlnOk) ? D : -1;

  // ---- traceback: O(1) per step via stored column states ----
  uint8_t* path = a.path + desc.path_offset;
  uint32_t plen = 0;
  int32_t i = n, j = m;
  while (st == kAlnOk && i > 0 && j > 0) {
    const int32_t babs = (i - 1) >> 6;
    const uint32_t k = (i - 1) & 63;
    const int32_t btj = btop_of<K>(j, step, nbt);
    const int32_t btj1 = btop_of<K>(j - 1, step, nbt);
    const int32_t rbj = babs - btj;
    const int32_t rbj1 = babs - btj1;
    if (rbj < 0 || rbj >= K || rbj1 < 0 || rbj1 >= K) {
      st = kAlnBandEdge;
      break;
    }
    const uint64_t oj = (static_cast<uint64_t>(j) * K + rbj) * 2;
    const uint64_t oj1 = (static_cast<uint64_t>(j - 1) * K + rbj1) * 2;
    const uint64_t Pvj = tb[(oj + 0) * kLanes + lane];
    const uint64_t Mvj = tb[(oj + 1) * kLanes + lane];
    const uint64_t Pvj1 = tb[(oj1 + 0) * kLanes + lane];
    const uint64_t Mvj1 = tb[(oj1 + 1) * kLanes + lane];
    const int32_t S1 = sb[(static_cast<uint64_t>(j - 1) * K + rbj1) * kLanes + lane];

    const int32_t vd = ((Pvj >> k) & 1) ? 1 : (((Mvj >> k) & 1) ? -1 : 0);
    const int32_t D_left = score_at(S1, Pvj1, Mvj1, k);
    const int32_t vd1 = ((Pvj1 >> k) & 1) ? 1 : (((Mvj1 >> k) & 1) ? -1 : 0);
    const int32_t D_diag = D_left - vd1;
    const int32_t sub = (base_code(q[i - 1]) != base_code(t[j - 1]) ||
                         base_code(q[i - 1]) >= 4)
                            ? 1
                            : 0;
    if (D_diag + sub == D) {
      path[plen++] = 0;  // M
      --i;
      --j;
      D = D_diag;
    } else if (D_left + 1 == D) {
      path[plen++] = 2;  // D (consume target)
      --j;
      D = D_left;
    } else if ((D - vd) + 1 == D) {
      path[plen++] = 1;  // I (consume query)
      --i;
      D = D - vd;
    } else {
      st = kAlnBandEdge;
      break;
    }
  }
  if (st == kAlnOk) {
    while (i > 0) {
      path[plen++] = 1;
      --i;
    }
    while (j > 0) {
      path[plen++] = 2;
      --j;
    }
  }
  a.path_len[idx] = (st == kAlnOk) ? plen : 0;
  a.status[idx] = st;
}

}  // namespace

void launch_aligner_kernel(const AlnDeviceArena& arena, uint32_t num_waves, uint32_t num_align,
                           uint32_t band_k, void* stream) {
  auto s = static_cast<hipStream_t>(stream);
  switch (band_k) {
    case 4:
      hipLaunchKernelGGL(myers_kernel<4>, dim3(num_waves), dim3(kLanes), 0, s, arena, num_align);
      break;
    case 8:
      hipLaunchKernelGGL(myers_kernel<8>, dim3(num_waves), dim3(kLanes), 0, s, arena, num_align);
      break;
    default:
      hipLaunchKernelGGL(myers_kernel<16>, dim3(num_waves), dim3(kLanes), 0, s, arena,
                         num_align);
      break;
  }
}

}  // namespace rga::hip
