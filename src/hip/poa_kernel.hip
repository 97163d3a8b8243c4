// HIP/CDNA4 POA mega-kernel v3: one 64-lane wavefront per window.
//
// Per window it loops over the layers: Needleman-Wunsch of the layer against
// the current POA graph (rows = graph nodes in topological order, columns =
// layer bases; the horizontal gap pass is a wave-wide max-scan with slope),
// move-byte traceback, threading the alignment into the graph, Kahn
// topological re-sort, and finally heaviest-bundle consensus with per-base
// coverage.
//
// Perf lineage. v1 streamed every DP value through HBM: 85% of wave-cycles
// parked on memory (measured, profiles/). v2 mirrored the graph arrays into
// 31 KB of LDS, which fixed the DP stalls but capped co-residency at 5
// blocks/CU — the still-global serial phases could no longer be hidden and
// wall time got worse. v3 keeps only the DP essentials in LDS (4-row ring +
// layer bases, 9 KB -> ~16 blocks/CU): predecessor rows are LDS hits unless
// the rank distance exceeds the ring (rare bubbles -> global matrix), the
// per-row scalar pointer-chase (sorted -> letter/in-count/first-edge ->
// pred rank) is replaced by one packed u64 `row_desc` per rank built
// LANE-PARALLEL after each topological sort, and the DP records a move byte
// per cell so the serial traceback is one byte load per step. The serial
// graph phases stay in global memory and are hidden by the deep window
// co-residency, as in v1.
//
// Semantics mirror the CPU engine (src/align/poa.cpp) so GPU results are
// deterministic and window-content-only (reference racon-gpu pins separate
// GPU goldens; so do we — topological order here is Kahn FIFO, not spoa DFS).
#include <hip/hip_runtime.h>

#include <cstdlib>

#include "hip/poa_types.hpp"

namespace rga::hip {

namespace {

constexpr int kLanes = 64;
constexpr int32_t kNegInf = -(1 << 28);
constexpr uint32_t kMaxW = 1024;  // LDS row width; matrix_width must fit
constexpr uint32_t kMaxN = 2048;  // LDS graph mirrors; max_nodes must fit
constexpr uint32_t kRing = 4;     // DP rows kept in LDS
constexpr uint32_t kMaxPre = 2;   // predecessor rows precomputed per row

// move byte encoding
constexpr uint8_t kMvDiag = 0;
constexpr uint8_t kMvUp = 1;
constexpr uint8_t kMvLeft = 2;  // (value 3 is reserved/invalid)

// Single-wavefront LDS visibility: waits only the LDS (lgkm) counter and
// stops compiler reordering. Unlike __syncthreads(), it does NOT drain the
// outstanding global (vm) stores of the row just written — those are
// fire-and-forget into HBM and must not sit on the per-row critical path.
__device__ inline void wave_lds_sync() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

// Inclusive max-scan over the 64-lane wavefront via DPP row ops (VALU-speed;
// the naive __shfl_up chain is 6 dependent ds_bpermute round-trips through
// the LDS unit and dominated the DP phase — measured with RGA_POA_TIMING).
// Sequence: shr1/2/4/8 within 16-lane rows, then bcast15 into rows 1&3 and
// bcast31 into rows 2&3 (the classic GCN prefix idiom with max).
__device__ inline int32_t wave_scan_max(int32_t v, int /*lane*/) {
  int32_t t;
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x111, 0xf, 0xf, false);  // row_shr:1
  v = max(v, t);
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x112, 0xf, 0xf, false);  // row_shr:2
  v = max(v, t);
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x114, 0xf, 0xf, false);  // row_shr:4
  v = max(v, t);
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x118, 0xf, 0xf, false);  // row_shr:8
  v = max(v, t);
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x142, 0xa, 0xf, false);  // row_bcast:15
  v = max(v, t);
  t = __builtin_amdgcn_update_dpp(kNegInf, v, 0x143, 0xc, 0xf, false);  // row_bcast:31
  v = max(v, t);
  return v;
}

// chunk range [klo, khi] of the static band for DP row R (1-based):
// center column follows the rank diagonal via a 16.16 slope; the band is
// rounded out to whole 64-column chunks (>= bw coverage). Full-width mode
// (bw == 0) is the caller's fast path and never calls this.
__device__ inline void band_chunks(uint32_t R, uint32_t slope16, uint32_t bw, uint32_t chunks,
                                   uint32_t* klo, uint32_t* khi) {
  const uint32_t center = (static_cast<uint64_t>(R) * slope16) >> 16;
  const uint32_t jlo = center > bw / 2 ? center - bw / 2 : 0;
  uint32_t lo = (jlo == 0) ? 0 : (jlo - 1) / 64;
  if (lo > chunks - 1) lo = chunks - 1;
  uint32_t hi = lo + bw / 64;
  if (hi > chunks - 1) hi = chunks - 1;
  *klo = lo;
  *khi = hi;
}

// packed row descriptor (one per topological rank); flags: bit0 = end node
// (no out-edges), bit1 = some successor is beyond the LDS ring, so this
// row's scores must go to global memory (matrix rows are otherwise
// write-only traffic nobody reads — the dominant HBM cost of the DP).
constexpr uint64_t kRdEnd = 1;
constexpr uint64_t kRdStore = 2;

__device__ inline uint64_t pack_rd(uint8_t letter, uint8_t nin, uint16_t node,
                                   uint16_t pred_row, uint8_t flags) {
  return static_cast<uint64_t>(letter) | (static_cast<uint64_t>(nin) << 8) |
         (static_cast<uint64_t>(node) << 16) | (static_cast<uint64_t>(pred_row) << 32) |
         (static_cast<uint64_t>(flags) << 48);
}

// LDS block state — cross-batch window co-residency is the dominant
// throughput lever (a 19 KB variant with full graph-array mirrors made each
// window ~10% faster but halved residency and lost 20% end to end), so only
// two things live here: the DP row ring, and — aliased into the same union
// because the phases never overlap — the Kahn scratch + FIFO queue, which
// kills the store-to-load round trip the serial sort otherwise does against
// HBM for every node. The queue doubles as the topological order read by
// the consensus phase. Everything else stays in the global slabs, hidden by
// co-residency.
//
// The ring width W is a template parameter: the kernel is instantiated per
// width bucket (384 / 576 / 1024 columns) so the ~500-column windows of a
// default w=500 run pay ~7 KB of LDS (23 blocks/CU, ~5.7 waves/SIMD)
// instead of the full-width 10.5 KB (15 blocks/CU, ~3.75/SIMD) — LDS, not
// VGPRs (60), is the occupancy limiter.
template <uint32_t W, uint32_t MN>
struct alignas(16) Shared {
  union {
    int16_t ring[kRing][W];  // DP rows (slot = row % kRing)
    struct {
      uint8_t work[MN];    // Kahn in-degree scratch (in-degree < 256)
      uint16_t queue[MN];  // Kahn FIFO == topological order
    } kahn;
  } u;
  // per-layer match bitvectors: bit l of match[c][k] says layer base
  // (1 + k*64 + l) equals code c — turns the per-cell seq comparison into
  // one register bit test instead of a byte load. One spare word so
  // cross-word extraction at the last chunk never reads out of bounds.
  uint64_t match[4][W / 64 + 1];
  uint64_t rd_block[64];  // row descriptors staged 64 at a time
};

__device__ inline int32_t poa_code(uint8_t b) {
  switch (b) {
    case 'A': return 0;
    case 'C': return 1;
    case 'G': return 2;
    case 'T': return 3;
    default: return -1;
  }
}

__device__ inline uint16_t out_edge_of(const struct WindowCtx& c, uint32_t node, uint32_t e);

struct WindowCtx {
  uint8_t* letters;
  uint8_t* in_cnt;
  uint8_t* out_cnt;
  uint8_t* ring_cnt;
  uint16_t* in_edges;
  int32_t* in_weights;
  uint16_t* out_edges;
  uint16_t* ring;
  uint16_t* nseq;
  uint16_t* rank;
  int64_t* hb_score;
  int32_t* hb_pred;
  int32_t* aln_nodes;
  int32_t* aln_seq;
  int16_t* matrix;
  uint8_t* moves;
  uint64_t* row_desc;

  const uint8_t* seq_base;
  const uint8_t* weight_base;
  const uint32_t* ends;   // layer end offsets (relative to window start)
  const uint32_t* spans;  // per layer: begin<<16|end backbone span, or
                          // 0xFFFFFFFF for window-spanning layers (full DP)
  uint32_t num_seqs;

  uint32_t ME;  // max edges
  uint32_t MR;  // max ring
  uint32_t MW;  // matrix width
  uint32_t MN;  // max nodes
  uint32_t bw;  // 0 = full width; else static band (columns) around diagonal
  int32_t m, x, g;

  uint32_t num_nodes;
  uint32_t seqs_in_graph;
  int32_t status;
};

// ---------- serial (lane 0) graph helpers ----------

__device__ inline uint16_t out_edge_of(const WindowCtx& c, uint32_t node, uint32_t e) {
  return c.out_edges[node * c.ME + e];
}

__device__ inline bool add_edge_d(WindowCtx& c, uint32_t a, uint32_t b, int32_t w) {
  uint32_t n_out = c.out_cnt[a];
  for (uint32_t e = 0; e < n_out; ++e) {
    if (out_edge_of(c, a, e) == b) {
      uint32_t n_in = c.in_cnt[b];
      for (uint32_t f = 0; f < n_in; ++f) {
        if (c.in_edges[b * c.ME + f] == a) {
          c.in_weights[b * c.ME + f] += w;
          return true;
        }
      }
      return true;  // unreachable for a consistent graph
    }
  }
  if (n_out >= c.ME || c.in_cnt[b] >= c.ME) {
    c.status = kPoaEdgeOverflow;
    return false;
  }
  c.out_edges[a * c.ME + n_out] = static_cast<uint16_t>(b);
  c.out_cnt[a] = static_cast<uint8_t>(n_out + 1);
  uint32_t n_in = c.in_cnt[b];
  c.in_edges[b * c.ME + n_in] = static_cast<uint16_t>(a);
  c.in_weights[b * c.ME + n_in] = w;
  c.in_cnt[b] = static_cast<uint8_t>(n_in + 1);
  return true;
}

__device__ inline int32_t add_node_d(WindowCtx& c, uint8_t letter) {
  if (c.num_nodes >= c.MN || c.num_nodes >= kMaxN) {
    c.status = kPoaNodeOverflow;
    return -1;
  }
  uint32_t id = c.num_nodes++;
  c.letters[id] = letter;
  c.in_cnt[id] = 0;
  c.out_cnt[id] = 0;
  c.ring_cnt[id] = 0;
  c.nseq[id] = 0;
  return static_cast<int32_t>(id);
}

// Threads the traceback path (stored reversed in aln_*) into the graph.
// Mirrors Graph::add_alignment (src/align/poa.cpp).
__device__ void add_alignment_d(WindowCtx& c, const uint8_t* seq,
                                const uint8_t* wts, uint32_t len, int32_t aln_len) {
  // first/last aligned sequence positions
  int32_t first_pos = -1, last_pos = -1;
  for (int32_t k = aln_len - 1; k >= 0; --k) {  // reversed storage -> forward walk
    if (c.aln_seq[k] != -1) {
      if (first_pos == -1) {
        first_pos = c.aln_seq[k];
      }
      last_pos = c.aln_seq[k];
    }
  }

  int32_t head = -1;
  int32_t prev_weight = 0;
  int32_t last_counted = -1;

  auto link = [&](int32_t a, int32_t b, int32_t w) {
    if (!add_edge_d(c, a, b, w)) {
      return;
    }
    if (a != last_counted) {
      ++c.nseq[a];
    }
    ++c.nseq[b];
    last_counted = b;
  };

  if (first_pos == -1) {
    // fully unaligned layer: append as a fresh chain
    first_pos = len;  // head chain covers the whole sequence below
  }

  // head chain: seq[0 .. first_pos)
  for (int32_t p = 0; p < first_pos; ++p) {
    int32_t id = add_node_d(c, seq[p]);
    if (id < 0) return;
    if (head != -1) {
      link(head, id, prev_weight + wts[p]);
    }
    head = id;
    prev_weight = wts[p];
  }

  // aligned middle
  for (int32_t k = aln_len - 1; k >= 0; --k) {
    int32_t spos = c.aln_seq[k];
    if (spos == -1) {
      continue;
    }
    uint8_t letter = seq[spos];
    int32_t node = c.aln_nodes[k];
    int32_t new_id;
    if (node == -1) {
      new_id = add_node_d(c, letter);
      if (new_id < 0) return;
    } else if (c.letters[node] == letter) {
      new_id = node;
    } else {
      new_id = -1;
      uint32_t nr = c.ring_cnt[node];
      for (uint32_t r = 0; r < nr; ++r) {
        uint16_t aid = c.ring[node * c.MR + r];
        if (c.letters[aid] == letter) {
          new_id = aid;
          break;
        }
      }
      if (new_id == -1) {
        new_id = add_node_d(c, letter);
        if (new_id < 0) return;
        // join the ring: new node linked with node and all its partners
        if (nr >= c.MR) {
          c.status = kPoaRingOverflow;
          return;
        }
        for (uint32_t r = 0; r < nr; ++r) {
          uint16_t aid = c.ring[node * c.MR + r];
          c.ring[new_id * c.MR + r] = aid;
          uint32_t arc = c.ring_cnt[aid];
          if (arc >= c.MR) {
            c.status = kPoaRingOverflow;
            return;
          }
          c.ring[aid * c.MR + arc] = static_cast<uint16_t>(new_id);
          c.ring_cnt[aid] = static_cast<uint8_t>(arc + 1);
        }
        c.ring[new_id * c.MR + nr] = static_cast<uint16_t>(node);
        c.ring_cnt[new_id] = static_cast<uint8_t>(nr + 1);
        uint32_t nrc = c.ring_cnt[node];
        c.ring[node * c.MR + nrc] = static_cast<uint16_t>(new_id);
        c.ring_cnt[node] = static_cast<uint8_t>(nrc + 1);
      }
    }
    if (head != -1) {
      link(head, new_id, prev_weight + wts[spos]);
    }
    head = new_id;
    prev_weight = wts[spos];
  }

  // tail chain: seq[last_pos+1 .. len)
  for (int32_t p = (last_pos == -1 ? len : last_pos + 1); p < static_cast<int32_t>(len); ++p) {
    int32_t id = add_node_d(c, seq[p]);
    if (id < 0) return;
    if (head != -1) {
      link(head, id, prev_weight + wts[p]);
    }
    head = id;
    prev_weight = wts[p];
  }

  ++c.seqs_in_graph;
}

// Kahn topological sort (FIFO, deterministic) over the LDS mirrors. The
// FIFO queue IS the final topological order (s.u.kahn.queue); rank goes to
// the global slab (read lane-parallel by build_row_desc).
// Cooperative: the in-degree staging and the rank writeback are
// lane-parallel; only the FIFO core (whose order is the deterministic
// topological order) stays on lane 0. Call from ALL lanes.
template <class SH>
__device__ void topo_sort_d(WindowCtx& c, SH& s, int lane) {
  const uint32_t n = c.num_nodes;
  for (uint32_t i = lane; i < n; i += kLanes) {
    s.u.kahn.work[i] = c.in_cnt[i];
  }
  __syncthreads();
  if (lane == 0) {
    uint32_t qhead = 0, qtail = 0;
    for (uint32_t i = 0; i < n; ++i) {
      if (s.u.kahn.work[i] == 0) {
        s.u.kahn.queue[qtail++] = static_cast<uint16_t>(i);
      }
    }
    while (qhead < qtail) {
      uint16_t u = s.u.kahn.queue[qhead++];
      uint32_t nout = c.out_cnt[u];
      for (uint32_t e = 0; e < nout; ++e) {
        uint16_t v = out_edge_of(c, u, e);
        if (--s.u.kahn.work[v] == 0) {
          s.u.kahn.queue[qtail++] = v;
        }
      }
    }
  }
  __syncthreads();
  for (uint32_t r = lane; r < n; r += kLanes) {
    c.rank[s.u.kahn.queue[r]] = static_cast<uint16_t>(r);
  }
  __syncthreads();  // rank stores -> visible to the cross-lane readers
                    // (build_row_desc, the DP's pred-rank loads)
}

// Heaviest-bundle consensus (mirrors Graph::traverse_heaviest_bundle).
// Returns consensus length written into out/cov (forward order), or -1.
template <class SH>
__device__ int32_t consensus_d(WindowCtx& c, SH& s, uint8_t* out, uint16_t* cov,
                               uint32_t max_out) {
  uint32_t n = c.num_nodes;
  for (uint32_t i = 0; i < n; ++i) {
    c.hb_score[i] = -1;
    c.hb_pred[i] = -1;
  }

  uint32_t max_id = s.u.kahn.queue[0];
  for (uint32_t r = 0; r < n; ++r) {
    uint16_t nid = s.u.kahn.queue[r];
    uint32_t nin = c.in_cnt[nid];
    for (uint32_t e = 0; e < nin; ++e) {
      uint16_t p = c.in_edges[nid * c.ME + e];
      int64_t w = c.in_weights[nid * c.ME + e];
      if (c.hb_score[nid] < w ||
          (c.hb_score[nid] == w && c.hb_pred[nid] != -1 &&
           c.hb_score[c.hb_pred[nid]] <= c.hb_score[p])) {
        c.hb_score[nid] = w;
        c.hb_pred[nid] = p;
      }
    }
    if (c.hb_pred[nid] != -1) {
      c.hb_score[nid] += c.hb_score[c.hb_pred[nid]];
    }
    if (c.hb_score[max_id] < c.hb_score[nid]) {
      max_id = nid;
    }
  }

  // branch completion until we end on a sink (bounded: the restart rank is
  // strictly increasing, so at most n rounds; all-zero-weight graphs would
  // otherwise spin forever — those windows fail over to the CPU instead)
  uint32_t guard = 0;
  uint32_t prev_rank = 0;
  while (c.out_cnt[max_id] != 0) {
    if (++guard > n || (guard > 1 && c.rank[max_id] <= prev_rank)) {
      c.status = kPoaConsensusOverflow;
      return -1;
    }
    prev_rank = c.rank[max_id];
    uint32_t rank0 = c.rank[max_id];
    // invalidate alternative branches
    uint32_t nout = c.out_cnt[max_id];
    for (uint32_t e = 0; e < nout; ++e) {
      uint16_t endn = out_edge_of(c, max_id, e);
      uint32_t nin = c.in_cnt[endn];
      for (uint32_t f = 0; f < nin; ++f) {
        uint16_t o = c.in_edges[endn * c.ME + f];
        if (o != max_id) {
          c.hb_score[o] = -1;
        }
      }
    }
    int64_t best = 0;
    uint32_t best_id = 0;
    for (uint32_t r = rank0 + 1; r < n; ++r) {
      uint16_t nid = s.u.kahn.queue[r];
      c.hb_score[nid] = -1;
      c.hb_pred[nid] = -1;
      uint32_t nin = c.in_cnt[nid];
      for (uint32_t e = 0; e < nin; ++e) {
        uint16_t p = c.in_edges[nid * c.ME + e];
        if (c.hb_score[p] == -1) {
          continue;
        }
        int64_t w = c.in_weights[nid * c.ME + e];
        if (c.hb_score[nid] < w ||
            (c.hb_score[nid] == w && c.hb_pred[nid] != -1 &&
             c.hb_score[c.hb_pred[nid]] <= c.hb_score[p])) {
          c.hb_score[nid] = w;
          c.hb_pred[nid] = p;
        }
      }
      if (c.hb_pred[nid] != -1) {
        c.hb_score[nid] += c.hb_score[c.hb_pred[nid]];
      }
      if (best < c.hb_score[nid]) {
        best = c.hb_score[nid];
        best_id = nid;
      }
    }
    max_id = best_id;
  }

  // backtrack: path length first
  int32_t path_len = 0;
  int32_t idw = static_cast<int32_t>(max_id);
  while (idw != -1) {
    ++path_len;
    idw = c.hb_pred[idw];
  }
  if (path_len > static_cast<int32_t>(max_out)) {
    c.status = kPoaConsensusOverflow;
    return -1;
  }
  idw = static_cast<int32_t>(max_id);
  for (int32_t k = path_len - 1; k >= 0; --k) {
    uint32_t node = static_cast<uint32_t>(idw);
    out[k] = c.letters[node];
    uint32_t covv = c.nseq[node];
    uint32_t nr = c.ring_cnt[node];
    for (uint32_t r = 0; r < nr; ++r) {
      covv += c.nseq[c.ring[node * c.MR + r]];
    }
    cov[k] = static_cast<uint16_t>(min(covv, 65535u));
    idw = c.hb_pred[idw];
  }
  return path_len;
}

// Lane-parallel: rebuild the packed per-rank row descriptors from the graph
// arrays (called after backbone init and after every topo sort).
__device__ void build_row_desc(WindowCtx& c, int lane) {
  const uint32_t n = c.num_nodes;
  for (uint32_t node = lane; node < n; node += kLanes) {
    const uint32_t r = c.rank[node];
    const uint8_t nin = c.in_cnt[node];
    uint16_t pred_row = 0;
    if (nin > 0) {
      pred_row = static_cast<uint16_t>(c.rank[c.in_edges[node * c.ME]] + 1);
    }
    const uint32_t nout = c.out_cnt[node];
    uint8_t flags = (nout == 0) ? kRdEnd : 0;
    for (uint32_t e = 0; e < nout; ++e) {
      const uint32_t sr = c.rank[out_edge_of(c, node, e)];
      if (sr - r >= kRing) {
        flags |= kRdStore;
        break;
      }
    }
    c.row_desc[r] = pack_rd(c.letters[node], nin, static_cast<uint16_t>(node), pred_row, flags);
  }
}

// ---------- the mega-kernel ----------

// MINWAVES is the occupancy the compiler must budget registers for
// (waves/SIMD floor): the 8-wide variants otherwise help themselves to the
// full 128-VGPR budget of 4 waves/SIMD and registers — not LDS — become
// the residency limiter.
template <bool TIMED, uint32_t WB, uint32_t MAXW, uint32_t MAXN = kMaxN,
          uint32_t MINWAVES = 4>
__launch_bounds__(kLanes, MINWAVES)
__global__ void poa_window_kernel(PoaDeviceArena a, uint32_t window_base,
                                  uint32_t num_windows) {
  static_assert(MAXW <= kMaxW, "ring width exceeds the slab matrix width");
  static_assert(MAXN <= kMaxN, "node cap exceeds the slab graph capacity");
  if (blockIdx.x >= num_windows) {
    return;
  }
  const uint32_t win = window_base + blockIdx.x;
  const int lane = threadIdx.x;
  const PoaWindowDesc desc = a.windows[win];
  // the capacity model is fixed (PoaLimits defaults; PoaBatch never alters
  // it) — compile-time limits keep slab addressing in immediates instead of
  // ~100 live SGPRs that were spilling into v_readlane/writelane traffic in
  // the DP row loop
  constexpr PoaLimits L{};
  const uint32_t slab = desc.scratch_idx;

  __shared__ Shared<MAXW, MAXN> s;

  WindowCtx c;
  c.letters = a.letters + static_cast<size_t>(slab) * L.max_nodes;
  c.in_cnt = a.in_cnt + static_cast<size_t>(slab) * L.max_nodes;
  c.out_cnt = a.out_cnt + static_cast<size_t>(slab) * L.max_nodes;
  c.ring_cnt = a.ring_cnt + static_cast<size_t>(slab) * L.max_nodes;
  c.in_edges = a.in_edges + static_cast<size_t>(slab) * L.max_nodes * L.max_edges;
  c.in_weights = a.in_weights + static_cast<size_t>(slab) * L.max_nodes * L.max_edges;
  c.out_edges = a.out_edges + static_cast<size_t>(slab) * L.max_nodes * L.max_edges;
  c.ring = a.ring + static_cast<size_t>(slab) * L.max_nodes * L.max_ring;
  c.nseq = a.nseq + static_cast<size_t>(slab) * L.max_nodes;
  c.rank = a.rank + static_cast<size_t>(slab) * L.max_nodes;
  c.hb_score = a.hb_score + static_cast<size_t>(slab) * L.max_nodes;
  c.hb_pred = a.hb_pred + static_cast<size_t>(slab) * L.max_nodes;
  c.aln_nodes = a.aln_nodes + static_cast<size_t>(slab) * (2 * L.matrix_width + L.max_nodes);
  c.aln_seq = a.aln_seq + static_cast<size_t>(slab) * (2 * L.matrix_width + L.max_nodes);
  c.matrix = a.matrix + static_cast<size_t>(slab) * (L.max_nodes + 1) * L.matrix_width;
  c.moves = a.moves + static_cast<size_t>(slab) * (L.max_nodes + 1) * L.matrix_width;
  c.row_desc = a.row_desc + static_cast<size_t>(slab) * L.max_nodes;

  unsigned long long* timing = a.timing + static_cast<size_t>(win) * 8;
  unsigned long long tick = TIMED ? wall_clock64() : 0;
  const unsigned long long t_start = tick;
  unsigned long long t_dp = 0, t_tb = 0, t_add = 0, t_topo = 0, t_rd = 0, t_cons = 0;
  unsigned long long layers_done = 0;
  auto lap = [&]() -> unsigned long long {
    if (!TIMED) {
      return 0;
    }
    unsigned long long now = wall_clock64();
    unsigned long long d = now - tick;
    tick = now;
    return d;
  };

  c.seq_base = a.seq_data + desc.seq_offset;
  c.weight_base = a.weight_data + desc.seq_offset;
  c.ends = a.layer_ends + a.layer_ends_index[win];
  c.spans = a.layer_spans + a.layer_ends_index[win];
  c.num_seqs = desc.num_seqs;
  c.ME = L.max_edges;
  c.MR = L.max_ring;
  c.MW = L.matrix_width;
  c.MN = min(L.max_nodes, MAXN - 1);  // variant Kahn capacity (see Shared)
  c.bw = a.band_width;
  c.m = a.match;
  c.x = a.mismatch;
  c.g = a.gap;
  c.status = (L.matrix_width <= kMaxW && L.max_nodes <= kMaxN) ? kPoaOk : kPoaNodeOverflow;

  // ---- init graph from the backbone (layer 0), lane-parallel ----
  const uint32_t bb_len = c.ends[0];
  const uint8_t* bb_seq = c.seq_base;
  const uint8_t* bb_wts = c.weight_base;
  for (uint32_t i = lane; i < bb_len; i += kLanes) {
    c.letters[i] = bb_seq[i];
    c.ring_cnt[i] = 0;
    c.nseq[i] = bb_len >= 2 ? 1 : 0;
    // queue holds the trivial order for the no-aligned-layer case (it is
    // clobbered by the DP ring and rebuilt by every topo sort afterwards)
    s.u.kahn.queue[i] = static_cast<uint16_t>(i);
    c.rank[i] = static_cast<uint16_t>(i);
    uint8_t nin = 0;
    if (i == 0) {
      c.in_cnt[i] = 0;
    } else {
      nin = 1;
      c.in_cnt[i] = 1;
      c.in_edges[i * c.ME] = static_cast<uint16_t>(i - 1);
      c.in_weights[i * c.ME] = static_cast<int32_t>(bb_wts[i - 1]) + bb_wts[i];
    }
    uint8_t flags = 0;
    if (i + 1 < bb_len) {
      c.out_cnt[i] = 1;
      c.out_edges[i * c.ME] = static_cast<uint16_t>(i + 1);
    } else {
      c.out_cnt[i] = 0;
      flags = kRdEnd;
    }
    c.row_desc[i] = pack_rd(bb_seq[i], nin, static_cast<uint16_t>(i),
                            static_cast<uint16_t>(i), flags);
  }
  c.num_nodes = bb_len;
  c.seqs_in_graph = 1;
  __syncthreads();  // graph writes -> visible to all lanes

  // ---- per-layer loop ----
  for (uint32_t layer = 1; layer < c.num_seqs && c.status == kPoaOk; ++layer) {
    const uint32_t beg = c.ends[layer - 1];
    const uint32_t len = c.ends[layer] - beg;
    const uint8_t* seq = c.seq_base + beg;
    const uint8_t* wts = c.weight_base + beg;
    if (len == 0) {
      continue;  // host should have filtered; skip defensively
    }
    if (len + 1 > c.MW || len + 1 > MAXW) {
      if (len + 1 > c.MW) {
        continue;  // over the slab matrix: host drops these (defensive)
      }
      c.status = kPoaWidthOverflow;  // mis-bucketed: fail to the CPU path
      break;
    }

    // build the per-layer match bitvectors straight from the (L2-resident)
    // global layer bytes, lane-parallel over chunk words; one spare zero
    // word so cross-word extraction never reads out of bounds
    {
      const uint32_t words = (len + kLanes - 1) / kLanes + 1;
      for (uint32_t w = lane; w < words * 4; w += kLanes) {
        const uint32_t k = w >> 2, cc = w & 3;
        uint64_t bits = 0;
        const uint32_t base = k * kLanes;
        if (base < len) {
          const uint32_t lim = min(kLanes, len - base);
          for (uint32_t l = 0; l < lim; ++l) {
            if (poa_code(seq[base + l]) == static_cast<int32_t>(cc)) {
              bits |= 1ull << l;
            }
          }
        }
        s.match[cc][k] = bits;
      }
    }
    __syncthreads();

    const uint32_t n = c.num_nodes;
    const uint32_t chunks = (len + kLanes - 1) / kLanes;

    // SUBGRAPH restriction (CPU parity, Window::generate_consensus): a
    // layer that does not span the window aligns only against the rank
    // window [rank(backbone[begin]), rank(backbone[end])]. Rows outside are
    // skipped, predecessor edges from below rank(begin) are ignored (rows
    // losing every predecessor become NW start rows), and the alignment
    // target is the span's end rank (plus true sinks inside the window).
    // The reference's cudapoa aligns every layer to the full graph instead,
    // which is what degrades its long-window GPU goldens (racon_test.cpp
    // pins 4168 vs CPU 1289 at w=1000); this engine keeps the CPU
    // windowing semantics on device.
    const uint32_t span_word = c.spans[layer];
    const bool sub = span_word != 0xFFFFFFFFu;
    uint32_t rlo = 0, rhi = n - 1;
    if (sub) {
      rlo = c.rank[span_word >> 16];
      rhi = c.rank[span_word & 0xffffu];
    }

    // banded mode (-b): static band around the rank diagonal of the
    // (possibly rank-restricted) row window
    const bool banded = (c.bw != 0) && (c.bw < len);
    const uint32_t slope16 =
        banded ? static_cast<uint32_t>((static_cast<uint64_t>(len) << 16) / (rhi - rlo + 1))
               : 0;

    int32_t best_score = kNegInf;
    uint32_t best_row = 0;
    (void)lap();

    // row 0 (all-gap) is arithmetic: H0[j] = j * g — never materialized.
    // Row descriptors are staged 64 at a time through LDS with one
    // coalesced load: a per-row dependent global read (~600 ns) was the
    // dominant per-row latency and neither scan nor store restructuring
    // moved it (measured via RGA_POA_TIMING).
    for (uint32_t rblk = rlo & ~(kLanes - 1); rblk <= rhi; rblk += kLanes) {
      if (rblk + lane < n) {
        s.rd_block[lane] = c.row_desc[rblk + lane];
      }
      wave_lds_sync();
      const uint32_t rlim = min(rhi + 1, rblk + kLanes);
    for (uint32_t r = max(rblk, rlo); r < rlim; ++r) {
      const uint64_t rd = s.rd_block[r - rblk];
      const uint8_t letter = static_cast<uint8_t>(rd);
      const int32_t letter_code = poa_code(letter);
      const uint32_t nin = static_cast<uint32_t>((rd >> 8) & 0xff);
      const uint32_t node = static_cast<uint32_t>((rd >> 16) & 0xffff);
      const uint32_t pred0 = static_cast<uint32_t>((rd >> 32) & 0xffff);
      const uint64_t flags = (rd >> 48) & 0xff;
      const bool is_end = (flags & kRdEnd) != 0;
      const bool store_row = (flags & kRdStore) != 0;
      int16_t* Hrow = c.matrix + static_cast<size_t>(r + 1) * c.MW;
      uint8_t* Mrow = c.moves + static_cast<size_t>(r + 1) * c.MW;
      int16_t* ring_row = s.u.ring[(r + 1) % kRing];

      // predecessor rows (e < kMaxPre precomputed; beyond that re-derived
      // in the chunk loop — nodes with >8 in-edges are rare)
      uint32_t pred_rows[kMaxPre];
      pred_rows[0] = pred0;
      const uint32_t npre = min(nin, kMaxPre);
      for (uint32_t e = 1; e < npre; ++e) {
        pred_rows[e] = c.rank[c.in_edges[node * c.ME + e]] + 1;
      }

      // usable predecessor edges: in subgraph mode edges from ranks below
      // the window are cut; a row losing every predecessor becomes an NW
      // start row (virtual row 0), matching the CPU subgraph's sources.
      // Edge index 63 in a move byte encodes that virtual start for the
      // traceback (real indices stop at max_edges-1 = 47).
      uint64_t usable = (1ull << nin) - 1;  // nin <= 48 < 64
      if (sub && nin != 0) {
        usable = 0;
        for (uint32_t e = 0; e < nin; ++e) {
          const uint32_t p = (e < kMaxPre) ? pred_rows[e]
                                           : c.rank[c.in_edges[node * c.ME + e]] + 1;
          if (p > rlo) {
            usable |= 1ull << e;
          }
        }
      }
      const bool vstart = usable == 0;  // nin == 0, or every pred cut

      uint32_t row_klo = 0, row_khi = chunks - 1;
      if (banded) {
        band_chunks(r + 1 - rlo, slope16, c.bw, chunks, &row_klo, &row_khi);
      }

      // fetch a predecessor row value: LDS ring if close, global otherwise;
      // in banded mode, columns outside the predecessor's band are -inf
      auto pred_val = [&](uint32_t p, uint32_t col) -> int32_t {
        if (p == 0) {
          return static_cast<int32_t>(col) * c.g;  // arithmetic row 0
        }
        if (banded && col != 0) {
          uint32_t pklo, pkhi;
          band_chunks(p - rlo, slope16, c.bw, chunks, &pklo, &pkhi);
          const uint32_t kc = (col - 1) / 64;
          if (kc < pklo || kc > pkhi) {
            return kNegInf;
          }
        } else if (banded && col == 0) {
          uint32_t pklo, pkhi;
          band_chunks(p - rlo, slope16, c.bw, chunks, &pklo, &pkhi);
          if (pklo != 0) {
            return kNegInf;
          }
        }
        if (r + 1 - p < kRing) {
          return s.u.ring[p % kRing][col];
        }
        return c.matrix[static_cast<size_t>(p) * c.MW + col];
      };

      // first column (j = 0): max over preds of Hp[0] + gap
      // (banded: only when this row's band includes column 0)
      int32_t h0 = kNegInf;
      uint32_t e0 = 0;
      if (row_klo == 0) {
        int32_t best0 = kNegInf;
        if (vstart) {
          best0 = 0;
          e0 = 63;  // virtual-start sentinel (see `usable`)
        } else {
          for (uint32_t e = 0; e < nin; ++e) {
            if (!((usable >> e) & 1)) {
              continue;
            }
            const uint32_t p = (e < kMaxPre) ? pred_rows[e]
                                             : c.rank[c.in_edges[node * c.ME + e]] + 1;
            const int32_t hp0 = pred_val(p, 0);
            if (hp0 > best0) {
              best0 = hp0;
              e0 = e;
            }
          }
        }
        h0 = best0 + c.g;
        if (lane == 0) {
          if (store_row) {
            Hrow[0] = static_cast<int16_t>(h0);
          }
          ring_row[0] = static_cast<int16_t>(h0);
          // column 0 is always a vertical chain through the argmax edge
          // (moves layout is shifted: col j lives at j-1, col 0 at MW-1,
          // so each lane's 8 move bytes form one aligned u64 store)
          Mrow[c.MW - 1] = static_cast<uint8_t>(kMvUp | (e0 << 2));
        }
      }

      // ---- lane-blocked columns ----
      // Lane l owns WB contiguous columns per pass; one register-local
      // inclusive scan + one DPP wave scan per pass replaces the previous
      // chunk-serial carry chain (8 dependent LDS+scan segments per row).
      const uint32_t j0 = banded ? row_klo * kLanes : 0;
      const uint32_t jend = banded ? min(len, (row_khi + 1) * kLanes) : len;
      int32_t carry_u = h0;  // u-space max through column j0 (h0 = -inf when
                             // the band excludes column 0)
      int32_t last_col_val = kNegInf;
      int32_t pass_tail = 0;  // lane 63's last clamped value of the previous
                              // pass (the shifted store's column `base`)

      for (uint32_t base = j0; base < jend; base += kLanes * WB) {
        const uint32_t cbase = base + lane * WB;  // own cols: cbase+1..cbase+WB
        const uint32_t nown =
            (cbase < jend) ? min(WB, jend - cbase) : 0;

        // substitution matches for own columns: bits w of mbits
        uint64_t mbits = 0;
        if (letter_code >= 0 && nown > 0) {
          const uint32_t bit0 = cbase & 63;
          const uint32_t w0 = cbase >> 6;
          mbits = s.match[letter_code][w0] >> bit0;
          if (bit0 != 0) {
            mbits |= s.match[letter_code][w0 + 1] << (64 - bit0);
          }
        } else if (nown > 0) {
          for (uint32_t w = 0; w < nown; ++w) {
            if (seq[cbase + w] == letter) {  // rare: non-ACGT row letter
              mbits |= 1ull << w;
            }
          }
        }

        int32_t bd[WB], bu[WB];  // best diagonal / vertical candidates
#pragma unroll
        for (uint32_t w = 0; w < WB; ++w) {
          bd[w] = kNegInf;
          bu[w] = kNegInf;
        }

        if (vstart) {
#pragma unroll
          for (uint32_t w = 0; w < WB; ++w) {
            if (w < nown) {
              const int32_t j = static_cast<int32_t>(cbase + 1 + w);
              const int32_t subst = ((mbits >> w) & 1) ? c.m : c.x;
              bd[w] = (j - 1) * c.g + subst;
              bu[w] = j * c.g + c.g;
            }
          }
        } else {
          for (uint32_t e = 0; e < nin; ++e) {
            if (!((usable >> e) & 1)) {
              continue;
            }
            const uint32_t p = (e < kMaxPre) ? pred_rows[e]
                                             : c.rank[c.in_edges[node * c.ME + e]] + 1;
            // gather pred row values pv[w] = H(p, cbase + w), w in 0..WB.
            // Loads are UNCONDITIONAL with a clamped index + VALU select:
            // per-element predication compiled to one exec-branched
            // flat_load each (generic pointer), serializing the row. The
            // LDS and global paths are separate loops so each keeps its
            // address space (ds_read vs global_load).
            int32_t pv[WB + 1];
            if (p == 0) {
#pragma unroll
              for (uint32_t w = 0; w <= WB; ++w) {
                pv[w] = static_cast<int32_t>(cbase + w) * c.g;
              }
            } else {
              uint32_t plo = 0, phi = len;  // valid column range of pred row
              bool p_has0 = true;
              if (banded) {
                uint32_t pklo, pkhi;
                band_chunks(p - rlo, slope16, c.bw, chunks, &pklo, &pkhi);
                plo = pklo * kLanes;  // valid cols: {0 if pklo==0} + [plo+1..phi]
                phi = min(len, (pkhi + 1) * kLanes);
                p_has0 = (pklo == 0);
              }
              // validity of the contiguous range [plo+1, phi] (plus col 0
              // when the pred row computed it) hoisted into ONE bitmask per
              // edge — per-element compare+select chains were ~10% of the
              // kernel's VALU issue
              const uint32_t lo_col = plo + 1;
              const uint32_t lo_w = lo_col > cbase ? min(lo_col - cbase, WB + 1) : 0;
              const uint32_t hi_w = (phi + 1 > cbase) ? min(phi + 1 - cbase, WB + 1) : 0;
              uint32_t okmask = (hi_w > lo_w) ? ((1u << hi_w) - (1u << lo_w)) : 0u;
              if (cbase == 0 && p_has0) {
                okmask |= 1u;  // col 0 sits outside [plo+1, phi] by construction
              }
              const uint32_t colmax = c.MW - 1;
              constexpr uint32_t kRingMax = MAXW - 1;  // LDS ring row bound
              // unpack a 16-byte vector load into pv[0..7] (cbase*2 is a
              // multiple of 16 when WB == 8, and both the ring rows and the
              // global matrix rows are 16-byte aligned); replaces 8 scalar
              // b16 loads — the LDS pipe is the congested unit at 20+
              // co-resident windows per CU (profiles/PARKED.md)
              auto unpack8 = [&pv](int4 v) {
                const int32_t ws[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
                for (uint32_t q = 0; q < 4; ++q) {
                  pv[2 * q] = static_cast<int32_t>(static_cast<int16_t>(ws[q] & 0xffff));
                  pv[2 * q + 1] = ws[q] >> 16;  // arithmetic: sign-extended
                }
              };
              if (r + 1 - p < kRing) {
                const uint32_t slot = p % kRing;
                if (WB == 8 && cbase + WB <= kRingMax) {
                  unpack8(*reinterpret_cast<const int4*>(&s.u.ring[slot][cbase]));
                  pv[WB] = s.u.ring[slot][cbase + WB];
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    pv[w] = ((okmask >> w) & 1u) ? pv[w] : kNegInf;
                  }
                } else if (cbase + WB <= kRingMax) {
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    const int32_t val = s.u.ring[slot][cbase + w];
                    pv[w] = ((okmask >> w) & 1u) ? val : kNegInf;
                  }
                } else {
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    const int32_t val = s.u.ring[slot][min(cbase + w, kRingMax)];
                    pv[w] = ((okmask >> w) & 1u) ? val : kNegInf;
                  }
                }
              } else {
                const int16_t* gsrc = c.matrix + static_cast<size_t>(p) * c.MW;
                if (WB == 8 && cbase + WB <= colmax) {
                  unpack8(*reinterpret_cast<const int4*>(gsrc + cbase));
                  pv[WB] = gsrc[cbase + WB];
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    pv[w] = ((okmask >> w) & 1u) ? pv[w] : kNegInf;
                  }
                } else if (cbase + WB <= colmax) {
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    const int32_t val = gsrc[cbase + w];
                    pv[w] = ((okmask >> w) & 1u) ? val : kNegInf;
                  }
                } else {
#pragma unroll
                  for (uint32_t w = 0; w <= WB; ++w) {
                    const int32_t val = gsrc[min(cbase + w, colmax)];
                    pv[w] = ((okmask >> w) & 1u) ? val : kNegInf;
                  }
                }
              }
            }
            // no nown guard: out-of-range pv entries are -inf and propagate
#pragma unroll
            for (uint32_t w = 0; w < WB; ++w) {
              const int32_t sub = ((mbits >> w) & 1) ? c.m : c.x;
              bd[w] = max(bd[w], pv[w] + sub);
              bu[w] = max(bu[w], pv[w + 1] + c.g);
            }
          }
        }

        // local inclusive u-space scan over own columns (out-of-range
        // candidates are already -inf; no per-element guard needed)
        int32_t us[WB];
        int32_t run = kNegInf;
#pragma unroll
        for (uint32_t w = 0; w < WB; ++w) {
          const int32_t j = static_cast<int32_t>(cbase + 1 + w);
          const int32_t u = max(bd[w], bu[w]) - j * c.g;
          run = max(run, u);
          us[w] = run;
        }
        // wave scan over lane totals -> exclusive prefix for this lane
        const int32_t incl = wave_scan_max(run, lane);
        int32_t excl =
            __builtin_amdgcn_update_dpp(kNegInf, incl, 0x138, 0xf, 0xf, false);  // wave_shr:1
        excl = max(excl, carry_u);
        // next pass's carry: wave-uniform max over everything <= this pass
        carry_u = max(carry_u, __builtin_amdgcn_readlane(incl, kLanes - 1));

        // finalize own columns: h, moves, stores. h is computed for every
        // slot (junk past the row end feeds only junk columns); the move
        // logic stays guarded.
        int32_t h_sel = kNegInf;
        uint64_t mvpack = 0;  // 8 move bytes -> one aligned store at Mrow[cbase]
        int32_t harr[WB];
#pragma unroll
        for (uint32_t w = 0; w < WB; ++w) {
          const uint32_t j = cbase + 1 + w;
          const int32_t v = max(bd[w], bu[w]);
          const int32_t h = max(us[w], excl) + static_cast<int32_t>(j) * c.g;
          harr[w] = h < -28000 ? -28000 : h;
          if (w < nown) {
            uint8_t mv;
            if (h != v) {
              mv = kMvLeft;
            } else if (vstart) {
              mv = static_cast<uint8_t>(((bd[w] >= bu[w]) ? kMvDiag : kMvUp) | (63u << 2));
            } else if (nin == 1) {
              mv = (bd[w] >= bu[w]) ? kMvDiag : kMvUp;
            } else {
              // rare multi-pred row: recover the argmax edge (first e wins)
              const uint8_t type = (bd[w] >= bu[w]) ? kMvDiag : kMvUp;
              const int32_t want = (type == kMvDiag) ? bd[w] : bu[w];
              uint32_t esel = 0;
              const int32_t sub = ((mbits >> w) & 1) ? c.m : c.x;
              for (uint32_t e = 0; e < nin; ++e) {
                if (!((usable >> e) & 1)) {
                  continue;
                }
                const uint32_t p = (e < kMaxPre)
                                       ? pred_rows[e]
                                       : c.rank[c.in_edges[node * c.ME + e]] + 1;
                const int32_t cand =
                    (type == kMvDiag) ? pred_val(p, j - 1) + sub : pred_val(p, j) + c.g;
                if (cand == want) {
                  esel = e;
                  break;
                }
              }
              mv = static_cast<uint8_t>(type | (esel << 2));
            }
            if (WB == 8) {
              mvpack |= static_cast<uint64_t>(mv) << (8 * w);
              if (a.vstore_mode != 1) {
                if (store_row) {
                  Hrow[j] = static_cast<int16_t>(harr[w]);
                }
                ring_row[j] = static_cast<int16_t>(harr[w]);
              }
            } else {
              if (store_row) {
                Hrow[j] = static_cast<int16_t>(harr[w]);
              }
              ring_row[j] = static_cast<int16_t>(harr[w]);
              Mrow[j - 1] = mv;  // shifted layout, per-byte for odd widths
            }
            if (j == len) {
              h_sel = h;
            }
          }
        }
        if (WB == 8 && a.vstore_mode != 0) {
          // Shifted 16-byte row stores: lane l writes columns
          // [cbase .. cbase+7] = [neighbor's last value | own h[0..6]] so
          // the store is one aligned b128 instead of 8 b16 ops (LDS-pipe
          // congestion is the bottleneck — profiles/PARKED.md). Column
          // `base` comes from lane 63 of the previous pass (or h0), the
          // pass-boundary column base+512 from the NEXT pass's lane 0, and
          // a row ending exactly on the boundary stores it explicitly.
          const int32_t tail_in = (base == j0) ? h0 : pass_tail;
          const int32_t shifted =
              __builtin_amdgcn_update_dpp(tail_in, harr[WB - 1], 0x138, 0xf, 0xf, false);
          pass_tail = __builtin_amdgcn_readlane(harr[WB - 1], kLanes - 1);
          if (cbase < jend) {
            int4 vec;
            vec.x = (shifted & 0xffff) | (harr[0] << 16);
            vec.y = (harr[1] & 0xffff) | (harr[2] << 16);
            vec.z = (harr[3] & 0xffff) | (harr[4] << 16);
            vec.w = (harr[5] & 0xffff) | (harr[6] << 16);
            *reinterpret_cast<int4*>(&ring_row[cbase]) = vec;
            if (store_row) {
              *reinterpret_cast<int4*>(Hrow + cbase) = vec;
            }
          }
          // When this pass's last covered column is `len` itself AND len
          // lands on a lane boundary ((len - base) % WB == 0), the lane
          // that would store it via its shifted slot has cbase == len and
          // is excluded by the cbase < jend guard — store it explicitly
          // from the owning lane's last value. (Missing this for ANY
          // multiple-of-8 row length left ring[len] uninitialized and made
          // results depend on stale LDS — caught by tools/probe_order.py.)
          const uint32_t rem = len > base ? len - base : 0;
          if (rem > 0 && rem <= kLanes * WB && (rem & (WB - 1)) == 0) {
            const int owner = static_cast<int>(rem / WB) - 1;
            const int32_t t16 = __builtin_amdgcn_readlane(harr[WB - 1], owner);
            if (lane == 0) {
              ring_row[len] = static_cast<int16_t>(t16);
              if (store_row) {
                Hrow[len] = static_cast<int16_t>(t16);
              }
            }
          }
        }
        if (WB == 8 && cbase < len) {
          if (cbase + WB < c.MW) {
            // one aligned u64 store covers the lane's 8 move bytes
            // (cbase is a multiple of 8 exactly when WB == 8); bytes beyond
            // nown encode columns past the row end and are never read back
            *reinterpret_cast<uint64_t*>(Mrow + cbase) = mvpack;
          } else {
            // near-full row: byte MW-1 is the shifted column-0 slot (stored
            // above as kMvUp); a packed store here would clobber it with the
            // padding byte 0 (kMvDiag) and derail the traceback at j==0.
            // Store only the live bytes individually.
            for (uint32_t w = 0; w < nown; ++w) {
              Mrow[cbase + w] = static_cast<uint8_t>(mvpack >> (8 * w));
            }
          }
        }
        if (len > base && len <= base + kLanes * WB) {
          const int owner = static_cast<int>((len - 1 - base) / WB);
          const int32_t lc = __builtin_amdgcn_readlane(h_sel, owner);
          last_col_val = lc;
        }
      }

      // ring writes must be visible to every lane before the next row;
      // deliberately NOT __syncthreads (would drain the global row stores)
      wave_lds_sync();

      // end-node max (strict >, first in topological order wins);
      // last_col_val is already wave-uniform (readlane). In subgraph mode
      // the span's end rank is the alignment sink (plus true sinks inside
      // the window).
      if ((is_end || (sub && r == rhi)) && last_col_val > best_score) {
        best_score = last_col_val;
        best_row = r + 1;
      }
    }
      wave_lds_sync();  // rows done before the next cooperative rd_block load
    }

    t_dp += lap();
    ++layers_done;

    // ---- serial phases on lane 0 ----
    if (lane == 0) {
      // move-byte traceback
      int32_t aln_len = 0;
      uint32_t i = best_row, j = len;
      while (!(i == 0 && j == 0)) {
        uint32_t prev_i = i, prev_j = j;
        int32_t rec_node = -1, rec_seq = -1;
        if (i == 0) {
          // row 0: all-gap prefix — consume remaining columns
          rec_seq = static_cast<int32_t>(j - 1);
          prev_j = j - 1;
        } else {
          const uint8_t mv =
              c.moves[static_cast<size_t>(i) * c.MW + (j != 0 ? j - 1 : c.MW - 1)];
          const uint8_t type = mv & 3;
          const uint32_t e = mv >> 2;
          const uint64_t rd = c.row_desc[i - 1];
          const uint32_t node = static_cast<uint32_t>((rd >> 16) & 0xffff);
          const uint32_t nin = static_cast<uint32_t>((rd >> 8) & 0xff);
          if (type == kMvLeft) {
            rec_seq = static_cast<int32_t>(j - 1);
            prev_j = j - 1;
          } else {
            uint32_t p = 0;  // virtual row 0: genuine start rows (nin==0)
                             // and subgraph sources (edge sentinel 63)
            if (nin != 0 && e != 63) {
              p = (e == 0) ? static_cast<uint32_t>((rd >> 32) & 0xffff)
                           : c.rank[c.in_edges[node * c.ME + e]] + 1;
            }
            if (type == kMvDiag) {
              rec_node = static_cast<int32_t>(node);
              rec_seq = static_cast<int32_t>(j - 1);
              prev_i = p;
              prev_j = j - 1;
            } else if (type == kMvUp) {
              rec_node = static_cast<int32_t>(node);
              prev_i = p;
            } else {
              c.status = kPoaConsensusOverflow;  // invalid move: fail window
              break;
            }
          }
        }
        c.aln_nodes[aln_len] = rec_node;
        c.aln_seq[aln_len] = rec_seq;
        ++aln_len;
        i = prev_i;
        j = prev_j;
      }

      t_tb += lap();
      if (c.status == kPoaOk) {
        add_alignment_d(c, seq, wts, len, aln_len);
      }
      t_add += lap();
    }

    // lane 0's graph updates must be visible to the whole wave
    __syncthreads();
    // broadcast updated scalars from lane 0 to the wave
    c.num_nodes = __shfl(c.num_nodes, 0, kLanes);
    c.seqs_in_graph = __shfl(c.seqs_in_graph, 0, kLanes);
    c.status = __shfl(c.status, 0, kLanes);

    (void)lap();
    if (c.status == kPoaOk) {
      topo_sort_d(c, s, lane);  // cooperative (internal barriers)
    }
    if (lane == 0) {
      t_topo += lap();
    }

    // rebuild the packed row descriptors for the grown graph, lane-parallel
    (void)lap();
    if (c.status == kPoaOk) {
      build_row_desc(c, lane);
    }
    __syncthreads();
    t_rd += lap();
  }

  // ---- consensus ----
  if (lane == 0) {
    uint8_t* out = a.consensus + static_cast<size_t>(win) * L.max_consensus;
    uint16_t* cov = a.coverage + static_cast<size_t>(win) * L.max_consensus;
    int32_t clen = -1;
    (void)lap();
    if (c.status == kPoaOk) {
      clen = consensus_d(c, s, out, cov, L.max_consensus);
    }
    t_cons = lap();
    a.consensus_len[win] = clen < 0 ? 0 : static_cast<uint32_t>(clen);
    a.status[win] = c.status;
    if (TIMED) {
      timing[0] = t_dp;
      timing[1] = t_tb;
      timing[2] = t_add;
      timing[3] = t_topo;
      timing[4] = t_rd;
      timing[5] = t_cons;
      timing[6] = wall_clock64() - t_start;
      timing[7] = layers_done;
    }
  }
}

}  // namespace

void launch_poa_kernel(const PoaDeviceArena& arena, uint32_t window_base,
                       uint32_t num_windows, uint32_t bucket, void* stream) {
  static const bool timed = getenv("RGA_POA_TIMING") != nullptr;
  auto st = static_cast<hipStream_t>(stream);
  const dim3 grid(num_windows), block(kLanes);
  // separate __global__ instantiations per (columns-per-lane, ring-width)
  // bucket: one kernel containing all variants pays the widest variant's
  // registers and LDS on every path (measured 3x occupancy collapse).
  // Measured: the power-of-two 8-wide variant (two passes for a 530-column
  // window) beats a 9-wide single-pass variant by ~8% — non-power-of-two
  // unrolls lose more in generated address code than the extra, mostly
  // empty pass costs. The ring width (LDS) sets occupancy; see Shared<W>.
  if (timed) {
    switch (bucket) {
      case 0:
        hipLaunchKernelGGL((poa_window_kernel<true, 5, 384>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      case 1:
        hipLaunchKernelGGL((poa_window_kernel<true, 8, 576, 1536, 5>), grid, block, 0, st,
                           arena, window_base, num_windows);
        break;
      case 3:
        hipLaunchKernelGGL((poa_window_kernel<true, 5, 576>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      case 4:
        hipLaunchKernelGGL((poa_window_kernel<true, 5, 1024>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      default:
        hipLaunchKernelGGL((poa_window_kernel<true, 8, 1024>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
    }
  } else {
    switch (bucket) {
      case 0:
        hipLaunchKernelGGL((poa_window_kernel<false, 5, 384>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      case 1:
        hipLaunchKernelGGL((poa_window_kernel<false, 8, 576, 1536, 5>), grid, block, 0, st,
                           arena, window_base, num_windows);
        break;
      case 3:
        hipLaunchKernelGGL((poa_window_kernel<false, 5, 576>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      case 4:
        hipLaunchKernelGGL((poa_window_kernel<false, 5, 1024>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
      default:
        hipLaunchKernelGGL((poa_window_kernel<false, 8, 1024>), grid, block, 0, st, arena,
                           window_base, num_windows);
        break;
    }
  }
}

}  // namespace rga::hip
