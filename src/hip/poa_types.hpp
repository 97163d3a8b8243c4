// Device-side batch layout for the HIP POA engine.
//
// Capability parity target: GenomeWorks cudapoa as driven by reference
// src/cuda/cudabatch.cpp (BatchConfig(1023, 200, 256, ...), add_poa_group /
// generate_poa / get_consensus contract) — re-designed for CDNA4: one 64-lane
// wavefront per window, full-width DP rows streamed through HBM as int16,
// serial graph phases on lane 0 hidden by 16-32 co-resident windows per CU.
#pragma once

#include <cstdint>

namespace rga::hip {

// Capacity model (per window). Windows exceeding any cap are rejected or
// failed on device and fall back to the CPU path (reference contract:
// cudapolisher.cpp:354-383).
struct PoaLimits {
  uint32_t max_seq_len = 1023;    // bases per layer (incl. backbone)
  uint32_t max_depth = 200;       // layers per window (MAX_DEPTH_PER_WINDOW)
  uint32_t max_nodes = 2047;      // POA graph nodes per window
  uint32_t max_edges = 48;        // in- or out-edges per node
  uint32_t max_ring = 4;          // aligned-ring partners per node
  uint32_t matrix_width = 1024;   // DP row width (max_seq_len + 1)
  uint32_t max_consensus = 2048;  // consensus output cap
};

// Window status codes produced by the kernel.
enum PoaStatus : int32_t {
  kPoaOk = 0,
  kPoaNodeOverflow = 1,
  kPoaEdgeOverflow = 2,
  kPoaRingOverflow = 3,
  kPoaConsensusOverflow = 4,
  kPoaNotRun = 5,
  kPoaWidthOverflow = 6,  // layer wider than the launched ring variant
};

// Per-window input descriptor (layers already sorted: backbone first, then
// by window start position — the same order as the CPU path).
struct PoaWindowDesc {
  uint32_t seq_offset;   // byte offset into the packed seq/weight arena
  uint32_t num_seqs;     // layers shipped to the device (<= max_depth + 1)
  uint32_t scratch_idx;  // which device slab this window uses
  uint32_t max_len;      // longest layer (columns) — selects the kernel's
                         // columns-per-lane instantiation
};

// Device arena pointers (one allocation, carved into slabs).
struct PoaDeviceArena {
  // inputs (packed tight, H2D once per batch)
  const uint8_t* seq_data;     // concatenated layer bases
  const uint8_t* weight_data;  // matching per-base weights
  const uint32_t* layer_ends;  // per layer: end offset within window's span
  // per layer: backbone span driving the SUBGRAPH-restricted alignment
  // (begin << 16 | inclusive end; 0xFFFFFFFF = spans the window, full DP).
  // CPU parity: layers not reaching within 1% of both window edges align
  // against a subgraph of their backbone range (Window::generate_consensus);
  // the device restricts DP rows to the [rank(begin), rank(end)] window —
  // a capability the reference's cudapoa lacks (it aligns every layer to
  // the full graph, which is what degrades its w=1000 GPU goldens).
  const uint32_t* layer_spans;
  const uint32_t* layer_ends_index;  // per window: first index into layer_ends
  const PoaWindowDesc* windows;

  // per-window graph slabs (device scratch, indexed by scratch_idx)
  uint8_t* letters;      // [max_nodes]
  uint8_t* in_cnt;       // [max_nodes]
  uint8_t* out_cnt;      // [max_nodes]
  uint8_t* ring_cnt;     // [max_nodes]
  uint16_t* in_edges;    // [max_nodes * max_edges]
  int32_t* in_weights;   // [max_nodes * max_edges]
  uint16_t* out_edges;   // [max_nodes * max_edges]
  uint16_t* ring;        // [max_nodes * max_ring]
  uint16_t* nseq;        // [max_nodes] sequences touching node (coverage)
  uint16_t* rank;        // [max_nodes] node -> rank (Kahn scratch/queue and
                         // the topological order itself live in LDS)
  int64_t* hb_score;     // [max_nodes] heaviest-bundle running scores
  int32_t* hb_pred;      // [max_nodes]
  int32_t* aln_nodes;    // [2 * matrix_width + max_nodes] alignment node ids
  int32_t* aln_seq;      // [same] alignment sequence positions
  int16_t* matrix;       // [(max_nodes + 1) * matrix_width] DP scores
  uint8_t* moves;        // [(max_nodes + 1) * matrix_width] DP move bytes:
                         // bits 0-1 {0=diag,1=up,2=left,3=invalid}, bits 2-7
                         // in-edge index of the chosen predecessor
  uint64_t* row_desc;    // [max_nodes] per-rank packed row descriptor, built
                         // lane-parallel after each topo sort: letter(8) |
                         // nin(8) | node(16) | pred_row(16) | is_end(8)

  // per-window phase timing (wall_clock64 deltas; slots: 0 DP rows,
  // 1 traceback, 2 add_alignment, 3 topo sort, 4 row_desc, 5 consensus,
  // 6 total, 7 layers processed) — aggregated by the host under
  // RGA_POA_TIMING for kernel-phase attribution
  unsigned long long* timing;  // [8] per window

  // outputs (D2H once per batch)
  uint8_t* consensus;     // [max_consensus] per window, reversed on host
  uint16_t* coverage;     // [max_consensus] per window
  uint32_t* consensus_len;  // [1] per window
  int32_t* status;          // [1] per window

  int8_t match, mismatch, gap;
  uint32_t vstore_mode;  // debug: 0 per-element row stores, 1 vector, 2 both
  uint32_t band_width;  // 0 = full-width DP; else static band (reference -b:
                        // BatchConfig band 256, src/cuda/cudabatch.cpp:56-59)
  PoaLimits limits;
};

// Launches one (columns-per-lane, LDS-ring-width) kernel variant for
// windows [window_base, window_base + num_windows) of the (bucket-sorted)
// desc array. bucket: 0 = WB5/384-wide (rows <= 320 columns, single pass),
// 1 = WB8/576-wide/1536-node (<= 575 and depth <= 96), 2 = WB8/1024-wide,
// 3 = WB5/576-wide (banded rows in a 576 matrix), 4 = WB5/1024-wide
// (banded rows in a full-width matrix).
void launch_poa_kernel(const PoaDeviceArena& arena, uint32_t window_base,
                       uint32_t num_windows, uint32_t bucket, void* stream);

}  // namespace rga::hip
