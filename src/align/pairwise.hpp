// Pairwise global (NW) edit-distance alignment with CIGAR traceback.
// Capability parity target: vendor/edlib as called from reference
// src/overlap.cpp:205-224 (EDLIB_MODE_NW + EDLIB_TASK_PATH +
// EDLIB_CIGAR_STANDARD). Implementation: Myers/Hyyro bit-parallel DP over
// 64-row blocks (full band), columns over the target, with a cell-value
// traceback that reproduces edlib's move priority (up, left, diagonal).
#pragma once

#include <cstdint>
#include <string>

namespace rga {

// Edit distance only (no path); used by tests as the golden-number metric.
int64_t edit_distance(const char* a, uint32_t a_len, const char* b, uint32_t b_len);

// Global alignment of q (query) vs t (target); returns a standard CIGAR
// ('M'/'I'/'D', match+mismatch merged into M; 'I' consumes query).
std::string align_global_cigar(const char* q, uint32_t q_len, const char* t, uint32_t t_len);

}  // namespace rga
