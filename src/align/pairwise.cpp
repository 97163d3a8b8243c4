#include "align/pairwise.hpp"

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <vector>

namespace rga {

namespace {

using Word = uint64_t;
constexpr int kWordBits = 64;
constexpr Word kHighBit = Word(1) << (kWordBits - 1);

// One Myers/Hyyro block step: updates (P, M) for a 64-row block given the
// match bit-vector Eq and the horizontal delta coming in from above (hin);
// returns the horizontal delta going out at the bottom (hout).
inline int myers_step(Word& P, Word& M, Word Eq, int hin) {
  Word Xv = Eq | M;
  if (hin < 0) {
    Eq |= Word(1);
  }
  Word Xh = (((Eq & P) + P) ^ P) | Eq;
  Word Ph = M | ~(Xh | P);
  Word Mh = P & Xh;

  int hout = 0;
  if (Ph & kHighBit) {
    hout = 1;
  } else if (Mh & kHighBit) {
    hout = -1;
  }

  Ph <<= 1;
  Mh <<= 1;
  if (hin < 0) {
    Mh |= Word(1);
  } else if (hin > 0) {
    Ph |= Word(1);
  }

  P = Mh | ~(Xv | Ph);
  M = Ph & Xv;
  return hout;
}

struct Peq {
  // eq[code][block]: bit i set when query row (block*64+i) matches code.
  std::vector<Word> eq;
  int num_blocks;
  uint8_t code_of[256];
  int num_codes;

  Peq(const char* q, uint32_t qn, const char* t, uint32_t tn) {
    num_blocks = static_cast<int>((qn + kWordBits - 1) / kWordBits);
    std::memset(code_of, 0xff, sizeof(code_of));
    num_codes = 0;
    auto intern = [&](unsigned char c) {
      if (code_of[c] == 0xff) {
        code_of[c] = static_cast<uint8_t>(num_codes++);
      }
    };
    for (uint32_t i = 0; i < qn; ++i) intern(static_cast<unsigned char>(q[i]));
    for (uint32_t i = 0; i < tn; ++i) intern(static_cast<unsigned char>(t[i]));

    eq.assign(static_cast<size_t>(num_codes) * num_blocks, 0);
    for (uint32_t i = 0; i < qn; ++i) {
      uint8_t code = code_of[static_cast<unsigned char>(q[i])];
      eq[static_cast<size_t>(code) * num_blocks + i / kWordBits] |= Word(1) << (i % kWordBits);
    }
    // Padding rows (>= qn) match every code so they never add cost.
    for (int c = 0; c < num_codes; ++c) {
      for (uint32_t i = qn; i < static_cast<uint32_t>(num_blocks) * kWordBits; ++i) {
        eq[static_cast<size_t>(c) * num_blocks + i / kWordBits] |= Word(1) << (i % kWordBits);
      }
    }
  }
};

struct ColumnStore {
  // P/M/score per (column, block); score is the value of the block's bottom row.
  std::vector<Word> P, M;
  std::vector<int32_t> score;
  int num_blocks = 0;

  void resize(uint32_t cols, int blocks) {
    num_blocks = blocks;
    P.resize(static_cast<size_t>(cols) * blocks);
    M.resize(static_cast<size_t>(cols) * blocks);
    score.resize(static_cast<size_t>(cols) * blocks);
  }
};

// Value of cell (row i, column c) reconstructed from the stored block data.
// Boundary convention: value(-1, c) = c + 1, value(i, -1) = i + 1.
inline int32_t cell_value(const ColumnStore& cs, int64_t c, int64_t i) {
  if (c < 0) {
    return static_cast<int32_t>(i + 1);
  }
  if (i < 0) {
    return static_cast<int32_t>(c + 1);
  }
  int b = static_cast<int>(i / kWordBits);
  int bit = static_cast<int>(i % kWordBits);
  size_t base = static_cast<size_t>(c) * cs.num_blocks + b;
  int32_t v = cs.score[base];
  Word P = cs.P[base];
  Word M = cs.M[base];
  for (int k = kWordBits - 1; k > bit; --k) {
    Word mask = Word(1) << k;
    if (P & mask) {
      --v;
    } else if (M & mask) {
      ++v;
    }
  }
  return v;
}

// Runs the full (unbanded) Myers NW over all blocks. When store != nullptr,
// saves every column's block state for traceback. Returns the NW score at
// (qn - 1, tn - 1).
int64_t myers_nw(const char* q, uint32_t qn, const char* t, uint32_t tn, ColumnStore* store) {
  Peq peq(q, qn, t, tn);
  int nb = peq.num_blocks;

  std::vector<Word> P(nb, ~Word(0));
  std::vector<Word> M(nb, 0);
  std::vector<int32_t> score(nb);
  for (int b = 0; b < nb; ++b) {
    score[b] = (b + 1) * kWordBits;
  }

  if (store != nullptr) {
    store->resize(tn, nb);
  }

  for (uint32_t c = 0; c < tn; ++c) {
    const Word* eq_col = &peq.eq[static_cast<size_t>(
                             peq.code_of[static_cast<unsigned char>(t[c])]) *
                         nb];
    int hin = 1;  // top boundary: value(-1, c) increases by 1 per column
    for (int b = 0; b < nb; ++b) {
      hin = myers_step(P[b], M[b], eq_col[b], hin);
      score[b] += hin;
    }
    if (store != nullptr) {
      size_t base = static_cast<size_t>(c) * nb;
      std::copy(P.begin(), P.end(), store->P.begin() + base);
      std::copy(M.begin(), M.end(), store->M.begin() + base);
      std::copy(score.begin(), score.end(), store->score.begin() + base);
    }
  }

  // Walk up the padding rows of the last block to read row qn-1.
  int32_t v = score[nb - 1];
  Word lp = P[nb - 1], lm = M[nb - 1];
  int last_bit = static_cast<int>((qn - 1) % kWordBits);
  for (int k = kWordBits - 1; k > last_bit; --k) {
    Word mask = Word(1) << k;
    if (lp & mask) {
      --v;
    } else if (lm & mask) {
      ++v;
    }
  }
  return v;
}

void append_run(std::string& cigar, uint32_t n, char op) {
  if (n == 0) {
    return;
  }
  char buf[16];
  int len = snprintf(buf, sizeof(buf), "%u%c", n, op);
  cigar.append(buf, len);
}

constexpr int32_t kOutOfBand = (1 << 28);  // "unreached" marker in score columns

// Scores of the FINAL column of NW(q[0..qn) vs t[0..tn)) for every query
// prefix length 0..qn, computed inside a Ukkonen band of half-width k
// around the main diagonal: out[i] = NW(q[0..i), t) for in-band rows,
// kOutOfBand otherwise. out[0] = tn (the boundary row) when in band.
//
// Exactness: a cell with true value <= k satisfies |i - j| <= k, so it lies
// in the band, and its optimal path does too; blocks entering the band get
// the chain-from-above initialization (P = all ones continuing the vertical
// +1 run), which upper-bounds the out-of-band cells — in-band values <= k
// are therefore exact, which is all the Hirschberg crossing search reads
// (crossing cells satisfy fwd + bwd == best, so both sides are <= best <= k).
void nw_last_column(const char* q, uint32_t qn, const char* t, uint32_t tn, int64_t k,
                    std::vector<int32_t>* out) {
  out->assign(qn + 1, kOutOfBand);
  if (static_cast<int64_t>(tn) <= k + 0) {
    (*out)[0] = static_cast<int32_t>(tn);
  }
  if (qn == 0) {
    return;
  }
  Peq peq(q, qn, t, tn);
  const int nb = peq.num_blocks;
  std::vector<Word> P(nb), M(nb);
  std::vector<int32_t> bottom(nb);

  // row band for column j (1-based target prefix j): rows |i - j| <= k
  auto lo_block = [&](int64_t j) {
    const int64_t lo_row = j - k;  // first in-band row (1-based)
    return static_cast<int>(std::max<int64_t>(0, (lo_row - 1) / kWordBits));
  };
  auto hi_block = [&](int64_t j) {
    const int64_t hi_row = std::min<int64_t>(qn, j + k);
    return static_cast<int>(std::min<int64_t>(nb - 1, (hi_row - 1) / kWordBits));
  };

  int first_b = 0;
  int last_b = hi_block(1);
  for (int b = first_b; b <= last_b; ++b) {
    P[b] = ~Word(0);
    M[b] = 0;
    bottom[b] = (b + 1) * kWordBits;
  }
  for (uint32_t c = 0; c < tn; ++c) {
    const int64_t j = static_cast<int64_t>(c) + 1;
    const Word* eq_col =
        &peq.eq[static_cast<size_t>(peq.code_of[static_cast<unsigned char>(t[c])]) * nb];
    // extend the bottom of the band with the virgin-chain initialization
    const int want_hi = hi_block(j);
    while (last_b < want_hi) {
      ++last_b;
      P[last_b] = ~Word(0);
      M[last_b] = 0;
      bottom[last_b] = bottom[last_b - 1] + kWordBits;
    }
    const int want_lo = lo_block(j);
    // hin at the band top: out-of-band rows above follow the +1-per-column
    // boundary chain, exactly like row 0
    int hin = 1;
    for (int b = std::max(first_b, want_lo); b <= last_b; ++b) {
      hin = myers_step(P[b], M[b], eq_col[b], hin);
      bottom[b] += hin;
    }
    first_b = std::max(first_b, want_lo);
  }
  // unpack in-band rows of the final column, walking up from block bottoms
  const int flo = lo_block(tn), fhi = hi_block(tn);
  const int64_t row_lo = std::max<int64_t>(0, static_cast<int64_t>(tn) - k);
  const int64_t row_hi = std::min<int64_t>(qn, static_cast<int64_t>(tn) + k);
  for (int b = std::max(first_b, flo); b <= std::min(last_b, fhi); ++b) {
    int32_t v = bottom[b];
    for (int bit = kWordBits - 1; bit >= 0; --bit) {
      const int64_t row = static_cast<int64_t>(b) * kWordBits + bit + 1;  // 1-based
      const Word mask = Word(1) << bit;
      if (row <= static_cast<int64_t>(qn) && row >= row_lo && row <= row_hi) {
        (*out)[row] = v;
      }
      if (P[b] & mask) {
        --v;
      } else if (M[b] & mask) {
        ++v;
      }
    }
  }
  if (row_lo == 0) {
    (*out)[0] = static_cast<int32_t>(tn);
  }
}

// Banded NW edit distance with iterative band doubling (edlib's top-level
// strategy): exact, ~O((d/64) * tn) for distance d instead of O((qn/64) * tn).
int64_t banded_distance(const char* q, uint32_t qn, const char* t, uint32_t tn) {
  const int64_t dmin = qn > tn ? qn - tn : tn - qn;
  int64_t k = std::max<int64_t>(kWordBits, dmin);
  std::vector<int32_t> col;
  while (true) {
    nw_last_column(q, qn, t, tn, k, &col);
    const int32_t v = col[qn];
    if (v != kOutOfBand && v <= k) {
      return v;
    }
    k *= 2;
    if (k >= static_cast<int64_t>(qn) + tn) {
      nw_last_column(q, qn, t, tn, k, &col);
      return col[qn];
    }
  }
}

// Backward walk over a fully stored DP (blocked bit-vectors), reproducing
// the move priority of edlib's obtainAlignmentTraceback: up first ('I',
// consume query), then left ('D', consume target), then diagonal. Appends
// ops in FORWARD order.
void traceback_small(const char* q, uint32_t qn, const char* t, uint32_t tn, std::string* ops) {
  ColumnStore cs;
  myers_nw(q, qn, t, tn, &cs);

  std::string rev_ops;
  rev_ops.reserve(qn + tn);
  int64_t i = qn - 1;
  int64_t c = tn - 1;
  int32_t v = cell_value(cs, c, i);
  while (i >= 0 || c >= 0) {
    if (i >= 0) {
      const int32_t up = cell_value(cs, c, i - 1);
      if (up + 1 == v) {
        rev_ops.push_back('I');
        --i;
        v = up;
        continue;
      }
    }
    if (c >= 0) {
      const int32_t left = cell_value(cs, c - 1, i);
      if (left + 1 == v) {
        rev_ops.push_back('D');
        --c;
        v = left;
        continue;
      }
    }
    rev_ops.push_back('M');  // match or mismatch; standard CIGAR merges both
    v = cell_value(cs, c - 1, i - 1);
    --i;
    --c;
  }
  ops->append(rev_ops.rbegin(), rev_ops.rend());
}

// Equal-cost path selection compatible with edlib's obtainAlignment
// (re-derived from the published algorithm; the reference's edlib submodule
// is an empty directory — see docs/PARITY.md). Small problems use the
// stored-DP traceback above; larger ones split the target in half
// (Hirschberg), score the middle column from both sides, and take the
// SMALLEST query row whose forward+reverse scores sum to the optimum —
// that first-crossing rule plus the I>D>M base priority is what pins
// edlib's choice among equal-cost alignments.
// rq/rt point at the reverses of q/t (rq[0] == q[qn-1], ...).
void obtain_ops(const char* q, const char* rq, uint32_t qn, const char* t, const char* rt,
                uint32_t tn, int64_t score, std::string* ops) {
  if (qn == 0) {
    ops->append(tn, 'D');
    return;
  }
  if (tn == 0) {
    ops->append(qn, 'I');
    return;
  }

  // edlib's dispatch rule: full traceback when the stored DP fits in 1 MB
  // ((2 words + 1 int) per block per target column)
  static const long long tb_limit = [] {
    const char* e = getenv("RGA_EDLIB_TB_LIMIT");
    return e != nullptr ? atoll(e) : 1024ll * 1024;
  }();
  const long long num_blocks = (qn + kWordBits - 1) / kWordBits;
  if ((2ll * sizeof(Word) + sizeof(int32_t)) * num_blocks * tn < tb_limit) {
    traceback_small(q, qn, t, tn, ops);
    return;
  }

  const uint32_t left_w = tn / 2;
  const uint32_t right_w = tn - left_w;

  // middle-column scores from both directions: fwd[i] = NW(q[0..i), left
  // half), bwd[j] = NW(q[qn-j..qn), right half). Banded at k = the
  // subproblem score: crossing cells satisfy fwd + bwd == score so both
  // sides are <= score and in-band values at that magnitude are exact;
  // out-of-band entries hold kOutOfBand and can never fake a crossing.
  std::vector<int32_t> fwd, bwd;
  nw_last_column(q, qn, t, left_w, score, &fwd);
  nw_last_column(rq, qn, rt, right_w, score, &bwd);

  // first (lowest) query row where the two halves meet at the optimum
  int64_t cross = -2;
  for (uint32_t r = 0; r < qn; ++r) {
    if (static_cast<int64_t>(fwd[r + 1]) + bwd[qn - 1 - r] == score) {
      cross = static_cast<int64_t>(r);
      break;
    }
  }
  if (cross == -2 && static_cast<int64_t>(fwd[0]) + bwd[qn] == score) {
    cross = -1;  // the optimum consumes the whole left half before any query
  }
  if (cross == -2) {
    // cannot happen for a correct score; fail loudly rather than mis-align
    fprintf(stderr, "[rga::align] error: no crossing row at the half-target column!\n");
    exit(1);
  }

  const uint32_t left_q = static_cast<uint32_t>(cross + 1);
  obtain_ops(q, rq + (qn - left_q), left_q, t, rt + right_w, left_w, fwd[left_q], ops);
  obtain_ops(q + left_q, rq, qn - left_q, t + left_w, rt, right_w, bwd[qn - left_q], ops);
}

}  // namespace

int64_t edit_distance(const char* a, uint32_t a_len, const char* b, uint32_t b_len) {
  if (a_len == 0) {
    return b_len;
  }
  if (b_len == 0) {
    return a_len;
  }
  return banded_distance(a, a_len, b, b_len);
}

std::string align_global_cigar(const char* q, uint32_t q_len, const char* t, uint32_t t_len) {
  std::string cigar;
  if (q_len == 0) {
    append_run(cigar, t_len, 'D');
    return cigar;
  }
  if (t_len == 0) {
    append_run(cigar, q_len, 'I');
    return cigar;
  }

  const int64_t score = banded_distance(q, q_len, t, t_len);

  std::string rq(q, q + q_len), rt(t, t + t_len);
  std::reverse(rq.begin(), rq.end());
  std::reverse(rt.begin(), rt.end());

  std::string ops;  // forward 'M'/'I'/'D' chars
  ops.reserve(q_len + t_len);
  obtain_ops(q, rq.data(), q_len, t, rt.data(), t_len, score, &ops);

  // collapse into CIGAR runs
  uint32_t run = 0;
  char run_op = 0;
  for (char op : ops) {
    if (op == run_op) {
      ++run;
    } else {
      append_run(cigar, run, run_op);
      run_op = op;
      run = 1;
    }
  }
  append_run(cigar, run, run_op);
  return cigar;
}

}  // namespace rga
