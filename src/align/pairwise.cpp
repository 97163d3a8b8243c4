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

}  // namespace

int64_t edit_distance(const char* a, uint32_t a_len, const char* b, uint32_t b_len) {
  if (a_len == 0) {
    return b_len;
  }
  if (b_len == 0) {
    return a_len;
  }
  return myers_nw(a, a_len, b, b_len, nullptr);
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

  ColumnStore cs;
  myers_nw(q, q_len, t, t_len, &cs);

  // Backward walk from (q_len-1, t_len-1) reproducing edlib's move priority.
  // Priority order is configurable for golden-parity tuning via
  // RGA_TRACEBACK_ORDER (a permutation of "IDM"); default "IDM" = up ('I',
  // consume query), then left ('D', consume target), then diagonal ('M').
  static const char* order_env = getenv("RGA_TRACEBACK_ORDER");
  const char* order = order_env != nullptr ? order_env : "IDM";

  std::string ops;  // reversed op chars
  ops.reserve(q_len + t_len);
  int64_t i = q_len - 1;
  int64_t c = t_len - 1;
  int32_t v = cell_value(cs, c, i);
  while (i >= 0 || c >= 0) {
    bool moved = false;
    for (const char* o = order; *o != '\0' && !moved; ++o) {
      switch (*o) {
        case 'I':
          if (i >= 0) {
            int32_t up = cell_value(cs, c, i - 1);
            if (up + 1 == v) {
              ops.push_back('I');
              --i;
              v = up;
              moved = true;
            }
          }
          break;
        case 'D':
          if (c >= 0) {
            int32_t left = cell_value(cs, c - 1, i);
            if (left + 1 == v) {
              ops.push_back('D');
              --c;
              v = left;
              moved = true;
            }
          }
          break;
        case 'M':
          if (i >= 0 && c >= 0) {
            int32_t diag = cell_value(cs, c - 1, i - 1);
            int32_t step = (q[i] == t[c]) ? 0 : 1;
            if (diag + step == v) {
              ops.push_back('M');  // match or mismatch; standard CIGAR merges both
              --i;
              --c;
              v = diag;
              moved = true;
            }
          }
          break;
      }
    }
    if (!moved) {
      // Fallback diagonal (cannot happen for a consistent DP).
      ops.push_back('M');
      --i;
      --c;
      v = cell_value(cs, c, i);
    }
  }

  // Collapse the reversed op string into CIGAR runs (forward order).
  uint32_t run = 0;
  char run_op = 0;
  for (auto it = ops.rbegin(); it != ops.rend(); ++it) {
    if (*it == run_op) {
      ++run;
    } else {
      append_run(cigar, run, run_op);
      run_op = *it;
      run = 1;
    }
  }
  append_run(cigar, run, run_op);
  return cigar;
}

}  // namespace rga
