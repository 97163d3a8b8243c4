#include "core/overlap.hpp"

#include <algorithm>
#include <cstdio>
#include <cstdlib>

#include "align/pairwise.hpp"
#include "core/sequence.hpp"

namespace rga {

namespace {

void span_metrics(uint32_t q_span, uint32_t t_span, uint32_t* length, double* error) {
  *length = std::max(q_span, t_span);
  *error = 1.0 - std::min(q_span, t_span) / static_cast<double>(*length);
}

}  // namespace

std::unique_ptr<Overlap> Overlap::from_mhap(uint64_t a_id, uint64_t b_id, uint32_t a_rc,
                                            uint32_t a_begin, uint32_t a_end, uint32_t a_length,
                                            uint32_t b_rc, uint32_t b_begin, uint32_t b_end,
                                            uint32_t b_length) {
  std::unique_ptr<Overlap> o(new Overlap());
  o->q_id_ = a_id - 1;  // MHAP ids are 1-based
  o->q_begin_ = a_begin;
  o->q_end_ = a_end;
  o->q_length_ = a_length;
  o->t_id_ = b_id - 1;
  o->t_begin_ = b_begin;
  o->t_end_ = b_end;
  o->t_length_ = b_length;
  o->strand_ = (a_rc ^ b_rc) != 0;
  span_metrics(a_end - a_begin, b_end - b_begin, &o->length_, &o->error_);
  return o;
}

std::unique_ptr<Overlap> Overlap::from_paf(std::string q_name, uint32_t q_length, uint32_t q_begin,
                                           uint32_t q_end, char orientation, std::string t_name,
                                           uint32_t t_length, uint32_t t_begin, uint32_t t_end) {
  std::unique_ptr<Overlap> o(new Overlap());
  o->q_name_ = std::move(q_name);
  o->q_begin_ = q_begin;
  o->q_end_ = q_end;
  o->q_length_ = q_length;
  o->t_name_ = std::move(t_name);
  o->t_begin_ = t_begin;
  o->t_end_ = t_end;
  o->t_length_ = t_length;
  o->strand_ = orientation == '-';
  span_metrics(q_end - q_begin, t_end - t_begin, &o->length_, &o->error_);
  return o;
}

std::unique_ptr<Overlap> Overlap::from_sam(std::string q_name, uint32_t flag, std::string t_name,
                                           uint32_t pos, std::string cigar) {
  std::unique_ptr<Overlap> o(new Overlap());
  o->q_name_ = std::move(q_name);
  o->t_name_ = std::move(t_name);
  o->t_begin_ = pos - 1;
  o->strand_ = (flag & 0x10) != 0;
  o->is_valid_ = !(flag & 0x4);
  o->cigar_ = std::move(cigar);

  if (o->cigar_.size() < 2 && o->is_valid_) {
    fprintf(stderr, "[rga::Overlap::from_sam] error: missing alignment from SAM record!\n");
    exit(1);
  }

  // leading clip length = query start on the stored (file) strand
  for (size_t i = 0; i < o->cigar_.size(); ++i) {
    char op = o->cigar_[i];
    if (op == 'S' || op == 'H') {
      o->q_begin_ = static_cast<uint32_t>(atoi(o->cigar_.c_str()));
      break;
    }
    if (op == 'M' || op == '=' || op == 'I' || op == 'D' || op == 'N' || op == 'P' || op == 'X') {
      break;
    }
  }

  uint32_t q_aln = 0, q_clip = 0, t_aln = 0;
  for (size_t i = 0, j = 0; i < o->cigar_.size(); ++i) {
    char op = o->cigar_[i];
    if (op == 'M' || op == '=' || op == 'X') {
      uint32_t n = static_cast<uint32_t>(atoi(o->cigar_.c_str() + j));
      j = i + 1;
      q_aln += n;
      t_aln += n;
    } else if (op == 'I') {
      q_aln += static_cast<uint32_t>(atoi(o->cigar_.c_str() + j));
      j = i + 1;
    } else if (op == 'D' || op == 'N') {
      t_aln += static_cast<uint32_t>(atoi(o->cigar_.c_str() + j));
      j = i + 1;
    } else if (op == 'S' || op == 'H') {
      q_clip += static_cast<uint32_t>(atoi(o->cigar_.c_str() + j));
      j = i + 1;
    } else if (op == 'P') {
      j = i + 1;
    }
  }

  o->q_end_ = o->q_begin_ + q_aln;
  o->q_length_ = q_clip + q_aln;
  if (o->strand_) {
    uint32_t tmp = o->q_begin_;
    o->q_begin_ = o->q_length_ - o->q_end_;
    o->q_end_ = o->q_length_ - tmp;
  }
  o->t_end_ = o->t_begin_ + t_aln;
  span_metrics(q_aln, t_aln, &o->length_, &o->error_);
  return o;
}

void Overlap::resolve_ids(const std::vector<std::unique_ptr<Sequence>>& sequences,
                          const SequenceIndex& index) {
  if (!is_valid_ || is_resolved_) {
    return;
  }

  if (!q_name_.empty()) {
    auto it = index.read_names.find(q_name_);
    if (it == index.read_names.end()) {
      is_valid_ = false;
      return;
    }
    q_id_ = it->second;
    std::string().swap(q_name_);
  } else if (q_id_ < index.read_ids.size()) {
    q_id_ = index.read_ids[q_id_];
  } else {
    is_valid_ = false;
    return;
  }

  if (q_length_ != sequences[q_id_]->data().size()) {
    fprintf(stderr,
            "[rga::Overlap::resolve_ids] error: unequal lengths in sequence "
            "and overlap file for sequence %s!\n",
            sequences[q_id_]->name().c_str());
    exit(1);
  }

  if (!t_name_.empty()) {
    auto it = index.target_names.find(t_name_);
    if (it == index.target_names.end()) {
      is_valid_ = false;
      return;
    }
    t_id_ = it->second;
    std::string().swap(t_name_);
  } else if (t_id_ < index.target_ids.size()) {
    t_id_ = index.target_ids[t_id_];
  } else {
    is_valid_ = false;
    return;
  }

  if (t_length_ != 0 && t_length_ != sequences[t_id_]->data().size()) {
    fprintf(stderr,
            "[rga::Overlap::resolve_ids] error: unequal lengths in target "
            "and overlap file for target %s!\n",
            sequences[t_id_]->name().c_str());
    exit(1);
  }
  t_length_ = static_cast<uint32_t>(sequences[t_id_]->data().size());  // SAM has no t_length

  is_resolved_ = true;
}

std::pair<const char*, uint32_t> Overlap::query_span(
    const std::vector<std::unique_ptr<Sequence>>& sequences) const {
  const char* q = !strand_ ? sequences[q_id_]->data().c_str() + q_begin_
                           : sequences[q_id_]->reverse_complement().c_str() + (q_length_ - q_end_);
  return {q, q_end_ - q_begin_};
}

std::pair<const char*, uint32_t> Overlap::target_span(
    const std::vector<std::unique_ptr<Sequence>>& sequences) const {
  return {sequences[t_id_]->data().c_str() + t_begin_, t_end_ - t_begin_};
}

void Overlap::find_breaking_points(const std::vector<std::unique_ptr<Sequence>>& sequences,
                                   uint32_t window_length) {
  if (!is_resolved_) {
    fprintf(stderr, "[rga::Overlap::find_breaking_points] error: overlap ids not resolved!\n");
    exit(1);
  }
  if (!breaking_points_.empty()) {
    return;
  }

  if (cigar_.empty()) {
    auto q = query_span(sequences);
    auto t = target_span(sequences);
    cigar_ = align_global_cigar(q.first, q.second, t.first, t.second);
  }

  find_breaking_points_from_cigar(window_length);
  std::string().swap(cigar_);
}

void Overlap::find_breaking_points_from_cigar(uint32_t window_length) {
  // window-end target positions covered by this overlap (ref overlap.cpp:229-235)
  std::vector<int32_t> window_ends;
  for (uint32_t i = 0; i < t_end_; i += window_length) {
    if (i > t_begin_) {
      window_ends.emplace_back(static_cast<int32_t>(i) - 1);
    }
  }
  window_ends.emplace_back(static_cast<int32_t>(t_end_) - 1);

  uint32_t w = 0;
  bool found_first_match = false;
  std::pair<uint32_t, uint32_t> first_match = {0, 0}, last_match = {0, 0};

  int32_t q_ptr = static_cast<int32_t>(strand_ ? (q_length_ - q_end_) : q_begin_) - 1;
  int32_t t_ptr = static_cast<int32_t>(t_begin_) - 1;

  auto close_window = [&]() {
    if (found_first_match) {
      breaking_points_.emplace_back(first_match);
      breaking_points_.emplace_back(last_match);
    }
    found_first_match = false;
    ++w;
  };

  for (size_t i = 0, j = 0; i < cigar_.size(); ++i) {
    char op = cigar_[i];
    if (op == 'M' || op == '=' || op == 'X') {
      uint32_t n = static_cast<uint32_t>(atoi(cigar_.c_str() + j));
      j = i + 1;
      for (uint32_t k = 0; k < n; ++k) {
        ++q_ptr;
        ++t_ptr;
        if (!found_first_match) {
          found_first_match = true;
          first_match = {static_cast<uint32_t>(t_ptr), static_cast<uint32_t>(q_ptr)};
        }
        last_match = {static_cast<uint32_t>(t_ptr) + 1, static_cast<uint32_t>(q_ptr) + 1};
        if (t_ptr == window_ends[w]) {
          close_window();
        }
      }
    } else if (op == 'I') {
      q_ptr += atoi(cigar_.c_str() + j);
      j = i + 1;
    } else if (op == 'D' || op == 'N') {
      uint32_t n = static_cast<uint32_t>(atoi(cigar_.c_str() + j));
      j = i + 1;
      for (uint32_t k = 0; k < n; ++k) {
        ++t_ptr;
        if (t_ptr == window_ends[w]) {
          close_window();
        }
      }
    } else if (op == 'S' || op == 'H' || op == 'P') {
      j = i + 1;
    }
  }
}

}  // namespace rga
