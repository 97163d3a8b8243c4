// One read-to-target overlap record, in any of the three input conventions.
// Behavioral parity with reference src/overlap.{hpp,cpp}:
//  - MHAP: 1-based ids, strand = a_rc ^ b_rc            (ref overlap.cpp:15-27)
//  - PAF: names, '+'/'-' orientation                    (ref overlap.cpp:29-42)
//  - SAM: flag 0x10 strand, 0x4 invalid, CIGAR walk     (ref overlap.cpp:44-108)
//  - error = 1 - min(qspan,tspan)/max(qspan,tspan)
//  - resolve_ids ("transmute") maps names/file ids to global sequence indices
//  - find_breaking_points computes per-window (target,query) match anchors by
//    walking the CIGAR against window boundaries          (ref overlap.cpp:226-292)
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <unordered_map>
#include <utility>
#include <vector>

namespace rga {

class Sequence;

// Lookup tables mapping overlap-file identifiers (names for PAF/SAM, file
// ordinals for MHAP) to global sequence indices. Reads whose name duplicates
// a target share the target's slot — the polisher dedups them on load.
struct SequenceIndex {
  std::unordered_map<std::string, uint64_t> read_names;
  std::unordered_map<std::string, uint64_t> target_names;
  std::vector<uint64_t> read_ids;    // file-order read ordinal -> global index
  std::vector<uint64_t> target_ids;  // file-order target ordinal -> global index
};

class Overlap {
 public:
  // MHAP record fields.
  static std::unique_ptr<Overlap> from_mhap(uint64_t a_id, uint64_t b_id, uint32_t a_rc,
                                            uint32_t a_begin, uint32_t a_end, uint32_t a_length,
                                            uint32_t b_rc, uint32_t b_begin, uint32_t b_end,
                                            uint32_t b_length);
  // PAF record fields.
  static std::unique_ptr<Overlap> from_paf(std::string q_name, uint32_t q_length, uint32_t q_begin,
                                           uint32_t q_end, char orientation, std::string t_name,
                                           uint32_t t_length, uint32_t t_begin, uint32_t t_end);
  // SAM record fields (pos is 1-based as in the file).
  static std::unique_ptr<Overlap> from_sam(std::string q_name, uint32_t flag, std::string t_name,
                                           uint32_t pos, std::string cigar);

  uint64_t q_id() const { return q_id_; }
  uint64_t t_id() const { return t_id_; }
  uint32_t q_begin() const { return q_begin_; }
  uint32_t q_end() const { return q_end_; }
  uint32_t q_length() const { return q_length_; }
  uint32_t t_begin() const { return t_begin_; }
  uint32_t t_end() const { return t_end_; }
  uint32_t t_length() const { return t_length_; }
  bool strand() const { return strand_; }
  uint32_t length() const { return length_; }
  double error() const { return error_; }
  bool is_valid() const { return is_valid_; }
  const std::string& cigar() const { return cigar_; }
  void set_cigar(std::string cigar) { cigar_ = std::move(cigar); }
  bool has_cigar() const { return !cigar_.empty(); }

  const std::vector<std::pair<uint32_t, uint32_t>>& breaking_points() const {
    return breaking_points_;
  }

  // Maps names / file-local ids to global sequence indices; marks the overlap
  // invalid when a name/id is unknown; validates lengths against sequences.
  void resolve_ids(const std::vector<std::unique_ptr<Sequence>>& sequences,
                   const SequenceIndex& index);

  // Aligns q vs t spans when no CIGAR is present (CPU Myers NW), then walks the
  // CIGAR to produce per-window breaking points; frees the CIGAR afterwards.
  void find_breaking_points(const std::vector<std::unique_ptr<Sequence>>& sequences,
                            uint32_t window_length);

  // CIGAR walk only (used by both CPU and GPU alignment paths).
  void find_breaking_points_from_cigar(uint32_t window_length);

  // The (query, target) character spans this overlap aligns, on the strand the
  // aligner consumes (query already reverse-complemented when strand_ is set).
  std::pair<const char*, uint32_t> query_span(
      const std::vector<std::unique_ptr<Sequence>>& sequences) const;
  std::pair<const char*, uint32_t> target_span(
      const std::vector<std::unique_ptr<Sequence>>& sequences) const;

 private:
  Overlap() = default;

  std::string q_name_;
  uint64_t q_id_ = 0;
  uint32_t q_begin_ = 0, q_end_ = 0, q_length_ = 0;
  std::string t_name_;
  uint64_t t_id_ = 0;
  uint32_t t_begin_ = 0, t_end_ = 0, t_length_ = 0;
  bool strand_ = false;
  uint32_t length_ = 0;
  double error_ = 0.0;
  std::string cigar_;
  bool is_valid_ = true;
  bool is_resolved_ = false;
  std::vector<std::pair<uint32_t, uint32_t>> breaking_points_;
};

}  // namespace rga
