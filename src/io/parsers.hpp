// Chunked, gzip-transparent parsers for FASTA/FASTQ (sequences) and
// MHAP/PAF/SAM (overlaps). Capability parity: vendor/bioparser as called
// from reference src/polisher.cpp:83-133,202-203,228-231,310-313 —
// parse(dst, max_bytes) appends records until the byte budget is reached,
// returning true while more data remains; reset() rewinds the file.
// Sequence names are truncated at the first whitespace.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "core/overlap.hpp"
#include "core/sequence.hpp"
#include "util/gzreader.hpp"

namespace rga {

class SequenceParser {
 public:
  virtual ~SequenceParser() = default;
  // Appends records to dst; stops after ~max_bytes of sequence data
  // (max_bytes == uint64(-1) parses the whole file). Returns true when more
  // records remain.
  virtual bool parse(std::vector<std::unique_ptr<Sequence>>& dst, uint64_t max_bytes) = 0;
  virtual void reset() = 0;
};

class OverlapParser {
 public:
  virtual ~OverlapParser() = default;
  virtual bool parse(std::vector<std::unique_ptr<Overlap>>& dst, uint64_t max_bytes) = 0;
  virtual void reset() = 0;
};

// Factory keyed on file extension; exits with the reference's error message
// shape when the extension is unsupported.
std::unique_ptr<SequenceParser> createSequenceParser(const std::string& path);
std::unique_ptr<OverlapParser> createOverlapParser(const std::string& path);

bool has_sequence_extension(const std::string& path);
bool has_overlap_extension(const std::string& path);

}  // namespace rga
