// Read/contig container.
// Behavioral parity with reference src/sequence.{hpp,cpp}: uppercase on parse,
// all-'!' quality strings dropped, lazy reverse complement + reversed quality,
// release() frees unused fields after overlap routing is known.
#pragma once

#include <cstdint>
#include <memory>
#include <string>

namespace rga {

class Sequence {
 public:
  // FASTA record (no quality).
  Sequence(const char* name, uint32_t name_len, const char* data, uint32_t data_len);
  // FASTQ record; quality kept only if any base has phred > 0 (ref sequence.cpp:34-42).
  Sequence(const char* name, uint32_t name_len, const char* data, uint32_t data_len,
           const char* quality, uint32_t quality_len);
  // Already-clean construction (polished output records).
  Sequence(std::string name, std::string data);

  const std::string& name() const { return name_; }
  const std::string& data() const { return data_; }
  const std::string& quality() const { return quality_; }
  const std::string& reverse_complement() const { return reverse_complement_; }
  const std::string& reverse_quality() const { return reverse_quality_; }

  // Materializes reverse complement (and reversed quality) once.
  void make_reverse_complement();

  // Frees fields that are no longer needed (ref sequence.cpp:86-100).
  void release(bool keep_name, bool keep_data, bool need_reverse_data);

 private:
  std::string name_;
  std::string data_;
  std::string reverse_complement_;
  std::string quality_;
  std::string reverse_quality_;
};

std::unique_ptr<Sequence> createSequence(std::string name, std::string data);

}  // namespace rga
