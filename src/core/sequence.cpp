#include "core/sequence.hpp"

namespace rga {

Sequence::Sequence(const char* name, uint32_t name_len, const char* data, uint32_t data_len)
    : name_(name, name_len) {
  data_.resize(data_len);
  // branchless ASCII uppercase (auto-vectorizes; libc toupper does not)
  for (uint32_t i = 0; i < data_len; ++i) {
    const char ch = data[i];
    data_[i] = (ch >= 'a' && ch <= 'z') ? static_cast<char>(ch - 32) : ch;
  }
}

Sequence::Sequence(const char* name, uint32_t name_len, const char* data, uint32_t data_len,
                   const char* quality, uint32_t quality_len)
    : Sequence(name, name_len, data, data_len) {
  uint64_t quality_sum = 0;
  for (uint32_t i = 0; i < quality_len; ++i) {
    quality_sum += static_cast<uint8_t>(quality[i]) - static_cast<uint8_t>('!');
  }
  if (quality_sum > 0) {
    quality_.assign(quality, quality_len);
  }
}

Sequence::Sequence(std::string name, std::string data)
    : name_(std::move(name)), data_(std::move(data)) {}

void Sequence::make_reverse_complement() {
  if (!reverse_complement_.empty()) {
    return;
  }
  reverse_complement_.resize(data_.size());
  for (size_t i = 0, n = data_.size(); i < n; ++i) {
    char c = data_[n - 1 - i];
    switch (c) {
      case 'A': c = 'T'; break;
      case 'T': c = 'A'; break;
      case 'C': c = 'G'; break;
      case 'G': c = 'C'; break;
      default: break;
    }
    reverse_complement_[i] = c;
  }
  reverse_quality_.assign(quality_.rbegin(), quality_.rend());
}

void Sequence::release(bool keep_name, bool keep_data, bool need_reverse_data) {
  if (!keep_name) {
    std::string().swap(name_);
  }
  if (need_reverse_data) {
    make_reverse_complement();
  }
  if (!keep_data) {
    std::string().swap(data_);
    std::string().swap(quality_);
  }
}

std::unique_ptr<Sequence> createSequence(std::string name, std::string data) {
  return std::make_unique<Sequence>(std::move(name), std::move(data));
}

}  // namespace rga
