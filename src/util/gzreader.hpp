// Buffered line reader over zlib gzFile (transparent for plain files).
// Capability parity: reference vendor/bioparser gz-aware chunked parsing
// (see /root/reference/src/polisher.cpp:83-133 call sites). New implementation.
#pragma once

#include <zlib.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace rga {

class GzReader {
 public:
  explicit GzReader(const std::string& path)
      : file_(gzopen(path.c_str(), "rb")), buf_(1u << 20), pos_(0), len_(0), eof_(false) {
    if (file_ == nullptr) {
      throw std::runtime_error("[rga::GzReader] error: unable to open file " + path);
    }
    gzbuffer(file_, 1u << 20);
  }

  GzReader(const GzReader&) = delete;
  GzReader& operator=(const GzReader&) = delete;

  ~GzReader() {
    if (file_ != nullptr) {
      gzclose(file_);
    }
  }

  void rewind() {
    gzrewind(file_);
    pos_ = len_ = 0;
    eof_ = false;
  }

  // Appends the next line (without terminating '\n' or '\r') to dst.
  // Returns false at end of file with nothing read.
  bool getline(std::string& dst) {
    dst.clear();
    bool got_any = false;
    while (true) {
      if (pos_ == len_) {
        if (!fill()) {
          return got_any;
        }
      }
      const char* start = buf_.data() + pos_;
      const char* nl = static_cast<const char*>(memchr(start, '\n', len_ - pos_));
      if (nl == nullptr) {
        dst.append(start, len_ - pos_);
        pos_ = len_;
        got_any = true;
        continue;
      }
      dst.append(start, nl - start);
      pos_ = (nl - buf_.data()) + 1;
      got_any = true;
      break;
    }
    while (!dst.empty() && (dst.back() == '\r' || dst.back() == '\n')) {
      dst.pop_back();
    }
    return true;
  }

 private:
  bool fill() {
    if (eof_) {
      return false;
    }
    int n = gzread(file_, buf_.data(), static_cast<unsigned>(buf_.size()));
    if (n < 0) {
      throw std::runtime_error("[rga::GzReader] error: gzread failed");
    }
    if (n == 0) {
      eof_ = true;
      return false;
    }
    pos_ = 0;
    len_ = static_cast<size_t>(n);
    return true;
  }

  gzFile file_;
  std::vector<char> buf_;
  size_t pos_, len_;
  bool eof_;
};

}  // namespace rga
