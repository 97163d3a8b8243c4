#include "io/parsers.hpp"

#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace rga {

namespace {

bool is_suffix(const std::string& src, const std::string& suffix) {
  return src.size() >= suffix.size() &&
         src.compare(src.size() - suffix.size(), suffix.size(), suffix) == 0;
}

// First whitespace-delimited token of a header line.
std::string header_name(const std::string& line, size_t offset) {
  size_t begin = offset;
  size_t end = begin;
  while (end < line.size() && !isspace(static_cast<unsigned char>(line[end]))) {
    ++end;
  }
  return line.substr(begin, end - begin);
}

// Splits a line on single-character delimiters (tab or space runs).
void tokenize(const std::string& line, char delim, std::vector<std::string>& dst) {
  dst.clear();
  size_t begin = 0;
  while (begin <= line.size()) {
    size_t end = line.find(delim, begin);
    if (end == std::string::npos) {
      dst.emplace_back(line.substr(begin));
      break;
    }
    dst.emplace_back(line.substr(begin, end - begin));
    begin = end + 1;
  }
}

class FastaParser : public SequenceParser {
 public:
  explicit FastaParser(const std::string& path) : reader_(path) {}

  bool parse(std::vector<std::unique_ptr<Sequence>>& dst, uint64_t max_bytes) override {
    uint64_t bytes = 0;
    std::string line;
    while (true) {
      if (pending_header_.empty()) {
        if (!reader_.getline(pending_header_)) {
          return false;
        }
        if (pending_header_.empty()) {
          pending_header_.clear();
          continue;
        }
        if (pending_header_[0] != '>') {
          fprintf(stderr, "[rga::FastaParser] error: invalid FASTA header!\n");
          exit(1);
        }
      }
      std::string name = header_name(pending_header_, 1);
      std::string data;
      pending_header_.clear();
      while (reader_.getline(line)) {
        if (!line.empty() && line[0] == '>') {
          pending_header_ = line;
          break;
        }
        data += line;
      }
      bytes += data.size();
      dst.emplace_back(std::make_unique<Sequence>(name.c_str(),
                                                  static_cast<uint32_t>(name.size()), data.c_str(),
                                                  static_cast<uint32_t>(data.size())));
      if (pending_header_.empty()) {
        return false;  // end of file
      }
      if (bytes >= max_bytes) {
        return true;
      }
    }
  }

  void reset() override {
    reader_.rewind();
    pending_header_.clear();
  }

 private:
  GzReader reader_;
  std::string pending_header_;
};

class FastqParser : public SequenceParser {
 public:
  explicit FastqParser(const std::string& path) : reader_(path) {}

  bool parse(std::vector<std::unique_ptr<Sequence>>& dst, uint64_t max_bytes) override {
    uint64_t bytes = 0;
    std::string header, line, data, quality;
    while (reader_.getline(header)) {
      if (header.empty()) {
        continue;
      }
      if (header[0] != '@') {
        fprintf(stderr, "[rga::FastqParser] error: invalid FASTQ header!\n");
        exit(1);
      }
      std::string name = header_name(header, 1);
      data.clear();
      quality.clear();
      // sequence lines until '+'
      while (reader_.getline(line)) {
        if (!line.empty() && line[0] == '+') {
          break;
        }
        data += line;
      }
      // quality lines until length matches
      while (quality.size() < data.size() && reader_.getline(line)) {
        quality += line;
      }
      if (quality.size() != data.size()) {
        fprintf(stderr, "[rga::FastqParser] error: unequal quality length!\n");
        exit(1);
      }
      bytes += data.size() + quality.size();
      dst.emplace_back(std::make_unique<Sequence>(
          name.c_str(), static_cast<uint32_t>(name.size()), data.c_str(),
          static_cast<uint32_t>(data.size()), quality.c_str(),
          static_cast<uint32_t>(quality.size())));
      if (bytes >= max_bytes) {
        return true;
      }
    }
    return false;
  }

  void reset() override { reader_.rewind(); }

 private:
  GzReader reader_;
};

class MhapParser : public OverlapParser {
 public:
  explicit MhapParser(const std::string& path) : reader_(path) {}

  bool parse(std::vector<std::unique_ptr<Overlap>>& dst, uint64_t max_bytes) override {
    uint64_t bytes = 0;
    std::string line;
    std::vector<std::string> f;
    while (reader_.getline(line)) {
      if (line.empty()) {
        continue;
      }
      tokenize(line, ' ', f);
      if (f.size() < 12) {
        fprintf(stderr, "[rga::MhapParser] error: invalid MHAP record!\n");
        exit(1);
      }
      bytes += line.size();
      dst.emplace_back(Overlap::from_mhap(
          strtoull(f[0].c_str(), nullptr, 10), strtoull(f[1].c_str(), nullptr, 10),
          static_cast<uint32_t>(atoi(f[4].c_str())), static_cast<uint32_t>(atoi(f[5].c_str())),
          static_cast<uint32_t>(atoi(f[6].c_str())), static_cast<uint32_t>(atoi(f[7].c_str())),
          static_cast<uint32_t>(atoi(f[8].c_str())), static_cast<uint32_t>(atoi(f[9].c_str())),
          static_cast<uint32_t>(atoi(f[10].c_str())),
          static_cast<uint32_t>(atoi(f[11].c_str()))));
      if (bytes >= max_bytes) {
        return true;
      }
    }
    return false;
  }

  void reset() override { reader_.rewind(); }

 private:
  GzReader reader_;
};

class PafParser : public OverlapParser {
 public:
  explicit PafParser(const std::string& path) : reader_(path) {}

  bool parse(std::vector<std::unique_ptr<Overlap>>& dst, uint64_t max_bytes) override {
    uint64_t bytes = 0;
    std::string line;
    std::vector<std::string> f;
    while (reader_.getline(line)) {
      if (line.empty()) {
        continue;
      }
      tokenize(line, '\t', f);
      if (f.size() < 12) {
        fprintf(stderr, "[rga::PafParser] error: invalid PAF record!\n");
        exit(1);
      }
      bytes += line.size();
      dst.emplace_back(Overlap::from_paf(
          f[0], static_cast<uint32_t>(atoi(f[1].c_str())),
          static_cast<uint32_t>(atoi(f[2].c_str())), static_cast<uint32_t>(atoi(f[3].c_str())),
          f[4].empty() ? '+' : f[4][0], f[5], static_cast<uint32_t>(atoi(f[6].c_str())),
          static_cast<uint32_t>(atoi(f[7].c_str())), static_cast<uint32_t>(atoi(f[8].c_str()))));
      if (bytes >= max_bytes) {
        return true;
      }
    }
    return false;
  }

  void reset() override { reader_.rewind(); }

 private:
  GzReader reader_;
};

class SamParser : public OverlapParser {
 public:
  explicit SamParser(const std::string& path) : reader_(path) {}

  bool parse(std::vector<std::unique_ptr<Overlap>>& dst, uint64_t max_bytes) override {
    uint64_t bytes = 0;
    std::string line;
    std::vector<std::string> f;
    while (reader_.getline(line)) {
      if (line.empty() || line[0] == '@') {
        continue;
      }
      tokenize(line, '\t', f);
      if (f.size() < 11) {
        fprintf(stderr, "[rga::SamParser] error: invalid SAM record!\n");
        exit(1);
      }
      bytes += line.size();
      dst.emplace_back(Overlap::from_sam(f[0], static_cast<uint32_t>(atoi(f[1].c_str())), f[2],
                                         static_cast<uint32_t>(atoi(f[3].c_str())), f[5]));
      if (bytes >= max_bytes) {
        return true;
      }
    }
    return false;
  }

  void reset() override { reader_.rewind(); }

 private:
  GzReader reader_;
};

}  // namespace

bool has_sequence_extension(const std::string& path) {
  for (const char* s : {".fasta", ".fasta.gz", ".fna", ".fna.gz", ".fa", ".fa.gz", ".fastq",
                        ".fastq.gz", ".fq", ".fq.gz"}) {
    if (is_suffix(path, s)) {
      return true;
    }
  }
  return false;
}

bool has_overlap_extension(const std::string& path) {
  for (const char* s : {".mhap", ".mhap.gz", ".paf", ".paf.gz", ".sam", ".sam.gz"}) {
    if (is_suffix(path, s)) {
      return true;
    }
  }
  return false;
}

std::unique_ptr<SequenceParser> createSequenceParser(const std::string& path) {
  if (is_suffix(path, ".fasta") || is_suffix(path, ".fasta.gz") || is_suffix(path, ".fna") ||
      is_suffix(path, ".fna.gz") || is_suffix(path, ".fa") || is_suffix(path, ".fa.gz")) {
    return std::make_unique<FastaParser>(path);
  }
  if (is_suffix(path, ".fastq") || is_suffix(path, ".fastq.gz") || is_suffix(path, ".fq") ||
      is_suffix(path, ".fq.gz")) {
    return std::make_unique<FastqParser>(path);
  }
  fprintf(stderr,
          "[rga::createPolisher] error: file %s has unsupported format extension (valid "
          "extensions: .fasta, .fasta.gz, .fna, .fna.gz, .fa, .fa.gz, .fastq, .fastq.gz, .fq, "
          ".fq.gz)!\n",
          path.c_str());
  exit(1);
}

std::unique_ptr<OverlapParser> createOverlapParser(const std::string& path) {
  if (is_suffix(path, ".mhap") || is_suffix(path, ".mhap.gz")) {
    return std::make_unique<MhapParser>(path);
  }
  if (is_suffix(path, ".paf") || is_suffix(path, ".paf.gz")) {
    return std::make_unique<PafParser>(path);
  }
  if (is_suffix(path, ".sam") || is_suffix(path, ".sam.gz")) {
    return std::make_unique<SamParser>(path);
  }
  fprintf(stderr,
          "[rga::createPolisher] error: file %s has unsupported format extension (valid "
          "extensions: .mhap, .mhap.gz, .paf, .paf.gz, .sam, .sam.gz)!\n",
          path.c_str());
  exit(1);
}

}  // namespace rga
