// rampler-equivalent sequence toolkit: `subsample` and `split` subcommands.
// Capability parity with the vendored rampler binary as invoked by the
// reference wrapper (/root/reference/scripts/racon_wrapper.py:62-63,87-88):
//   rampler [-o dir] subsample <sequences> <reference_length> <coverage>
//       -> <dir>/<base>_<coverage>x.<fasta|fastq>
//   rampler [-o dir] split <sequences> <chunk_size_bytes>
//       -> <dir>/<base>_<i>.<fasta|fastq>
// New implementation: streaming two-pass reservoir-free subsampling with a
// fixed RNG seed (deterministic; override with RAMPLER_SEED), single-pass
// greedy split on sequence-byte budget.
#include <getopt.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <random>
#include <string>
#include <vector>

#include "util/gzreader.hpp"

namespace {

struct Record {
  std::string name;   // full header line without leading sentinel
  std::string data;
  std::string qual;   // empty for FASTA
};

// Streaming FASTA/FASTQ record reader over gz (format sniffed from first byte).
class RecordReader {
 public:
  explicit RecordReader(const std::string& path) : in_(path), is_fastq_(false), primed_(false) {}

  bool is_fastq() {
    prime();
    return is_fastq_;
  }

  bool next(Record& r) {
    prime();
    r.name.clear();
    r.data.clear();
    r.qual.clear();
    if (is_fastq_) {
      if (pending_.empty() && !in_.getline(pending_)) return false;
      if (pending_.empty() || pending_[0] != '@') return false;
      r.name = pending_.substr(1);
      pending_.clear();
      std::string line;
      // data lines until '+'
      while (in_.getline(line)) {
        if (!line.empty() && line[0] == '+') break;
        r.data += line;
      }
      // quality lines until length matches
      while (r.qual.size() < r.data.size() && in_.getline(line)) {
        r.qual += line;
      }
      return !r.data.empty();
    }
    if (pending_.empty() && !in_.getline(pending_)) return false;
    if (pending_.empty() || pending_[0] != '>') return false;
    r.name = pending_.substr(1);
    pending_.clear();
    std::string line;
    while (in_.getline(line)) {
      if (!line.empty() && line[0] == '>') {
        pending_ = line;
        break;
      }
      r.data += line;
    }
    return !r.data.empty();
  }

 private:
  void prime() {
    if (primed_) return;
    primed_ = true;
    if (in_.getline(pending_)) {
      is_fastq_ = !pending_.empty() && pending_[0] == '@';
    }
  }

  rga::GzReader in_;
  std::string pending_;
  bool is_fastq_;
  bool primed_;
};

void write_record(FILE* f, const Record& r, bool fastq) {
  if (fastq) {
    fprintf(f, "@%s\n%s\n+\n%s\n", r.name.c_str(), r.data.c_str(), r.qual.c_str());
  } else {
    fprintf(f, ">%s\n%s\n", r.name.c_str(), r.data.c_str());
  }
}

// "<dir>/<basename up to first '.'>"
std::string base_path(const std::string& out_dir, const std::string& input) {
  size_t slash = input.find_last_of('/');
  std::string base = (slash == std::string::npos) ? input : input.substr(slash + 1);
  size_t dot = base.find('.');
  if (dot != std::string::npos) base = base.substr(0, dot);
  return out_dir + "/" + base;
}

void help() {
  printf(
      "usage: rampler [options ...] <mode>\n"
      "\n"
      "    <mode>\n"
      "        subsample <sequences> <reference length> <coverage>\n"
      "            subsample sequences to desired coverage of the reference\n"
      "        split <sequences> <chunk size>\n"
      "            split sequences into chunks of desired size in bytes\n"
      "\n"
      "    <sequences> FASTA/FASTQ, optionally gzipped\n"
      "\n"
      "    options:\n"
      "        -o, --out-directory <dir>  default: current directory\n"
      "        --version                  prints the version\n"
      "        -h, --help                 prints the usage\n");
}

int subsample(const std::string& out_dir, const std::string& path, uint64_t ref_len,
              uint32_t coverage) {
  // pass 1: total bases
  uint64_t total = 0;
  {
    RecordReader in(path);
    Record r;
    while (in.next(r)) total += r.data.size();
  }
  if (total == 0) {
    fprintf(stderr, "[rampler::subsample] error: empty sequences file\n");
    return 1;
  }
  const double want = static_cast<double>(ref_len) * coverage;
  const double keep_p = want >= static_cast<double>(total) ? 1.0 : want / total;

  uint64_t seed = 20250913;
  if (const char* s = getenv("RAMPLER_SEED")) seed = strtoull(s, nullptr, 10);
  std::mt19937_64 rng(seed);
  std::uniform_real_distribution<double> uni(0.0, 1.0);

  RecordReader in(path);
  const bool fastq = in.is_fastq();
  std::string out = base_path(out_dir, path) + "_" + std::to_string(coverage) + "x" +
                    (fastq ? ".fastq" : ".fasta");
  FILE* f = fopen(out.c_str(), "w");
  if (f == nullptr) {
    fprintf(stderr, "[rampler::subsample] error: unable to open %s\n", out.c_str());
    return 1;
  }
  Record r;
  uint64_t kept = 0, kept_bp = 0;
  while (in.next(r)) {
    if (keep_p >= 1.0 || uni(rng) < keep_p) {
      write_record(f, r, fastq);
      ++kept;
      kept_bp += r.data.size();
    }
  }
  fclose(f);
  fprintf(stderr, "[rampler::subsample] kept %llu sequences (%llu bp, target %.0f bp) -> %s\n",
          (unsigned long long)kept, (unsigned long long)kept_bp, want, out.c_str());
  return 0;
}

int split(const std::string& out_dir, const std::string& path, uint64_t chunk_size) {
  RecordReader in(path);
  const bool fastq = in.is_fastq();
  const std::string base = base_path(out_dir, path);
  const std::string ext = fastq ? ".fastq" : ".fasta";

  Record r;
  FILE* f = nullptr;
  uint64_t in_chunk = 0, idx = 0;
  while (in.next(r)) {
    if (f == nullptr || (in_chunk > 0 && in_chunk + r.data.size() > chunk_size)) {
      if (f != nullptr) fclose(f);
      std::string out = base + "_" + std::to_string(idx++) + ext;
      f = fopen(out.c_str(), "w");
      if (f == nullptr) {
        fprintf(stderr, "[rampler::split] error: unable to open %s\n", out.c_str());
        return 1;
      }
      in_chunk = 0;
    }
    write_record(f, r, fastq);
    in_chunk += r.data.size();
  }
  if (f != nullptr) fclose(f);
  fprintf(stderr, "[rampler::split] wrote %llu chunks\n", (unsigned long long)idx);
  return 0;
}

}  // namespace

int main(int argc, char** argv) {
  std::string out_dir = ".";
  option longopts[] = {{"out-directory", required_argument, nullptr, 'o'},
                       {"version", no_argument, nullptr, 'v'},
                       {"help", no_argument, nullptr, 'h'},
                       {nullptr, 0, nullptr, 0}};
  int c;
  while ((c = getopt_long(argc, argv, "o:h", longopts, nullptr)) != -1) {
    switch (c) {
      case 'o': out_dir = optarg; break;
      case 'v': printf("v1.0.0\n"); return 0;
      case 'h': help(); return 0;
      default: help(); return 1;
    }
  }
  if (optind >= argc) {
    help();
    return 1;
  }
  std::string mode = argv[optind++];
  try {
    if (mode == "subsample") {
      if (argc - optind < 3) {
        help();
        return 1;
      }
      return subsample(out_dir, argv[optind], strtoull(argv[optind + 1], nullptr, 10),
                       static_cast<uint32_t>(strtoul(argv[optind + 2], nullptr, 10)));
    }
    if (mode == "split") {
      if (argc - optind < 2) {
        help();
        return 1;
      }
      return split(out_dir, argv[optind], strtoull(argv[optind + 1], nullptr, 10));
    }
  } catch (const std::exception& e) {
    fprintf(stderr, "[rampler] error: %s\n", e.what());
    return 1;
  }
  fprintf(stderr, "[rampler] error: unknown mode %s\n", mode.c_str());
  help();
  return 1;
}
