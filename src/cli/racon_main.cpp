// racon CLI: drop-in flag parity with reference src/main.cpp:23-43 (same
// option names, defaults, positional arguments and FASTA-on-stdout contract).
// The GPU flags select the HIP/MI355X pipeline instead of CUDA batches.
#include <getopt.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>

#include "core/polisher.hpp"
#include "version.hpp"

namespace {

constexpr int32_t kAlignerBatchesOpt = 10000;
constexpr int32_t kAlignerBandWidthOpt = 10001;

struct option long_options[] = {
    {"include-unpolished", no_argument, nullptr, 'u'},
    {"fragment-correction", no_argument, nullptr, 'f'},
    {"window-length", required_argument, nullptr, 'w'},
    {"quality-threshold", required_argument, nullptr, 'q'},
    {"error-threshold", required_argument, nullptr, 'e'},
    {"no-trimming", no_argument, nullptr, 'T'},
    {"match", required_argument, nullptr, 'm'},
    {"mismatch", required_argument, nullptr, 'x'},
    {"gap", required_argument, nullptr, 'g'},
    {"threads", required_argument, nullptr, 't'},
    {"version", no_argument, nullptr, 'v'},
    {"help", no_argument, nullptr, 'h'},
    {"cudapoa-batches", optional_argument, nullptr, 'c'},
    {"cuda-banded-alignment", no_argument, nullptr, 'b'},
    {"cudaaligner-batches", required_argument, nullptr, kAlignerBatchesOpt},
    {"cudaaligner-band-width", required_argument, nullptr, kAlignerBandWidthOpt},
    {nullptr, 0, nullptr, 0}};

void help() {
  printf(
      "usage: racon [options ...] <sequences> <overlaps> <target sequences>\n"
      "\n"
      "    #default output is stdout\n"
      "    <sequences>\n"
      "        input file in FASTA/FASTQ format (can be compressed with gzip)\n"
      "        containing sequences used for correction\n"
      "    <overlaps>\n"
      "        input file in MHAP/PAF/SAM format (can be compressed with gzip)\n"
      "        containing overlaps between sequences and target sequences\n"
      "    <target sequences>\n"
      "        input file in FASTA/FASTQ format (can be compressed with gzip)\n"
      "        containing sequences which will be corrected\n"
      "\n"
      "    options:\n"
      "        -u, --include-unpolished\n"
      "            output unpolished target sequences\n"
      "        -f, --fragment-correction\n"
      "            perform fragment correction instead of contig polishing\n"
      "            (overlaps file should contain dual/self overlaps!)\n"
      "        -w, --window-length <int>\n"
      "            default: 500\n"
      "            size of window on which POA is performed\n"
      "        -q, --quality-threshold <float>\n"
      "            default: 10.0\n"
      "            threshold for average base quality of windows used in POA\n"
      "        -e, --error-threshold <float>\n"
      "            default: 0.3\n"
      "            maximum allowed error rate used for filtering overlaps\n"
      "        --no-trimming\n"
      "            disables consensus trimming at window ends\n"
      "        -m, --match <int>\n"
      "            default: 3\n"
      "            score for matching bases\n"
      "        -x, --mismatch <int>\n"
      "            default: -5\n"
      "            score for mismatching bases\n"
      "        -g, --gap <int>\n"
      "            default: -4\n"
      "            gap penalty (must be negative)\n"
      "        -t, --threads <int>\n"
      "            default: 1\n"
      "            number of threads\n"
      "        --version\n"
      "            prints the version number\n"
      "        -h, --help\n"
      "            prints the usage\n"
      "        -c, --cudapoa-batches <int>\n"
      "            default: 0\n"
      "            number of batches for HIP accelerated polishing per GPU\n"
      "        -b, --cuda-banded-alignment\n"
      "            use banding approximation for alignment on GPU\n"
      "        --cudaaligner-batches <int>\n"
      "            default: 0\n"
      "            number of batches for HIP accelerated alignment per GPU\n"
      "        --cudaaligner-band-width <int>\n"
      "            default: 0\n"
      "            Band width for HIP alignment. Must be >= 0. Non-zero allows user defined \n"
      "            band width, whereas 0 implies auto band width determination.\n");
}

}  // namespace

int main(int argc, char** argv) {
  rga::PolisherConfig config;
  bool drop_unpolished_sequences = true;

  const char* optstring = "ufw:q:e:m:x:g:t:hbc::";
  int32_t argument;
  while ((argument = getopt_long(argc, argv, optstring, long_options, nullptr)) != -1) {
    switch (argument) {
      case 'u': drop_unpolished_sequences = false; break;
      case 'f': config.type = rga::PolisherType::kF; break;
      case 'w': config.window_length = atoi(optarg); break;
      case 'q': config.quality_threshold = atof(optarg); break;
      case 'e': config.error_threshold = atof(optarg); break;
      case 'T': config.trim = false; break;
      case 'm': config.match = atoi(optarg); break;
      case 'x': config.mismatch = atoi(optarg); break;
      case 'g': config.gap = atoi(optarg); break;
      case 't': config.num_threads = atoi(optarg); break;
      case 'v': printf("%s\n", RACON_MI355X_VERSION); exit(0);
      case 'h': help(); exit(0);
      case 'c':
        // -c with no attached value: consume a following bare number, else 1
        config.poa_batches = 1;
        if (optarg == nullptr && argv[optind] != nullptr && argv[optind][0] != '-') {
          config.poa_batches = atoi(argv[optind++]);
        }
        if (optarg != nullptr) {
          config.poa_batches = atoi(optarg);
        }
        break;
      case 'b': config.banded_poa = true; break;
      case kAlignerBatchesOpt: config.aligner_batches = atoi(optarg); break;
      case kAlignerBandWidthOpt: config.aligner_band_width = atoi(optarg); break;
      default: exit(1);
    }
  }

  std::vector<std::string> input_paths;
  for (int32_t i = optind; i < argc; ++i) {
    input_paths.emplace_back(argv[i]);
  }

  if (input_paths.size() < 3) {
    fprintf(stderr, "[racon::] error: missing input file(s)!\n");
    help();
    exit(1);
  }

  auto polisher = rga::createPolisher(input_paths[0], input_paths[1], input_paths[2], config);
  polisher->initialize();

  std::vector<std::unique_ptr<rga::Sequence>> polished;
  polisher->polish(polished, drop_unpolished_sequences);

  for (const auto& it : polished) {
    fprintf(stdout, ">%s\n%s\n", it->name().c_str(), it->data().c_str());
  }
  return 0;
}
