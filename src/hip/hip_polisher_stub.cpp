// Placeholder until the HIP pipeline object files are linked in; the real
// implementation lives in src/hip/hip_polisher.cpp (GPU builds replace this
// translation unit). Fails loudly rather than silently falling back to CPU.
#include <cstdio>
#include <cstdlib>
#include <memory>
#include <string>
#include <tuple>
#include <utility>
#include <vector>

#include "core/polisher.hpp"

namespace rga {

namespace hip {
int runtime_device_count() { return 0; }
std::vector<std::tuple<std::string, int32_t, int32_t>> align_pairs(
    const std::vector<std::pair<std::string, std::string>>&, uint32_t) {
  fprintf(stderr, "[rga::hip::align_pairs] error: no HIP backend in this build!\n");
  exit(1);
}
}  // namespace hip

std::unique_ptr<Polisher> createHipPolisher(std::unique_ptr<SequenceParser>,
                                            std::unique_ptr<OverlapParser>,
                                            std::unique_ptr<SequenceParser>, PolisherConfig) {
  fprintf(stderr,
          "[racon::createPolisher] error: HIP pipeline requested but this build "
          "does not include the HIP backend!\n");
  exit(1);
}

}  // namespace rga
