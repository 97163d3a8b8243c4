// Placeholder until the HIP pipeline object files are linked in; the real
// implementation lives in src/hip/hip_polisher.cpp (GPU builds replace this
// translation unit). Fails loudly rather than silently falling back to CPU.
#include <cstdio>
#include <cstdlib>
#include <memory>

#include "core/polisher.hpp"

namespace rga {

namespace hip {
int runtime_device_count() { return 0; }
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
