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
#include "hip/comm.hpp"

namespace rga {

namespace comm {
Communicator::~Communicator() {}
void Communicator::init(int, int, const std::string&, int, bool) {
  fprintf(stderr, "[rga::comm] error: no HIP backend in this build!\n");
  exit(1);
}
void Communicator::finalize() {}
std::vector<std::string> Communicator::gather(const std::string&, int) { return {}; }
double Communicator::allreduce_max(double v) { return v; }
double Communicator::allreduce_sum(double v) { return v; }
void Communicator::barrier() {}
void Communicator::ctl_send(int, const void*, size_t) {}
void Communicator::ctl_recv(int, void*, size_t) {}
std::vector<std::string> Communicator::ctl_gather(const std::string&, int) { return {}; }
void Communicator::ctl_bcast(void*, size_t, int) {}
Communicator& world_comm() {
  static Communicator c;
  return c;
}
}  // namespace comm

namespace hip {
int runtime_device_count() { return 0; }
void runtime_device_synchronize() {}
std::vector<std::pair<std::string, bool>> poa_windows_gpu(
    const std::vector<std::vector<std::tuple<std::string, std::string, uint32_t, uint32_t>>>&,
    int8_t, int8_t, int8_t, bool, bool, bool) {
  fprintf(stderr, "[rga::hip::poa_windows_gpu] error: no HIP backend in this build!\n");
  exit(1);
}
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
