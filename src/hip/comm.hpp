// First-class distributed-communication component (north star: window-batch
// scatter / polished-contig gather on RCCL over xGMI, PyTorch-free).
//
// One process per GPU (the launcher — torchrun or a plain spawner — provides
// RANK / WORLD_SIZE / MASTER_ADDR envs). Two planes:
//   - control plane: TCP on the loopback/master address. Bootstraps the RCCL
//     unique id, carries sizes and small host payloads, and IS the data
//     plane in CPU-only runs (keeps the multi-process path testable on
//     machines without GPUs).
//   - data plane: an RCCL communicator over xGMI for the variable-length
//     byte gathers (length-prefix exchange on the control plane, then
//     ncclSend/ncclRecv point-to-point — ring collectives are per-link
//     bound on xGMI and the payloads are naturally p2p; reference analog:
//     host-staged merge in cudapolisher.cpp:385-411, which never scaled
//     past one process).
//
// RCCL refuses duplicate devices in one communicator, so single-GPU
// multi-rank rehearsals set RGA_COMM_FORCE_TCP=1 to run the full
// multi-process flow with the TCP data plane.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

namespace rga::comm {

class Communicator {
 public:
  Communicator() = default;
  ~Communicator();

  Communicator(const Communicator&) = delete;
  Communicator& operator=(const Communicator&) = delete;

  // Collective across all ranks. `use_gpu` selects the RCCL data plane
  // (each rank must own a distinct visible device). Root listens on
  // host:port; other ranks connect with retries (the launcher may start
  // them in any order).
  void init(int rank, int world, const std::string& host, int port, bool use_gpu);
  void finalize();

  bool initialized() const { return world_ > 0; }
  int rank() const { return rank_; }
  int world() const { return world_; }

  // Variable-length byte gather to root; non-root ranks get an empty vector.
  std::vector<std::string> gather(const std::string& payload, int root);

  double allreduce_max(double v);
  double allreduce_sum(double v);
  void barrier();

 private:
  // control plane (TCP)
  void ctl_send(int to_rank, const void* data, size_t bytes);
  void ctl_recv(int from_rank, void* data, size_t bytes);
  std::vector<std::string> ctl_gather(const std::string& payload, int root);
  void ctl_bcast(void* data, size_t bytes, int root);

  int rank_ = 0;
  int world_ = 0;
  bool use_gpu_ = false;
  std::vector<int> fds_;   // root: fd per rank (own slot -1); others: fds_[root]
  void* nccl_comm_ = nullptr;
  void* stream_ = nullptr;
};

// Process-wide communicator used by the benchmark/pipeline entry points.
Communicator& world_comm();

}  // namespace rga::comm
