#include "hip/comm.hpp"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <thread>

#include "hip/hip_common.hpp"

namespace rga::comm {

namespace {

constexpr int kConnectTimeoutSec = 120;

#define RGA_NCCL_CHECK(expr)                                                          \
  do {                                                                                \
    ncclResult_t rc_ = (expr);                                                        \
    if (rc_ != ncclSuccess) {                                                         \
      char msg_[256];                                                                 \
      snprintf(msg_, sizeof(msg_), "[rga::comm] RCCL error %s at %s:%d",              \
               ncclGetErrorString(rc_), __FILE__, __LINE__);                          \
      throw std::runtime_error(msg_);                                                 \
    }                                                                                 \
  } while (0)

void send_all(int fd, const void* data, size_t bytes) {
  const char* p = static_cast<const char*>(data);
  while (bytes > 0) {
    ssize_t n = ::send(fd, p, bytes, 0);
    if (n <= 0) {
      throw std::runtime_error("[rga::comm] control-plane send failed");
    }
    p += n;
    bytes -= static_cast<size_t>(n);
  }
}

void recv_all(int fd, void* data, size_t bytes) {
  char* p = static_cast<char*>(data);
  while (bytes > 0) {
    ssize_t n = ::recv(fd, p, bytes, 0);
    if (n <= 0) {
      throw std::runtime_error("[rga::comm] control-plane recv failed");
    }
    p += n;
    bytes -= static_cast<size_t>(n);
  }
}

sockaddr_in make_addr(const std::string& host, int port) {
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(static_cast<uint16_t>(port));
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
    // hostname that is not a dotted quad: the pool's single-node contract
    // pins the master to loopback
    inet_pton(AF_INET, "127.0.0.1", &addr.sin_addr);
  }
  return addr;
}

}  // namespace

Communicator::~Communicator() { finalize(); }

void Communicator::init(int rank, int world, const std::string& host, int port, bool use_gpu) {
  if (initialized()) {
    throw std::runtime_error("[rga::comm] communicator already initialized");
  }
  if (world < 2) {
    throw std::runtime_error("[rga::comm] init needs world >= 2");
  }
  rank_ = rank;
  world_ = world;
  use_gpu_ = use_gpu;
  fds_.assign(world, -1);

  // ---- control plane: rank 0 accepts world-1 loopback connections ----
  if (rank == 0) {
    int lfd = ::socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(lfd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr = make_addr("0.0.0.0", port);
    addr.sin_addr.s_addr = INADDR_ANY;
    if (bind(lfd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0 ||
        listen(lfd, world) != 0) {
      close(lfd);
      throw std::runtime_error("[rga::comm] root failed to bind/listen on the comm port");
    }
    for (int i = 1; i < world; ++i) {
      int fd = accept(lfd, nullptr, nullptr);
      if (fd < 0) {
        close(lfd);
        throw std::runtime_error("[rga::comm] accept failed");
      }
      int one2 = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one2, sizeof(one2));
      int32_t peer = -1;
      recv_all(fd, &peer, sizeof(peer));
      if (peer < 1 || peer >= world || fds_[peer] != -1) {
        close(fd);
        close(lfd);
        throw std::runtime_error("[rga::comm] bad peer rank during bootstrap");
      }
      fds_[peer] = fd;
    }
    close(lfd);
  } else {
    sockaddr_in addr = make_addr(host, port);
    const auto deadline =
        std::chrono::steady_clock::now() + std::chrono::seconds(kConnectTimeoutSec);
    int fd = -1;
    while (true) {
      fd = ::socket(AF_INET, SOCK_STREAM, 0);
      if (connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0) {
        break;
      }
      close(fd);
      fd = -1;
      if (std::chrono::steady_clock::now() > deadline) {
        throw std::runtime_error("[rga::comm] could not reach the root rank");
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    int32_t me = rank;
    send_all(fd, &me, sizeof(me));
    fds_[0] = fd;
  }

  // ---- data plane: RCCL communicator over the visible device ----
  if (use_gpu_) {
    ncclUniqueId id;
    if (rank == 0) {
      RGA_NCCL_CHECK(ncclGetUniqueId(&id));
    }
    ctl_bcast(&id, sizeof(id), 0);
    RGA_HIP_CHECK(hipSetDevice(0));  // one process per GPU: HIP_VISIBLE_DEVICES pins it
    hipStream_t s;
    RGA_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    stream_ = s;
    ncclComm_t comm;
    RGA_NCCL_CHECK(ncclCommInitRank(&comm, world, id, rank));
    nccl_comm_ = comm;
  }
}

void Communicator::finalize() {
  if (nccl_comm_ != nullptr) {
    (void)ncclCommDestroy(static_cast<ncclComm_t>(nccl_comm_));
    nccl_comm_ = nullptr;
  }
  if (stream_ != nullptr) {
    (void)hipStreamDestroy(static_cast<hipStream_t>(stream_));
    stream_ = nullptr;
  }
  for (int fd : fds_) {
    if (fd >= 0) {
      close(fd);
    }
  }
  fds_.clear();
  world_ = 0;
}

void Communicator::ctl_send(int to_rank, const void* data, size_t bytes) {
  send_all(fds_[to_rank], data, bytes);
}

void Communicator::ctl_recv(int from_rank, void* data, size_t bytes) {
  recv_all(fds_[from_rank], data, bytes);
}

// Control-plane broadcast: star topology through rank 0.
void Communicator::ctl_bcast(void* data, size_t bytes, int root) {
  if (rank_ == 0) {
    if (root != 0) {
      ctl_recv(root, data, bytes);
    }
    for (int r = 1; r < world_; ++r) {
      if (r != root) {
        ctl_send(r, data, bytes);
      }
    }
  } else {
    if (rank_ == root) {
      ctl_send(0, data, bytes);
    } else {
      ctl_recv(0, data, bytes);
    }
  }
}

std::vector<std::string> Communicator::ctl_gather(const std::string& payload, int root) {
  std::vector<std::string> out;
  // star through rank 0 (the only rank holding every control connection)
  if (rank_ == 0) {
    std::vector<std::string> all(world_);
    all[0] = payload;
    for (int r = 1; r < world_; ++r) {
      uint64_t n = 0;
      ctl_recv(r, &n, sizeof(n));
      all[r].resize(n);
      if (n > 0) {
        ctl_recv(r, all[r].data(), n);
      }
    }
    if (root == 0) {
      return all;
    }
    for (auto& s : all) {
      uint64_t n = s.size();
      ctl_send(root, &n, sizeof(n));
      if (n > 0) {
        ctl_send(root, s.data(), n);
      }
    }
  } else {
    uint64_t n = payload.size();
    ctl_send(0, &n, sizeof(n));
    if (n > 0) {
      ctl_send(0, payload.data(), n);
    }
    if (rank_ == root) {
      out.resize(world_);
      for (int r = 0; r < world_; ++r) {
        uint64_t m = 0;
        ctl_recv(0, &m, sizeof(m));
        out[r].resize(m);
        if (m > 0) {
          ctl_recv(0, out[r].data(), m);
        }
      }
    }
  }
  return rank_ == root ? out : std::vector<std::string>();
}

std::vector<std::string> Communicator::gather(const std::string& payload, int root) {
  if (!initialized()) {
    throw std::runtime_error("[rga::comm] gather before init");
  }
  if (nccl_comm_ == nullptr) {
    return ctl_gather(payload, root);
  }

  // length-prefix exchange on the control plane, payload bytes p2p on RCCL
  // (each sender/receiver pair is its own xGMI route; no padded ring)
  uint64_t my_size = payload.size();
  std::string sizes_blob(reinterpret_cast<char*>(&my_size), sizeof(my_size));
  std::vector<std::string> size_msgs = ctl_gather(sizes_blob, root);

  auto comm = static_cast<ncclComm_t>(nccl_comm_);
  auto stream = static_cast<hipStream_t>(stream_);
  std::vector<std::string> out;

  if (rank_ == root) {
    std::vector<uint64_t> sizes(world_);
    for (int r = 0; r < world_; ++r) {
      std::memcpy(&sizes[r], size_msgs[r].data(), sizeof(uint64_t));
    }
    const uint64_t total = [&] {
      uint64_t t = 0;
      for (int r = 0; r < world_; ++r) {
        if (r != root) {
          t += sizes[r];
        }
      }
      return t;
    }();
    void* dbuf = nullptr;
    if (total > 0) {
      RGA_HIP_CHECK(hipMalloc(&dbuf, total));
    }
    RGA_NCCL_CHECK(ncclGroupStart());
    uint64_t off = 0;
    for (int r = 0; r < world_; ++r) {
      if (r != root && sizes[r] > 0) {
        RGA_NCCL_CHECK(ncclRecv(static_cast<uint8_t*>(dbuf) + off, sizes[r], ncclUint8, r,
                                comm, stream));
        off += sizes[r];
      }
    }
    RGA_NCCL_CHECK(ncclGroupEnd());
    RGA_HIP_CHECK(hipStreamSynchronize(stream));

    out.resize(world_);
    out[root] = payload;
    off = 0;
    for (int r = 0; r < world_; ++r) {
      if (r != root) {
        out[r].resize(sizes[r]);
        if (sizes[r] > 0) {
          RGA_HIP_CHECK(hipMemcpy(out[r].data(), static_cast<uint8_t*>(dbuf) + off, sizes[r],
                                  hipMemcpyDeviceToHost));
          off += sizes[r];
        }
      }
    }
    if (dbuf != nullptr) {
      RGA_HIP_CHECK(hipFree(dbuf));
    }
  } else {
    void* dbuf = nullptr;
    if (my_size > 0) {
      RGA_HIP_CHECK(hipMalloc(&dbuf, my_size));
      RGA_HIP_CHECK(hipMemcpy(dbuf, payload.data(), my_size, hipMemcpyHostToDevice));
      RGA_NCCL_CHECK(ncclSend(dbuf, my_size, ncclUint8, root, comm, stream));
      RGA_HIP_CHECK(hipStreamSynchronize(stream));
      RGA_HIP_CHECK(hipFree(dbuf));
    }
  }
  return out;
}

double Communicator::allreduce_max(double v) {
  if (nccl_comm_ != nullptr) {
    auto comm = static_cast<ncclComm_t>(nccl_comm_);
    auto stream = static_cast<hipStream_t>(stream_);
    double* d = nullptr;
    RGA_HIP_CHECK(hipMalloc(&d, sizeof(double)));
    RGA_HIP_CHECK(hipMemcpy(d, &v, sizeof(double), hipMemcpyHostToDevice));
    RGA_NCCL_CHECK(ncclAllReduce(d, d, 1, ncclDouble, ncclMax, comm, stream));
    RGA_HIP_CHECK(hipStreamSynchronize(stream));
    RGA_HIP_CHECK(hipMemcpy(&v, d, sizeof(double), hipMemcpyDeviceToHost));
    RGA_HIP_CHECK(hipFree(d));
    return v;
  }
  std::string blob(reinterpret_cast<char*>(&v), sizeof(v));
  auto all = ctl_gather(blob, 0);
  double result = v;
  if (rank_ == 0) {
    for (auto& s : all) {
      double x;
      std::memcpy(&x, s.data(), sizeof(x));
      result = std::max(result, x);
    }
  }
  ctl_bcast(&result, sizeof(result), 0);
  return result;
}

double Communicator::allreduce_sum(double v) {
  if (nccl_comm_ != nullptr) {
    auto comm = static_cast<ncclComm_t>(nccl_comm_);
    auto stream = static_cast<hipStream_t>(stream_);
    double* d = nullptr;
    RGA_HIP_CHECK(hipMalloc(&d, sizeof(double)));
    RGA_HIP_CHECK(hipMemcpy(d, &v, sizeof(double), hipMemcpyHostToDevice));
    RGA_NCCL_CHECK(ncclAllReduce(d, d, 1, ncclDouble, ncclSum, comm, stream));
    RGA_HIP_CHECK(hipStreamSynchronize(stream));
    RGA_HIP_CHECK(hipMemcpy(&v, d, sizeof(double), hipMemcpyDeviceToHost));
    RGA_HIP_CHECK(hipFree(d));
    return v;
  }
  std::string blob(reinterpret_cast<char*>(&v), sizeof(v));
  auto all = ctl_gather(blob, 0);
  double result = 0.0;
  if (rank_ == 0) {
    for (auto& s : all) {
      double x;
      std::memcpy(&x, s.data(), sizeof(x));
      result += x;
    }
  }
  ctl_bcast(&result, sizeof(result), 0);
  return result;
}

void Communicator::barrier() {
  // the sum doubles as the rendezvous on both planes
  (void)allreduce_sum(0.0);
}

Communicator& world_comm() {
  static Communicator c;
  return c;
}

}  // namespace rga::comm
