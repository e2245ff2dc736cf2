// TcpComm: localhost socket star (rank 0 = hub) for true multi-process
// CPU data-parallel tests (SURVEY.md §4 T4 complement). Host buffers only;
// the GPU path uses RCCL (rccl_comm.cc).
#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <cstring>

#include "src/comm/comm.h"
#include "src/util/util.h"

namespace nz {
namespace {

void send_all(int fd, const void* buf, size_t n) {
  const char* p = (const char*)buf;
  while (n) {
    ssize_t k = ::send(fd, p, n, 0);
    NZ_CHECK(k > 0, std::string("send failed: ") + strerror(errno));
    p += k;
    n -= k;
  }
}
void recv_all(int fd, void* buf, size_t n) {
  char* p = (char*)buf;
  while (n) {
    ssize_t k = ::recv(fd, p, n, 0);
    NZ_CHECK(k > 0, std::string("recv failed: ") + strerror(errno));
    p += k;
    n -= k;
  }
}

class TcpComm : public Comm {
 public:
  TcpComm(int nranks, int rank, const char* host, int port)
      : nranks_(nranks), rank_(rank) {
    if (nranks == 1) return;
    if (rank == 0) {
      int srv = socket(AF_INET, SOCK_STREAM, 0);
      int one = 1;
      setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_addr.s_addr = INADDR_ANY;
      addr.sin_port = htons((uint16_t)port);
      NZ_CHECK(bind(srv, (sockaddr*)&addr, sizeof(addr)) == 0,
               std::string("bind failed: ") + strerror(errno));
      NZ_CHECK(listen(srv, nranks) == 0, "listen failed");
      fds_.resize(nranks, -1);
      for (int i = 1; i < nranks; ++i) {
        int fd = accept(srv, nullptr, nullptr);
        NZ_CHECK(fd >= 0, "accept failed");
        int r;
        recv_all(fd, &r, 4);
        fds_[r] = fd;
        int on = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
      }
      close(srv);
    } else {
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_port = htons((uint16_t)port);
      inet_pton(AF_INET, host, &addr.sin_addr);
      int fd = -1;
      for (int attempt = 0; attempt < 200; ++attempt) {  // hub may start late
        fd = socket(AF_INET, SOCK_STREAM, 0);
        if (connect(fd, (sockaddr*)&addr, sizeof(addr)) == 0) break;
        close(fd);
        fd = -1;
        usleep(50 * 1000);
      }
      NZ_CHECK(fd >= 0, "connect to hub failed");
      send_all(fd, &rank_, 4);
      int on = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
      hub_fd_ = fd;
    }
  }
  ~TcpComm() override {
    if (hub_fd_ >= 0) close(hub_fd_);
    for (int fd : fds_)
      if (fd >= 0) close(fd);
  }
  int rank() const override { return rank_; }
  int nranks() const override { return nranks_; }

  void all_reduce(void* data, int64_t count, DType dt, ReduceOp op,
                  Stream*) override {
    if (nranks_ == 1) return;
    size_t bytes = size_t(count) * dtype_size(dt);
    if (rank_ == 0) {
      std::vector<char> tmp(bytes);
      std::vector<float> acc(count);
      to_f32(data, dt, acc.data(), count);
      for (int r = 1; r < nranks_; ++r) {
        recv_all(fds_[r], tmp.data(), bytes);
        std::vector<float> other(count);
        to_f32(tmp.data(), dt, other.data(), count);
        for (int64_t i = 0; i < count; ++i)
          acc[i] = op == ReduceOp::SUM ? acc[i] + other[i]
                                       : std::max(acc[i], other[i]);
      }
      from_f32(acc.data(), dt, data, count);
      for (int r = 1; r < nranks_; ++r) send_all(fds_[r], data, bytes);
    } else {
      send_all(hub_fd_, data, bytes);
      recv_all(hub_fd_, data, bytes);
    }
  }

  void broadcast(void* data, int64_t count, DType dt, int root, Stream*) override {
    if (nranks_ == 1) return;
    NZ_CHECK(root == 0, "TcpComm broadcast supports root 0 only");
    size_t bytes = size_t(count) * dtype_size(dt);
    if (rank_ == 0) {
      for (int r = 1; r < nranks_; ++r) send_all(fds_[r], data, bytes);
    } else {
      recv_all(hub_fd_, data, bytes);
    }
  }

  void barrier() override { (void)all_reduce_scalar(0.f, ReduceOp::SUM); }

  float all_reduce_scalar(float v, ReduceOp op) override {
    float buf = v;
    all_reduce(&buf, 1, DType::F32, op, nullptr);
    return buf;
  }

 private:
  static void to_f32(const void* src, DType dt, float* dst, int64_t n) {
    if (dt == DType::F32) std::memcpy(dst, src, n * 4);
    else
      for (int64_t i = 0; i < n; ++i) dst[i] = bf16_to_f32(((const uint16_t*)src)[i]);
  }
  static void from_f32(const float* src, DType dt, void* dst, int64_t n) {
    if (dt == DType::F32) std::memcpy(dst, src, n * 4);
    else
      for (int64_t i = 0; i < n; ++i) ((uint16_t*)dst)[i] = f32_to_bf16(src[i]);
  }
  int nranks_, rank_;
  int hub_fd_ = -1;
  std::vector<int> fds_;  // rank 0 only
};

}  // namespace

std::unique_ptr<Comm> make_tcp_comm(int nranks, int rank, const char* host, int port) {
  return std::make_unique<TcpComm>(nranks, rank, host, port);
}

}  // namespace nz
