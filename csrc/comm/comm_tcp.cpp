// TCP transport: rendezvous + hub-and-spoke collectives through rank 0.
//
// This is the CPU fallback and the RCCL bootstrap (comm.h).  Message
// volumes on this path are one flux array per batch or a handful of
// bootstrap bytes, so hub-and-spoke latency is irrelevant; what matters
// is that it works with zero external dependencies (no MPI, no gloo) so
// a C++ host app gets multi-process tallies from the library alone.
#include "comm.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <thread>

namespace pumitally {

void Comm::allreduce_sum_device(double *, int64_t) {
  throw std::runtime_error("device collectives require the RCCL comm");
}
int64_t Comm::alltoallv_device(const double *, const std::vector<int64_t> &,
                               const std::vector<int64_t> &, double **) {
  throw std::runtime_error("device collectives require the RCCL comm");
}

namespace {

[[noreturn]] void die(const std::string &msg) {
  throw std::runtime_error("TcpComm: " + msg + " (" + strerror(errno) + ")");
}

void send_all(int fd, const void *buf, int64_t n) {
  const char *p = (const char *)buf;
  while (n > 0) {
    const ssize_t k = ::send(fd, p, (size_t)n, MSG_NOSIGNAL);
    if (k < 0 && errno == EINTR) continue;
    if (k <= 0) die("send failed");
    p += k;
    n -= k;
  }
}

void recv_all(int fd, void *buf, int64_t n) {
  char *p = (char *)buf;
  while (n > 0) {
    const ssize_t k = ::recv(fd, p, (size_t)n, 0);
    if (k < 0 && errno == EINTR) continue;
    if (k <= 0) die("recv failed (peer closed?)");
    p += k;
    n -= k;
  }
}

// Every comm a process creates rendezvouses on its own derived port:
// ranks construct comms in identical program order, so the Nth comm of
// every rank agrees on port+13N and two back-to-back comms can never
// cross-wire their handshakes (a fast rank's connection for comm N+1
// arriving at comm N's listener).
std::atomic<int> g_comm_seq{0};

int connect_retry(const std::string &addr, int port, double timeout_s) {
  addrinfo hints{};
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  addrinfo *res = nullptr;
  const std::string ps = std::to_string(port);
  if (getaddrinfo(addr.c_str(), ps.c_str(), &hints, &res) != 0 || !res)
    die("getaddrinfo(" + addr + ") failed");
  const auto t0 = std::chrono::steady_clock::now();
  for (;;) {
    const int fd = socket(res->ai_family, res->ai_socktype, res->ai_protocol);
    if (fd < 0) die("socket failed");
    if (::connect(fd, res->ai_addr, res->ai_addrlen) == 0) {
      freeaddrinfo(res);
      const int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
      return fd;
    }
    ::close(fd);
    const double el = std::chrono::duration<double>(
                          std::chrono::steady_clock::now() - t0)
                          .count();
    if (el > timeout_s) {
      freeaddrinfo(res);
      die("connect to " + addr + ":" + ps + " timed out after " +
          std::to_string((int)timeout_s) + "s");
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(200));
  }
}

class TcpComm final : public Comm {
public:
  TcpComm(int rank, int world, const std::string &addr, int port)
      : rank_(rank), world_(world) {
    if (world_ < 2) return; // degenerate; all ops become no-ops
    port += 13 * (g_comm_seq.fetch_add(1) % 97);
    if (rank_ == 0) {
      const int lfd = socket(AF_INET, SOCK_STREAM, 0);
      if (lfd < 0) die("socket failed");
      const int one = 1;
      setsockopt(lfd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
      sockaddr_in sa{};
      sa.sin_family = AF_INET;
      sa.sin_addr.s_addr = htonl(INADDR_ANY);
      sa.sin_port = htons((uint16_t)port);
      if (bind(lfd, (sockaddr *)&sa, sizeof sa) != 0)
        die("bind to port " + std::to_string(port) +
            " failed (set PUMITALLY_PORT to a free port)");
      if (listen(lfd, world_) != 0) die("listen failed");
      peer_.assign(world_, -1);
      for (int i = 1; i < world_; ++i) {
        const int fd = accept(lfd, nullptr, nullptr);
        if (fd < 0) die("accept failed");
        const int one2 = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one2, sizeof one2);
        int32_t r = -1;
        recv_all(fd, &r, 4);
        if (r < 1 || r >= world_ || peer_[r] != -1)
          throw std::runtime_error("TcpComm: bad peer rank handshake");
        peer_[r] = fd;
      }
      ::close(lfd);
    } else {
      hub_ = connect_retry(addr, port, 120.0);
      const int32_t r = rank_;
      send_all(hub_, &r, 4);
    }
  }

  ~TcpComm() override {
    for (int fd : peer_)
      if (fd >= 0) ::close(fd);
    if (hub_ >= 0) ::close(hub_);
  }

  int rank() const override { return rank_; }
  int world() const override { return world_; }

  void barrier() override {
    char b = 0;
    if (world_ < 2) return;
    if (rank_ == 0) {
      for (int i = 1; i < world_; ++i) recv_all(peer_[i], &b, 1);
      for (int i = 1; i < world_; ++i) send_all(peer_[i], &b, 1);
    } else {
      send_all(hub_, &b, 1);
      recv_all(hub_, &b, 1);
    }
  }

  template <class T> void allreduce_sum_t(T *data, int64_t n) {
    if (world_ < 2) return;
    if (rank_ == 0) {
      std::vector<T> tmp(n);
      for (int i = 1; i < world_; ++i) {
        recv_all(peer_[i], tmp.data(), n * (int64_t)sizeof(T));
        for (int64_t k = 0; k < n; ++k) data[k] += tmp[k];
      }
      for (int i = 1; i < world_; ++i)
        send_all(peer_[i], data, n * (int64_t)sizeof(T));
    } else {
      send_all(hub_, data, n * (int64_t)sizeof(T));
      recv_all(hub_, data, n * (int64_t)sizeof(T));
    }
  }

  void allreduce_sum(double *data, int64_t n) override {
    allreduce_sum_t(data, n);
  }
  void allreduce_sum(int64_t *data, int64_t n) override {
    allreduce_sum_t(data, n);
  }

  void allreduce_max(double *data, int64_t n) override {
    if (world_ < 2) return;
    if (rank_ == 0) {
      std::vector<double> tmp(n);
      for (int i = 1; i < world_; ++i) {
        recv_all(peer_[i], tmp.data(), n * 8);
        for (int64_t k = 0; k < n; ++k)
          if (tmp[k] > data[k]) data[k] = tmp[k];
      }
      for (int i = 1; i < world_; ++i) send_all(peer_[i], data, n * 8);
    } else {
      send_all(hub_, data, n * 8);
      recv_all(hub_, data, n * 8);
    }
  }

  void bcast(void *data, int64_t bytes, int root) override {
    if (world_ < 2) return;
    if (rank_ == 0) {
      if (root != 0) recv_all(peer_[root], data, bytes);
      for (int i = 1; i < world_; ++i)
        if (i != root) send_all(peer_[i], data, bytes);
    } else if (rank_ == root) {
      send_all(hub_, data, bytes);
    } else {
      recv_all(hub_, data, bytes);
    }
  }

  std::vector<int64_t> allgather(int64_t v) override {
    std::vector<int64_t> out(world_, v);
    if (world_ < 2) return out;
    if (rank_ == 0) {
      for (int i = 1; i < world_; ++i) recv_all(peer_[i], &out[i], 8);
      for (int i = 1; i < world_; ++i)
        send_all(peer_[i], out.data(), world_ * 8);
    } else {
      send_all(hub_, &v, 8);
      recv_all(hub_, out.data(), world_ * 8);
    }
    return out;
  }

  std::vector<double> alltoallv(
      const double *send, const std::vector<int64_t> &send_counts) override {
    if ((int)send_counts.size() != world_)
      throw std::runtime_error("alltoallv: send_counts size != world");
    if (world_ < 2)
      return {send, send + (send_counts.empty() ? 0 : send_counts[0])};
    if (rank_ == 0) {
      // gather every source's (counts row, data)
      std::vector<std::vector<int64_t>> cnt(world_);
      std::vector<std::vector<double>> dat(world_);
      cnt[0] = send_counts;
      {
        int64_t tot = 0;
        for (int64_t c : send_counts) tot += c;
        dat[0].assign(send, send + tot);
      }
      for (int i = 1; i < world_; ++i) {
        cnt[i].resize(world_);
        recv_all(peer_[i], cnt[i].data(), world_ * 8);
        int64_t tot = 0;
        for (int64_t c : cnt[i]) tot += c;
        dat[i].resize(tot);
        if (tot) recv_all(peer_[i], dat[i].data(), tot * 8);
      }
      // redistribute: to dest r, the per-source counts then the blocks
      std::vector<double> mine;
      for (int r = 0; r < world_; ++r) {
        std::vector<int64_t> rc(world_);
        for (int s = 0; s < world_; ++s) rc[s] = cnt[s][r];
        std::vector<double> block;
        for (int s = 0; s < world_; ++s) {
          int64_t off = 0;
          for (int q = 0; q < r; ++q) off += cnt[s][q];
          block.insert(block.end(), dat[s].begin() + off,
                       dat[s].begin() + off + cnt[s][r]);
        }
        if (r == 0) {
          mine = std::move(block);
        } else {
          send_all(peer_[r], rc.data(), world_ * 8);
          if (!block.empty())
            send_all(peer_[r], block.data(), (int64_t)block.size() * 8);
        }
      }
      return mine;
    }
    send_all(hub_, send_counts.data(), world_ * 8);
    int64_t tot = 0;
    for (int64_t c : send_counts) tot += c;
    if (tot) send_all(hub_, send, tot * 8);
    std::vector<int64_t> rc(world_);
    recv_all(hub_, rc.data(), world_ * 8);
    int64_t rtot = 0;
    for (int64_t c : rc) rtot += c;
    std::vector<double> out(rtot);
    if (rtot) recv_all(hub_, out.data(), rtot * 8);
    return out;
  }

private:
  int rank_, world_;
  int hub_ = -1;              // nonzero ranks: socket to rank 0
  std::vector<int> peer_;     // rank 0: socket per peer rank
};

} // namespace

std::unique_ptr<Comm> make_tcp_comm(int rank, int world,
                                    const std::string &addr, int port) {
  return std::make_unique<TcpComm>(rank, world, addr, port);
}

// Weak fallback so CPU-only builds (sanitizer jobs, clang-tidy compile
// database) link without the HIP TU; the strong definition in
// comm_rccl.hip wins whenever it is in the link.
__attribute__((weak)) std::unique_ptr<Comm>
make_rccl_comm(int, int, const std::string &, int, int) {
  return nullptr;
}

EnvComm comm_env() {
  EnvComm e;
  const char *r = getenv("RANK");
  const char *w = getenv("WORLD_SIZE");
  e.rank = r ? atoi(r) : 0;
  e.world = w ? atoi(w) : 1;
  const char *lr = getenv("LOCAL_RANK");
  e.local_rank = lr ? atoi(lr) : e.rank;
  const char *a = getenv("MASTER_ADDR");
  e.addr = a ? a : "127.0.0.1";
  const char *pp = getenv("PUMITALLY_PORT");
  if (pp) {
    e.port = atoi(pp);
  } else {
    const char *mp = getenv("MASTER_PORT");
    // offset so the comm port never collides with a torchrun rendezvous
    // holding MASTER_PORT itself
    e.port = (mp ? atoi(mp) : 29500) + 371;
  }
  return e;
}

std::unique_ptr<Comm> make_comm_from_env(bool want_gpu, int device) {
  const EnvComm e = comm_env();
  if (e.world <= 1) return nullptr;
  const char *f = getenv("PUMITALLY_COMM");
  const bool force_tcp = f && std::string(f) == "tcp";
  if (want_gpu && !force_tcp) {
    auto c = make_rccl_comm(e.rank, e.world, e.addr, e.port, device);
    if (c) return c;
  }
  return make_tcp_comm(e.rank, e.world, e.addr, e.port);
}

} // namespace pumitally
