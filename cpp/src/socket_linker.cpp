/*! migbm TCP socket linker — standalone distributed collectives for CPU training
 *  with no external runtime (capability parity: reference src/network/
 *  linkers_socket.cpp + network.cpp ring collectives; fresh implementation).
 *  Full TCP mesh between the machines listed in `machines` ("ip:port,ip:port,..."),
 *  ring allgather on top, plugged into the same migbm::Network seam the injected
 *  (gloo) backend uses. Exposed as LGBM_NetworkInit for reference C-API parity;
 *  the CLI (`num_machines>1` + machine_list) and Dask (`_train_part`) use it. */
#include "migbm/network.h"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

namespace migbm {

namespace {

struct Peer {
  std::string ip;
  int port = 0;
};

class SocketLinker {
 public:
  static SocketLinker& Get() {
    static SocketLinker l;
    return l;
  }

  /*! machines = "ip:port,ip:port,..." (also accepts "ip port" entries).
   *  This machine is the entry whose port == local_listen_port (and, when
   *  several entries share the port, whose ip is local). */
  void Init(const std::string& machines, int local_listen_port, int timeout_sec,
            int num_machines) {
    Free();
    std::vector<Peer> peers;
    std::string tok;
    auto flush = [&]() {
      if (tok.empty()) return;
      Peer p;
      size_t sep = tok.find_first_of(": ");
      if (sep == std::string::npos) Log::Fatal("Bad machine entry '%s'", tok.c_str());
      p.ip = tok.substr(0, sep);
      p.port = atoi(tok.c_str() + sep + 1);
      peers.push_back(p);
      tok.clear();
    };
    for (char c : machines + ",") {
      if (c == ',' || c == '\n') flush();
      else tok += c;
    }
    if (static_cast<int>(peers.size()) < num_machines)
      Log::Fatal("machines lists %d entries but num_machines=%d",
                 static_cast<int>(peers.size()), num_machines);
    peers.resize(num_machines);
    world_ = num_machines;
    rank_ = -1;
    for (int r = 0; r < world_; ++r) {
      if (peers[r].port == local_listen_port) {
        rank_ = r;
        break;
      }
    }
    if (rank_ < 0)
      Log::Fatal("local_listen_port=%d not found in the machine list", local_listen_port);

    // listen socket
    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = INADDR_ANY;
    addr.sin_port = htons(static_cast<uint16_t>(local_listen_port));
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0)
      Log::Fatal("Cannot bind listen port %d", local_listen_port);
    if (listen(listen_fd_, world_) != 0) Log::Fatal("listen() failed");

    fds_.assign(world_, -1);
    const auto deadline =
        std::chrono::steady_clock::now() + std::chrono::seconds(std::max(5, timeout_sec));
    // accept from lower ranks on a helper thread while connecting to higher ones
    std::thread acceptor([&] {
      for (int need = rank_; need > 0; --need) {
        int fd = accept(listen_fd_, nullptr, nullptr);
        if (fd < 0) return;
        int peer_rank = -1;
        if (RecvAll(fd, reinterpret_cast<char*>(&peer_rank), sizeof(int)) &&
            peer_rank >= 0 && peer_rank < world_) {
          SetNoDelay(fd);
          fds_[peer_rank] = fd;
        } else {
          close(fd);
          ++need;  // retry this slot
        }
      }
    });
    bool ok = true;
    for (int r = rank_ + 1; r < world_ && ok; ++r) {
      int fd = -1;
      for (;;) {
        fd = socket(AF_INET, SOCK_STREAM, 0);
        sockaddr_in pa{};
        pa.sin_family = AF_INET;
        pa.sin_port = htons(static_cast<uint16_t>(peers[r].port));
        inet_pton(AF_INET, peers[r].ip.c_str(), &pa.sin_addr);
        if (connect(fd, reinterpret_cast<sockaddr*>(&pa), sizeof(pa)) == 0) break;
        close(fd);
        fd = -1;
        if (std::chrono::steady_clock::now() > deadline) {
          ok = false;
          break;
        }
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
      }
      if (fd >= 0) {
        SendAll(fd, reinterpret_cast<const char*>(&rank_), sizeof(int));
        SetNoDelay(fd);
        fds_[r] = fd;
      }
    }
    acceptor.join();
    close(listen_fd_);
    listen_fd_ = -1;
    if (!ok) Log::Fatal("Socket mesh connect timed out (%d s)", timeout_sec);
    for (int r = 0; r < world_; ++r)
      if (r != rank_ && fds_[r] < 0)
        Log::Fatal("Socket mesh incomplete: no link to rank %d", r);
    Log::Info("Socket mesh up: rank %d / %d machines", rank_, world_);
  }

  void Free() {
    for (int fd : fds_)
      if (fd >= 0) close(fd);
    fds_.clear();
    if (listen_fd_ >= 0) close(listen_fd_);
    listen_fd_ = -1;
    world_ = 1;
    rank_ = 0;
  }

  bool active() const { return world_ > 1; }
  int world() const { return world_; }
  int rank() const { return rank_; }

  /*! ring allgather: world-1 steps, send to (rank+1), recv from (rank-1);
   *  full-duplex via a writer thread so large blocks cannot deadlock. */
  void Allgather(const char* input, int size, char* output) {
    memcpy(output + static_cast<size_t>(rank_) * size, input, size);
    const int next = (rank_ + 1) % world_;
    const int prev = (rank_ + world_ - 1) % world_;
    for (int s = 0; s < world_ - 1; ++s) {
      const int send_block = (rank_ - s + world_) % world_;
      const int recv_block = (rank_ - s - 1 + world_) % world_;
      const char* src = output + static_cast<size_t>(send_block) * size;
      char* dst = output + static_cast<size_t>(recv_block) * size;
      std::thread writer([&] { SendAll(fds_[next], src, size); });
      if (!RecvAll(fds_[prev], dst, size)) Log::Fatal("Socket recv failed");
      writer.join();
    }
  }

 private:
  static void SetNoDelay(int fd) {
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  }
  static bool SendAll(int fd, const char* buf, size_t n) {
    size_t off = 0;
    while (off < n) {
      ssize_t w = send(fd, buf + off, n - off, 0);
      if (w <= 0) return false;
      off += static_cast<size_t>(w);
    }
    return true;
  }
  static bool RecvAll(int fd, char* buf, size_t n) {
    size_t off = 0;
    while (off < n) {
      ssize_t r = recv(fd, buf + off, n - off, 0);
      if (r <= 0) return false;
      off += static_cast<size_t>(r);
    }
    return true;
  }

  int world_ = 1;
  int rank_ = 0;
  int listen_fd_ = -1;
  std::vector<int> fds_;
};

void SocketAllgather(const char* input, int size, char* output) {
  SocketLinker::Get().Allgather(input, size, output);
}

}  // namespace

void NetworkInitSockets(const std::string& machines, int local_listen_port,
                        int timeout_sec, int num_machines) {
  SocketLinker::Get().Init(machines, local_listen_port, timeout_sec, num_machines);
  Network::Init(num_machines, SocketLinker::Get().rank(), SocketAllgather);
}

void NetworkFreeSockets() { SocketLinker::Get().Free(); }

}  // namespace migbm
