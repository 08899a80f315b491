// rcclx — native RCCL-over-xGMI backend for dist_tuto_pth_amd.
//
// Owns the layer the reference tutorial inherits from PyTorch 0.x THD
// (tuto.md:404-419): a TCP key-value store for rendezvous (the
// master/worker handshake of tuto.md:409-418 — rank 0 hosts, workers
// connect and exchange through it), ncclUniqueId broadcast,
// ncclCommInitRank (one rank per MI355X), the six collectives
// (tuto.md:197-202), point-to-point send/recv (tuto.md:87-112), the four
// reduce ops (tuto.md:190-193) and sub-communicators via ncclCommSplit
// (tuto.md:182-184).
//
// Pure HIP + RCCL + POSIX sockets; no torch headers (tensors arrive as
// raw device pointers + dtype/count from the Python wrapper), no CUDA
// compatibility paths.  Build: hipcc --offload-arch=gfx950 (build.py).

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <sys/select.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    if (e_ != hipSuccess)                                                     \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e_) + " @ " #cmd);           \
  } while (0)

#define NCCL_CHECK(cmd)                                                       \
  do {                                                                        \
    ncclResult_t r_ = (cmd);                                                  \
    if (r_ != ncclSuccess)                                                    \
      throw std::runtime_error(std::string("RCCL error: ") +                  \
                               ncclGetErrorString(r_) + " @ " #cmd);          \
  } while (0)

// ---------------------------------------------------------------------------
// TCP store: the rendezvous the tutorial describes in prose
// (tuto.md:409-419) — rank 0 is the master holding a key/value table;
// every rank (incl. 0) connects as a client.  Wire protocol:
//   SET : u8(1) u32 klen key u32 vlen val            -> u8(1)
//   GET : u8(2) u32 klen key u32 timeout_ms          -> u32 vlen val
//         (vlen == 0xFFFFFFFF signals timeout)
//   ADD : u8(3) u32 klen key i64 delta               -> i64 new_value
// ---------------------------------------------------------------------------
namespace {

void send_all(int fd, const void* buf, size_t n) {
  const char* p = static_cast<const char*>(buf);
  while (n) {
    ssize_t k = ::send(fd, p, n, MSG_NOSIGNAL);
    if (k <= 0) throw std::runtime_error("store: send failed");
    p += k;
    n -= static_cast<size_t>(k);
  }
}

bool recv_all(int fd, void* buf, size_t n) {
  char* p = static_cast<char*>(buf);
  while (n) {
    ssize_t k = ::recv(fd, p, n, 0);
    if (k <= 0) return false;
    p += k;
    n -= static_cast<size_t>(k);
  }
  return true;
}

}  // namespace

class StoreServer {
 public:
  StoreServer(const std::string& host, int port, int expected_clients)
      : stop_(false) {
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("store: socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(port));
    addr.sin_addr.s_addr = INADDR_ANY;
    if (::bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)))
      throw std::runtime_error("store: bind failed on port " +
                               std::to_string(port));
    if (::listen(listen_fd_, 128))
      throw std::runtime_error("store: listen failed");
    accept_thread_ = std::thread([this] { accept_loop(); });
    (void)expected_clients;
  }

  ~StoreServer() {
    stop_ = true;
    cv_.notify_all();  // unblock pending GET waits
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> lk(threads_mu_);
      for (int fd : client_fds_) ::shutdown(fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    for (auto& t : client_threads_)
      if (t.joinable()) t.join();
  }

 private:
  void accept_loop() {
    while (!stop_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (stop_) break;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      std::lock_guard<std::mutex> lk(threads_mu_);
      client_fds_.push_back(fd);
      client_threads_.emplace_back([this, fd] { serve(fd); });
    }
  }

  void serve(int fd) {
    for (;;) {
      uint8_t op;
      if (!recv_all(fd, &op, 1)) break;
      uint32_t klen;
      if (!recv_all(fd, &klen, 4)) break;
      std::string key(klen, '\0');
      if (!recv_all(fd, key.data(), klen)) break;
      if (op == 1) {  // SET
        uint32_t vlen;
        if (!recv_all(fd, &vlen, 4)) break;
        std::vector<char> val(vlen);
        if (vlen && !recv_all(fd, val.data(), vlen)) break;
        {
          std::lock_guard<std::mutex> lk(mu_);
          table_[key] = std::move(val);
        }
        cv_.notify_all();
        uint8_t ok = 1;
        send_all(fd, &ok, 1);
      } else if (op == 2) {  // GET (blocking w/ timeout)
        uint32_t timeout_ms;
        if (!recv_all(fd, &timeout_ms, 4)) break;
        std::vector<char> val;
        bool found = false;
        {
          std::unique_lock<std::mutex> lk(mu_);
          cv_.wait_for(
              lk, std::chrono::milliseconds(timeout_ms), [&] {
                return stop_ || table_.count(key) > 0;
              });
          found = table_.count(key) > 0;
          if (found) val = table_[key];
        }
        uint32_t vlen = found ? static_cast<uint32_t>(val.size())
                              : 0xFFFFFFFFu;
        send_all(fd, &vlen, 4);
        if (found && !val.empty()) send_all(fd, val.data(), val.size());
      } else if (op == 3) {  // ADD
        int64_t delta;
        if (!recv_all(fd, &delta, 8)) break;
        int64_t nv;
        {
          std::lock_guard<std::mutex> lk(mu_);
          int64_t cur = 0;
          auto it = table_.find(key);
          if (it != table_.end() && it->second.size() == 8)
            memcpy(&cur, it->second.data(), 8);
          nv = cur + delta;
          std::vector<char> val(8);
          memcpy(val.data(), &nv, 8);
          table_[key] = std::move(val);
        }
        cv_.notify_all();
        send_all(fd, &nv, 8);
      } else {
        break;
      }
    }
    ::close(fd);
  }

  int listen_fd_;
  std::atomic<bool> stop_;
  std::thread accept_thread_;
  std::mutex threads_mu_;
  std::vector<std::thread> client_threads_;
  std::vector<int> client_fds_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::map<std::string, std::vector<char>> table_;
};

class TcpStore {
 public:
  TcpStore(const std::string& host, int port, int rank, int world,
           bool is_server, int timeout_ms)
      : timeout_ms_(timeout_ms) {
    if (is_server)
      server_.reset(new StoreServer(host, port, world));
    // connect as client (with retries: workers may race the master,
    // tuto.md:414 "workers wait for the master")
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::milliseconds(timeout_ms);
    for (;;) {
      fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_port = htons(static_cast<uint16_t>(port));
      hostent* he = gethostbyname(host.c_str());
      if (he == nullptr)
        throw std::runtime_error("store: cannot resolve " + host);
      memcpy(&addr.sin_addr, he->h_addr_list[0], he->h_length);
      if (::connect(fd_, reinterpret_cast<sockaddr*>(&addr),
                    sizeof(addr)) == 0)
        break;
      ::close(fd_);
      if (std::chrono::steady_clock::now() > deadline)
        throw std::runtime_error("store: connect timed out to " + host +
                                 ":" + std::to_string(port));
      std::this_thread::sleep_for(std::chrono::milliseconds(50));
    }
    int one = 1;
    setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  }

  ~TcpStore() { ::close(fd_); }

  void set(const std::string& key, const py::bytes& value) {
    std::string v = value;
    std::lock_guard<std::mutex> lk(mu_);
    uint8_t op = 1;
    uint32_t klen = key.size(), vlen = v.size();
    send_all(fd_, &op, 1);
    send_all(fd_, &klen, 4);
    send_all(fd_, key.data(), klen);
    send_all(fd_, &vlen, 4);
    if (vlen) send_all(fd_, v.data(), vlen);
    uint8_t ok;
    if (!recv_all(fd_, &ok, 1)) throw std::runtime_error("store: set failed");
  }

  py::bytes get(const std::string& key) {
    std::vector<char> val;
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      uint8_t op = 2;
      uint32_t klen = key.size(), to = timeout_ms_;
      send_all(fd_, &op, 1);
      send_all(fd_, &klen, 4);
      send_all(fd_, key.data(), klen);
      send_all(fd_, &to, 4);
      uint32_t vlen;
      if (!recv_all(fd_, &vlen, 4))
        throw std::runtime_error("store: get failed");
      if (vlen == 0xFFFFFFFFu)
        throw std::runtime_error("store: get('" + key + "') timed out");
      val.resize(vlen);
      if (vlen && !recv_all(fd_, val.data(), vlen))
        throw std::runtime_error("store: get payload failed");
    }
    return py::bytes(val.data(), val.size());
  }

  int64_t add(const std::string& key, int64_t delta) {
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(mu_);
    uint8_t op = 3;
    uint32_t klen = key.size();
    send_all(fd_, &op, 1);
    send_all(fd_, &klen, 4);
    send_all(fd_, key.data(), klen);
    send_all(fd_, &delta, 8);
    int64_t nv;
    if (!recv_all(fd_, &nv, 8)) throw std::runtime_error("store: add failed");
    return nv;
  }

 private:
  std::unique_ptr<StoreServer> server_;
  int fd_;
  int timeout_ms_;
  std::mutex mu_;
};

// ---------------------------------------------------------------------------
// RCCL communicator
// ---------------------------------------------------------------------------
static ncclDataType_t as_dtype(int d) { return static_cast<ncclDataType_t>(d); }
static ncclRedOp_t as_op(int o) { return static_cast<ncclRedOp_t>(o); }

class Comm {
 public:
  Comm(int nranks, int rank, const py::bytes& uid_bytes, int device)
      : nranks_(nranks), rank_(rank), device_(device) {
    std::string uid_s = uid_bytes;
    if (uid_s.size() != sizeof(ncclUniqueId))
      throw std::runtime_error("bad ncclUniqueId size");
    ncclUniqueId uid;
    memcpy(&uid, uid_s.data(), sizeof(uid));
    py::gil_scoped_release nogil;
    HIP_CHECK(hipSetDevice(device));
    NCCL_CHECK(ncclCommInitRank(&comm_, nranks, uid, rank));
  }

  Comm(ncclComm_t c, int nranks, int rank, int device)
      : comm_(c), nranks_(nranks), rank_(rank), device_(device) {}

  void destroy() {
    if (comm_) {
      py::gil_scoped_release nogil;
      ncclCommDestroy(comm_);
      comm_ = nullptr;
    }
  }

  Comm* split(int color, int key) {
    ncclComm_t nc = nullptr;
    {
      py::gil_scoped_release nogil;
      NCCL_CHECK(ncclCommSplit(comm_,
                               color < 0 ? NCCL_SPLIT_NOCOLOR : color, key,
                               &nc, nullptr));
    }
    if (color < 0) return nullptr;
    int n, r;
    NCCL_CHECK(ncclCommCount(nc, &n));
    NCCL_CHECK(ncclCommUserRank(nc, &r));
    return new Comm(nc, n, r, device_);
  }

  // ---- collectives (tuto.md:197-202) --------------------------------
  void all_reduce(uintptr_t send, uintptr_t recv, size_t count, int dtype,
                  int op, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclAllReduce(reinterpret_cast<void*>(send),
                             reinterpret_cast<void*>(recv), count,
                             as_dtype(dtype), as_op(op), comm_,
                             reinterpret_cast<hipStream_t>(stream)));
  }

  void broadcast(uintptr_t send, uintptr_t recv, size_t count, int dtype,
                 int root, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclBroadcast(reinterpret_cast<void*>(send),
                             reinterpret_cast<void*>(recv), count,
                             as_dtype(dtype), root, comm_,
                             reinterpret_cast<hipStream_t>(stream)));
  }

  void reduce(uintptr_t send, uintptr_t recv, size_t count, int dtype,
              int op, int root, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclReduce(reinterpret_cast<void*>(send),
                          reinterpret_cast<void*>(recv), count,
                          as_dtype(dtype), as_op(op), root, comm_,
                          reinterpret_cast<hipStream_t>(stream)));
  }

  void all_gather(uintptr_t send, uintptr_t recv, size_t sendcount,
                  int dtype, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclAllGather(reinterpret_cast<void*>(send),
                             reinterpret_cast<void*>(recv), sendcount,
                             as_dtype(dtype), comm_,
                             reinterpret_cast<hipStream_t>(stream)));
  }

  void reduce_scatter(uintptr_t send, uintptr_t recv, size_t recvcount,
                      int dtype, int op, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclReduceScatter(reinterpret_cast<void*>(send),
                                 reinterpret_cast<void*>(recv), recvcount,
                                 as_dtype(dtype), as_op(op), comm_,
                                 reinterpret_cast<hipStream_t>(stream)));
  }

  void gather(uintptr_t send, uintptr_t recv, size_t sendcount, int dtype,
              int root, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclGather(reinterpret_cast<void*>(send),
                          reinterpret_cast<void*>(recv), sendcount,
                          as_dtype(dtype), root, comm_,
                          reinterpret_cast<hipStream_t>(stream)));
  }

  void scatter(uintptr_t send, uintptr_t recv, size_t recvcount, int dtype,
               int root, uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclScatter(reinterpret_cast<void*>(send),
                           reinterpret_cast<void*>(recv), recvcount,
                           as_dtype(dtype), root, comm_,
                           reinterpret_cast<hipStream_t>(stream)));
  }

  void all_to_all(uintptr_t send, uintptr_t recv, size_t count, int dtype,
                  uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclAllToAll(reinterpret_cast<void*>(send),
                            reinterpret_cast<void*>(recv), count,
                            as_dtype(dtype), comm_,
                            reinterpret_cast<hipStream_t>(stream)));
  }

  // ---- p2p (tuto.md:87-112; ring transport C5-C8) --------------------
  void send(uintptr_t buf, size_t count, int dtype, int peer,
            uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclSend(reinterpret_cast<void*>(buf), count,
                        as_dtype(dtype), peer, comm_,
                        reinterpret_cast<hipStream_t>(stream)));
  }

  void recv(uintptr_t buf, size_t count, int dtype, int peer,
            uintptr_t stream) {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclRecv(reinterpret_cast<void*>(buf), count,
                        as_dtype(dtype), peer, comm_,
                        reinterpret_cast<hipStream_t>(stream)));
  }

  void group_start() { NCCL_CHECK(ncclGroupStart()); }
  void group_end() {
    py::gil_scoped_release nogil;
    NCCL_CHECK(ncclGroupEnd());
  }

  int rank() const { return rank_; }
  int nranks() const { return nranks_; }
  int device() const { return device_; }

 private:
  ncclComm_t comm_ = nullptr;
  int nranks_, rank_, device_;
};

// ---------------------------------------------------------------------------
// TcpMesh — the full-mesh CPU p2p layer of the native "tcp" backend.
// This is the part of THD the tutorial documents in prose
// (tuto.md:409-419): every pair of ranks holds one TCP connection,
// established master/worker-style (rank i accepts from all j > i and
// connects to all j < i; addresses travel through the TCP store).
// Messages are length-framed; targeted recv reads the pair socket
// (per-peer FIFO ordering), any-source recv polls all sockets.  Reads
// and writes move raw bytes to/from caller-provided pointers (zero
// copy from contiguous CPU tensors).
// ---------------------------------------------------------------------------
class TcpMesh {
 public:
  TcpMesh(TcpStore* store, const std::string& tag, int rank, int world,
          int timeout_ms)
      : rank_(rank), world_(world) {
    fds_.assign(world, -1);
    send_mu_ = std::vector<std::mutex>(world);
    recv_mu_ = std::vector<std::mutex>(world);
    if (world == 1) return;
    // bind an ephemeral listening socket and publish its port
    int lfd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (lfd < 0) throw std::runtime_error("mesh: socket() failed");
    int one = 1;
    setsockopt(lfd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = 0;
    addr.sin_addr.s_addr = INADDR_ANY;
    if (::bind(lfd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)))
      throw std::runtime_error("mesh: bind failed");
    socklen_t alen = sizeof(addr);
    getsockname(lfd, reinterpret_cast<sockaddr*>(&addr), &alen);
    int my_port = ntohs(addr.sin_port);
    if (::listen(lfd, world))
      throw std::runtime_error("mesh: listen failed");
    {
      py::gil_scoped_acquire gil;  // store API touches py::bytes
      store->set(tag + ":addr:" + std::to_string(rank),
                 py::bytes(std::to_string(my_port)));
    }
    // connect to lower ranks (each connection opens with our rank id)
    for (int j = 0; j < rank; ++j) {
      std::string port_s;
      {
        py::gil_scoped_acquire gil;
        port_s = std::string(store->get(tag + ":addr:" + std::to_string(j)));
      }
      int port = std::stoi(port_s);
      auto deadline = std::chrono::steady_clock::now() +
                      std::chrono::milliseconds(timeout_ms);
      int fd;
      for (;;) {
        fd = ::socket(AF_INET, SOCK_STREAM, 0);
        sockaddr_in a{};
        a.sin_family = AF_INET;
        a.sin_port = htons(static_cast<uint16_t>(port));
        a.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
        if (::connect(fd, reinterpret_cast<sockaddr*>(&a), sizeof(a)) == 0)
          break;
        ::close(fd);
        if (std::chrono::steady_clock::now() > deadline)
          throw std::runtime_error("mesh: connect to rank " +
                                   std::to_string(j) + " timed out");
        std::this_thread::sleep_for(std::chrono::milliseconds(20));
      }
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      uint32_t me = static_cast<uint32_t>(rank);
      send_all(fd, &me, 4);
      fds_[j] = fd;
    }
    // accept from higher ranks
    for (int n = 0; n < world - 1 - rank; ++n) {
      int fd = ::accept(lfd, nullptr, nullptr);
      if (fd < 0) throw std::runtime_error("mesh: accept failed");
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      uint32_t peer;
      if (!recv_all(fd, &peer, 4))
        throw std::runtime_error("mesh: peer id read failed");
      if (peer >= static_cast<uint32_t>(world) || fds_[peer] != -1)
        throw std::runtime_error("mesh: bad peer id");
      fds_[peer] = fd;
    }
    ::close(lfd);
  }

  ~TcpMesh() {
    for (int fd : fds_)
      if (fd >= 0) ::close(fd);
  }

  void send(int peer, uintptr_t ptr, uint64_t nbytes) {
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(send_mu_[peer]);
    int fd = fds_.at(peer);
    if (fd < 0) throw std::runtime_error("mesh: no link to peer");
    send_all(fd, &nbytes, 8);
    if (nbytes) send_all(fd, reinterpret_cast<void*>(ptr), nbytes);
  }

  void recv(int peer, uintptr_t ptr, uint64_t nbytes) {
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(recv_mu_[peer]);
    int fd = fds_.at(peer);
    if (fd < 0) throw std::runtime_error("mesh: no link to peer");
    read_frame(fd, ptr, nbytes, peer);
  }

  // any-source receive (tuto.md:90 semantics): poll every pair socket,
  // take the first message that arrives.  Returns the source rank.
  int recv_any(uintptr_t ptr, uint64_t nbytes, int timeout_ms) {
    py::gil_scoped_release nogil;
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::milliseconds(timeout_ms);
    for (;;) {
      fd_set rset;
      FD_ZERO(&rset);
      int maxfd = -1;
      for (int p = 0; p < world_; ++p) {
        if (fds_[p] >= 0) {
          FD_SET(fds_[p], &rset);
          maxfd = std::max(maxfd, fds_[p]);
        }
      }
      timeval tv{0, 200 * 1000};  // 200 ms poll slices
      int n = ::select(maxfd + 1, &rset, nullptr, nullptr, &tv);
      if (n > 0) {
        for (int p = 0; p < world_; ++p) {
          if (fds_[p] >= 0 && FD_ISSET(fds_[p], &rset)) {
            std::unique_lock<std::mutex> lk(recv_mu_[p],
                                            std::try_to_lock);
            if (!lk.owns_lock()) continue;  // someone else is reading p
            // a peer that exited leaves an EOF-readable socket: peek
            // the header so a drained link is skipped, not mistaken
            // for a message
            uint64_t hdr;
            ssize_t k = ::recv(fds_[p], &hdr, 8, MSG_PEEK);
            if (k == 0) {
              ::close(fds_[p]);
              fds_[p] = -1;
              continue;
            }
            if (k < 8) continue;  // header still arriving
            if (hdr != nbytes) continue;
            // ^ a frame of a DIFFERENT size is not ours: a peer that
            // already delivered its message may queue traffic for a
            // later operation on the same pair socket (e.g. its
            // destroy() barrier token, 4 B) while this any-source
            // receive still waits on other ranks.  Leave it queued
            // for the targeted recv that will want it — consuming it
            // here corrupted the barrier (caught by
            // tests/test_tcp_backend.py::test_any_source_recv under
            // repetition).
            read_frame(fds_[p], ptr, nbytes, p);
            return p;
          }
        }
      }
      if (std::chrono::steady_clock::now() > deadline)
        throw std::runtime_error("mesh: recv(any) timed out");
    }
  }

  int rank() const { return rank_; }
  int world() const { return world_; }

 private:
  void read_frame(int fd, uintptr_t ptr, uint64_t nbytes, int peer) {
    uint64_t len;
    if (!recv_all(fd, &len, 8))
      throw std::runtime_error("mesh: recv header failed from rank " +
                               std::to_string(peer));
    if (len != nbytes)
      throw std::runtime_error(
          "mesh: size mismatch from rank " + std::to_string(peer) +
          ": sent " + std::to_string(len) + "B, receiving " +
          std::to_string(nbytes) + "B");
    if (len && !recv_all(fd, reinterpret_cast<void*>(ptr), len))
      throw std::runtime_error("mesh: recv payload failed");
  }

  int rank_, world_;
  std::vector<int> fds_;
  std::vector<std::mutex> send_mu_, recv_mu_;
};

// ---------------------------------------------------------------------------
// IPC peer-copy transport (the self-owned wire of the hand-tuned
// all-reduce — BASELINE's "hipMemcpyPeerAsync across the 7 xGMI links").
// A DeviceBuffer is a raw hipMalloc allocation (NOT torch caching-
// allocator memory: hipIpcGetMemHandle must see the real allocation
// base), exported through hipIpcGetMemHandle; peers exchange the
// 64-byte handles through the TCP store and open them with
// hipIpcOpenMemHandle, after which a plain stream-ordered
// hipMemcpyAsync into the mapped peer pointer IS the xGMI push.
// Requires dmabuf IPC (HSA_ENABLE_IPC_MODE_LEGACY=0, exported by the
// environment).
// ---------------------------------------------------------------------------
class DeviceBuffer {
 public:
  DeviceBuffer(size_t nbytes, int device) : nbytes_(nbytes) {
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipMalloc(&ptr_, nbytes));
  }
  ~DeviceBuffer() {
    if (ptr_) (void)hipFree(ptr_);
  }
  uintptr_t ptr() const { return reinterpret_cast<uintptr_t>(ptr_); }
  size_t nbytes() const { return nbytes_; }
  py::bytes ipc_handle() const {
    hipIpcMemHandle_t h;
    HIP_CHECK(hipIpcGetMemHandle(&h, ptr_));
    return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
  }

 private:
  void* ptr_ = nullptr;
  size_t nbytes_;
};

static uintptr_t ipc_open(const py::bytes& handle_bytes) {
  std::string s = handle_bytes;
  if (s.size() != sizeof(hipIpcMemHandle_t))
    throw std::runtime_error("bad hipIpcMemHandle size");
  hipIpcMemHandle_t h;
  memcpy(&h, s.data(), sizeof(h));
  void* p = nullptr;
  py::gil_scoped_release nogil;
  HIP_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess));
  return reinterpret_cast<uintptr_t>(p);
}

static void ipc_close(uintptr_t ptr) {
  py::gil_scoped_release nogil;
  HIP_CHECK(hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr)));
}

static void memcpy_async(uintptr_t dst, uintptr_t src, size_t nbytes,
                         uintptr_t stream) {
  // unified addressing: works for local DtoD and for IPC-mapped peer
  // destinations (the xGMI push)
  py::gil_scoped_release nogil;
  HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                           reinterpret_cast<void*>(src), nbytes,
                           hipMemcpyDeviceToDevice,
                           reinterpret_cast<hipStream_t>(stream)));
}

// ---------------------------------------------------------------------------
// small HIP utilities for the Python side
// ---------------------------------------------------------------------------
static py::bytes get_unique_id() {
  ncclUniqueId uid;
  NCCL_CHECK(ncclGetUniqueId(&uid));
  return py::bytes(reinterpret_cast<const char*>(&uid), sizeof(uid));
}

static uintptr_t record_event(uintptr_t stream) {
  hipEvent_t ev;
  HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  HIP_CHECK(hipEventRecord(ev, reinterpret_cast<hipStream_t>(stream)));
  return reinterpret_cast<uintptr_t>(ev);
}

static void event_wait(uintptr_t ev) {
  py::gil_scoped_release nogil;
  hipEvent_t e = reinterpret_cast<hipEvent_t>(ev);
  HIP_CHECK(hipEventSynchronize(e));
  HIP_CHECK(hipEventDestroy(e));
}

static void stream_sync(uintptr_t stream) {
  py::gil_scoped_release nogil;
  HIP_CHECK(hipStreamSynchronize(reinterpret_cast<hipStream_t>(stream)));
}

static int device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

PYBIND11_MODULE(_rcclx, m) {
  m.doc() = "native RCCL-over-xGMI backend (TCP store + communicator)";

  py::class_<TcpStore>(m, "TcpStore")
      .def(py::init<const std::string&, int, int, int, bool, int>(),
           py::arg("host"), py::arg("port"), py::arg("rank"),
           py::arg("world"), py::arg("is_server"),
           py::arg("timeout_ms") = 300000)
      .def("set", &TcpStore::set)
      .def("get", &TcpStore::get)
      .def("add", &TcpStore::add);

  py::class_<Comm>(m, "Comm")
      .def(py::init<int, int, const py::bytes&, int>(), py::arg("nranks"),
           py::arg("rank"), py::arg("uid"), py::arg("device"))
      .def("destroy", &Comm::destroy)
      .def("split", &Comm::split, py::return_value_policy::take_ownership)
      .def("all_reduce", &Comm::all_reduce)
      .def("broadcast", &Comm::broadcast)
      .def("reduce", &Comm::reduce)
      .def("all_gather", &Comm::all_gather)
      .def("reduce_scatter", &Comm::reduce_scatter)
      .def("gather", &Comm::gather)
      .def("scatter", &Comm::scatter)
      .def("all_to_all", &Comm::all_to_all)
      .def("send", &Comm::send)
      .def("recv", &Comm::recv)
      .def("group_start", &Comm::group_start)
      .def("group_end", &Comm::group_end)
      .def("rank", &Comm::rank)
      .def("nranks", &Comm::nranks)
      .def("device", &Comm::device);

  py::class_<TcpMesh>(m, "TcpMesh")
      .def(py::init<TcpStore*, const std::string&, int, int, int>(),
           py::arg("store"), py::arg("tag"), py::arg("rank"),
           py::arg("world"), py::arg("timeout_ms") = 120000,
           py::keep_alive<1, 2>())
      .def("send", &TcpMesh::send)
      .def("recv", &TcpMesh::recv)
      .def("recv_any", &TcpMesh::recv_any, py::arg("ptr"),
           py::arg("nbytes"), py::arg("timeout_ms") = 120000)
      .def("rank", &TcpMesh::rank)
      .def("world", &TcpMesh::world);

  py::class_<DeviceBuffer>(m, "DeviceBuffer")
      .def(py::init<size_t, int>(), py::arg("nbytes"), py::arg("device"))
      .def("ptr", &DeviceBuffer::ptr)
      .def("nbytes", &DeviceBuffer::nbytes)
      .def("ipc_handle", &DeviceBuffer::ipc_handle);

  m.def("ipc_open", &ipc_open);
  m.def("ipc_close", &ipc_close);
  m.def("memcpy_async", &memcpy_async);

  m.def("get_unique_id", &get_unique_id);
  m.def("record_event", &record_event);
  m.def("event_wait", &event_wait);
  m.def("stream_sync", &stream_sync);
  m.def("device_count", &device_count);
}
