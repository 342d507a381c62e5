// KV client: async zero-copy push/pull over TCP (from-scratch ps-lite
// replacement — reference KVWorker<char>::ZPush/ZPull call sites,
// common/core_loops.cc:538-618).
//
// One socket per (worker, server) pair.  A dedicated sender thread drains
// a request queue using writev (header + payload from the caller's pinned
// staging buffer — no copies); a receiver thread demultiplexes replies by
// sequence number, writing pull payloads straight into the caller's
// buffer.  Python calls enter with the GIL released; completion is a
// mutex+condvar ticket (the reference instead busy-span 1 µs sleeps,
// common/core_loops.cc:185).

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/mman.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/uio.h>
#include <unistd.h>

#include <hip/hip_runtime_api.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "kv.h"
#include "rdma_abi.h"

namespace py = pybind11;

namespace bpsamd {
namespace {

void write_all(int fd, const void* buf, size_t n) {
  const char* p = (const char*)buf;
  while (n > 0) {
    ssize_t w = ::write(fd, p, n);
    if (w < 0) {
      if (errno == EINTR) continue;
      throw std::runtime_error("kv: write failed: " +
                               std::string(strerror(errno)));
    }
    p += w;
    n -= (size_t)w;
  }
}

bool read_all(int fd, void* buf, size_t n) {
  char* p = (char*)buf;
  while (n > 0) {
    ssize_t r = ::read(fd, p, n);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    if (r == 0) return false;  // peer closed
    p += r;
    n -= (size_t)r;
  }
  return true;
}

int connect_to(const std::string& host, int port) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) throw std::runtime_error("kv: socket() failed");
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons((uint16_t)port);
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
    throw std::runtime_error("kv: bad address " + host);
  // retry for up to ~30 s — server may still be starting
  for (int attempt = 0;; ++attempt) {
    if (::connect(fd, (sockaddr*)&addr, sizeof(addr)) == 0) break;
    if (attempt > 300)
      throw std::runtime_error("kv: cannot connect to " + host + ":" +
                               std::to_string(port));
    ::usleep(100 * 1000);
  }
  return fd;
}

struct Request {
  MsgHeader hdr;
  const void* send_payload = nullptr;  // for push
  void* recv_buf = nullptr;            // for pull
  size_t recv_cap = 0;
  bool use_ext = false;                // IPC: body is the 16-byte IpcExt
  IpcExt ext{0, 0};
  // completion
  std::mutex mu;
  std::condition_variable cv;
  bool done = false;
  uint64_t reply_len = 0;
  uint64_t reply_aux = 0;
  std::string error;
};

// One shm region shared with a colocated server (the client's staging
// arena IS the transport: push payloads are read by the server in place,
// pull replies are written by the server straight into the staging the
// GPU H2D reads from — zero socket copies for gradient bytes).
struct Region {
  std::string name;
  char* base = nullptr;
  uint64_t size = 0;
  uint64_t used = 0;
  bool hip_registered = false;
  bpsrdma::ibv_mr* mr = nullptr;   // RDMA lane: registered MR
};

class ServerConn {
 public:
  ServerConn(const std::string& host, int port, uint32_t rank)
      : rank_(rank) {
    fd_ = connect_to(host, port);
    const char* e = getenv("BPS_ENABLE_IPC");
    ipc_enabled_ = !(e && (e[0] == '0' || e[0] == 'n' || e[0] == 'N'));
    const char* rm = getenv("BPS_IPC_REGION_MB");
    default_region_ = (rm ? std::max(1L, atol(rm)) : 64L) * (1ULL << 20);
    sender_ = std::thread([this] { this->send_loop(); });
    receiver_ = std::thread([this] { this->recv_loop(); });
  }

  // Allocate staging bytes inside a shm region shared with the server.
  // Returns the mapped address, or 0 when IPC is unavailable (remote
  // server / disabled) — caller falls back to private pinned staging.
  uintptr_t ipc_alloc(uint64_t nbytes) {
    if (!ipc_enabled_ || closed_) return 0;
    std::lock_guard<std::mutex> lk(regions_mu_);
    if (!ipc_enabled_) return 0;
    nbytes = (nbytes + 255) & ~255ULL;
    for (auto& r : regions_) {
      if (r.size - r.used >= nbytes) {
        uintptr_t p = (uintptr_t)(r.base + r.used);
        r.used += nbytes;
        return p;
      }
    }
    Region r;
    r.size = std::max<uint64_t>(default_region_, nbytes);
    static std::atomic<uint32_t> ctr{0};
    char nm[96];
    snprintf(nm, sizeof(nm), "/bpsamd-%d-%u-%u", (int)getpid(), rank_,
             ctr.fetch_add(1));
    r.name = nm;
    int fd = shm_open(nm, O_CREAT | O_EXCL | O_RDWR, 0600);
    if (fd < 0 || ftruncate(fd, (off_t)r.size) != 0) {
      if (fd >= 0) { ::close(fd); shm_unlink(nm); }
      ipc_enabled_ = false;
      return 0;
    }
    r.base = (char*)mmap(nullptr, r.size, PROT_READ | PROT_WRITE,
                         MAP_SHARED, fd, 0);
    ::close(fd);
    if (r.base == MAP_FAILED) {
      shm_unlink(nm);
      ipc_enabled_ = false;
      return 0;
    }
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) == hipSuccess && ndev > 0) {
      // pin for fast async D2H/H2D; ignore failure (stays pageable)
      r.hip_registered =
          hipHostRegister(r.base, r.size, hipHostRegisterPortable) ==
          hipSuccess;
      (void)hipGetLastError();
    }
    // hello handshake: the server shm_opens the name — success proves
    // colocation (a remote server cannot see this host's /dev/shm)
    uint32_t region_id = (uint32_t)regions_.size();
    auto req = submit(kIpcHello, region_id, r.name.c_str(),
                      r.name.size() + 1, nullptr, 0, 0, r.size);
    uint64_t aux = ~0ULL;
    {
      std::unique_lock<std::mutex> rlk(req->mu);
      req->cv.wait(rlk, [&] { return req->done; });
      if (req->error.empty()) aux = req->reply_aux;
    }
    shm_unlink(nm);  // server holds its own mapping (or failed)
    if (aux == ~0ULL) {
      // not colocated: try the RDMA lane — register this region as an
      // MR and announce {addr, rkey} instead of a shm name (the
      // register-once reuse of reference server/server.cc:39-80)
      if (try_rdma_region(r)) {
        uintptr_t p = (uintptr_t)r.base;
        r.used = nbytes;
        regions_.push_back(r);
        return p;
      }
      if (r.hip_registered) (void)hipHostUnregister(r.base);
      munmap(r.base, r.size);
      ipc_enabled_ = false;
      return 0;
    }
    uintptr_t p = (uintptr_t)r.base;
    r.used = nbytes;
    regions_.push_back(r);
    return p;
  }

  // Bootstrap the RC QP (once) and announce a registered region.
  // Returns false when RDMA is unavailable/denied → caller falls back.
  bool try_rdma_region(Region& r) {
    using namespace bpsrdma;
    if (!rdma_available()) return false;
    if (!rdma_) {
      RdmaConn* c = rdma_conn_create();
      if (!c) return false;
      RdmaPeerInfo mine = rdma_conn_local_info(c);
      RdmaPeerInfo peer{};
      auto req = submit(kRdmaConnect, 0, &mine, sizeof(mine), &peer,
                        sizeof(peer), 0, 0);
      uint64_t aux = ~0ULL, rlen = 0;
      {
        std::unique_lock<std::mutex> rlk(req->mu);
        req->cv.wait(rlk, [&] { return req->done; });
        if (req->error.empty()) {
          aux = req->reply_aux;
          rlen = req->reply_len;
        }
      }
      if (aux == ~0ULL || rlen != sizeof(peer) ||
          !rdma_conn_connect(c, peer)) {
        rdma_conn_destroy(c);
        return false;
      }
      rdma_ = c;
    }
    ibv_mr* mr = rdma_conn_reg(rdma_, r.base, r.size);
    if (!mr) return false;
    RdmaRegionInfo ri{(uint64_t)(uintptr_t)r.base, r.size, mr->rkey, 0};
    uint32_t region_id = (uint32_t)regions_.size();
    auto req = submit(kRdmaHello, region_id, &ri, sizeof(ri), nullptr, 0,
                      0, r.size);
    uint64_t aux = ~0ULL;
    {
      std::unique_lock<std::mutex> rlk(req->mu);
      req->cv.wait(rlk, [&] { return req->done; });
      if (req->error.empty()) aux = req->reply_aux;
    }
    if (aux == ~0ULL) {
      rdma_mr_dereg(mr);
      return false;
    }
    r.mr = mr;
    return true;
  }

  bool ipc_active() {
    std::lock_guard<std::mutex> lk(regions_mu_);
    return ipc_enabled_ && !regions_.empty();
  }

  // locate a pointer range inside a shared region → wire locator
  bool locate(const void* p, uint64_t len, uint64_t* loc) {
    if (!p) return false;
    std::lock_guard<std::mutex> lk(regions_mu_);
    const char* c = (const char*)p;
    for (size_t i = 0; i < regions_.size(); ++i) {
      const Region& r = regions_[i];
      if (c >= r.base && c + len <= r.base + r.size) {
        *loc = make_locator((uint32_t)i, (uint64_t)(c - r.base));
        return true;
      }
    }
    return false;
  }

  ~ServerConn() { close(); }

  void close() {
    bool expected = false;
    if (!closed_.compare_exchange_strong(expected, true)) return;
    {
      std::lock_guard<std::mutex> lk(q_mu_);
      q_cv_.notify_all();
    }
    ::shutdown(fd_, SHUT_RDWR);
    if (sender_.joinable()) sender_.join();
    if (receiver_.joinable()) receiver_.join();
    ::close(fd_);
    // fail anything still outstanding
    {
      std::lock_guard<std::mutex> lk(inflight_mu_);
      for (auto& kv : inflight_)
        complete(kv.second, 0, 0, "connection closed");
      inflight_.clear();
    }
    std::lock_guard<std::mutex> rlk(regions_mu_);
    for (auto& r : regions_) {
      if (r.mr) bpsrdma::rdma_mr_dereg(r.mr);
      if (r.hip_registered) (void)hipHostUnregister(r.base);
      munmap(r.base, r.size);
    }
    regions_.clear();
    if (rdma_) {
      bpsrdma::rdma_conn_destroy(rdma_);
      rdma_ = nullptr;
    }
  }

  std::shared_ptr<Request> submit(uint32_t op, uint64_t key,
                                  const void* payload, uint64_t len,
                                  void* recv_buf, size_t recv_cap,
                                  uint32_t cmd, uint64_t aux) {
    auto req = std::make_shared<Request>();
    req->hdr = MsgHeader{kMagic, op, key,
                         (op == kPull || op == kBarrier) ? 0 : len,
                         aux, rank_, cmd, seq_.fetch_add(1)};
    req->send_payload = payload;
    req->recv_buf = recv_buf;
    req->recv_cap = recv_cap;
    // colocated fast path: payload / receive buffer inside a shared
    // region → ship a 16-byte locator instead of the bytes
    uint64_t loc;
    if (op == kPush && len > 0 && locate(payload, len, &loc)) {
      req->use_ext = true;
      req->ext = IpcExt{loc, 0};
      req->hdr.cmd |= kCmdIpcPayload;
    } else if (op == kPull && recv_buf &&
               locate(recv_buf, recv_cap, &loc)) {
      req->use_ext = true;
      req->ext = IpcExt{loc, recv_cap};
      req->hdr.cmd |= kCmdIpcPayload;
    }
    {
      std::lock_guard<std::mutex> lk(inflight_mu_);
      inflight_[req->hdr.seq] = req;
    }
    {
      std::lock_guard<std::mutex> lk(q_mu_);
      queue_.push_back(req);
    }
    q_cv_.notify_one();
    return req;
  }

 private:
  void send_loop() {
    for (;;) {
      std::shared_ptr<Request> req;
      {
        std::unique_lock<std::mutex> lk(q_mu_);
        q_cv_.wait(lk, [this] { return closed_ || !queue_.empty(); });
        if (closed_ && queue_.empty()) return;
        if (queue_.empty()) continue;
        req = queue_.front();
        queue_.pop_front();
      }
      try {
        if (req->use_ext) {
          char frame[sizeof(MsgHeader) + sizeof(IpcExt)];
          std::memcpy(frame, &req->hdr, sizeof(MsgHeader));
          std::memcpy(frame + sizeof(MsgHeader), &req->ext, sizeof(IpcExt));
          write_all(fd_, frame, sizeof(frame));
        } else if (req->hdr.len > 0 && req->send_payload) {
          iovec iov[2];
          iov[0].iov_base = &req->hdr;
          iov[0].iov_len = sizeof(MsgHeader);
          iov[1].iov_base = const_cast<void*>(req->send_payload);
          iov[1].iov_len = req->hdr.len;
          size_t total = iov[0].iov_len + iov[1].iov_len;
          size_t sent = 0;
          while (sent < total) {
            ssize_t w = ::writev(fd_, iov, 2);
            if (w < 0) {
              if (errno == EINTR) continue;
              throw std::runtime_error("kv: writev failed");
            }
            sent += (size_t)w;
            // adjust iov
            size_t off = (size_t)w;
            for (int i = 0; i < 2 && off > 0; ++i) {
              size_t take = off < iov[i].iov_len ? off : iov[i].iov_len;
              iov[i].iov_base = (char*)iov[i].iov_base + take;
              iov[i].iov_len -= take;
              off -= take;
            }
          }
        } else {
          write_all(fd_, &req->hdr, sizeof(MsgHeader));
        }
      } catch (const std::exception& e) {
        fail(req->hdr.seq, e.what());
      }
    }
  }

  void recv_loop() {
    std::vector<char> scratch;
    for (;;) {
      MsgHeader h;
      if (!read_all(fd_, &h, sizeof(h))) return;
      if (h.magic != kMagic) return;
      std::shared_ptr<Request> req;
      {
        std::lock_guard<std::mutex> lk(inflight_mu_);
        auto it = inflight_.find(h.seq);
        if (it != inflight_.end()) {
          req = it->second;
          inflight_.erase(it);
        }
      }
      if (h.cmd & kCmdIpcPayload) {
        // reply body is the 16-byte locator echo; the payload is already
        // in the shared region (= the caller's receive buffer)
        IpcExt ext;
        if (!read_all(fd_, &ext, sizeof(ext))) return;
      } else if (h.len > 0) {
        void* dst = nullptr;
        if (req && req->recv_buf && h.len <= req->recv_cap) {
          dst = req->recv_buf;  // zero-copy into caller's pinned buffer
        } else {
          scratch.resize(h.len);
          dst = scratch.data();
        }
        if (!read_all(fd_, dst, h.len)) return;
      }
      if (req) complete(req, h.len, h.aux, "");
    }
  }

  void fail(uint64_t seq, const char* what) {
    std::shared_ptr<Request> req;
    {
      std::lock_guard<std::mutex> lk(inflight_mu_);
      auto it = inflight_.find(seq);
      if (it == inflight_.end()) return;
      req = it->second;
      inflight_.erase(it);
    }
    complete(req, 0, 0, what);
  }

  static void complete(const std::shared_ptr<Request>& req, uint64_t len,
                       uint64_t aux, const std::string& err) {
    std::lock_guard<std::mutex> lk(req->mu);
    req->reply_len = len;
    req->reply_aux = aux;
    req->error = err;
    req->done = true;
    req->cv.notify_all();
  }

  int fd_;
  uint32_t rank_;
  std::atomic<uint64_t> seq_{1};
  std::atomic<bool> closed_{false};
  bool ipc_enabled_ = false;
  bpsrdma::RdmaConn* rdma_ = nullptr;
  uint64_t default_region_ = 64ULL << 20;
  std::mutex regions_mu_;
  std::vector<Region> regions_;
  std::mutex q_mu_;
  std::condition_variable q_cv_;
  std::deque<std::shared_ptr<Request>> queue_;
  std::mutex inflight_mu_;
  std::unordered_map<uint64_t, std::shared_ptr<Request>> inflight_;
  std::thread sender_, receiver_;
};

class KVClient {
 public:
  KVClient(uint32_t rank, const std::vector<std::string>& uris) : rank_(rank) {
    for (const auto& uri : uris) {
      auto pos = uri.rfind(':');
      if (pos == std::string::npos)
        throw std::runtime_error("kv: server uri must be host:port, got " +
                                 uri);
      conns_.emplace_back(std::make_unique<ServerConn>(
          uri.substr(0, pos), std::stoi(uri.substr(pos + 1)), rank));
    }
  }

  uint64_t submit(int server, uint32_t op, uint64_t key, uintptr_t payload,
                  uint64_t len, uintptr_t recv_buf, uint64_t recv_cap,
                  uint32_t cmd, uint64_t aux) {
    auto req = conns_.at(server)->submit(
        op, key, reinterpret_cast<const void*>(payload), len,
        reinterpret_cast<void*>(recv_buf), recv_cap, cmd, aux);
    std::lock_guard<std::mutex> lk(tickets_mu_);
    uint64_t id = next_ticket_++;
    tickets_[id] = req;
    return id;
  }

  // returns (reply_len, reply_aux); throws on transport error
  std::pair<uint64_t, uint64_t> wait(uint64_t ticket) {
    std::shared_ptr<Request> req;
    {
      std::lock_guard<std::mutex> lk(tickets_mu_);
      auto it = tickets_.find(ticket);
      if (it == tickets_.end())
        throw std::runtime_error("kv: bad ticket");
      req = it->second;
      tickets_.erase(it);
    }
    std::unique_lock<std::mutex> lk(req->mu);
    req->cv.wait(lk, [&] { return req->done; });
    if (!req->error.empty()) throw std::runtime_error("kv: " + req->error);
    return {req->reply_len, req->reply_aux};
  }

  bool test(uint64_t ticket) {
    std::shared_ptr<Request> req;
    {
      std::lock_guard<std::mutex> lk(tickets_mu_);
      auto it = tickets_.find(ticket);
      if (it == tickets_.end()) return true;
      req = it->second;
    }
    std::lock_guard<std::mutex> lk(req->mu);
    return req->done;
  }

  void close() {
    for (auto& c : conns_) c->close();
  }

  // colocated IPC staging: returns the shm-backed address (0 = fall back
  // to private pinned staging)
  uintptr_t ipc_alloc(int server, uint64_t nbytes) {
    return conns_.at(server)->ipc_alloc(nbytes);
  }

  bool ipc_active(int server) { return conns_.at(server)->ipc_active(); }

  int num_servers() const { return (int)conns_.size(); }

 private:
  uint32_t rank_;
  std::vector<std::unique_ptr<ServerConn>> conns_;
  std::mutex tickets_mu_;
  uint64_t next_ticket_ = 1;
  std::unordered_map<uint64_t, std::shared_ptr<Request>> tickets_;
};

}  // namespace
}  // namespace bpsamd

void init_kv(py::module_& m) {
  using bpsamd::KVClient;
  py::class_<KVClient>(m, "KVClient")
      .def(py::init<uint32_t, const std::vector<std::string>&>(),
           py::arg("rank"), py::arg("server_uris"),
           py::call_guard<py::gil_scoped_release>())
      .def("submit", &KVClient::submit, py::arg("server"), py::arg("op"),
           py::arg("key"), py::arg("payload"), py::arg("len"),
           py::arg("recv_buf"), py::arg("recv_cap"), py::arg("cmd"),
           py::arg("aux"), py::call_guard<py::gil_scoped_release>())
      .def("wait", &KVClient::wait,
           py::call_guard<py::gil_scoped_release>())
      .def("test", &KVClient::test,
           py::call_guard<py::gil_scoped_release>())
      .def("close", &KVClient::close,
           py::call_guard<py::gil_scoped_release>())
      .def("ipc_alloc", &KVClient::ipc_alloc,
           py::call_guard<py::gil_scoped_release>())
      .def("ipc_active", &KVClient::ipc_active,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("num_servers", &KVClient::num_servers);

  m.attr("OP_PUSH") = (uint32_t)bpsamd::kPush;
  m.attr("OP_PULL") = (uint32_t)bpsamd::kPull;
  m.attr("OP_INIT") = (uint32_t)bpsamd::kInit;
  m.attr("OP_BARRIER") = (uint32_t)bpsamd::kBarrier;
  m.def("make_cmd", &bpsamd::make_cmd);
}
