// byteps_amd parameter server (CPU) — from-scratch equivalent of the
// reference's server process (reference server/server.cc:458-531,
// server/queue.h), over the kv.h TCP protocol instead of ps-lite.
//
// Design (mirrors the reference's semantics, new implementation):
//  - accept thread + one reader thread per worker connection;
//  - N engine threads, each with a scheduling queue; keys are sharded to
//    engine threads by accumulated load (reference server/server.h:154-178);
//    with BPS_SERVER_ENABLE_SCHEDULE the queue pops the key with the
//    fewest total pushes first — earliest layers first (reference
//    server/queue.h:91-97);
//  - per-key state machine: first push of a round COPIES into the fp32
//    accumulator, later pushes SUM; when all expected pushers arrived the
//    merge is versioned and queued pulls flush (reference
//    server/server.cc:82-203,295-409);
//  - compressed pushes are decompressed on the fly (CPU codecs shared
//    with the HIP kernels via common.h RNG); the merged result is
//    re-compressed once per round for the pull replies;
//  - async mode sums straight into the store and answers pulls
//    immediately (reference server/server.cc:315-319).

#include <arpa/inet.h>
#include <fcntl.h>
#include <sched.h>
#include <cstdio>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/mman.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <queue>
#include <set>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/pybind11.h>

#ifdef _OPENMP
#include <omp.h>
#endif

#include "common.h"
#include "kv.h"
#include "rdma_abi.h"

namespace py = pybind11;

// CPU codec entry points (cpu_reducer.cc)
extern "C" {
int bps_cpu_sum(void* dst, const void* src, int64_t n, int dtype);
int bps_cpu_onebit_compress(const float* x, int64_t n, uint64_t* bits,
                            float* scale_sum);
int bps_cpu_onebit_decompress(const uint64_t* bits, float scale_sum, int64_t n,
                              float* out);
int bps_cpu_onebit_accumulate(const uint64_t* bits, float scale_sum,
                              int64_t n, float* acc, int first);
int bps_cpu_onebit_reply_pack(const float* acc, const float* err, int64_t n,
                              uint64_t* bits, float* scale_sum, float* comp);
int bps_cpu_onebit_err_update(const float* comp, int64_t n, float scale_sum,
                              float* err);
int bps_cpu_sparse_accumulate(const int32_t* idx, const float* val, int64_t k,
                              float* acc);
int bps_cpu_dithering_compress(const float* x, int64_t n, int s, uint64_t seed,
                               int natural, float norm, int8_t* code);
int bps_cpu_dithering_decompress(const int8_t* code, int64_t n, int s,
                                 int natural, float norm, float* out);
int bps_cpu_dithering_accumulate(const int8_t* code, int64_t n, int s,
                                 int natural, float norm, float* acc,
                                 int first);
int bps_cpu_dither_encode(const int8_t* code, int64_t n, uint8_t* out,
                          int64_t out_cap, int64_t* out_len);
int bps_cpu_dither_decode(const uint8_t* in, int64_t in_len, int64_t n,
                          int8_t* code);
int bps_cpu_dither_compensate_norm(const float* acc, const float* err,
                                   int64_t n, int natural, float* comp,
                                   float* out_norm);
int bps_cpu_dithering_compress_fast(const float* x, int64_t n, int s,
                                    uint64_t seed, int natural, float norm,
                                    int8_t* code);
int bps_cpu_topk_select(const float* x, int64_t n, int64_t k, int32_t* idx,
                        float* val);
int bps_cpu_fp8_compress(const float* x, int64_t n, float amax,
                         uint8_t* code);
int bps_cpu_fp8_accumulate(const uint8_t* code, int64_t n, float amax,
                           float* acc, int first);
int bps_cpu_fp8_decompress(const uint8_t* code, int64_t n, float amax,
                           float* out);
float bps_cpu_norm(const float* x, int64_t n, int mode);
}

namespace bpsamd {
namespace {

bool read_all_fd(int fd, void* buf, size_t n) {
  char* p = (char*)buf;
  while (n > 0) {
    ssize_t r = ::read(fd, p, n);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    if (r == 0) return false;
    p += r;
    n -= (size_t)r;
  }
  return true;
}

struct RdmaRemoteRegion {
  uint64_t addr = 0;
  uint64_t size = 0;
  uint32_t rkey = 0;
};

struct Conn {
  int fd;
  std::mutex write_mu;
  // colocated IPC: shm regions announced by this worker (kIpcHello);
  // reader thread installs, engine threads look up
  std::mutex regions_mu;
  std::unordered_map<uint32_t, std::pair<char*, size_t>> regions;
  // RDMA lane (remote worker): RC endpoint + the worker's registered
  // regions; bounce buffers are this server's registered staging
  bpsrdma::RdmaConn* rdma = nullptr;
  std::unordered_map<uint32_t, RdmaRemoteRegion> rdma_regions;
  char* rd_bounce = nullptr;       // reader-thread READ target
  size_t rd_cap = 0;
  bpsrdma::ibv_mr* rd_mr = nullptr;
  char* wr_bounce = nullptr;       // reply WRITE source (under write_mu)
  size_t wr_cap = 0;
  bpsrdma::ibv_mr* wr_mr = nullptr;

  ~Conn() {
    for (auto& kv : regions) munmap(kv.second.first, kv.second.second);
    if (rd_mr) bpsrdma::rdma_mr_dereg(rd_mr);
    if (wr_mr) bpsrdma::rdma_mr_dereg(wr_mr);
    free(rd_bounce);
    free(wr_bounce);
    if (rdma) bpsrdma::rdma_conn_destroy(rdma);
  }

  bool rdma_lookup(uint64_t locator, uint64_t len, uint64_t* raddr,
                   uint32_t* rkey) {
    std::lock_guard<std::mutex> lk(regions_mu);
    auto it = rdma_regions.find(bpsamd::locator_region(locator));
    if (it == rdma_regions.end()) return false;
    uint64_t off = bpsamd::locator_off(locator);
    if (off + len > it->second.size) return false;
    *raddr = it->second.addr + off;
    *rkey = it->second.rkey;
    return true;
  }

  bool ensure_bounce(char** buf, size_t* cap, bpsrdma::ibv_mr** mr,
                     size_t need) {
    if (*cap >= need) return true;
    size_t ncap = need < (8u << 20) ? (8u << 20) : need;
    char* nb = (char*)aligned_alloc(4096, (ncap + 4095) & ~4095UL);
    if (!nb) return false;
    bpsrdma::ibv_mr* nmr = bpsrdma::rdma_conn_reg(rdma, nb, ncap);
    if (!nmr) {
      free(nb);
      return false;
    }
    if (*mr) bpsrdma::rdma_mr_dereg(*mr);
    free(*buf);
    *buf = nb;
    *cap = ncap;
    *mr = nmr;
    return true;
  }

  char* resolve(uint64_t locator, uint64_t len) {
    std::lock_guard<std::mutex> lk(regions_mu);
    auto it = regions.find(bpsamd::locator_region(locator));
    if (it == regions.end()) return nullptr;
    uint64_t off = bpsamd::locator_off(locator);
    if (off + len > it->second.second) return nullptr;  // bounds check
    return it->second.first + off;
  }

  // header + 16-byte IpcExt frame (reply for region-backed payloads)
  void send_ext(const MsgHeader& h, const bpsamd::IpcExt& ext) {
    char frame[sizeof(MsgHeader) + sizeof(bpsamd::IpcExt)];
    std::memcpy(frame, &h, sizeof(MsgHeader));
    std::memcpy(frame + sizeof(MsgHeader), &ext, sizeof(ext));
    std::lock_guard<std::mutex> lk(write_mu);
    const char* p = frame;
    size_t n = sizeof(frame);
    while (n > 0) {
      ssize_t w = ::write(fd, p, n);
      if (w < 0) {
        if (errno == EINTR) continue;
        return;
      }
      p += w;
      n -= (size_t)w;
    }
  }

  void send(const MsgHeader& h, const void* payload) {
    std::lock_guard<std::mutex> lk(write_mu);
    const char* pl = (const char*)payload;
    size_t n = sizeof(MsgHeader);
    const char* p = (const char*)&h;
    while (n > 0) {
      ssize_t w = ::write(fd, p, n);
      if (w < 0) {
        if (errno == EINTR) continue;
        return;
      }
      p += w;
      n -= (size_t)w;
    }
    n = h.len;
    while (n > 0) {
      ssize_t w = ::write(fd, pl, n);
      if (w < 0) {
        if (errno == EINTR) continue;
        return;
      }
      pl += w;
      n -= (size_t)w;
    }
  }
};

struct InitPayload {
  uint64_t nelem;
  uint32_t expected;
  uint32_t levels;  // dithering s / sparse k
  // optional extension (new workers send 24 bytes; 16-byte payloads get
  // flags = 0)
  uint32_t flags;   // bit0: server-side error feedback on the merged reply
  uint32_t pad;
};

struct PendingPull {
  std::shared_ptr<Conn> conn;
  MsgHeader hdr;
  bool ipc = false;
  IpcExt ext{0, 0};
};

struct KeyState {
  std::mutex mu;
  std::vector<float> store;      // in-progress fp32 accumulator
  std::vector<float> published;  // snapshot served to pulls (sync raw mode):
                                 // without it a fast worker's next-round
                                 // COPY overwrites the store while a slow
                                 // worker's pull of the previous round is
                                 // still outstanding
  std::vector<char> reply;       // compressed merged (codec mode)
  uint64_t nelem = 0;
  uint32_t expected = 1;
  uint32_t codec = kRaw;
  uint32_t levels = 64;
  bool async_mode = false;
  std::set<uint32_t> round_senders;
  uint64_t version = 0;
  std::vector<PendingPull> pending;
  uint64_t push_total = 0;       // scheduling signal
  std::vector<float> scratch;    // decompress workspace
  std::vector<int8_t> code_scratch;  // dithering dense-code workspace
  bool server_ef = false;        // error-feedback on the merged reply
  bool dense_reply = false;      // colocated worker: skip Elias coding
  std::vector<float> ef_err;     // residual of the previous reply
  std::vector<float> ef_comp;    // compensated merge workspace
};

// NUMA topology from sysfs (north star: "NUMA-aware CPU server" —
// reference used libnuma-bound shm per PCIe switch, shared_memory.cc:52-82
// and numa_bind, global.cc:194-205.  Here: engine threads are pinned
// round-robin across nodes and each key's store is first-touched by its
// engine thread, so merge traffic stays node-local).
std::vector<std::vector<int>> numa_cpu_nodes() {
  std::vector<std::vector<int>> nodes;
  for (int nid = 0;; ++nid) {
    char path[96];
    snprintf(path, sizeof(path), "/sys/devices/system/node/node%d/cpulist",
             nid);
    FILE* f = fopen(path, "r");
    if (!f) break;
    char buf[4096];
    std::vector<int> cpus;
    if (fgets(buf, sizeof(buf), f)) {
      // parse "0-7,16-23" style ranges
      char* save = nullptr;
      for (char* tok = strtok_r(buf, ",\n", &save); tok;
           tok = strtok_r(nullptr, ",\n", &save)) {
        int lo, hi;
        if (sscanf(tok, "%d-%d", &lo, &hi) == 2)
          for (int c = lo; c <= hi; ++c) cpus.push_back(c);
        else if (sscanf(tok, "%d", &lo) == 1)
          cpus.push_back(lo);
      }
    }
    fclose(f);
    if (!cpus.empty()) nodes.push_back(std::move(cpus));
  }
  return nodes;
}

void pin_to_cpus(const std::vector<int>& cpus) {
  if (cpus.empty()) return;
  cpu_set_t set;
  CPU_ZERO(&set);
  for (int c : cpus) CPU_SET(c, &set);
  sched_setaffinity(0, sizeof(set), &set);
}

struct Task {
  std::shared_ptr<Conn> conn;
  MsgHeader hdr;
  std::vector<char> payload;
  KeyState* ks;
  const char* ipc_payload = nullptr;  // payload lives in the worker's shm

  const char* data() const {
    return ipc_payload ? ipc_payload : payload.data();
  }
};

class Server {
 public:
  Server(int port, int engine_threads, bool enable_schedule)
      : engine_threads_(std::max(1, engine_threads)),
        enable_schedule_(enable_schedule) {
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("server: socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons((uint16_t)port);
    if (::bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("server: bind failed on port " +
                               std::to_string(port));
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, (sockaddr*)&addr, &alen);
    port_ = ntohs(addr.sin_port);
    if (::listen(listen_fd_, 64) != 0)
      throw std::runtime_error("server: listen failed");
  }

  ~Server() { stop(); }

  int port() const { return port_; }

  void start() {
    running_ = true;
    for (int t = 0; t < engine_threads_; ++t) {
      queues_.emplace_back(std::make_unique<EngineQueue>());
      workers_.emplace_back([this, t] { engine_loop(t); });
    }
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  void stop() {
    bool expected = true;
    if (!running_.compare_exchange_strong(expected, false)) return;
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> lk(conns_mu_);
      for (auto& c : conns_) ::shutdown(c->fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    for (auto& q : queues_) q->close();
    for (auto& w : workers_) {
      if (w.joinable()) w.join();
    }
    {
      std::lock_guard<std::mutex> lk(readers_mu_);
      for (auto& r : readers_) {
        if (r.joinable()) r.join();
      }
    }
    {
      std::lock_guard<std::mutex> lk(conns_mu_);
      for (auto& c : conns_) ::close(c->fd);
      conns_.clear();
    }
  }

 private:
  struct EngineQueue {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<Task> fifo;
    bool closed = false;

    void push(Task&& t) {
      {
        std::lock_guard<std::mutex> lk(mu);
        fifo.push_back(std::move(t));
      }
      cv.notify_one();
    }
    bool pop(Task& out, bool schedule) {
      std::unique_lock<std::mutex> lk(mu);
      cv.wait(lk, [this] { return closed || !fifo.empty(); });
      if (fifo.empty()) return false;
      if (!schedule) {
        out = std::move(fifo.front());
        fifo.pop_front();
      } else {
        // fewest-pushes-first (earliest layer) — reference server
        // PriorityQueue (server/queue.h:91-97)
        auto best = fifo.begin();
        for (auto it = fifo.begin(); it != fifo.end(); ++it)
          if (it->ks->push_total < best->ks->push_total) best = it;
        out = std::move(*best);
        fifo.erase(best);
      }
      return true;
    }
    void close() {
      {
        std::lock_guard<std::mutex> lk(mu);
        closed = true;
      }
      cv.notify_all();
    }
  };

  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (errno == EINTR) continue;
        break;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      auto conn = std::make_shared<Conn>();
      conn->fd = fd;
      {
        std::lock_guard<std::mutex> lk(conns_mu_);
        conns_.push_back(conn);
      }
      {
        std::lock_guard<std::mutex> lk(readers_mu_);
        readers_.emplace_back([this, conn] { reader_loop(conn); });
      }
    }
  }

  KeyState* key_state(uint64_t key) {
    std::lock_guard<std::mutex> lk(keys_mu_);
    auto it = keys_.find(key);
    return it == keys_.end() ? nullptr : it->second.get();
  }

  void reader_loop(std::shared_ptr<Conn> conn) {
    while (running_) {
      MsgHeader h;
      if (!read_all_fd(conn->fd, &h, sizeof(h))) return;
      if (h.magic != kMagic) return;
      const bool ipc = (h.cmd & kCmdIpcPayload) != 0;
      IpcExt ext{0, 0};
      std::vector<char> payload;
      const char* ipc_ptr = nullptr;
      if (ipc) {
        // body is the 16-byte locator; payload lives in the worker's shm
        if (!read_all_fd(conn->fd, &ext, sizeof(ext))) return;
        if (h.op == kPush) {
          ipc_ptr = conn->resolve(ext.locator, h.len);
          if (!ipc_ptr && conn->rdma) {
            // RDMA lane: pull the payload with a one-sided READ from
            // the worker's registered staging into this reader's
            // bounce, then process inline (the bounce is reused per
            // frame, so the merge must complete before the next read)
            uint64_t raddr;
            uint32_t rkey;
            if (conn->rdma_lookup(ext.locator, h.len, &raddr, &rkey) &&
                conn->ensure_bounce(&conn->rd_bounce, &conn->rd_cap,
                                    &conn->rd_mr, h.len) &&
                bpsrdma::rdma_conn_read(conn->rdma, conn->rd_bounce,
                                        conn->rd_mr->lkey, raddr, rkey,
                                        (uint32_t)h.len)) {
              KeyState* ks2 = key_state(h.key);
              if (!ks2) {
                reply_err(conn, h);
                continue;
              }
              {
                std::lock_guard<std::mutex> lk(ks2->mu);
                ks2->push_total++;
              }
              Task t2{conn, h, {}, ks2};
              t2.ipc_payload = conn->rd_bounce;
              process_push(t2);
              continue;
            }
            reply_err(conn, h);
            continue;
          }
          if (!ipc_ptr) {
            reply_err(conn, h);
            continue;
          }
        }
      } else if (h.len > 0) {
        payload.resize(h.len);
        if (!read_all_fd(conn->fd, payload.data(), h.len)) return;
      }

      switch (h.op) {
        case kIpcHello:
          handle_ipc_hello(conn, h, payload);
          break;
        case kRdmaConnect:
          handle_rdma_connect(conn, h, payload);
          break;
        case kRdmaHello:
          handle_rdma_hello(conn, h, payload);
          break;
        case kInit:
          handle_init(conn, h, payload);
          break;
        case kPush: {
          KeyState* ks = key_state(h.key);
          if (!ks) {
            reply_err(conn, h);
            break;
          }
          {
            std::lock_guard<std::mutex> lk(ks->mu);
            ks->push_total++;
          }
          int tid = engine_of(h.key);
          Task t{conn, h, std::move(payload), ks};
          t.ipc_payload = ipc_ptr;
          queues_[tid]->push(std::move(t));
          break;
        }
        case kPull: {
          KeyState* ks = key_state(h.key);
          if (!ks) {
            reply_err(conn, h);
            break;
          }
          handle_pull(conn, h, ks, ipc, ext);
          break;
        }
        case kBarrier:
          handle_barrier(conn, h);
          break;
        case kShutdown:
          return;
        default:
          return;
      }
    }
  }

  void handle_ipc_hello(const std::shared_ptr<Conn>& conn, const MsgHeader& h,
                        const std::vector<char>& payload) {
    MsgHeader r = h;
    r.op = kIpcHelloReply;
    r.len = 0;
    // shm_open succeeding proves colocation: a remote worker's region
    // name does not exist in this host's /dev/shm
    char* base = nullptr;
    if (!payload.empty() && payload.back() == '\0' && h.aux > 0) {
      int fd = shm_open(payload.data(), O_RDWR, 0600);
      if (fd >= 0) {
        base = (char*)mmap(nullptr, h.aux, PROT_READ | PROT_WRITE,
                           MAP_SHARED, fd, 0);
        ::close(fd);
        if (base == MAP_FAILED) base = nullptr;
      }
    }
    if (base) {
      std::lock_guard<std::mutex> lk(conn->regions_mu);
      conn->regions[(uint32_t)h.key] = {base, (size_t)h.aux};
      r.aux = h.key;
    } else {
      r.aux = ~0ULL;  // not colocated / cannot map
    }
    conn->send(r, nullptr);
  }

  void reply_err(const std::shared_ptr<Conn>& conn, const MsgHeader& req) {
    MsgHeader h = req;
    h.len = 0;
    h.aux = ~0ULL;  // error marker
    h.cmd &= ~kCmdIpcPayload;  // no ext follows
    h.op = req.op == kPull ? kPullReply
         : req.op == kInit ? kInitReply : kPushReply;
    conn->send(h, nullptr);
  }

  void handle_init(const std::shared_ptr<Conn>& conn, const MsgHeader& h,
                   const std::vector<char>& payload) {
    bool mismatch = false;
    if (payload.size() >= 16) {
      InitPayload ip{};
      std::memcpy(&ip, payload.data(),
                  std::min(payload.size(), sizeof(ip)));
      std::lock_guard<std::mutex> lk(keys_mu_);
      auto& slot = keys_[h.key];
      if (slot) {
        // re-init (elastic resume): the config must match the stored
        // state — stale sizes would later cause out-of-bounds merges.
        // A changed world (expected) or levels is adopted; a changed
        // nelem/codec is an error the worker must see.
        std::lock_guard<std::mutex> klk(slot->mu);
        if (slot->nelem != ip.nelem || slot->codec != cmd_codec(h.cmd)) {
          mismatch = true;
        } else {
          slot->expected = std::max(1u, ip.expected);
          slot->levels = std::max(1u, ip.levels);
          slot->async_mode = cmd_async(h.cmd);
          slot->server_ef = (ip.flags & 1u) != 0;
          slot->dense_reply = (ip.flags & 2u) != 0;
        }
      }
      if (!slot) {
        slot = std::make_unique<KeyState>();
        slot->nelem = ip.nelem;
        slot->expected = std::max(1u, ip.expected);
        slot->codec = cmd_codec(h.cmd);
        slot->levels = std::max(1u, ip.levels);
        slot->async_mode = cmd_async(h.cmd);
        slot->server_ef = (ip.flags & 1u) != 0;
        slot->dense_reply = (ip.flags & 2u) != 0;
        // store allocation is DEFERRED to the first push so the
        // engine thread that owns this key first-touches the pages on
        // its own NUMA node (reference pre-allocated page-aligned at
        // init, server/server.cc:266-294 — NUMA placement came from
        // numactl there)
        // engine assignment by least accumulated load (reference
        // server/server.h:154-178)
        int best = 0;
        for (int t = 1; t < engine_threads_; ++t)
          if (engine_load_[t] < engine_load_[best]) best = t;
        engine_of_[h.key] = best;
        engine_load_[best] += (int64_t)ip.nelem;
      }
    }
    if (mismatch) {
      reply_err(conn, h);
      return;
    }
    MsgHeader r = h;
    r.op = kInitReply;
    r.len = 0;
    conn->send(r, nullptr);
  }

  void handle_rdma_connect(const std::shared_ptr<Conn>& conn,
                           const MsgHeader& h,
                           const std::vector<char>& payload) {
    MsgHeader r = h;
    r.op = kRdmaConnectReply;
    r.len = 0;
    r.aux = ~0ULL;
    if (bpsrdma::rdma_available() &&
        payload.size() >= sizeof(bpsrdma::RdmaPeerInfo) && !conn->rdma) {
      bpsrdma::RdmaPeerInfo peer{};
      std::memcpy(&peer, payload.data(), sizeof(peer));
      bpsrdma::RdmaConn* c = bpsrdma::rdma_conn_create();
      if (c) {
        bpsrdma::RdmaPeerInfo mine = bpsrdma::rdma_conn_local_info(c);
        if (bpsrdma::rdma_conn_connect(c, peer)) {
          conn->rdma = c;
          r.aux = 0;
          r.len = sizeof(mine);
          conn->send(r, &mine);
          return;
        }
        bpsrdma::rdma_conn_destroy(c);
      }
    }
    conn->send(r, nullptr);
  }

  void handle_rdma_hello(const std::shared_ptr<Conn>& conn,
                         const MsgHeader& h,
                         const std::vector<char>& payload) {
    MsgHeader r = h;
    r.op = kRdmaHelloReply;
    r.len = 0;
    if (conn->rdma && payload.size() >= sizeof(bpsrdma::RdmaRegionInfo)) {
      bpsrdma::RdmaRegionInfo ri{};
      std::memcpy(&ri, payload.data(), sizeof(ri));
      std::lock_guard<std::mutex> lk(conn->regions_mu);
      conn->rdma_regions[(uint32_t)h.key] =
          RdmaRemoteRegion{ri.addr, ri.size, ri.rkey};
      r.aux = h.key;
    } else {
      r.aux = ~0ULL;
    }
    conn->send(r, nullptr);
  }

  int engine_of(uint64_t key) {
    std::lock_guard<std::mutex> lk(keys_mu_);
    auto it = engine_of_.find(key);
    return it == engine_of_.end() ? 0 : it->second;
  }

  void handle_pull(const std::shared_ptr<Conn>& conn, const MsgHeader& h,
                   KeyState* ks, bool ipc, const IpcExt& ext) {
    std::unique_lock<std::mutex> lk(ks->mu);
    if (ks->store.size() != ks->nelem)
      ks->store.assign(ks->nelem, 0.0f);   // pull-before-push: zeros
    uint64_t want_version = h.aux;
    PendingPull p{conn, h, ipc, ext};
    if (ks->async_mode || ks->version >= want_version) {
      send_pull_reply(p, ks);
    } else {
      ks->pending.push_back(std::move(p));
    }
  }

  void send_pull_reply(const PendingPull& p, KeyState* ks) {
    // ks->mu held
    MsgHeader r = p.hdr;
    r.op = kPullReply;
    r.aux = ks->version;
    const void* src;
    uint64_t len;
    if (ks->codec == kRaw) {
      const std::vector<float>& v =
          (ks->async_mode || ks->published.empty()) ? ks->store
                                                    : ks->published;
      src = v.data();
      len = v.size() * sizeof(float);
    } else {
      src = ks->reply.data();
      len = ks->reply.size();
    }
    r.len = len;
    if (p.ipc) {
      // colocated: write straight into the worker's recv staging —
      // the only copy on the reply path
      char* dst = len <= p.ext.cap
                      ? p.conn->resolve(p.ext.locator, len) : nullptr;
      if (dst) {
        std::memcpy(dst, src, len);
        p.conn->send_ext(r, p.ext);
        return;
      }
      if (p.conn->rdma && len <= p.ext.cap) {
        // remote worker over verbs: stage into the registered reply
        // bounce and RDMA WRITE it into the worker's recv staging,
        // then ship the header (write_mu serializes bounce use)
        uint64_t raddr;
        uint32_t rkey;
        if (p.conn->rdma_lookup(p.ext.locator, len, &raddr, &rkey)) {
          std::lock_guard<std::mutex> wlk(p.conn->write_mu);
          if (p.conn->ensure_bounce(&p.conn->wr_bounce, &p.conn->wr_cap,
                                    &p.conn->wr_mr, len)) {
            std::memcpy(p.conn->wr_bounce, src, len);
            if (bpsrdma::rdma_conn_write(p.conn->rdma, p.conn->wr_bounce,
                                         p.conn->wr_mr->lkey, raddr, rkey,
                                         (uint32_t)len)) {
              // header without payload; ext echoed (mutex already held
              // → inline the send)
              char frame[sizeof(MsgHeader) + sizeof(IpcExt)];
              std::memcpy(frame, &r, sizeof(MsgHeader));
              std::memcpy(frame + sizeof(MsgHeader), &p.ext,
                          sizeof(IpcExt));
              const char* q = frame;
              size_t left = sizeof(frame);
              while (left > 0) {
                ssize_t w = ::write(p.conn->fd, q, left);
                if (w < 0) {
                  if (errno == EINTR) continue;
                  return;
                }
                q += w;
                left -= (size_t)w;
              }
              return;
            }
          }
        }
      }
      // capacity/region mismatch: fall back to inline (clears the IPC
      // bit so the client reads the payload from the socket)
      r.cmd &= ~kCmdIpcPayload;
    }
    p.conn->send(r, src);
  }

  void engine_loop(int tid) {
    // NUMA: pin this engine (and the OMP team it spawns — workers
    // inherit the creating thread's mask) to one node, round-robin.
    // Single-node boxes: no-op.  BPS_SERVER_NUMA=0 disables.
    static std::vector<std::vector<int>> nodes = numa_cpu_nodes();
    const char* nenv = getenv("BPS_SERVER_NUMA");
    bool numa_on = !(nenv && nenv[0] == '0');
    if (numa_on && nodes.size() > 1)
      pin_to_cpus(nodes[tid % nodes.size()]);
#ifdef _OPENMP
    // Cap this engine thread's OMP team: with N engine threads each
    // spawning a default (all-cores) team, the server oversubscribes
    // catastrophically — measured 85-99 ms per 4 MB push on a colocated
    // server (reference's equivalent knob: BYTEPS_OMP_THREAD_PER_GPU=4,
    // common/cpu_reducer.cc:41-45).
    int per = 4;
    if (const char* e = getenv("BPS_OMP_THREAD_PER_ENGINE"))
      per = std::max(1, atoi(e));
    else if (const char* e2 = getenv("BYTEPS_OMP_THREAD_PER_GPU"))
      per = std::max(1, atoi(e2));
    omp_set_num_threads(per);
#endif
    Task task;
    while (queues_[tid]->pop(task, enable_schedule_)) {
      process_push(task);
    }
  }

  void process_push(Task& t) {
    KeyState* ks = t.ks;
    std::unique_lock<std::mutex> lk(ks->mu);
    const uint32_t codec = cmd_codec(t.hdr.cmd);
    const bool first = ks->async_mode ? false : ks->round_senders.empty();
    const int64_t n = (int64_t)ks->nelem;
    if (ks->store.size() != ks->nelem)
      ks->store.assign(ks->nelem, 0.0f);   // first-touch on THIS node
    float* acc = ks->store.data();

    switch (codec) {
      case kRaw: {
        const float* src = (const float*)t.data();
        if (first)
          std::memcpy(acc, src, n * sizeof(float));
        else
          bps_cpu_sum(acc, src, n, 0);
        break;
      }
      case kOnebit: {
        int64_t nwords = (n + 63) >> 6;
        float scale_sum;
        std::memcpy(&scale_sum, t.data() + nwords * 8, 4);
        // fused decode→accumulate: one pass instead of
        // decompress-to-scratch + copy/sum
        bps_cpu_onebit_accumulate((const uint64_t*)t.data(), scale_sum, n,
                                  acc, first ? 1 : 0);
        break;
      }
      case kTopk:
      case kRandomk: {
        int64_t k = (int64_t)t.hdr.aux & 0xFFFFFFFF;
        const int32_t* idx = (const int32_t*)t.data();
        const float* val = (const float*)(t.data() + k * 4);
        if (first) std::memset(acc, 0, n * sizeof(float));
        bps_cpu_sparse_accumulate(idx, val, k, acc);
        break;
      }
      case kDitherLinear:
      case kDitherNatural: {
        // wire: [norm f32][flag u8][dense int8 | Elias-delta stream]
        float norm;
        std::memcpy(&norm, t.data(), 4);
        uint8_t flag = (uint8_t)t.data()[4];
        const int8_t* code;
        if (flag == 1) {
          ks->code_scratch.resize(n);
          if (bps_cpu_dither_decode((const uint8_t*)t.data() + 5,
                                    (int64_t)t.hdr.len - 5, n,
                                    ks->code_scratch.data()) != 0)
            break;  // malformed — drop (worker sees stalled round)
          code = ks->code_scratch.data();
        } else {
          code = (const int8_t*)(t.data() + 5);
        }
        bps_cpu_dithering_accumulate(code, n, (int)ks->levels,
                                     codec == kDitherNatural, norm, acc,
                                     first ? 1 : 0);
        break;
      }
      case kFp8: {
        float amax;
        std::memcpy(&amax, t.data(), 4);
        const uint8_t* code = (const uint8_t*)(t.data() + 4);
        // fused decode→accumulate, one pass
        bps_cpu_fp8_accumulate(code, n, amax, acc, first ? 1 : 0);
        break;
      }
      default:
        break;
    }

    // push ack (clear the IPC bit: no 16-byte ext follows the header)
    MsgHeader ack = t.hdr;
    ack.op = kPushReply;
    ack.len = 0;
    ack.cmd &= ~kCmdIpcPayload;
    t.conn->send(ack, nullptr);

    if (ks->async_mode) return;

    ks->round_senders.insert(t.hdr.sender);
    if ((uint32_t)ks->round_senders.size() >= ks->expected) {
      // ALL_RECV: finalize merge (reference server/server.cc:348-370)
      ks->version++;
      ks->round_senders.clear();
      if (ks->codec != kRaw) {
        compress_reply(ks);
      } else {
        // publish the merge; next round's COPY reuses the old buffer
        ks->published.swap(ks->store);
        if (ks->store.size() != ks->nelem)
          ks->store.assign(ks->nelem, 0.0f);
      }
      auto pending = std::move(ks->pending);
      ks->pending.clear();
      for (auto& p : pending) {
        if (p.hdr.aux <= ks->version) send_pull_reply(p, ks);
        else ks->pending.push_back(p);
      }
      if (ks->server_ef && ks->codec != kRaw) {
        // deferred EF residual: only needed by NEXT round's pack, so it
        // runs after the queued pulls were answered (off the pull
        // critical path)
        if (ks->codec == kOnebit) {
          int64_t nn = (int64_t)ks->nelem;
          int64_t nwords = (nn + 63) >> 6;
          float sc;
          std::memcpy(&sc, ks->reply.data() + nwords * 8, 4);
          bps_cpu_onebit_err_update(ks->ef_comp.data(), nn, sc,
                                    ks->ef_err.data());
        } else {
          deferred_generic_ef(ks);
        }
      }
    }
  }

  void compress_reply(KeyState* ks) {
    const int64_t n = (int64_t)ks->nelem;
    const float* acc = ks->store.data();
    if (ks->server_ef && ks->codec == kOnebit) {
      // fused EF + sign-pack: two passes instead of the generic
      // compensate/compress/decompress/subtract chain
      if (ks->ef_err.empty()) ks->ef_err.assign(n, 0.0f);
      ks->ef_comp.resize(n);
      int64_t nwords = (n + 63) >> 6;
      ks->reply.resize(nwords * 8 + 8);
      float scale_sum = 0.0f;
      bps_cpu_onebit_reply_pack(acc, ks->ef_err.data(), n,
                                (uint64_t*)ks->reply.data(), &scale_sum,
                                ks->ef_comp.data());
      std::memcpy(ks->reply.data() + nwords * 8, &scale_sum, 4);
      std::memset(ks->reply.data() + nwords * 8 + 4, 0, 4);
      return;
    }
    // server-side error feedback (reference server compressor mirror,
    // server/server.cc:228-257 + vanilla EF): compensate the merge with
    // the previous reply's residual before compressing, then store the
    // new residual.
    const bool dither = ks->codec == kDitherLinear ||
                        ks->codec == kDitherNatural;
    if (ks->server_ef && !dither) {   // dithering fuses compensation
      if (ks->ef_err.empty()) ks->ef_err.assign(n, 0.0f);
      ks->ef_comp.resize(n);
      const float* err = ks->ef_err.data();
      float* comp = ks->ef_comp.data();
#pragma omp parallel for
      for (int64_t i = 0; i < n; ++i) comp[i] = acc[i] + err[i];
      acc = comp;
    }
    switch (ks->codec) {
      case kOnebit: {
        int64_t nwords = (n + 63) >> 6;
        ks->reply.resize(nwords * 8 + 8);
        float scale_sum = 0.0f;
        bps_cpu_onebit_compress(acc, n, (uint64_t*)ks->reply.data(),
                                &scale_sum);
        std::memcpy(ks->reply.data() + nwords * 8, &scale_sum, 4);
        std::memset(ks->reply.data() + nwords * 8 + 4, 0, 4);
        break;
      }
      case kTopk: {
        // k for topk is carried in `levels` at init; parallel per-thread
        // heap select (cpu_reducer.cc) — was a serial O(n log k)
        // partial_sort, the slowest codec on the merge path
        int64_t k = std::max<int64_t>(1, std::min<int64_t>(n, (int64_t)ks->levels));
        ks->reply.resize(k * 8);
        bps_cpu_topk_select(acc, n, k, (int32_t*)ks->reply.data(),
                            (float*)(ks->reply.data() + k * 4));
        break;
      }
      case kRandomk: {
        int64_t k = std::max<int64_t>(1, std::min<int64_t>(n, (int64_t)ks->levels));
        uint64_t seed = splitmix64(ks->version * 0x9E3779B97F4A7C15ULL + 11);
        ks->reply.resize(k * 8);
        int32_t* idx = (int32_t*)ks->reply.data();
        float* val = (float*)(ks->reply.data() + k * 4);
        for (int64_t j = 0; j < k; ++j) {
          idx[j] = (int32_t)rand_index(seed, (uint64_t)j, (uint64_t)n);
          val[j] = acc[idx[j]];
        }
        break;
      }
      case kDitherLinear:
      case kDitherNatural: {
        // reply pack in two passes: fused compensate+norm, then a
        // cheap-RNG quantize (cpu_reducer.cc) — EF compensation is
        // folded here (acc already points at ef_comp when server_ef,
        // but the fused version recomputes with the norm in one pass)
        bool natural = ks->codec == kDitherNatural;
        uint64_t seed = splitmix64(ks->version * 0xD6E8FEB86659FD93ULL + 5);
        float norm = 0.0f;
        ks->code_scratch.resize(n);
        if (ks->server_ef) {
          // fused compensate + norm (pass 1), cheap-RNG quantize (pass 2)
          if (ks->ef_err.empty()) ks->ef_err.assign(n, 0.0f);
          ks->ef_comp.resize(n);
          bps_cpu_dither_compensate_norm(ks->store.data(),
                                         ks->ef_err.data(), n,
                                         natural ? 1 : 0,
                                         ks->ef_comp.data(), &norm);
          acc = ks->ef_comp.data();
        } else {
          norm = bps_cpu_norm(acc, n, natural ? 1 : 2);
        }
        bps_cpu_dithering_compress_fast(acc, n, (int)ks->levels, seed,
                                        natural, norm,
                                        ks->code_scratch.data());
        ks->reply.resize(5 + n);
        std::memcpy(ks->reply.data(), &norm, 4);
        int64_t wlen = 0;
        if (!ks->dense_reply &&
            bps_cpu_dither_encode(ks->code_scratch.data(), n,
                                  (uint8_t*)ks->reply.data() + 5, n,
                                  &wlen) == 0) {
          ks->reply[4] = 1;          // Elias sparse wire
          ks->reply.resize(5 + wlen);
        } else {
          ks->reply[4] = 0;          // dense fallback (stream ≥ dense)
          std::memcpy(ks->reply.data() + 5, ks->code_scratch.data(), n);
        }
        break;
      }
      case kFp8: {
        float amax = bps_cpu_norm(acc, n, 2);
        ks->reply.resize(4 + n);
        std::memcpy(ks->reply.data(), &amax, 4);
        bps_cpu_fp8_compress(acc, n, amax,
                             (uint8_t*)(ks->reply.data() + 4));
        break;
      }
      default:
        break;
    }

  }

  // EF residual for the generic (non-onebit) codecs — runs AFTER the
  // queued pulls flushed (only the NEXT round reads ef_err)
  void deferred_generic_ef(KeyState* ks) {
    const int64_t n = (int64_t)ks->nelem;
    const float* acc = ks->ef_comp.data();
    {
      // new residual = compensated merge − decompress(reply)
      ks->scratch.assign(n, 0.0f);
      float* dec = ks->scratch.data();
      switch (ks->codec) {
        case kOnebit: {
          int64_t nwords = (n + 63) >> 6;
          float sc;
          std::memcpy(&sc, ks->reply.data() + nwords * 8, 4);
          bps_cpu_onebit_decompress((const uint64_t*)ks->reply.data(), sc, n,
                                    dec);
          break;
        }
        case kTopk:
        case kRandomk: {
          int64_t k = (int64_t)(ks->reply.size() / 8);
          const int32_t* idx = (const int32_t*)ks->reply.data();
          const float* val = (const float*)(ks->reply.data() + k * 4);
          for (int64_t j = 0; j < k; ++j) dec[idx[j]] = val[j];
          break;
        }
        case kDitherLinear:
        case kDitherNatural: {
          float norm;
          std::memcpy(&norm, ks->reply.data(), 4);
          // dense codes of the reply are still in code_scratch
          bps_cpu_dithering_decompress(
              ks->code_scratch.data(), n, (int)ks->levels,
              ks->codec == kDitherNatural, norm, dec);
          break;
        }
        case kFp8: {
          float amax;
          std::memcpy(&amax, ks->reply.data(), 4);
          bps_cpu_fp8_decompress((const uint8_t*)(ks->reply.data() + 4), n,
                                 amax, dec);
          break;
        }
        default:
          break;
      }
      float* err = ks->ef_err.data();
#pragma omp parallel for
      for (int64_t i = 0; i < n; ++i) err[i] = acc[i] - dec[i];
    }
  }


  void handle_barrier(const std::shared_ptr<Conn>& conn, const MsgHeader& h) {
    std::lock_guard<std::mutex> lk(barrier_mu_);
    barrier_waiters_.push_back(PendingPull{conn, h});
    uint32_t expected = (uint32_t)h.aux;
    if (barrier_waiters_.size() >= expected) {
      for (auto& w : barrier_waiters_) {
        MsgHeader r = w.hdr;
        r.op = kBarrierReply;
        r.len = 0;
        w.conn->send(r, nullptr);
      }
      barrier_waiters_.clear();
    }
  }

  int listen_fd_;
  int port_;
  int engine_threads_;
  bool enable_schedule_;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::mutex readers_mu_;
  std::vector<std::thread> readers_;
  std::vector<std::thread> workers_;
  std::vector<std::unique_ptr<EngineQueue>> queues_;
  std::mutex conns_mu_;
  std::vector<std::shared_ptr<Conn>> conns_;
  std::mutex keys_mu_;
  std::map<uint64_t, std::unique_ptr<KeyState>> keys_;
  std::unordered_map<uint64_t, int> engine_of_;
  std::map<int, int64_t> engine_load_;
  std::mutex barrier_mu_;
  std::vector<PendingPull> barrier_waiters_;
};

}  // namespace
}  // namespace bpsamd

void init_server(py::module_& m) {
  using bpsamd::Server;
  py::class_<Server>(m, "Server")
      .def(py::init<int, int, bool>(), py::arg("port"),
           py::arg("engine_threads") = 4, py::arg("enable_schedule") = false)
      .def("start", &Server::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &Server::stop,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &Server::port);
}
