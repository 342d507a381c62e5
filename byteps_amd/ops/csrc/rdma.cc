// RDMA data plane for the KV transport (reference ps-lite's RDMA van;
// north-star: "inter-node push/pull goes to a NUMA-aware CPU server over
// RDMA").
//
// Design — the locator protocol over verbs:
//   The TCP connection stays as the CONTROL channel (48-byte headers +
//   16-byte locator extensions, identical framing to the colocated IPC
//   lane, kv.h).  Gradient bytes move by one-sided verbs against the
//   client's registered staging regions:
//     - region announce (kRdmaHello): client registers each staging
//       region as an MR once and ships {addr, rkey, size} — the
//       register-once reuse that made the reference's RDMA server fast
//       (response-MR caching, reference server/server.cc:39-80);
//     - push: client sends header+locator only; the SERVER posts an
//       RDMA READ from (client_region + offset) into its merge buffer —
//       server-driven flow control, no remote arena management;
//     - pull reply: server posts an RDMA WRITE into the client's
//       receive staging at the locator, then sends the header.
//   QPs are RC, bootstrapped by exchanging RdmaPeerInfo blobs over the
//   TCP channel (kRdmaConnect).
//
// VALIDATION STATUS: this file compiles into _core.so everywhere, binds
// libibverbs.so.1 at runtime via dlopen, and is DISABLED unless
// BPS_ENABLE_RDMA=1 and a verbs device exists.  The build/CI environment
// for this repo has no RDMA NIC and no rxe; the verbs code path has not
// executed — see docs/rdma.md for the bring-up checklist.

#include <dlfcn.h>
#include <string.h>
#include <unistd.h>

#include <atomic>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include "rdma_abi.h"

namespace bpsrdma {

RdmaLib& RdmaLib::get() {
  static RdmaLib lib = [] {
    RdmaLib l;
    const char* e = getenv("BPS_ENABLE_RDMA");
    if (!(e && e[0] == '1')) return l;   // default off
    void* h = dlopen("libibverbs.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) return l;
#define BIND(field, sym)                                       \
  l.field = reinterpret_cast<decltype(l.field)>(dlsym(h, sym)); \
  if (!l.field) {                                              \
    dlclose(h);                                                \
    return RdmaLib{};                                          \
  }
    BIND(get_device_list, "ibv_get_device_list")
    BIND(free_device_list, "ibv_free_device_list")
    BIND(get_device_name, "ibv_get_device_name")
    BIND(open_device, "ibv_open_device")
    BIND(close_device, "ibv_close_device")
    BIND(alloc_pd, "ibv_alloc_pd")
    BIND(dealloc_pd, "ibv_dealloc_pd")
    BIND(reg_mr, "ibv_reg_mr")
    BIND(dereg_mr, "ibv_dereg_mr")
    BIND(create_cq, "ibv_create_cq")
    BIND(destroy_cq, "ibv_destroy_cq")
    BIND(create_qp, "ibv_create_qp")
    BIND(destroy_qp, "ibv_destroy_qp")
    BIND(modify_qp, "ibv_modify_qp")
    BIND(query_port, "ibv_query_port")
    BIND(query_gid, "ibv_query_gid")
    BIND(post_send, "ibv_post_send")
    BIND(post_recv, "ibv_post_recv")
    BIND(poll_cq, "ibv_poll_cq")
#undef BIND
    l.handle = h;
    return l;
  }();
  return lib;
}

// Per-process verbs device context (first active port of first device).
class RdmaDevice {
 public:
  static RdmaDevice* instance() {
    static RdmaDevice* dev = [stat = 0]() mutable -> RdmaDevice* {
      (void)stat;
      RdmaLib& lib = RdmaLib::get();
      if (!lib.ok()) return nullptr;
      auto* d = new RdmaDevice();
      if (!d->open()) {
        delete d;
        return nullptr;
      }
      return d;
    }();
    return dev;
  }

  ibv_context* ctx = nullptr;
  ibv_pd* pd = nullptr;
  uint8_t port = 1;
  uint16_t lid = 0;
  ibv_gid gid{};
  uint8_t gid_index = 0;
  int active_mtu = IBV_MTU_1024;

 private:
  bool open() {
    RdmaLib& lib = RdmaLib::get();
    int n = 0;
    ibv_device** list = lib.get_device_list(&n);
    if (!list || n == 0) return false;
    ctx = lib.open_device(list[0]);
    lib.free_device_list(list);
    if (!ctx) return false;
    ibv_port_attr_raw pa{};
    if (lib.query_port(ctx, port, &pa) != 0 || pa.state != 4 /*ACTIVE*/) {
      lib.close_device(ctx);
      ctx = nullptr;
      return false;
    }
    lid = pa.lid;
    active_mtu = pa.active_mtu;
    // RoCE: GID index 0 may be link-local; prefer 1 (RoCEv2) when set.
    // Overridable: BPS_RDMA_GID_INDEX.
    gid_index = 0;
    if (const char* g = getenv("BPS_RDMA_GID_INDEX"))
      gid_index = (uint8_t)atoi(g);
    else if (pa.link_layer == 2 /*Ethernet*/)
      gid_index = 1;
    if (lib.query_gid(ctx, port, gid_index, &gid) != 0) {
      gid_index = 0;
      lib.query_gid(ctx, port, gid_index, &gid);
    }
    pd = lib.alloc_pd(ctx);
    return pd != nullptr;
  }
};

// One RC connection: QP + CQ + peer info.
class RdmaConn {
 public:
  bool init() {
    dev_ = RdmaDevice::instance();
    if (!dev_) return false;
    RdmaLib& lib = RdmaLib::get();
    cq_ = lib.create_cq(dev_->ctx, 256, nullptr, nullptr, 0);
    if (!cq_) return false;
    ibv_qp_init_attr a{};
    a.send_cq = cq_;
    a.recv_cq = cq_;
    a.cap.max_send_wr = 128;
    a.cap.max_recv_wr = 16;
    a.cap.max_send_sge = 1;
    a.cap.max_recv_sge = 1;
    a.qp_type = IBV_QPT_RC;
    qp_ = lib.create_qp(dev_->pd, &a);
    if (!qp_) return false;
    // INIT
    ibv_qp_attr qa{};
    qa.qp_state = IBV_QPS_INIT;
    qa.pkey_index = 0;
    qa.port_num = dev_->port;
    qa.qp_access_flags =
        IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_READ |
        IBV_ACCESS_REMOTE_WRITE;
    return lib.modify_qp(qp_, &qa,
                         IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT |
                             IBV_QP_ACCESS_FLAGS) == 0;
  }

  RdmaPeerInfo local_info() const {
    RdmaPeerInfo pi{};
    pi.qpn = qp_->qp_num;
    pi.psn = 0x123456 & 0xFFFFFF;
    pi.lid = dev_->lid;
    memcpy(pi.gid, dev_->gid.raw, 16);
    pi.gid_index = dev_->gid_index;
    return pi;
  }

  bool connect(const RdmaPeerInfo& peer) {
    RdmaLib& lib = RdmaLib::get();
    ibv_qp_attr qa{};
    qa.qp_state = IBV_QPS_RTR;
    qa.path_mtu = dev_->active_mtu;
    qa.dest_qp_num = peer.qpn;
    qa.rq_psn = peer.psn;
    qa.max_dest_rd_atomic = 4;
    qa.min_rnr_timer = 12;
    qa.ah_attr.dlid = peer.lid;
    qa.ah_attr.sl = 0;
    qa.ah_attr.src_path_bits = 0;
    qa.ah_attr.port_num = dev_->port;
    bool roce = peer.lid == 0;
    if (roce) {
      qa.ah_attr.is_global = 1;
      memcpy(qa.ah_attr.grh.dgid.raw, peer.gid, 16);
      qa.ah_attr.grh.sgid_index = dev_->gid_index;
      qa.ah_attr.grh.hop_limit = 64;
    }
    if (lib.modify_qp(qp_, &qa,
                      IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU |
                          IBV_QP_DEST_QPN | IBV_QP_RQ_PSN |
                          IBV_QP_MAX_DEST_RD_ATOMIC |
                          IBV_QP_MIN_RNR_TIMER) != 0)
      return false;
    ibv_qp_attr qs{};
    qs.qp_state = IBV_QPS_RTS;
    qs.timeout = 14;
    qs.retry_cnt = 7;
    qs.rnr_retry = 7;
    qs.sq_psn = 0x123456 & 0xFFFFFF;
    qs.max_rd_atomic = 4;
    return lib.modify_qp(qp_, &qs,
                         IBV_QP_STATE | IBV_QP_TIMEOUT | IBV_QP_RETRY_CNT |
                             IBV_QP_RNR_RETRY | IBV_QP_SQ_PSN |
                             IBV_QP_MAX_QP_RD_ATOMIC) == 0;
  }

  // one-sided ops against a remote region; completion polled inline
  bool read(void* local, uint32_t lkey, uint64_t raddr, uint32_t rkey,
            uint32_t len) {
    return post_one(IBV_WR_RDMA_READ, local, lkey, raddr, rkey, len);
  }
  bool write(const void* local, uint32_t lkey, uint64_t raddr, uint32_t rkey,
             uint32_t len) {
    return post_one(IBV_WR_RDMA_WRITE, const_cast<void*>(local), lkey, raddr,
                    rkey, len);
  }

  ibv_mr* reg(void* addr, size_t len) {
    return RdmaLib::get().reg_mr(
        dev_->pd, addr, len,
        IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_READ |
            IBV_ACCESS_REMOTE_WRITE);
  }

  ~RdmaConn() {
    RdmaLib& lib = RdmaLib::get();
    if (qp_ && lib.destroy_qp) lib.destroy_qp(qp_);
    if (cq_ && lib.destroy_cq) lib.destroy_cq(cq_);
  }

 private:
  bool post_one(int opcode, void* local, uint32_t lkey, uint64_t raddr,
                uint32_t rkey, uint32_t len) {
    RdmaLib& lib = RdmaLib::get();
    ibv_sge sge{(uint64_t)(uintptr_t)local, len, lkey};
    ibv_send_wr wr{};
    wr.wr_id = ++wrid_;
    wr.sg_list = &sge;
    wr.num_sge = 1;
    wr.opcode = opcode;
    wr.send_flags = IBV_SEND_SIGNALED;
    wr.wr.rdma.remote_addr = raddr;
    wr.wr.rdma.rkey = rkey;
    ibv_send_wr* bad = nullptr;
    if (lib.post_send(qp_, &wr, &bad) != 0) return false;
    // poll to completion (data-plane threads are dedicated; the TCP
    // header that races this op is only sent after return)
    for (;;) {
      ibv_wc wc{};
      int n = lib.poll_cq(cq_, 1, &wc);
      if (n < 0) return false;
      if (n == 1) return wc.status == IBV_WC_SUCCESS;
    }
  }

  RdmaDevice* dev_ = nullptr;
  ibv_cq* cq_ = nullptr;
  ibv_qp* qp_ = nullptr;
  uint64_t wrid_ = 0;
};

bool rdma_available() {
  return RdmaDevice::instance() != nullptr;
}

RdmaConn* rdma_conn_create() {
  auto* c = new RdmaConn();
  if (!c->init()) {
    delete c;
    return nullptr;
  }
  return c;
}

void rdma_conn_destroy(RdmaConn* c) { delete c; }

RdmaPeerInfo rdma_conn_local_info(RdmaConn* c) { return c->local_info(); }

bool rdma_conn_connect(RdmaConn* c, const RdmaPeerInfo& peer) {
  return c->connect(peer);
}

ibv_mr* rdma_conn_reg(RdmaConn* c, void* addr, size_t len) {
  return c->reg(addr, len);
}

void rdma_mr_dereg(ibv_mr* mr) {
  if (mr && RdmaLib::get().dereg_mr) RdmaLib::get().dereg_mr(mr);
}

bool rdma_conn_read(RdmaConn* c, void* local, uint32_t lkey, uint64_t raddr,
                    uint32_t rkey, uint32_t len) {
  return c->read(local, lkey, raddr, rkey, len);
}

bool rdma_conn_write(RdmaConn* c, const void* local, uint32_t lkey,
                     uint64_t raddr, uint32_t rkey, uint32_t len) {
  return c->write(local, lkey, raddr, rkey, len);
}

}  // namespace bpsrdma
