// Minimal libibverbs ABI, loaded via dlopen at runtime.
//
// WHY NOT <infiniband/verbs.h>: the build image carries no rdma-core, so
// the backend declares the stable ibverbs ABI subset it needs and binds
// the EXPORTED symbols (ibv_post_send/ibv_poll_cq etc. exist as real
// functions in libibverbs for non-inline callers) with dlsym.  On hosts
// without libibverbs.so.1 the loader reports unavailable and the KV
// layer stays on TCP/IPC — nothing links against RDMA at build time.
//
// Scope: reliable-connection (RC) QPs, RDMA READ/WRITE + SEND, one CQ
// per connection — the minimum for the push/pull locator protocol
// (reference ps-lite RDMA van; response-MR caching idea from reference
// server/server.cc:39-80 maps to the register-once region MRs here).
//
// CAUTION: struct layouts below mirror rdma-core's stable user ABI.
// This backend is compiled everywhere but has NOT been executed against
// real hardware in this environment (no NIC, no rxe) — see
// docs/rdma.md for validation status.
#pragma once

#include <cstdint>
#include <cstddef>

namespace bpsrdma {

// -- POD structs we must lay out exactly (rdma-core stable ABI) -------------

union ibv_gid {
  uint8_t raw[16];
  struct {
    uint64_t subnet_prefix;
    uint64_t interface_id;
  } global;
};

enum ibv_qp_type { IBV_QPT_RC = 2 };
enum ibv_qp_state {
  IBV_QPS_RESET = 0, IBV_QPS_INIT = 1, IBV_QPS_RTR = 2, IBV_QPS_RTS = 3
};
enum ibv_mtu { IBV_MTU_1024 = 3, IBV_MTU_4096 = 5 };

enum ibv_access_flags {
  IBV_ACCESS_LOCAL_WRITE = 1,
  IBV_ACCESS_REMOTE_WRITE = 2,
  IBV_ACCESS_REMOTE_READ = 4,
};

enum ibv_wr_opcode {
  IBV_WR_RDMA_WRITE = 0,
  IBV_WR_SEND = 2,
  IBV_WR_RDMA_READ = 4,
};

enum ibv_send_flags { IBV_SEND_SIGNALED = 2, IBV_SEND_INLINE = 8 };

enum ibv_wc_status { IBV_WC_SUCCESS = 0 };

struct ibv_sge {
  uint64_t addr;
  uint32_t length;
  uint32_t lkey;
};

struct ibv_send_wr {
  uint64_t wr_id;
  ibv_send_wr* next;
  ibv_sge* sg_list;
  int num_sge;
  int opcode;           // ibv_wr_opcode
  unsigned int send_flags;
  union {
    uint32_t imm_data;
    uint32_t invalidate_rkey;
  };
  union {
    struct {
      uint64_t remote_addr;
      uint32_t rkey;
    } rdma;
    struct {
      uint64_t remote_addr;
      uint64_t compare_add;
      uint64_t swap;
      uint32_t rkey;
    } atomic;
    struct {
      void* ah;
      uint32_t remote_qpn;
      uint32_t remote_qkey;
    } ud;
  } wr;
  // qp_type / bind_mw / tso tail members are not touched by RC
  // SEND/READ/WRITE paths; the kernel/provider never reads past
  // num_sge-described fields for these opcodes, but keep padding so a
  // provider memcpy of sizeof(ibv_send_wr) stays in bounds
  uint64_t reserved_tail[6];
};

struct ibv_recv_wr {
  uint64_t wr_id;
  ibv_recv_wr* next;
  ibv_sge* sg_list;
  int num_sge;
};

struct ibv_wc {
  uint64_t wr_id;
  int status;           // ibv_wc_status
  int opcode;
  uint32_t vendor_err;
  uint32_t byte_len;
  union {
    uint32_t imm_data;
    uint32_t invalidated_rkey;
  };
  uint32_t qp_num;
  uint32_t src_qp;
  unsigned int wc_flags;
  uint16_t pkey_index;
  uint16_t slid;
  uint8_t sl;
  uint8_t dlid_path_bits;
};

struct ibv_global_route {
  ibv_gid dgid;
  uint32_t flow_label;
  uint8_t sgid_index;
  uint8_t hop_limit;
  uint8_t traffic_class;
};

struct ibv_ah_attr {
  ibv_global_route grh;
  uint16_t dlid;
  uint8_t sl;
  uint8_t src_path_bits;
  uint8_t static_rate;
  uint8_t is_global;
  uint8_t port_num;
};

struct ibv_qp_cap {
  uint32_t max_send_wr;
  uint32_t max_recv_wr;
  uint32_t max_send_sge;
  uint32_t max_recv_sge;
  uint32_t max_inline_data;
};

struct ibv_qp_init_attr {
  void* qp_context;
  void* send_cq;
  void* recv_cq;
  void* srq;
  ibv_qp_cap cap;
  int qp_type;          // ibv_qp_type
  int sq_sig_all;
};

struct ibv_qp_attr {
  int qp_state;
  int cur_qp_state;
  int path_mtu;
  int path_mig_state;
  uint32_t qkey;
  uint32_t rq_psn;
  uint32_t sq_psn;
  uint32_t dest_qp_num;
  unsigned int qp_access_flags;
  ibv_qp_cap cap;
  ibv_ah_attr ah_attr;
  ibv_ah_attr alt_ah_attr;
  uint16_t pkey_index;
  uint16_t alt_pkey_index;
  uint8_t en_sqd_async_notify;
  uint8_t sq_draining;
  uint8_t max_rd_atomic;
  uint8_t max_dest_rd_atomic;
  uint8_t min_rnr_timer;
  uint8_t port_num;
  uint8_t timeout;
  uint8_t retry_cnt;
  uint8_t rnr_retry;
  uint8_t alt_port_num;
  uint8_t alt_timeout;
  uint32_t rate_limit;
};

enum ibv_qp_attr_mask {
  IBV_QP_STATE = 1 << 0,
  IBV_QP_CUR_STATE = 1 << 1,
  IBV_QP_ACCESS_FLAGS = 1 << 3,
  IBV_QP_PKEY_INDEX = 1 << 4,
  IBV_QP_PORT = 1 << 5,
  IBV_QP_QKEY = 1 << 6,
  IBV_QP_AV = 1 << 7,
  IBV_QP_PATH_MTU = 1 << 8,
  IBV_QP_TIMEOUT = 1 << 9,
  IBV_QP_RETRY_CNT = 1 << 10,
  IBV_QP_RNR_RETRY = 1 << 11,
  IBV_QP_RQ_PSN = 1 << 12,
  IBV_QP_MAX_QP_RD_ATOMIC = 1 << 13,
  IBV_QP_MIN_RNR_TIMER = 1 << 15,
  IBV_QP_SQ_PSN = 1 << 16,
  IBV_QP_MAX_DEST_RD_ATOMIC = 1 << 17,
  IBV_QP_DEST_QPN = 1 << 20,
};

// port attr: only lid/state/active_mtu are read; allocate generously and
// index by the stable offsets
struct ibv_port_attr_raw {
  int state;            // ibv_port_state (4 = ACTIVE)
  int max_mtu;
  int active_mtu;
  int gid_tbl_len;
  uint32_t port_cap_flags;
  uint32_t max_msg_sz;
  uint32_t bad_pkey_cntr;
  uint32_t qkey_viol_cntr;
  uint16_t pkey_tbl_len;
  uint16_t lid;
  uint16_t sm_lid;
  uint8_t lmc;
  uint8_t max_vl_num;
  uint8_t sm_sl;
  uint8_t subnet_timeout;
  uint8_t init_type_reply;
  uint8_t active_width;
  uint8_t active_speed;
  uint8_t phys_state;
  uint8_t link_layer;   // 1 = IB, 2 = Ethernet (RoCE)
  uint8_t flags;
  uint16_t port_cap_flags2;
  uint32_t active_speed_ex;
};

// opaque handles (only pointer identity is used; fields accessed through
// exported functions, never inline)
struct ibv_device;
struct ibv_context;
struct ibv_pd;
struct ibv_cq;
struct ibv_comp_channel;

// ibv_mr: lkey/rkey must be read from the struct (no accessor exported).
// Stable prefix of rdma-core's ibv_mr:
struct ibv_mr {
  ibv_context* context;
  ibv_pd* pd;
  void* addr;
  size_t length;
  uint32_t handle;
  uint32_t lkey;
  uint32_t rkey;
};

// ibv_qp: qp_num read from the struct.  Stable prefix:
struct ibv_qp {
  ibv_context* context;
  void* qp_context;
  ibv_pd* pd;
  ibv_cq* send_cq;
  ibv_cq* recv_cq;
  void* srq;
  uint32_t handle;
  uint32_t qp_num;
  int state;
  int qp_type;
};

// -- loader ------------------------------------------------------------------

struct RdmaLib {
  void* handle = nullptr;

  ibv_device** (*get_device_list)(int*) = nullptr;
  void (*free_device_list)(ibv_device**) = nullptr;
  const char* (*get_device_name)(ibv_device*) = nullptr;
  ibv_context* (*open_device)(ibv_device*) = nullptr;
  int (*close_device)(ibv_context*) = nullptr;
  ibv_pd* (*alloc_pd)(ibv_context*) = nullptr;
  int (*dealloc_pd)(ibv_pd*) = nullptr;
  ibv_mr* (*reg_mr)(ibv_pd*, void*, size_t, int) = nullptr;
  int (*dereg_mr)(ibv_mr*) = nullptr;
  ibv_cq* (*create_cq)(ibv_context*, int, void*, ibv_comp_channel*, int) =
      nullptr;
  int (*destroy_cq)(ibv_cq*) = nullptr;
  ibv_qp* (*create_qp)(ibv_pd*, ibv_qp_init_attr*) = nullptr;
  int (*destroy_qp)(ibv_qp*) = nullptr;
  int (*modify_qp)(ibv_qp*, ibv_qp_attr*, int) = nullptr;
  int (*query_port)(ibv_context*, uint8_t, ibv_port_attr_raw*) = nullptr;
  int (*query_gid)(ibv_context*, uint8_t, int, ibv_gid*) = nullptr;
  int (*post_send)(ibv_qp*, ibv_send_wr*, ibv_send_wr**) = nullptr;
  int (*post_recv)(ibv_qp*, ibv_recv_wr*, ibv_recv_wr**) = nullptr;
  int (*poll_cq)(ibv_cq*, int, ibv_wc*) = nullptr;

  bool ok() const { return handle != nullptr; }
  static RdmaLib& get();   // singleton loader (rdma.cc)
};

// QP bootstrap blob exchanged over the TCP control channel
struct RdmaPeerInfo {
  uint32_t qpn;
  uint32_t psn;
  uint16_t lid;
  uint8_t gid[16];
  uint8_t gid_index;
  uint8_t pad[3];
};

// region announce blob (kRdmaHello payload)
struct RdmaRegionInfo {
  uint64_t addr;
  uint64_t size;
  uint32_t rkey;
  uint32_t pad;
};

// -- narrow interface used by kv.cc / server.cc (implemented in rdma.cc) ----

class RdmaConn;
bool rdma_available();
RdmaConn* rdma_conn_create();
void rdma_conn_destroy(RdmaConn*);
RdmaPeerInfo rdma_conn_local_info(RdmaConn*);
bool rdma_conn_connect(RdmaConn*, const RdmaPeerInfo& peer);
ibv_mr* rdma_conn_reg(RdmaConn*, void* addr, size_t len);
void rdma_mr_dereg(ibv_mr*);
bool rdma_conn_read(RdmaConn*, void* local, uint32_t lkey, uint64_t raddr,
                    uint32_t rkey, uint32_t len);
bool rdma_conn_write(RdmaConn*, const void* local, uint32_t lkey,
                     uint64_t raddr, uint32_t rkey, uint32_t len);

}  // namespace bpsrdma
