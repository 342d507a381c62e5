// Wire protocol for the byteps_amd KV transport (from-scratch replacement
// for ps-lite's Van/KVWorker/KVServer — reference SURVEY §2 C14).  TCP
// with length-prefixed frames; the framing (fixed header + zero-copy
// payload into pre-registered buffers) is RDMA-ready: header maps to an
// RDMA immediate + SEND, payload to a WRITE into the registered region.
#pragma once

#include <cstdint>

namespace bpsamd {

constexpr uint32_t kMagic = 0xB1A5B1A5;

enum MsgOp : uint32_t {
  kPush = 1,
  kPull = 2,
  kPushReply = 3,
  kPullReply = 4,
  kInit = 5,        // declare key: len, expected pushers, dtype/codec
  kInitReply = 6,
  kBarrier = 7,
  kBarrierReply = 8,
  kShutdown = 9,
  kIpcHello = 10,   // announce a shm region: payload = shm name,
                    // aux = size, key = region id (colocated fast path)
  kIpcHelloReply = 11,
  kRdmaConnect = 12,      // payload = RdmaPeerInfo; reply carries server's
  kRdmaConnectReply = 13,
  kRdmaHello = 14,        // announce a REGISTERED region: payload =
                          // {u64 addr, u64 size, u32 rkey}, key = region id
  kRdmaHelloReply = 15,
};

// Colocated IPC fast path: when cmd has kCmdIpcPayload set, the frame
// body is a fixed 16-byte IpcExt (locator + capacity) instead of the
// payload — the payload itself lives in a shm region the client
// announced with kIpcHello, so gradient bytes never cross the socket
// (reference ps-lite had an IPC van for the same reason,
// docs/best-practice.md:32).
constexpr uint32_t kCmdIpcPayload = 1u << 24;

struct IpcExt {
  uint64_t locator;  // region_id << 48 | byte offset
  uint64_t cap;      // pull request: receive capacity at locator
};

inline uint64_t make_locator(uint32_t region, uint64_t off) {
  return ((uint64_t)region << 48) | (off & ((1ULL << 48) - 1));
}
inline uint32_t locator_region(uint64_t l) { return (uint32_t)(l >> 48); }
inline uint64_t locator_off(uint64_t l) { return l & ((1ULL << 48) - 1); }

// codec ids on the wire
enum WireCodec : uint32_t {
  kRaw = 0,
  kOnebit = 1,
  kTopk = 2,
  kRandomk = 3,
  kDitherLinear = 4,
  kDitherNatural = 5,
  kFp8 = 6,            // OCP e4m3fn with per-partition amax scale
};

// cmd encoding: low 8 bits codec, next 8 bits dtype, bit 16 async-mode
inline uint32_t make_cmd(uint32_t codec, uint32_t dtype, bool async_mode) {
  return (codec & 0xFF) | ((dtype & 0xFF) << 8) | (async_mode ? 1u << 16 : 0);
}
inline uint32_t cmd_codec(uint32_t cmd) { return cmd & 0xFF; }
inline uint32_t cmd_dtype(uint32_t cmd) { return (cmd >> 8) & 0xFF; }
inline bool cmd_async(uint32_t cmd) { return (cmd >> 16) & 1; }

struct MsgHeader {
  uint32_t magic;
  uint32_t op;
  uint64_t key;
  uint64_t len;      // payload bytes following this header
  uint64_t aux;      // kInit: uncompressed bytes; kPush(randomk): k; else seed/version
  uint32_t sender;   // global worker rank
  uint32_t cmd;
  uint64_t seq;      // request id for reply matching
};

static_assert(sizeof(MsgHeader) == 48, "header must be 48 bytes");

}  // namespace bpsamd
