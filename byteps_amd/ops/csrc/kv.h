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
};

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
