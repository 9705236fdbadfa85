// Wire frame format for the moolib_amd transport.
//
// Capability parity with the reference's framed-buffer transport
// (src/transports/ipc.cc:60-228): one contiguous "head" (control fields +
// serialized payload + tensor metadata) followed by each tensor's storage as
// its own iovec — tensors are sent zero-copy and received into freshly
// allocated ATen storage, never memcpy'd through an intermediate buffer.
//
// Layout on the wire (little endian):
//   u64 totalLen   (bytes after this field)
//   u32 headLen
//   head[headLen]: u8 kind | u64 rid | u64 fid | u32 nTensors
//                  | nTensors * { u8 dtype, u8 ndim, i64 sizes[ndim] }
//                  | u64 payloadLen | payload bytes
//   blobs: each tensor's contiguous data, in order
#pragma once

#include <ATen/ATen.h>

#include <string>
#include <vector>

#include "common.h"

namespace mrl {

enum class FrameKind : uint8_t {
  greeting = 1,      // payload: proto magic, name, uid, addr list
  request = 2,       // rid/fid set, payload = serialized args
  response = 3,      // rid set, payload = serialized result
  errorResponse = 4, // rid set, payload = error string
  // (5 was a request-ack that was never part of the shipped protocol;
  //  reliability is resend-on-reconnect + receiver dedupe + response acks)
  responseAck = 6,   // rid set: "I received your response"
  keepalive = 7,
  findPeer = 8,      // payload: peer name being looked for
  peerInfo = 9,      // payload: peer name + addr list (gossip answer)
};

struct Frame {
  FrameKind kind = FrameKind::keepalive;
  uint64_t rid = 0;
  uint64_t fid = 0;
  std::string payload;
  std::vector<at::Tensor> tensors;  // CPU, contiguous
};

// dtype <-> wire code. Own stable mapping (do not rely on ATen enum values).
inline uint8_t dtypeToWire(at::ScalarType t) {
  switch (t) {
    case at::kFloat: return 1;
    case at::kDouble: return 2;
    case at::kHalf: return 3;
    case at::kBFloat16: return 4;
    case at::kLong: return 5;
    case at::kInt: return 6;
    case at::kShort: return 7;
    case at::kChar: return 8;
    case at::kByte: return 9;
    case at::kBool: return 10;
    default: throw RpcError("unsupported tensor dtype for rpc");
  }
}

inline at::ScalarType wireToDtype(uint8_t c) {
  switch (c) {
    case 1: return at::kFloat;
    case 2: return at::kDouble;
    case 3: return at::kHalf;
    case 4: return at::kBFloat16;
    case 5: return at::kLong;
    case 6: return at::kInt;
    case 7: return at::kShort;
    case 8: return at::kChar;
    case 9: return at::kByte;
    case 10: return at::kBool;
    default: throw RpcError("bad tensor dtype code on wire");
  }
}

// Serialize the frame head (everything but blobs), with the 12-byte
// totalLen/headLen prefix at the front.
inline std::string encodeFrameHead(const Frame& f) {
  WireWriter w;
  w.u64(0);  // totalLen placeholder
  w.u32(0);  // headLen placeholder
  w.u8(static_cast<uint8_t>(f.kind));
  w.u64(f.rid);
  w.u64(f.fid);
  w.u32(static_cast<uint32_t>(f.tensors.size()));
  uint64_t blobBytes = 0;
  for (const auto& t : f.tensors) {
    w.u8(dtypeToWire(t.scalar_type()));
    w.u8(static_cast<uint8_t>(t.dim()));
    for (int64_t s : t.sizes()) w.i64(s);
    blobBytes += static_cast<uint64_t>(t.nbytes());
  }
  w.u64(static_cast<uint64_t>(f.payload.size()));
  w.raw(f.payload.data(), f.payload.size());
  uint32_t headLen = static_cast<uint32_t>(w.out.size() - 12);
  uint64_t totalLen = 4 + headLen + blobBytes;
  std::memcpy(w.out.data(), &totalLen, 8);
  std::memcpy(w.out.data() + 8, &headLen, 4);
  return std::move(w.out);
}

// Parse a frame head (without the 12-byte prefix). Allocates CPU tensors
// whose storage the socket layer will readv blob data into.
inline Frame decodeFrameHead(std::string_view head) {
  Frame f;
  WireReader r(head);
  f.kind = static_cast<FrameKind>(r.u8());
  f.rid = r.u64();
  f.fid = r.u64();
  uint32_t nT = r.u32();
  if (nT > 65536) throw RpcError("wire: absurd tensor count");
  f.tensors.reserve(nT);
  for (uint32_t i = 0; i < nT; ++i) {
    at::ScalarType dt = wireToDtype(r.u8());
    uint8_t ndim = r.u8();
    std::vector<int64_t> sizes(ndim);
    for (auto& s : sizes) {
      s = r.i64();
      if (s < 0) throw RpcError("wire: negative tensor dim");
    }
    f.tensors.push_back(at::empty(sizes, at::TensorOptions().dtype(dt)));
  }
  uint64_t plen = r.u64();
  if (plen != r.remaining()) throw RpcError("wire: payload length mismatch");
  f.payload.assign(r.p, plen);
  return f;
}

}  // namespace mrl
