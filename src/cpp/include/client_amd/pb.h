// Minimal protobuf wire-format reader/writer (proto3 subset: varint,
// 64-bit, length-delimited, 32-bit). The C++ gRPC client hand-encodes
// the KServe-v2 messages with these helpers — no protoc, no libprotobuf
// (field numbers per the schema the reference vendors at
// src/rust/triton-client/proto/grpc_service.proto).
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace client_amd {
namespace pb {

enum WireType { VARINT = 0, I64 = 1, LEN = 2, I32 = 5 };

class Writer {
 public:
  std::string out;

  void varint(uint64_t v) {
    while (v >= 0x80) {
      out.push_back((char)(0x80 | (v & 0x7F)));
      v >>= 7;
    }
    out.push_back((char)v);
  }
  void tag(int field, int wire) { varint(((uint64_t)field << 3) | wire); }
  void put_uint(int field, uint64_t v) {
    if (v == 0) return;  // proto3 default elision
    tag(field, VARINT);
    varint(v);
  }
  void put_uint_always(int field, uint64_t v) {
    tag(field, VARINT);
    varint(v);
  }
  void put_int(int field, int64_t v) { put_uint(field, (uint64_t)v); }
  void put_bool(int field, bool v) {
    if (v) put_uint_always(field, 1);
  }
  void put_str(int field, const std::string& s) {
    if (s.empty()) return;
    tag(field, LEN);
    varint(s.size());
    out.append(s);
  }
  void put_bytes(int field, const void* data, size_t n) {
    tag(field, LEN);
    varint(n);
    out.append((const char*)data, n);
  }
  void put_msg(int field, const std::string& encoded) {
    tag(field, LEN);
    varint(encoded.size());
    out.append(encoded);
  }
  void put_packed_i64(int field, const std::vector<int64_t>& vals) {
    if (vals.empty()) return;
    Writer tmp;
    for (int64_t v : vals) tmp.varint((uint64_t)v);
    put_msg(field, tmp.out);
  }
  void put_double(int field, double v) {
    if (v == 0) return;
    tag(field, I64);
    uint64_t bits;
    memcpy(&bits, &v, 8);
    for (int i = 0; i < 8; ++i) out.push_back((char)((bits >> (8 * i)) & 0xFF));
  }
};

class Reader {
 public:
  const uint8_t* p;
  const uint8_t* end;
  bool ok = true;

  Reader(const void* data, size_t n)
      : p((const uint8_t*)data), end((const uint8_t*)data + n) {}

  bool done() const { return p >= end || !ok; }

  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (p < end && shift < 64) {
      uint8_t b = *p++;
      v |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
    }
    ok = false;
    return v;
  }

  bool next(int* field, int* wire) {
    if (done()) return false;
    uint64_t key = varint();
    if (!ok) return false;
    *field = (int)(key >> 3);
    *wire = (int)(key & 7);
    return true;
  }

  // For LEN fields: view of the payload (advances past it).
  std::pair<const uint8_t*, size_t> bytes() {
    uint64_t n = varint();
    if (!ok || (uint64_t)(end - p) < n) {
      ok = false;
      return {nullptr, 0};
    }
    const uint8_t* start = p;
    p += n;
    return {start, (size_t)n};
  }

  std::string str() {
    auto [ptr, n] = bytes();
    return ptr ? std::string((const char*)ptr, n) : std::string();
  }

  uint64_t fixed64() {
    if ((size_t)(end - p) < 8) {
      ok = false;
      return 0;
    }
    uint64_t v = 0;
    for (int i = 0; i < 8; ++i) v |= (uint64_t)p[i] << (8 * i);
    p += 8;
    return v;
  }

  uint32_t fixed32() {
    if ((size_t)(end - p) < 4) {
      ok = false;
      return 0;
    }
    uint32_t v = 0;
    for (int i = 0; i < 4; ++i) v |= (uint32_t)p[i] << (8 * i);
    p += 4;
    return v;
  }

  void skip(int wire) {
    switch (wire) {
      case VARINT: varint(); break;
      case I64: fixed64(); break;
      case LEN: bytes(); break;
      case I32: fixed32(); break;
      default: ok = false;
    }
  }

  std::vector<int64_t> packed_i64() {
    auto [ptr, n] = bytes();
    std::vector<int64_t> vals;
    Reader sub(ptr, n);
    while (!sub.done()) vals.push_back((int64_t)sub.varint());
    return vals;
  }
};

}  // namespace pb
}  // namespace client_amd
