// Meta <-> flat bytes serialization for the TCP van and shm rings.
//
// Reference parity: ps-lite src/van.cc:689-831 PackMeta/UnpackMeta over
// src/meta.h POD structs. Re-designed as an explicit little-endian
// byte-stream writer/reader (no protobuf, no struct-punning).
#pragma once

#include <cstring>
#include <string>
#include <vector>

#include "message.h"

namespace xps {

class ByteWriter {
 public:
  explicit ByteWriter(std::string* out) : out_(out) {}
  void U8(uint8_t v) { out_->push_back(static_cast<char>(v)); }
  void I32(int32_t v) { Raw(&v, 4); }
  void U64(uint64_t v) { Raw(&v, 8); }
  void I64(int64_t v) { Raw(&v, 8); }
  void Str(const std::string& s) {
    U64(s.size());
    out_->append(s);
  }
  void Raw(const void* p, size_t n) { out_->append(static_cast<const char*>(p), n); }

 private:
  std::string* out_;
};

class ByteReader {
 public:
  ByteReader(const char* p, size_t n) : p_(p), end_(p + n) {}
  uint8_t U8() { uint8_t v; Raw(&v, 1); return v; }
  int32_t I32() { int32_t v; Raw(&v, 4); return v; }
  uint64_t U64() { uint64_t v; Raw(&v, 8); return v; }
  int64_t I64() { int64_t v; Raw(&v, 8); return v; }
  std::string Str() {
    uint64_t n = U64();
    // compare against the remaining length, never p_ + n (which can wrap
    // for a hostile/corrupt length and defeat the bounds check)
    XPS_CHECK_LE(n, static_cast<uint64_t>(end_ - p_)) << "wire: truncated string";
    std::string s(p_, n);
    p_ += n;
    return s;
  }
  void Raw(void* out, size_t n) {
    XPS_CHECK_LE(n, static_cast<size_t>(end_ - p_)) << "wire: truncated field";
    memcpy(out, p_, n);
    p_ += n;
  }
  bool AtEnd() const { return p_ == end_; }

 private:
  const char* p_;
  const char* end_;
};

void PackMeta(const Meta& meta, std::string* out);
void UnpackMeta(const char* buf, size_t len, Meta* meta);

}  // namespace xps
