// SArray<V>: zero-copy shared-ownership array with a device tag.
//
// Reference parity: ps-lite include/ps/sarray.h (SArray, segment, reset,
// src/dst device fields). Re-designed MI355X-first: the device tag is a
// plain HIP device ordinal (-1 = host); payloads that live on a GPU come
// from the process-wide HBM pool (hip_pool.h) so they are hipIpc-shareable
// and never re-registered per message.
#pragma once

#include <algorithm>
#include <functional>
#include <memory>
#include <vector>

#include "base.h"

namespace xps {

static const int kCPU = -1;  // device ordinal for host memory

template <typename V>
class SArray {
 public:
  SArray() = default;

  // Allocate host memory of `count` elements.
  explicit SArray(size_t count) { Resize(count); }
  SArray(size_t count, V init) {
    Resize(count);
    std::fill(ptr_, ptr_ + count, init);
  }

  // Borrow external memory without taking ownership.
  SArray(V* data, size_t count, int device = kCPU) {
    ptr_ = data;
    size_ = count;
    device_ = device;
    own_ = std::shared_ptr<void>(data, [](void*) {});
  }

  // Take shared ownership of external memory with a custom deleter.
  SArray(V* data, size_t count, std::function<void(V*)> deleter, int device = kCPU) {
    ptr_ = data;
    size_ = count;
    device_ = device;
    own_ = std::shared_ptr<void>(data, [deleter](void* p) { deleter(static_cast<V*>(p)); });
  }

  SArray(const std::vector<V>& vec) {  // NOLINT: implicit for ergonomics
    Resize(vec.size());
    std::copy(vec.begin(), vec.end(), ptr_);
  }

  SArray(const std::initializer_list<V>& l) {  // NOLINT
    Resize(l.size());
    std::copy(l.begin(), l.end(), ptr_);
  }

  // Zero-copy reinterpret from another element type (sizes must divide).
  template <typename W>
  static SArray<V> View(const SArray<W>& other) {
    SArray<V> out;
    out.ptr_ = reinterpret_cast<V*>(other.data());
    out.size_ = other.size() * sizeof(W) / sizeof(V);
    out.device_ = other.device();
    out.own_ = other.ownership();
    return out;
  }

  void Resize(size_t count) {
    V* p = static_cast<V*>(malloc(std::max<size_t>(count, 1) * sizeof(V)));
    XPS_CHECK_NOTNULL(p);
    ptr_ = p;
    size_ = count;
    device_ = kCPU;
    own_ = std::shared_ptr<void>(p, [](void* q) { free(q); });
  }

  void CopyFrom(const V* src, size_t count) {
    Resize(count);
    std::copy(src, src + count, ptr_);
  }

  // Zero-copy sub-range view [begin, end).
  SArray<V> Segment(size_t begin, size_t end) const {
    XPS_CHECK_LE(begin, end);
    XPS_CHECK_LE(end, size_);
    SArray<V> out;
    out.ptr_ = ptr_ + begin;
    out.size_ = end - begin;
    out.device_ = device_;
    out.own_ = own_;
    return out;
  }

  V* data() const { return ptr_; }
  size_t size() const { return size_; }
  size_t nbytes() const { return size_ * sizeof(V); }
  bool empty() const { return size_ == 0; }
  int device() const { return device_; }
  void set_device(int d) { device_ = d; }
  bool on_device() const { return device_ >= 0; }
  const std::shared_ptr<void>& ownership() const { return own_; }

  V& operator[](size_t i) const { return ptr_[i]; }
  V* begin() const { return ptr_; }
  V* end() const { return ptr_ + size_; }

  void clear() {
    ptr_ = nullptr;
    size_ = 0;
    device_ = kCPU;
    own_.reset();
  }

 private:
  template <typename W>
  friend class SArray;
  V* ptr_ = nullptr;
  size_t size_ = 0;
  int device_ = kCPU;
  std::shared_ptr<void> own_;
};

}  // namespace xps
