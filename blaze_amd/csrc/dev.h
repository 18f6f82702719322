// dev.h — HIP device-memory RAII + error plumbing for the engine.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <utility>

namespace auron {

struct HipError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

inline void hip_check(hipError_t e, const char* what) {
  if (e != hipSuccess) {
    throw HipError(std::string(what) + ": " + hipGetErrorString(e));
  }
}

#define AURON_HIP(call) ::auron::hip_check((call), #call)

// Owned device buffer. Plain hipMalloc/hipFree; the working set per task is a
// handful of large buffers, so a pool buys little at this stage.
class DevBuf {
 public:
  DevBuf() = default;
  explicit DevBuf(size_t size) { alloc(size); }
  DevBuf(const DevBuf&) = delete;
  DevBuf& operator=(const DevBuf&) = delete;
  DevBuf(DevBuf&& o) noexcept : ptr_(o.ptr_), size_(o.size_) {
    o.ptr_ = nullptr;
    o.size_ = 0;
  }
  DevBuf& operator=(DevBuf&& o) noexcept {
    if (this != &o) {
      free();
      ptr_ = o.ptr_;
      size_ = o.size_;
      o.ptr_ = nullptr;
      o.size_ = 0;
    }
    return *this;
  }
  ~DevBuf() { free(); }

  void alloc(size_t size) {
    free();
    if (size == 0) return;
    AURON_HIP(hipMalloc(&ptr_, size));
    size_ = size;
  }
  void free() {
    if (ptr_) {
      (void)hipFree(ptr_);
      ptr_ = nullptr;
      size_ = 0;
    }
  }
  void* release() { return std::exchange(ptr_, nullptr); }

  template <typename T = void>
  T* get() const {
    return static_cast<T*>(ptr_);
  }
  size_t size() const { return size_; }
  explicit operator bool() const { return ptr_ != nullptr; }

 private:
  void* ptr_ = nullptr;
  size_t size_ = 0;
};

// Pinned (page-locked) host buffer for small metadata readbacks — pageable
// D2H copies cost milliseconds per call on this stack.
class PinnedBuf {
 public:
  PinnedBuf() = default;
  explicit PinnedBuf(size_t size) { alloc(size); }
  PinnedBuf(const PinnedBuf&) = delete;
  PinnedBuf& operator=(const PinnedBuf&) = delete;
  ~PinnedBuf() { free(); }

  void alloc(size_t size) {
    free();
    if (size == 0) return;
    AURON_HIP(hipHostMalloc(&ptr_, size));
    size_ = size;
  }
  void free() {
    if (ptr_) {
      (void)hipHostFree(ptr_);
      ptr_ = nullptr;
      size_ = 0;
    }
  }
  template <typename T = void>
  T* get() const {
    return static_cast<T*>(ptr_);
  }
  size_t size() const { return size_; }

 private:
  void* ptr_ = nullptr;
  size_t size_ = 0;
};

}  // namespace auron
