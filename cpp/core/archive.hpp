// grapehip — byte-stream archives used as the message wire format.
// Reference parity: grape/serialization/{in_archive,out_archive}.h.
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace grapehip {

// Append-only byte buffer (reference InArchive).
class InArchive {
 public:
  template <typename T>
  void add(const T& v) {
    static_assert(std::is_trivially_copyable<T>::value, "POD only");
    size_t off = buf_.size();
    buf_.resize(off + sizeof(T));
    std::memcpy(buf_.data() + off, &v, sizeof(T));
  }
  void add_bytes(const void* p, size_t n) {
    size_t off = buf_.size();
    buf_.resize(off + n);
    std::memcpy(buf_.data() + off, p, n);
  }
  const char* data() const { return buf_.data(); }
  size_t size() const { return buf_.size(); }
  void clear() { buf_.clear(); }
  std::string release() { return std::move(buf_); }
  std::string& buffer() { return buf_; }

 private:
  std::string buf_;
};

// Consuming reader (reference OutArchive).
class OutArchive {
 public:
  OutArchive() = default;
  explicit OutArchive(std::string blob) : buf_(std::move(blob)) {}
  void reset(std::string blob) {
    buf_ = std::move(blob);
    pos_ = 0;
  }
  template <typename T>
  bool get(T* v) {
    static_assert(std::is_trivially_copyable<T>::value, "POD only");
    if (pos_ + sizeof(T) > buf_.size()) return false;
    std::memcpy(v, buf_.data() + pos_, sizeof(T));
    pos_ += sizeof(T);
    return true;
  }
  bool empty() const { return pos_ >= buf_.size(); }
  size_t remaining() const { return buf_.size() - pos_; }
  const char* cursor() const { return buf_.data() + pos_; }
  void skip(size_t n) { pos_ += n; }

 private:
  std::string buf_;
  size_t pos_ = 0;
};

}  // namespace grapehip
