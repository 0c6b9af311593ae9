// llmlb-gw — common utilities for the MI355X-native gateway data plane.
// Design: epoll event loops (one per core, SO_REUSEPORT), non-blocking
// sockets, single-buffer forwarding (scan-in-place SSE tap, no per-chunk
// copies).  Mirrors the behavior of the reference gateway's hot path
//  with a zero-copy pump.
#pragma once

#include <algorithm>
#include <cassert>
#include <cerrno>
#include <chrono>
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <string_view>
#include <vector>

namespace gw {

// ---------------------------------------------------------------- time
inline int64_t now_us() {
  using namespace std::chrono;
  return duration_cast<microseconds>(steady_clock::now().time_since_epoch())
      .count();
}
inline int64_t wall_ms() {
  using namespace std::chrono;
  return duration_cast<milliseconds>(system_clock::now().time_since_epoch())
      .count();
}
// RFC3339 UTC timestamp
std::string iso8601_now();

// ---------------------------------------------------------------- logging
enum class LogLevel { kDebug = 0, kInfo, kWarn, kError };
extern LogLevel g_log_level;
void log_init(const std::string& level, const std::string& jsonl_dir,
              int retention_days = 7);
void log_write(LogLevel lv, const char* fmt, ...)
    __attribute__((format(printf, 2, 3)));

#define LOG_DEBUG(...)                                     \
  do {                                                     \
    if (gw::g_log_level <= gw::LogLevel::kDebug)           \
      gw::log_write(gw::LogLevel::kDebug, __VA_ARGS__);    \
  } while (0)
#define LOG_INFO(...) gw::log_write(gw::LogLevel::kInfo, __VA_ARGS__)
#define LOG_WARN(...) gw::log_write(gw::LogLevel::kWarn, __VA_ARGS__)
#define LOG_ERROR(...) gw::log_write(gw::LogLevel::kError, __VA_ARGS__)

// ---------------------------------------------------------------- buffer
// Growable byte buffer with a consumed prefix — the unit of the
// zero-copy pump: bytes are read once into a Buffer, scanned in place,
// and written out from the same storage.
class Buffer {
 public:
  const char* data() const { return buf_.data() + off_; }
  size_t size() const { return len_ - off_; }
  bool empty() const { return size() == 0; }
  std::string_view view() const { return {data(), size()}; }

  void append(const void* p, size_t n) {
    ensure(n);
    memcpy(buf_.data() + len_, p, n);
    len_ += n;
  }
  void append(std::string_view s) { append(s.data(), s.size()); }
  // reserve n writable bytes at the end; returns pointer (commit() to
  // make written bytes visible — no zero-fill of the window)
  char* reserve(size_t n) {
    ensure(n);
    return buf_.data() + len_;
  }
  void commit(size_t n) {
    len_ += n;
    assert(len_ <= buf_.size());
  }
  void consume(size_t n) {
    off_ += n;
    assert(off_ <= len_);
    if (off_ == len_) {
      off_ = len_ = 0;
    } else if (off_ > 65536 && off_ > len_ / 2) {
      memmove(buf_.data(), buf_.data() + off_, len_ - off_);
      len_ -= off_;
      off_ = 0;
    }
  }
  void clear() { off_ = len_ = 0; }

 private:
  void ensure(size_t n) {
    if (len_ + n <= buf_.size()) return;
    // compact the dead prefix first if it buys enough
    if (off_ > 0 && len_ - off_ + n <= buf_.size() && off_ >= n) {
      memmove(buf_.data(), buf_.data() + off_, len_ - off_);
      len_ -= off_;
      off_ = 0;
      if (len_ + n <= buf_.size()) return;
    }
    size_t want = std::max(buf_.size() * 2, len_ + n);
    if (want < 4096) want = 4096;
    buf_.resize(want);
  }
  std::vector<char> buf_;
  size_t off_ = 0;   // consumed prefix
  size_t len_ = 0;   // valid bytes
};

// ---------------------------------------------------------------- strings
inline bool ieq(std::string_view a, std::string_view b) {
  if (a.size() != b.size()) return false;
  for (size_t i = 0; i < a.size(); ++i)
    if (tolower((unsigned char)a[i]) != tolower((unsigned char)b[i]))
      return false;
  return true;
}
inline bool starts_with(std::string_view s, std::string_view p) {
  return s.size() >= p.size() && s.substr(0, p.size()) == p;
}
inline std::string_view trim(std::string_view s) {
  while (!s.empty() && (s.front() == ' ' || s.front() == '\t'))
    s.remove_prefix(1);
  while (!s.empty() && (s.back() == ' ' || s.back() == '\t' ||
                        s.back() == '\r' || s.back() == '\n'))
    s.remove_suffix(1);
  return s;
}
std::string to_lower(std::string_view s);
std::vector<std::string> split(std::string_view s, char sep);
std::string url_decode(std::string_view s);

// uuid v4 (for endpoint / request ids)
std::string uuid4();
// fills buf with n bytes of OS entropy (getrandom/urandom); false on
// failure.  Use for all key/token material.
bool secure_random_bytes(void* buf, size_t n);
// true for 127.0.0.0/8, ::1 and their IPv4-mapped forms
bool is_loopback_ip(std::string_view ip);
inline bool ends_with(std::string_view s, std::string_view sfx) {
  return s.size() >= sfx.size() &&
         s.compare(s.size() - sfx.size(), sfx.size(), sfx) == 0;
}

// base64 / base64url
std::string b64_encode(std::string_view in, bool url_safe = false,
                       bool pad = true);
std::string b64_decode(std::string_view in);

std::string hex_encode(const uint8_t* p, size_t n);

}  // namespace gw
