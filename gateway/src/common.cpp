#include "common.h"

#include <errno.h>
#include <fcntl.h>
#include <dirent.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/types.h>
#include <unistd.h>

#include <atomic>
#include <ctime>
#include <random>

namespace gw {

LogLevel g_log_level = LogLevel::kInfo;
static FILE* g_log_file = nullptr;
static std::mutex g_log_mu;
static std::string g_log_dir;
static int g_log_retention_days = 7;
static int g_log_day = -1;   // yyyymmdd of the open file

std::string iso8601_now() {
  time_t t = time(nullptr);
  struct tm tmv;
  gmtime_r(&t, &tmv);
  char buf[40];
  strftime(buf, sizeof buf, "%Y-%m-%dT%H:%M:%SZ", &tmv);
  return buf;
}

static int today_yyyymmdd() {
  time_t t = time(nullptr);
  struct tm tmv;
  gmtime_r(&t, &tmv);
  return (tmv.tm_year + 1900) * 10000 + (tmv.tm_mon + 1) * 100 +
         tmv.tm_mday;
}

// delete llmlb-gw-*.jsonl older than the retention window
// (reference logging.rs retention; mtime-based, best effort)
static void purge_old_logs() {
  if (g_log_dir.empty()) return;
  DIR* d = opendir(g_log_dir.c_str());
  if (!d) return;
  const time_t cutoff =
      time(nullptr) - (time_t)g_log_retention_days * 86400;
  struct dirent* e;
  while ((e = readdir(d)) != nullptr) {
    std::string name = e->d_name;
    if (name.rfind("llmlb-gw-", 0) != 0 ||
        name.find(".jsonl") == std::string::npos)
      continue;
    std::string path = g_log_dir + "/" + name;
    struct stat st;
    if (stat(path.c_str(), &st) == 0 && st.st_mtime < cutoff)
      unlink(path.c_str());
  }
  closedir(d);
}

// open (or roll to) today's dated file; caller holds g_log_mu
static void open_log_for_today() {
  const int day = today_yyyymmdd();
  if (day == g_log_day && g_log_file) return;
  if (g_log_file) fclose(g_log_file);
  char name[64];
  snprintf(name, sizeof name, "/llmlb-gw-%08d.jsonl", day);
  g_log_file = fopen((g_log_dir + name).c_str(), "a");
  g_log_day = day;
  purge_old_logs();
}

void log_init(const std::string& level, const std::string& jsonl_dir,
              int retention_days) {
  if (level == "debug") g_log_level = LogLevel::kDebug;
  else if (level == "info") g_log_level = LogLevel::kInfo;
  else if (level == "warn" || level == "warning") g_log_level = LogLevel::kWarn;
  else if (level == "error") g_log_level = LogLevel::kError;
  if (!jsonl_dir.empty()) {
    ::mkdir(jsonl_dir.c_str(), 0755);
    g_log_dir = jsonl_dir;
    g_log_retention_days = retention_days > 0 ? retention_days : 7;
    std::lock_guard<std::mutex> lk(g_log_mu);
    open_log_for_today();
  }
}

void log_write(LogLevel lv, const char* fmt, ...) {
  if (lv < g_log_level) return;
  static const char* names[] = {"DEBUG", "INFO", "WARN", "ERROR"};
  char msg[2048];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(msg, sizeof msg, fmt, ap);
  va_end(ap);
  std::string ts = iso8601_now();
  std::lock_guard<std::mutex> lk(g_log_mu);
  fprintf(stderr, "%s %-5s %s\n", ts.c_str(), names[(int)lv], msg);
  if (!g_log_dir.empty()) open_log_for_today();   // daily rotation
  if (g_log_file) {
    // minimal JSONL sink (stdout + file, reference logging.rs:18)
    fprintf(g_log_file, "{\"ts\":\"%s\",\"level\":\"%s\",\"msg\":\"", ts.c_str(),
            names[(int)lv]);
    for (const char* p = msg; *p; ++p) {
      if (*p == '"' || *p == '\\') fputc('\\', g_log_file);
      if (*p == '\n') { fputs("\\n", g_log_file); continue; }
      fputc(*p, g_log_file);
    }
    fputs("\"}\n", g_log_file);
    fflush(g_log_file);
  }
}

std::string to_lower(std::string_view s) {
  std::string r(s);
  for (auto& c : r) c = tolower((unsigned char)c);
  return r;
}

std::vector<std::string> split(std::string_view s, char sep) {
  std::vector<std::string> out;
  size_t start = 0;
  for (size_t i = 0; i <= s.size(); ++i) {
    if (i == s.size() || s[i] == sep) {
      out.emplace_back(s.substr(start, i - start));
      start = i + 1;
    }
  }
  return out;
}

std::string url_decode(std::string_view s) {
  std::string out;
  out.reserve(s.size());
  for (size_t i = 0; i < s.size(); ++i) {
    if (s[i] == '%' && i + 2 < s.size()) {
      auto hex = [](char c) -> int {
        if (c >= '0' && c <= '9') return c - '0';
        if (c >= 'a' && c <= 'f') return c - 'a' + 10;
        if (c >= 'A' && c <= 'F') return c - 'A' + 10;
        return -1;
      };
      int h = hex(s[i + 1]), l = hex(s[i + 2]);
      if (h >= 0 && l >= 0) {
        out.push_back((char)(h * 16 + l));
        i += 2;
        continue;
      }
    }
    out.push_back(s[i] == '+' ? ' ' : s[i]);
  }
  return out;
}

// OS entropy (getrandom(2), /dev/urandom fallback).  API keys and the
// uuid4 seed come from here: the reference generates token material
// with a CSPRNG (rand::rng ThreadRng), and mt19937 output alone is
// predictable/recoverable.
bool secure_random_bytes(void* buf, size_t n) {
#ifdef SYS_getrandom
  size_t off = 0;
  while (off < n) {
    long r = syscall(SYS_getrandom, (char*)buf + off, n - off, 0);
    if (r > 0) { off += (size_t)r; continue; }
    if (r < 0 && errno == EINTR) continue;
    break;
  }
  if (off == n) return true;
#endif
  int fd = open("/dev/urandom", O_RDONLY | O_CLOEXEC);
  if (fd < 0) return false;
  size_t off2 = 0;
  while (off2 < n) {
    ssize_t r = read(fd, (char*)buf + off2, n - off2);
    if (r <= 0) { if (r < 0 && errno == EINTR) continue; break; }
    off2 += (size_t)r;
  }
  close(fd);
  return off2 == n;
}

bool is_loopback_ip(std::string_view ip) {
  if (ip == "::1" || ip == "localhost") return true;
  if (starts_with(ip, "::ffff:")) ip.remove_prefix(7);
  return starts_with(ip, "127.");
}

std::string uuid4() {
  // ids only (request/message ids) — key material uses
  // secure_random_bytes directly.  Seeded from OS entropy so even id
  // streams are not recoverable from a weak seed.
  static thread_local std::mt19937_64 rng([] {
    uint64_t seed[2] = {0, 0};
    if (!secure_random_bytes(seed, sizeof seed))
      seed[0] = std::random_device{}() ^ (uint64_t)(uintptr_t)&seed;
    return seed[0] ^ (seed[1] << 1);
  }());
  uint64_t a = rng(), b = rng();
  a = (a & 0xffffffffffff0fffULL) | 0x4000ULL;         // version 4
  b = (b & 0x3fffffffffffffffULL) | 0x8000000000000000ULL;  // variant
  char buf[40];
  snprintf(buf, sizeof buf, "%08x-%04x-%04x-%04x-%012llx",
           (unsigned)(a >> 32), (unsigned)((a >> 16) & 0xffff),
           (unsigned)(a & 0xffff), (unsigned)(b >> 48),
           (unsigned long long)(b & 0xffffffffffffULL));
  return buf;
}

static const char kB64[] =
    "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
static const char kB64Url[] =
    "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";

std::string b64_encode(std::string_view in, bool url_safe, bool pad) {
  const char* tbl = url_safe ? kB64Url : kB64;
  std::string out;
  out.reserve((in.size() + 2) / 3 * 4);
  size_t i = 0;
  for (; i + 3 <= in.size(); i += 3) {
    uint32_t v = ((uint8_t)in[i] << 16) | ((uint8_t)in[i + 1] << 8) |
                 (uint8_t)in[i + 2];
    out += tbl[(v >> 18) & 63];
    out += tbl[(v >> 12) & 63];
    out += tbl[(v >> 6) & 63];
    out += tbl[v & 63];
  }
  size_t rem = in.size() - i;
  if (rem == 1) {
    uint32_t v = (uint8_t)in[i] << 16;
    out += tbl[(v >> 18) & 63];
    out += tbl[(v >> 12) & 63];
    if (pad) out += "==";
  } else if (rem == 2) {
    uint32_t v = ((uint8_t)in[i] << 16) | ((uint8_t)in[i + 1] << 8);
    out += tbl[(v >> 18) & 63];
    out += tbl[(v >> 12) & 63];
    out += tbl[(v >> 6) & 63];
    if (pad) out += '=';
  }
  return out;
}

std::string b64_decode(std::string_view in) {
  auto val = [](char c) -> int {
    if (c >= 'A' && c <= 'Z') return c - 'A';
    if (c >= 'a' && c <= 'z') return c - 'a' + 26;
    if (c >= '0' && c <= '9') return c - '0' + 52;
    if (c == '+' || c == '-') return 62;
    if (c == '/' || c == '_') return 63;
    return -1;
  };
  std::string out;
  uint32_t acc = 0;
  int bits = 0;
  for (char c : in) {
    if (c == '=') break;
    int v = val(c);
    if (v < 0) continue;
    acc = (acc << 6) | v;
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      out.push_back((char)((acc >> bits) & 0xff));
    }
  }
  return out;
}

std::string hex_encode(const uint8_t* p, size_t n) {
  static const char* hx = "0123456789abcdef";
  std::string out;
  out.reserve(n * 2);
  for (size_t i = 0; i < n; ++i) {
    out += hx[p[i] >> 4];
    out += hx[p[i] & 15];
  }
  return out;
}

}  // namespace gw
