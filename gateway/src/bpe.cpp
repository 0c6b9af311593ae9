#include "bpe.h"

#include <limits.h>
#include <unistd.h>

#include <cstdlib>
#include <fstream>

#include "common.h"

namespace gw {

// base64 decode (standard alphabet, padded) — table files only
static bool b64dec(std::string_view in, std::string* out) {
  auto val = [](char c) -> int {
    if (c >= 'A' && c <= 'Z') return c - 'A';
    if (c >= 'a' && c <= 'z') return c - 'a' + 26;
    if (c >= '0' && c <= '9') return c - '0' + 52;
    if (c == '+') return 62;
    if (c == '/') return 63;
    return -1;
  };
  out->clear();
  unsigned buf = 0;
  int bits = 0;
  for (char c : in) {
    if (c == '=') break;
    int v = val(c);
    if (v < 0) return false;
    buf = (buf << 6) | (unsigned)v;
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      out->push_back((char)((buf >> bits) & 0xff));
    }
  }
  return true;
}

bool BpeTokenizer::load(const std::string& path) {
  std::ifstream f(path);
  if (!f.good()) return false;
  ranks_.clear();
  std::string line, tok;
  while (std::getline(f, line)) {
    if (line.empty()) continue;
    size_t sp = line.find(' ');
    if (sp == std::string::npos) continue;
    if (!b64dec(std::string_view(line).substr(0, sp), &tok)) continue;
    ranks_[tok] = atoi(line.c_str() + sp + 1);
  }
  return !ranks_.empty();
}

// GPT-2 regex pre-split, ASCII-exact (bytes >= 0x80 are classed as
// letters — a safe approximation for UTF-8 text):
//   's|'t|'re|'ve|'m|'ll|'d | ?L+ | ?N+ | ?P+ | \s+(?!\S) | \s+
long long BpeTokenizer::count(std::string_view t) const {
  auto is_sp = [](unsigned char c) {
    return c == ' ' || c == '\t' || c == '\n' || c == '\r' ||
           c == '\f' || c == '\v';
  };
  auto is_letter = [](unsigned char c) {
    return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') || c >= 0x80;
  };
  auto is_digit = [](unsigned char c) { return c >= '0' && c <= '9'; };

  long long total = 0;
  size_t i = 0, n = t.size();
  while (i < n) {
    unsigned char c = t[i];
    // contractions (lowercase per the GPT-2 regex)
    if (c == '\'' && i + 1 < n) {
      std::string_view rest = t.substr(i + 1);
      size_t clen = 0;
      if (starts_with(rest, "re") || starts_with(rest, "ve") ||
          starts_with(rest, "ll"))
        clen = 2;
      else if (rest[0] == 's' || rest[0] == 't' || rest[0] == 'm' ||
               rest[0] == 'd')
        clen = 1;
      if (clen) {
        total += count_piece(t.substr(i, clen + 1));
        i += clen + 1;
        continue;
      }
    }
    size_t start = i;
    bool led_space = false;
    if (c == ' ' && i + 1 < n && !is_sp(t[i + 1])) {
      led_space = true;
      ++i;
      c = t[i];
    }
    if (is_letter(c)) {
      while (i < n && is_letter((unsigned char)t[i])) ++i;
      total += count_piece(t.substr(start, i - start));
      continue;
    }
    if (is_digit(c)) {
      while (i < n && is_digit((unsigned char)t[i])) ++i;
      total += count_piece(t.substr(start, i - start));
      continue;
    }
    if (!is_sp(c)) {
      // punctuation run (not space/letter/digit); apostrophes that did
      // not match a contraction fall in here too
      while (i < n) {
        unsigned char d = t[i];
        if (is_sp(d) || is_letter(d) || is_digit(d)) break;
        if (d == '\'' && i + 1 < n) {
          // stop before a contraction start
          std::string_view rest = t.substr(i + 1);
          if (starts_with(rest, "re") || starts_with(rest, "ve") ||
              starts_with(rest, "ll") || rest[0] == 's' ||
              rest[0] == 't' || rest[0] == 'm' || rest[0] == 'd')
            break;
        }
        ++i;
      }
      if (i == start && led_space) {
        // lone trailing space only: emit as whitespace below
        i = start;
      } else if (i > start || led_space) {
        total += count_piece(t.substr(start, i - start));
        continue;
      }
    }
    if (led_space) {
      // space followed by whitespace-only tail: rewind, treat as run
      i = start;
      c = t[i];
    }
    // whitespace run; if followed by non-space and the run's last char
    // is ' ', leave that space for the next word piece
    size_t ws = i;
    while (ws < n && is_sp((unsigned char)t[ws])) ++ws;
    size_t end = ws;
    if (ws < n && t[ws - 1] == ' ' && ws - i >= 1) {
      end = ws - 1;  // last space binds to the following piece
    }
    if (end > i) {
      total += count_piece(t.substr(i, end - i));
      i = end;
    } else {
      // unreachable by construction (a lone space before a word is
      // taken by led_space) — consume one byte rather than risk a
      // non-progressing loop
      total += count_piece(t.substr(i, 1));
      ++i;
    }
  }
  return total;
}

long long BpeTokenizer::count_piece(std::string_view piece) const {
  if (piece.empty()) return 0;
  if (piece.size() == 1) return 1;
  // whole piece already a token?
  if (ranks_.count(std::string(piece))) return 1;
  // symbols as (offset, len) over piece; greedy lowest-rank merging
  std::vector<std::pair<size_t, size_t>> sym;
  sym.reserve(piece.size());
  for (size_t i = 0; i < piece.size(); ++i) sym.push_back({i, 1});
  std::string key;
  while (sym.size() > 1) {
    int best_rank = INT_MAX;
    size_t best_i = 0;
    for (size_t i = 0; i + 1 < sym.size(); ++i) {
      key.assign(piece.data() + sym[i].first,
                 sym[i].second + sym[i + 1].second);
      auto it = ranks_.find(key);
      if (it != ranks_.end() && it->second < best_rank) {
        best_rank = it->second;
        best_i = i;
      }
    }
    if (best_rank == INT_MAX) break;
    sym[best_i].second += sym[best_i + 1].second;
    sym.erase(sym.begin() + (long)best_i + 1);
  }
  return (long long)sym.size();
}

const BpeTokenizer* BpeTokenizer::default_instance() {
  // magic-static init: several worker threads can hit the first
  // estimation concurrently, so the load must be once-only and
  // thread-safe
  static const BpeTokenizer* inst = []() -> const BpeTokenizer* {
    static BpeTokenizer tok;
    bool ok = false;
    const char* env = getenv("LLMLB_TIKTOKEN_FILE");
    if (env && *env && tok.load(env)) {
      ok = true;
    } else {
      // <exe>/../data/llmlb.tiktoken, then ./gateway/data/...
      char exe[PATH_MAX];
      ssize_t m = readlink("/proc/self/exe", exe, sizeof exe - 1);
      if (m > 0) {
        exe[m] = 0;
        std::string dir(exe);
        size_t slash = dir.rfind('/');
        if (slash != std::string::npos) dir.resize(slash);
        if (tok.load(dir + "/../data/llmlb.tiktoken")) ok = true;
      }
      if (!ok && tok.load("gateway/data/llmlb.tiktoken")) ok = true;
    }
    if (ok)
      LOG_INFO("BPE token estimator loaded (%zu tokens)",
               tok.ranks_.size());
    else
      LOG_WARN("no BPE table found: token estimation falls back to "
               "chars/4 (set LLMLB_TIKTOKEN_FILE)");
    return ok ? &tok : nullptr;
  }();
  return inst;
}

}  // namespace gw
