// HTTP/1.1 parsing for the gateway: in-place request/response header
// parsing (string_views into the connection buffer — no copies), a
// stateful chunked-transfer scanner that reports payload ranges without
// rewriting them (the zero-copy SSE pump consumes these), and an SSE
// tap that counts tokens / extracts usage while bytes are forwarded
// verbatim (the reference's per-chunk lossy-utf8 copy in
// api/proxy.rs:227 is exactly what this design eliminates).
#pragma once

#include <string>
#include <string_view>
#include <vector>

#include "common.h"

namespace gw {

struct Header {
  std::string_view name;
  std::string_view value;
};

struct HttpRequest {
  std::string_view method;
  std::string_view target;   // path?query
  std::string_view path;
  std::string_view query;
  std::vector<Header> headers;
  size_t header_len = 0;     // bytes of request line + headers + CRLF
  long long content_length = 0;
  bool chunked = false;
  bool keep_alive = true;

  std::string_view header(std::string_view name) const {
    for (auto& h : headers)
      if (ieq(h.name, name)) return h.value;
    return {};
  }
};

struct HttpResponse {
  int status = 0;
  std::string_view reason;
  std::vector<Header> headers;
  size_t header_len = 0;
  long long content_length = -1;  // -1 = none given (read to EOF)
  bool chunked = false;
  bool keep_alive = true;

  std::string_view header(std::string_view name) const {
    for (auto& h : headers)
      if (ieq(h.name, name)) return h.value;
    return {};
  }
};

// Returns 1 = complete headers parsed, 0 = need more data, -1 = malformed.
int parse_request_head(std::string_view buf, HttpRequest& out);
int parse_response_head(std::string_view buf, HttpResponse& out);

// --------------------------------------------------------------------
// Chunked-transfer scanner: feed raw bytes, get back payload ranges
// (offset/len into the fed buffer) plus completion, WITHOUT modifying
// the bytes (they are forwarded verbatim downstream).
// --------------------------------------------------------------------
class ChunkedScanner {
 public:
  struct Result {
    // payload byte ranges within the buffer passed to scan()
    std::vector<std::pair<size_t, size_t>> payload;
    size_t consumed = 0;  // bytes of input consumed (all, unless error)
    bool complete = false;
    bool error = false;
  };
  Result scan(std::string_view data);
  bool complete() const { return state_ == State::kDone; }

 private:
  enum class State { kSize, kSizeLf, kData, kDataCr, kDataLf, kTrailer,
                     kDone } state_ = State::kSize;
  long long remaining_ = 0;
  std::string sizeline_;
  int trailer_crlf_ = 0;
};

// --------------------------------------------------------------------
// Body reader: wraps content-length / chunked / read-to-EOF semantics.
// --------------------------------------------------------------------
class BodyReader {
 public:
  void init(const HttpResponse& resp) {
    mode_ = resp.chunked ? Mode::kChunked
            : resp.content_length >= 0 ? Mode::kLength
                                       : Mode::kEof;
    remaining_ = resp.content_length;
    done_ = (mode_ == Mode::kLength && remaining_ == 0);
  }
  // Feed raw body bytes; returns payload ranges + whether response done.
  // `consumed` = bytes that belong to this response (rest = next resp).
  struct Out {
    std::vector<std::pair<size_t, size_t>> payload;
    size_t consumed = 0;
    bool done = false;
  };
  Out feed(std::string_view data) {
    Out o;
    if (done_) {
      o.done = true;
      return o;
    }
    switch (mode_) {
      case Mode::kLength: {
        size_t take = data.size();
        if ((long long)take >= remaining_) take = (size_t)remaining_;
        if (take) o.payload.push_back({0, take});
        remaining_ -= take;
        o.consumed = take;
        if (remaining_ == 0) done_ = true;
        break;
      }
      case Mode::kChunked: {
        auto r = chunked_.scan(data);
        o.payload = std::move(r.payload);
        o.consumed = r.consumed;
        if (r.complete) done_ = true;
        if (r.error) done_ = true;  // treat as connection-fatal upstream
        break;
      }
      case Mode::kEof:
        if (!data.empty()) o.payload.push_back({0, data.size()});
        o.consumed = data.size();
        break;
    }
    o.done = done_;
    return o;
  }
  void finish_eof() { done_ = true; }
  bool done() const { return done_; }
  bool eof_terminated() const { return mode_ == Mode::kEof; }

 private:
  enum class Mode { kLength, kChunked, kEof } mode_ = Mode::kEof;
  long long remaining_ = 0;
  bool done_ = false;
  ChunkedScanner chunked_;
};

// --------------------------------------------------------------------
// SSE tap: fed decoded payload ranges; scans `data:` lines in place to
// count completion events and capture the final `usage` object.  Only
// a partial trailing line is carried over (tiny copy); full lines are
// scanned as views.
// --------------------------------------------------------------------
class SseTap {
 public:
  void feed(std::string_view payload);
  long long completion_tokens() const {
    return usage_completion_ >= 0 ? usage_completion_ : event_tokens_;
  }
  // true when the count came from the upstream usage object (exact);
  // false = callers should BPE-estimate from text() (token/mod.rs:217)
  bool usage_seen() const { return usage_completion_ >= 0; }
  // accumulated delta content (capped) for the BPE estimate
  const std::string& text() const { return text_; }
  long long prompt_tokens() const { return usage_prompt_; }
  bool saw_done() const { return saw_done_; }
  long long events() const { return events_; }

 private:
  void line(std::string_view l);
  std::string carry_;
  std::string text_;            // unescaped delta content, <= 256 KiB
  long long events_ = 0;
  long long event_tokens_ = 0;  // fallback estimate: one token per delta
  long long usage_prompt_ = -1;
  long long usage_completion_ = -1;
  bool saw_done_ = false;
};

// appends the FIRST top-level-ish "key":"..." string value of `body`,
// JSON-unescaped, to *out (capped at `cap` bytes total); returns true
// when a value was found
bool append_json_string_field(std::string_view body, std::string_view key,
                              std::string* out, size_t cap);

// helpers
std::string http_date_now();
void append_status_line(Buffer& b, int status, std::string_view reason);
std::string_view status_reason(int status);

}  // namespace gw
