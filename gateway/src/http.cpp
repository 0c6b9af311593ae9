#include "http.h"

#include "json.h"

namespace gw {

static int parse_headers_common(std::string_view buf, size_t line_end,
                                std::vector<Header>& headers,
                                size_t& header_len) {
  size_t pos = line_end + 2;
  while (true) {
    size_t eol = buf.find("\r\n", pos);
    if (eol == std::string_view::npos) return 0;
    if (eol == pos) {  // empty line: end of headers
      header_len = eol + 2;
      return 1;
    }
    std::string_view line = buf.substr(pos, eol - pos);
    size_t colon = line.find(':');
    if (colon == std::string_view::npos) return -1;
    Header h;
    h.name = trim(line.substr(0, colon));
    h.value = trim(line.substr(colon + 1));
    headers.push_back(h);
    if (headers.size() > 128) return -1;
    pos = eol + 2;
  }
}

int parse_request_head(std::string_view buf, HttpRequest& out) {
  size_t eol = buf.find("\r\n");
  if (eol == std::string_view::npos)
    return buf.size() > 16384 ? -1 : 0;
  std::string_view line = buf.substr(0, eol);
  size_t sp1 = line.find(' ');
  if (sp1 == std::string_view::npos) return -1;
  size_t sp2 = line.rfind(' ');
  if (sp2 == sp1) return -1;
  out.method = line.substr(0, sp1);
  out.target = line.substr(sp1 + 1, sp2 - sp1 - 1);
  std::string_view ver = line.substr(sp2 + 1);
  size_t qpos = out.target.find('?');
  out.path = qpos == std::string_view::npos ? out.target
                                            : out.target.substr(0, qpos);
  out.query = qpos == std::string_view::npos
                  ? std::string_view{}
                  : out.target.substr(qpos + 1);
  out.headers.clear();
  int rc = parse_headers_common(buf, eol, out.headers, out.header_len);
  if (rc != 1) return rc;
  out.content_length = 0;
  out.chunked = false;
  out.keep_alive = !(ver == "HTTP/1.0");
  for (auto& h : out.headers) {
    if (ieq(h.name, "content-length")) {
      out.content_length = atoll(std::string(h.value).c_str());
    } else if (ieq(h.name, "transfer-encoding")) {
      if (to_lower(h.value).find("chunked") != std::string::npos)
        out.chunked = true;
    } else if (ieq(h.name, "connection")) {
      std::string v = to_lower(h.value);
      if (v.find("close") != std::string::npos) out.keep_alive = false;
      else if (v.find("keep-alive") != std::string::npos)
        out.keep_alive = true;
    }
  }
  return 1;
}

int parse_response_head(std::string_view buf, HttpResponse& out) {
  size_t eol = buf.find("\r\n");
  if (eol == std::string_view::npos)
    return buf.size() > 16384 ? -1 : 0;
  std::string_view line = buf.substr(0, eol);
  if (!starts_with(line, "HTTP/1.")) return -1;
  size_t sp1 = line.find(' ');
  if (sp1 == std::string_view::npos || sp1 + 4 > line.size()) return -1;
  out.status = atoi(std::string(line.substr(sp1 + 1, 3)).c_str());
  size_t sp2 = line.find(' ', sp1 + 1);
  out.reason = sp2 == std::string_view::npos ? std::string_view{}
                                             : line.substr(sp2 + 1);
  out.headers.clear();
  int rc = parse_headers_common(buf, eol, out.headers, out.header_len);
  if (rc != 1) return rc;
  out.content_length = -1;
  out.chunked = false;
  out.keep_alive = line.substr(0, 8) != "HTTP/1.0";
  for (auto& h : out.headers) {
    if (ieq(h.name, "content-length")) {
      out.content_length = atoll(std::string(h.value).c_str());
    } else if (ieq(h.name, "transfer-encoding")) {
      if (to_lower(h.value).find("chunked") != std::string::npos)
        out.chunked = true;
    } else if (ieq(h.name, "connection")) {
      std::string v = to_lower(h.value);
      if (v.find("close") != std::string::npos) out.keep_alive = false;
    }
  }
  if (out.status == 204 || out.status == 304 ||
      (out.status >= 100 && out.status < 200))
    out.content_length = out.content_length < 0 ? 0 : out.content_length;
  return 1;
}

// ------------------------------------------------------------------ chunked
ChunkedScanner::Result ChunkedScanner::scan(std::string_view data) {
  Result r;
  size_t i = 0;
  const size_t n = data.size();
  while (i < n) {
    switch (state_) {
      case State::kSize: {
        char c = data[i];
        if (c == '\r') {
          // parse accumulated size line (may contain extensions)
          remaining_ = strtoll(sizeline_.c_str(), nullptr, 16);
          sizeline_.clear();
          state_ = State::kSizeLf;
        } else {
          if (sizeline_.size() > 32) { r.error = true; r.consumed = i; return r; }
          sizeline_ += c;
        }
        ++i;
        break;
      }
      case State::kSizeLf:
        if (data[i] != '\n') { r.error = true; r.consumed = i; return r; }
        ++i;
        state_ = remaining_ == 0 ? State::kTrailer : State::kData;
        trailer_crlf_ = 0;
        break;
      case State::kData: {
        size_t take = n - i;
        if ((long long)take > remaining_) take = (size_t)remaining_;
        if (take) r.payload.push_back({i, take});
        i += take;
        remaining_ -= take;
        if (remaining_ == 0) state_ = State::kDataCr;
        break;
      }
      case State::kDataCr:
        if (data[i] != '\r') { r.error = true; r.consumed = i; return r; }
        ++i;
        state_ = State::kDataLf;
        break;
      case State::kDataLf:
        if (data[i] != '\n') { r.error = true; r.consumed = i; return r; }
        ++i;
        state_ = State::kSize;
        break;
      case State::kTrailer:
        // consume trailer lines until the blank CRLF
        if (data[i] == '\r') {
          ++i;
        } else if (data[i] == '\n') {
          ++i;
          ++trailer_crlf_;
          // blank line directly after the 0-size line ends the body;
          // a trailer line resets via non-CRLF chars below
          state_ = State::kDone;
          r.complete = true;
          r.consumed = i;
          return r;
        } else {
          // trailer header line: consume to its LF then continue
          size_t lf = data.find('\n', i);
          if (lf == std::string_view::npos) {
            r.consumed = n;
            return r;
          }
          i = lf + 1;
        }
        break;
      case State::kDone:
        r.complete = true;
        r.consumed = i;
        return r;
    }
  }
  r.consumed = n;
  r.complete = (state_ == State::kDone);
  return r;
}

// ------------------------------------------------------------------ SSE tap
void SseTap::feed(std::string_view payload) {
  size_t start = 0;
  while (true) {
    size_t nl = payload.find('\n', start);
    if (nl == std::string_view::npos) {
      carry_.append(payload.substr(start));
      if (carry_.size() > 1 << 20) carry_.clear();  // pathological line
      return;
    }
    if (!carry_.empty()) {
      carry_.append(payload.substr(start, nl - start));
      line(carry_);
      carry_.clear();
    } else {
      line(payload.substr(start, nl - start));
    }
    start = nl + 1;
  }
}

void SseTap::line(std::string_view l) {
  while (!l.empty() && (l.back() == '\r')) l.remove_suffix(1);
  if (!starts_with(l, "data:")) return;
  std::string_view body = trim(l.substr(5));
  if (body == "[DONE]") {
    saw_done_ = true;
    return;
  }
  ++events_;
  // token estimate: one completion token per delta event that carries
  // content; corrected by the final usage object when present, or by
  // a BPE count over the accumulated text (tap keeps <= 256 KiB)
  if (body.find("\"content\"") != std::string_view::npos) {
    ++event_tokens_;
    append_json_string_field(body, "content", &text_, 256 * 1024);
  } else if (body.find("\"text\"") != std::string_view::npos) {
    ++event_tokens_;
    append_json_string_field(body, "text", &text_, 256 * 1024);
  }
  size_t upos = body.find("\"usage\"");
  if (upos != std::string_view::npos) {
    // cheap scan for prompt_tokens / completion_tokens numbers
    auto grab = [&](std::string_view key) -> long long {
      size_t kp = body.find(key, upos);
      if (kp == std::string_view::npos) return -1;
      kp = body.find(':', kp);
      if (kp == std::string_view::npos) return -1;
      ++kp;
      while (kp < body.size() && (body[kp] == ' ')) ++kp;
      return atoll(std::string(body.substr(kp, 20)).c_str());
    };
    long long pt = grab("\"prompt_tokens\"");
    long long ct = grab("\"completion_tokens\"");
    if (pt < 0) pt = grab("\"input_tokens\"");    // Responses shape
    if (ct < 0) ct = grab("\"output_tokens\"");
    if (pt >= 0) usage_prompt_ = pt;
    if (ct >= 0) usage_completion_ = ct;
  }
}

bool append_json_string_field(std::string_view body, std::string_view key,
                              std::string* out, size_t cap) {
  std::string pat = "\"";
  pat.append(key);
  pat += "\"";
  size_t kp = body.find(pat);
  if (kp == std::string_view::npos) return false;
  size_t p = body.find_first_not_of(" \t", kp + pat.size());
  if (p == std::string_view::npos || body[p] != ':') return false;
  p = body.find_first_not_of(" \t", p + 1);
  if (p == std::string_view::npos || body[p] != '"') return false;
  ++p;
  while (p < body.size() && out->size() < cap) {
    char c = body[p];
    if (c == '"') return true;
    if (c == '\\' && p + 1 < body.size()) {
      char e = body[p + 1];
      p += 2;
      switch (e) {
        case 'n': out->push_back('\n'); break;
        case 't': out->push_back('\t'); break;
        case 'r': out->push_back('\r'); break;
        case 'b': out->push_back('\b'); break;
        case 'f': out->push_back('\f'); break;
        case 'u': {
          if (p + 4 <= body.size()) {
            unsigned v = 0;
            for (int i = 0; i < 4; ++i) {
              char h = body[p + i];
              v <<= 4;
              if (h >= '0' && h <= '9') v |= (unsigned)(h - '0');
              else if (h >= 'a' && h <= 'f') v |= (unsigned)(h - 'a' + 10);
              else if (h >= 'A' && h <= 'F') v |= (unsigned)(h - 'A' + 10);
            }
            p += 4;
            // UTF-8 encode (surrogates approximated as-is)
            if (v < 0x80) {
              out->push_back((char)v);
            } else if (v < 0x800) {
              out->push_back((char)(0xC0 | (v >> 6)));
              out->push_back((char)(0x80 | (v & 0x3F)));
            } else {
              out->push_back((char)(0xE0 | (v >> 12)));
              out->push_back((char)(0x80 | ((v >> 6) & 0x3F)));
              out->push_back((char)(0x80 | (v & 0x3F)));
            }
          }
          break;
        }
        default: out->push_back(e);
      }
      continue;
    }
    out->push_back(c);
    ++p;
  }
  return true;
}

// ------------------------------------------------------------------ misc
std::string http_date_now() {
  time_t t = time(nullptr);
  struct tm tmv;
  gmtime_r(&t, &tmv);
  char buf[64];
  strftime(buf, sizeof buf, "%a, %d %b %Y %H:%M:%S GMT", &tmv);
  return buf;
}

std::string_view status_reason(int status) {
  switch (status) {
    case 200: return "OK";
    case 201: return "Created";
    case 202: return "Accepted";
    case 204: return "No Content";
    case 301: return "Moved Permanently";
    case 302: return "Found";
    case 304: return "Not Modified";
    case 400: return "Bad Request";
    case 401: return "Unauthorized";
    case 403: return "Forbidden";
    case 404: return "Not Found";
    case 405: return "Method Not Allowed";
    case 408: return "Request Timeout";
    case 409: return "Conflict";
    case 411: return "Length Required";
    case 413: return "Payload Too Large";
    case 422: return "Unprocessable Entity";
    case 429: return "Too Many Requests";
    case 500: return "Internal Server Error";
    case 502: return "Bad Gateway";
    case 503: return "Service Unavailable";
    case 504: return "Gateway Timeout";
    default: return "Unknown";
  }
}

void append_status_line(Buffer& b, int status, std::string_view reason) {
  char line[96];
  int n = snprintf(line, sizeof line, "HTTP/1.1 %d %.*s\r\n", status,
                   (int)reason.size(), reason.data());
  b.append(line, n);
}

}  // namespace gw
