// Mock OpenAI-compatible echo backend — BASELINE config 1's backend and
// the integration-test fake (the reference tests use wiremock servers
// the same way).
//
//   llmlb-mock [--port N] [--models a,b] [--delay-ms D] [--workers N]
//              [--stream-chunks K] [--fail-rate F]
//
// Serves: GET /v1/models, POST /v1/chat/completions (stream +
// non-stream), /v1/completions, /v1/embeddings, /v1/responses,
// /v1/audio/*, /v1/images/*.
#include <csignal>

#include <atomic>
#include <thread>

#include "http.h"
#include "json.h"
#include "net.h"

using namespace gw;

struct MockConfig {
  int port = 9100;
  int workers = 2;
  std::vector<std::string> models = {"mock-model"};
  int delay_ms = 0;          // simulated inference latency
  bool delay_inference_only = false;
  double health_tps = 0;     // advertise tokens_per_sec in /api/health
  int stream_chunks = 8;     // SSE deltas per streaming response
  double fail_rate = 0.0;    // inject failures (config 5 testing)
  bool include_usage = true; // --no-usage: exercise the gateway's BPE
                             // token-estimation fallback
  std::string engine;        // --engine ollama: emulate the ollama
                             // control surface (tags/show/pull/delete)
                             // for the per-engine adapter tests
  int completion_tokens = 8;
};
static MockConfig g_cfg;
static std::atomic<long long> g_reqs{0};

struct MockWorker;

struct MockConn : FdHandler {
  MockWorker* w;
  int fd;
  Buffer in, out;
  HttpRequest req;
  bool have_head = false;
  uint32_t events = EPOLLIN;
  bool closed = false;
  bool want_close_after_flush = false;

  MockConn(MockWorker* w_, int fd_);
  ~MockConn() override;
  void on_event(uint32_t ev) override;
  void respond();
  void flush();
  void close_now();
  void want(uint32_t ev);
};

struct MockWorker {
  EventLoop loop;
  int listen_fd = -1;
  std::thread thread;
  struct Acceptor : FdHandler {
    MockWorker* w;
    void on_event(uint32_t) override {
      while (true) {
        int cfd = accept4(w->listen_fd, nullptr, nullptr,
                          SOCK_NONBLOCK | SOCK_CLOEXEC);
        if (cfd < 0) return;
        set_nodelay(cfd);
        new MockConn(w, cfd);
      }
    }
  } acceptor;
};

MockConn::MockConn(MockWorker* w_, int fd_) : w(w_), fd(fd_) {
  w->loop.add(fd, EPOLLIN, this);
}
MockConn::~MockConn() {
  if (fd >= 0) {
    w->loop.del(fd);
    ::close(fd);
  }
}
void MockConn::close_now() {
  if (closed) return;
  closed = true;
  w->loop.defer_delete([this] { delete this; });
}
void MockConn::want(uint32_t ev) {
  if (ev != events) {
    events = ev;
    w->loop.mod(fd, ev, this);
  }
}
void MockConn::flush() {
  while (!out.empty()) {
    ssize_t n = ::send(fd, out.data(), out.size(), MSG_NOSIGNAL);
    if (n > 0) out.consume((size_t)n);
    else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      want(EPOLLIN | EPOLLOUT);
      return;
    } else {
      close_now();
      return;
    }
  }
  want(EPOLLIN);
  if (want_close_after_flush) close_now();
}

void MockConn::on_event(uint32_t ev) {
  if (closed) return;
  if (ev & (EPOLLERR | EPOLLHUP)) {
    if (!(ev & EPOLLIN)) {
      close_now();
      return;
    }
  }
  if (ev & EPOLLOUT) flush();
  if (closed) return;
  if (!(ev & EPOLLIN)) return;
  while (true) {
    char* p = in.reserve(32768);
    ssize_t n = ::recv(fd, p, 32768, 0);
    if (n > 0) {
      in.commit((size_t)n);
    } else if (n == 0) {
      close_now();
      return;
    } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
      break;
    } else {
      close_now();
      return;
    }
    if ((size_t)n < 32768) break;
  }
  // serve all complete pipelined requests (head re-parsed per read:
  // its views must not outlive a buffer reallocation)
  while (!closed) {
    int rc = parse_request_head(in.view(), req);
    if (rc == 0) return;
    if (rc < 0) {
      close_now();
      return;
    }
    size_t need = req.header_len + (size_t)req.content_length;
    if (in.size() < need) return;
    bool keep = req.keep_alive;
    respond();
    in.consume(need);
    if (!keep) {
      // honor Connection: close once the response is flushed
      if (out.empty()) close_now();
      else want_close_after_flush = true;
      return;
    }
  }
}

static void canned_chat(MockConn* c, std::string_view body, bool echo_ids) {
  std::string model = raw_string_field(body, "model");
  if (model.empty()) model = g_cfg.models[0];
  bool stream = body.find("\"stream\":true") != std::string_view::npos ||
                body.find("\"stream\": true") != std::string_view::npos;
  long long id = g_reqs.fetch_add(1);
  if (g_cfg.fail_rate > 0 &&
      (double)(id % 1000) / 1000.0 < g_cfg.fail_rate) {
    std::string_view b = "{\"error\":{\"message\":\"injected failure\","
                         "\"type\":\"server_error\"}}";
    append_status_line(c->out, 500, "Internal Server Error");
    char h[128];
    int n = snprintf(h, sizeof h,
                     "Content-Type: application/json\r\nContent-Length: "
                     "%zu\r\n\r\n",
                     b.size());
    c->out.append(h, n);
    c->out.append(b);
    return;
  }
  if (!stream) {
    char bodybuf[1024];
    int bl;
    if (g_cfg.include_usage)
      bl = snprintf(
          bodybuf, sizeof bodybuf,
          "{\"id\":\"chatcmpl-%lld\",\"object\":\"chat.completion\","
          "\"created\":%lld,\"model\":\"%s\",\"choices\":[{\"index\":0,"
          "\"message\":{\"role\":\"assistant\",\"content\":\"Hello from the "
          "mock backend.\"},\"finish_reason\":\"stop\"}],\"usage\":{"
          "\"prompt_tokens\":12,\"completion_tokens\":%d,\"total_tokens\":"
          "%d}}",
          id, (long long)(wall_ms() / 1000), model.c_str(),
          g_cfg.completion_tokens, 12 + g_cfg.completion_tokens);
    else
      bl = snprintf(
          bodybuf, sizeof bodybuf,
          "{\"id\":\"chatcmpl-%lld\",\"object\":\"chat.completion\","
          "\"created\":%lld,\"model\":\"%s\",\"choices\":[{\"index\":0,"
          "\"message\":{\"role\":\"assistant\",\"content\":\"Hello from the "
          "mock backend.\"},\"finish_reason\":\"stop\"}]}",
          id, (long long)(wall_ms() / 1000), model.c_str());
    append_status_line(c->out, 200, "OK");
    char h[128];
    int n = snprintf(h, sizeof h,
                     "Content-Type: application/json\r\nContent-Length: "
                     "%d\r\n\r\n",
                     bl);
    c->out.append(h, n);
    c->out.append(bodybuf, bl);
    return;
  }
  // streaming SSE with chunked framing
  append_status_line(c->out, 200, "OK");
  c->out.append("Content-Type: text/event-stream\r\nCache-Control: "
                "no-cache\r\nTransfer-Encoding: chunked\r\n\r\n");
  auto chunk = [&](std::string_view payload) {
    char lenb[16];
    int n = snprintf(lenb, sizeof lenb, "%zx\r\n", payload.size());
    c->out.append(lenb, n);
    c->out.append(payload);
    c->out.append("\r\n");
  };
  char ev[512];
  for (int i = 0; i < g_cfg.stream_chunks; ++i) {
    int n = snprintf(
        ev, sizeof ev,
        "data: {\"id\":\"chatcmpl-%lld\",\"object\":\"chat.completion."
        "chunk\",\"model\":\"%s\",\"choices\":[{\"index\":0,\"delta\":{"
        "\"content\":\"tok%d \"},\"finish_reason\":null}]}\n\n",
        id, model.c_str(), i);
    chunk(std::string_view(ev, n));
  }
  int n;
  if (g_cfg.include_usage) {
    n = snprintf(
        ev, sizeof ev,
        "data: {\"id\":\"chatcmpl-%lld\",\"object\":\"chat.completion."
        "chunk\",\"model\":\"%s\",\"choices\":[{\"index\":0,\"delta\":{},"
        "\"finish_reason\":\"stop\"}],\"usage\":{\"prompt_tokens\":12,"
        "\"completion_tokens\":%d,\"total_tokens\":%d}}\n\n",
        id, model.c_str(), g_cfg.stream_chunks, 12 + g_cfg.stream_chunks);
  } else {
    n = snprintf(
        ev, sizeof ev,
        "data: {\"id\":\"chatcmpl-%lld\",\"object\":\"chat.completion."
        "chunk\",\"model\":\"%s\",\"choices\":[{\"index\":0,\"delta\":{},"
        "\"finish_reason\":\"stop\"}]}\n\n",
        id, model.c_str());
  }
  chunk(std::string_view(ev, n));
  chunk("data: [DONE]\n\n");
  c->out.append("0\r\n\r\n");
}

void MockConn::respond() {
  std::string_view body =
      in.view().substr(req.header_len, (size_t)req.content_length);
  std::string path(req.path);
  auto do_respond = [this, path, bodystr = std::string(body)]() {
    std::string_view body2 = bodystr;
    auto send_json = [this](int status, const std::string& b) {
      append_status_line(out, status,
                         status == 200 ? "OK" : "Not Found");
      char h[128];
      int n = snprintf(h, sizeof h,
                       "Content-Type: application/json\r\n"
                       "Content-Length: %zu\r\n\r\n", b.size());
      out.append(h, n);
      out.append(b);
    };
    if (g_cfg.engine == "ollama" && starts_with(path, "/api/")) {
      // minimal ollama control surface (adapter tests; the reference
      // talks to these endpoints for sync/metadata/pull/delete)
      if (path == "/api/tags") {
        std::string b = "{\"models\":[";
        for (size_t i = 0; i < g_cfg.models.size(); ++i) {
          if (i) b += ',';
          b += "{\"name\":\"" + g_cfg.models[i] +
               "\",\"size\":123456789}";
        }
        b += "]}";
        send_json(200, b);
      } else if (path == "/api/show") {
        std::string name = raw_string_field(body2, "name");
        bool have = false;
        for (auto& m : g_cfg.models) have |= (m == name);
        if (!have) send_json(404, "{\"error\":\"model not found\"}");
        else
          send_json(200,
                    "{\"modelfile\":\"FROM " + name +
                        "\",\"details\":{\"family\":\"llama\","
                        "\"parameter_size\":\"8B\","
                        "\"quantization_level\":\"Q4_K_M\"},"
                        "\"model_info\":{\"general.parameter_count\":"
                        "8030000000}}");
      } else if (path == "/api/pull") {
        std::string name = raw_string_field(body2, "name");
        if (!name.empty()) {
          bool have = false;
          for (auto& m : g_cfg.models) have |= (m == name);
          if (!have) g_cfg.models.push_back(name);
        }
        send_json(200, "{\"status\":\"success\"}");
      } else if (path == "/api/delete") {
        std::string name = raw_string_field(body2, "name");
        auto& ms = g_cfg.models;
        size_t before = ms.size();
        ms.erase(std::remove(ms.begin(), ms.end(), name), ms.end());
        send_json(before != ms.size() ? 200 : 404,
                  before != ms.size() ? "{}"
                                      : "{\"error\":\"not found\"}");
      } else {
        send_json(404, "{}");
      }
      flush();
      return;
    }
    if (path == "/v1/models" || path == "/api/v1/models") {
      std::string b = "{\"object\":\"list\",\"data\":[";
      for (size_t i = 0; i < g_cfg.models.size(); ++i) {
        if (i) b += ',';
        b += "{\"id\":\"" + g_cfg.models[i] +
             "\",\"object\":\"model\",\"owned_by\":\"mock\"}";
      }
      b += "]}";
      append_status_line(out, 200, "OK");
      char h[128];
      int n = snprintf(h, sizeof h,
                       "Content-Type: application/json\r\nContent-Length: "
                       "%zu\r\n\r\n",
                       b.size());
      out.append(h, n);
      out.append(b);
    } else if (path == "/v1/chat/completions" || path == "/v1/responses" ||
               path == "/v1/completions") {
      canned_chat(this, body2, false);
    } else if (path == "/v1/embeddings") {
      std::string b =
          "{\"object\":\"list\",\"data\":[{\"object\":\"embedding\","
          "\"index\":0,\"embedding\":[0.1,0.2,0.3,0.4]}],\"model\":\"" +
          g_cfg.models[0] +
          "\",\"usage\":{\"prompt_tokens\":4,\"total_tokens\":4}}";
      append_status_line(out, 200, "OK");
      char h[128];
      int n = snprintf(h, sizeof h,
                       "Content-Type: application/json\r\nContent-Length: "
                       "%zu\r\n\r\n",
                       b.size());
      out.append(h, n);
      out.append(b);
    } else if (starts_with(path, "/v1/audio/") ||
               starts_with(path, "/v1/images/")) {
      std::string b = starts_with(path, "/v1/audio/transcriptions")
                          ? "{\"text\":\"mock transcription\"}"
                          : starts_with(path, "/v1/audio/")
                                ? "{\"audio\":\"bW9jaw==\"}"
                                : "{\"created\":1,\"data\":[{\"b64_json\":"
                                  "\"bW9jaw==\"}]}";
      append_status_line(out, 200, "OK");
      char h[128];
      int n = snprintf(h, sizeof h,
                       "Content-Type: application/json\r\nContent-Length: "
                       "%zu\r\n\r\n",
                       b.size());
      out.append(h, n);
      out.append(b);
    } else if (path == "/api/health" && g_cfg.health_tps > 0) {
      char b[256];
      int bn = snprintf(b, sizeof b,
                        "{\"status\":\"ok\",\"model\":\"%s\","
                        "\"tokens_per_sec\":%.1f}",
                        g_cfg.models.empty() ? "m"
                                             : g_cfg.models[0].c_str(),
                        g_cfg.health_tps);
      char h[128];
      int hn = snprintf(h, sizeof h,
                        "Content-Type: application/json\r\n"
                        "Content-Length: %d\r\n\r\n", bn);
      append_status_line(out, 200, "OK");
      out.append(h, hn);
      out.append(b, bn);
    } else if (path == "/health") {
      append_status_line(out, 200, "OK");
      out.append("Content-Length: 2\r\n\r\nok");
    } else {
      append_status_line(out, 404, "Not Found");
      out.append("Content-Length: 0\r\n\r\n");
    }
    flush();
  };
  // --delay-inference-only: control/health/models respond instantly
  // so registration+sync work while inference stalls (timeout tests)
  const bool inference = starts_with(path, "/v1/chat") ||
                         starts_with(path, "/v1/completions") ||
                         starts_with(path, "/v1/embeddings") ||
                         starts_with(path, "/v1/responses");
  if (g_cfg.delay_ms > 0 && (!g_cfg.delay_inference_only || inference)) {
    w->loop.add_timer(g_cfg.delay_ms, do_respond);
  } else {
    do_respond();
  }
}

int main(int argc, char** argv) {
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> const char* {
      return i + 1 < argc ? argv[++i] : "";
    };
    if (a == "--port") g_cfg.port = atoi(next());
    else if (a == "--workers") g_cfg.workers = atoi(next());
    else if (a == "--models") g_cfg.models = split(next(), ',');
    else if (a == "--delay-ms") g_cfg.delay_ms = atoi(next());
    else if (a == "--delay-inference-only")
      g_cfg.delay_inference_only = true;
    else if (a == "--health-tps") g_cfg.health_tps = atof(next());
    else if (a == "--stream-chunks") g_cfg.stream_chunks = atoi(next());
    else if (a == "--fail-rate") g_cfg.fail_rate = atof(next());
    else if (a == "--no-usage") g_cfg.include_usage = false;
    else if (a == "--engine") g_cfg.engine = next();
  }
  signal(SIGPIPE, SIG_IGN);
  std::vector<std::unique_ptr<MockWorker>> ws;
  for (int i = 0; i < g_cfg.workers; ++i) {
    auto w = std::make_unique<MockWorker>();
    w->acceptor.w = w.get();
    w->listen_fd = make_listener("0.0.0.0", g_cfg.port, true);
    if (w->listen_fd < 0) {
      fprintf(stderr, "cannot listen on %d\n", g_cfg.port);
      return 1;
    }
    w->loop.add(w->listen_fd, EPOLLIN, &w->acceptor);
    ws.push_back(std::move(w));
  }
  fprintf(stderr, "llmlb-mock listening on %d (%d workers)\n", g_cfg.port,
          g_cfg.workers);
  for (auto& w : ws)
    w->thread = std::thread([wp = w.get()] { wp->loop.run(); });
  for (auto& w : ws) w->thread.join();
  return 0;
}
