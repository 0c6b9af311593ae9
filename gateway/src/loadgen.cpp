// llmlb-bench: wrk-style HTTP load generator (no external deps; wrk is
// not installed in this environment).  Drives POST (or GET) at a fixed
// connection count and reports the reference benchmark CSV schema:
//   label,rps,p50_ms,p75_ms,p90_ms,p95_ms,p99_ms,non2xx,socket_errors,
//   requests,duration_s            (benchmarks/README.md)
//
//   llmlb-bench --host H --port P --path /v1/chat/completions \
//       --threads 10 --connections 50 --duration 30 --body-file f.json \
//       [--header 'Authorization: Bearer k'] [--label local] [--json]
#include <csignal>

#include <algorithm>
#include <atomic>
#include <thread>

#include "http.h"
#include "net.h"

using namespace gw;

struct BenchConfig {
  std::string host = "127.0.0.1";
  int port = 32768;
  std::string path = "/v1/chat/completions";
  std::string method = "POST";
  std::string body =
      "{\"model\":\"mock-model\",\"messages\":[{\"role\":\"user\","
      "\"content\":\"hi\"}]}";
  std::vector<std::string> headers;
  int threads = 10;
  int connections = 50;
  double duration_s = 30;
  std::string label = "local";
  bool json_out = false;
};
static BenchConfig g_cfg;

// latency histogram: 1 us buckets to 100 ms, then 100 us to 10 s
struct Histo {
  static constexpr int kFine = 100000;          // 1 us buckets
  static constexpr int kCoarse = 99000;         // 100 us buckets
  std::vector<uint32_t> fine, coarse;
  uint64_t overflow = 0, count = 0;
  uint64_t sum_us = 0;
  Histo() : fine(kFine, 0), coarse(kCoarse, 0) {}
  void add(int64_t us) {
    ++count;
    sum_us += (uint64_t)us;
    if (us < kFine) ++fine[us];
    else if (us < kFine + kCoarse * 100) ++coarse[(us - kFine) / 100];
    else ++overflow;
  }
  void merge(const Histo& o) {
    for (int i = 0; i < kFine; ++i) fine[i] += o.fine[i];
    for (int i = 0; i < kCoarse; ++i) coarse[i] += o.coarse[i];
    overflow += o.overflow;
    count += o.count;
    sum_us += o.sum_us;
  }
  double percentile(double p) const {
    if (!count) return 0;
    uint64_t target = (uint64_t)(p * (double)count);
    uint64_t seen = 0;
    for (int i = 0; i < kFine; ++i) {
      seen += fine[i];
      if (seen > target) return i / 1000.0;
    }
    for (int i = 0; i < kCoarse; ++i) {
      seen += coarse[i];
      if (seen > target) return (kFine + i * 100) / 1000.0;
    }
    return 10000.0;
  }
};

struct ThreadStats {
  Histo histo;
  Histo ttfb;       // time to first response byte (streaming TTFT proxy)
  uint64_t non2xx = 0;
  uint64_t socket_errors = 0;
};

static std::atomic<bool> g_stop{false};

struct BenchConn : FdHandler {
  EventLoop* loop;
  ThreadStats* stats;
  const std::string* request;
  int fd = -1;
  Buffer in;
  size_t out_off = 0;
  bool connected = false;
  int64_t sent_us = 0;
  bool ttfb_recorded = false;
  HttpResponse resp;
  bool have_head = false;
  BodyReader body;
  uint32_t events = 0;

  void open() {
    fd = start_connect(g_cfg.host, g_cfg.port);
    if (fd < 0) {
      ++stats->socket_errors;
      return;
    }
    connected = false;
    have_head = false;
    in.clear();
    out_off = 0;
    events = EPOLLIN | EPOLLOUT;
    loop->add(fd, events, this);
  }
  void reopen() {
    if (fd >= 0) {
      loop->del(fd);
      ::close(fd);
      fd = -1;
    }
    if (!g_stop.load(std::memory_order_relaxed)) open();
  }
  void want(uint32_t ev) {
    if (ev != events) {
      events = ev;
      loop->mod(fd, ev, this);
    }
  }
  void send_request() {
    sent_us = now_us();
    ttfb_recorded = false;
    out_off = 0;
    have_head = false;
    resp = HttpResponse{};
    try_send();
  }
  void try_send() {
    while (out_off < request->size()) {
      ssize_t n = ::send(fd, request->data() + out_off,
                         request->size() - out_off, MSG_NOSIGNAL);
      if (n > 0) out_off += (size_t)n;
      else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        want(EPOLLIN | EPOLLOUT);
        return;
      } else {
        ++stats->socket_errors;
        reopen();
        return;
      }
    }
    want(EPOLLIN);
  }
  void on_event(uint32_t ev) override {
    if (g_stop.load(std::memory_order_relaxed)) return;
    if (ev & (EPOLLERR | EPOLLHUP)) {
      if (!(ev & EPOLLIN)) {
        ++stats->socket_errors;
        reopen();
        return;
      }
    }
    if (ev & EPOLLOUT) {
      if (!connected) {
        int err = 0;
        socklen_t len = sizeof err;
        getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &len);
        if (err != 0) {
          ++stats->socket_errors;
          reopen();
          return;
        }
        connected = true;
        send_request();
        if (fd < 0) return;
      } else {
        try_send();
      }
    }
    if (!(ev & EPOLLIN) || fd < 0) return;
    while (true) {
      char* p = in.reserve(32768);
      ssize_t n = ::recv(fd, p, 32768, 0);
      if (n > 0) {
        if (!ttfb_recorded) {
          stats->ttfb.add(now_us() - sent_us);
          ttfb_recorded = true;
        }
        in.commit((size_t)n);
      } else if (n == 0) {
        ++stats->socket_errors;
        reopen();
        return;
      } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
        return;
      } else {
        ++stats->socket_errors;
        reopen();
        return;
      }
      // parse as many responses as are complete
      while (true) {
        if (!have_head) {
          int rc = parse_response_head(in.view(), resp);
          if (rc == 0) break;
          if (rc < 0) {
            ++stats->socket_errors;
            reopen();
            return;
          }
          have_head = true;
          body.init(resp);
          in.consume(resp.header_len);
          body = BodyReader{};
          body.init(resp);
        }
        auto out = body.feed(in.view());
        in.consume(out.consumed);
        if (!out.done) break;
        // one response complete
        stats->histo.add(now_us() - sent_us);
        if (resp.status / 100 != 2) ++stats->non2xx;
        if (!resp.keep_alive) {
          reopen();
          return;
        }
        send_request();
        if (fd < 0) return;
      }
      if ((size_t)n < 32768) return;
    }
  }
};

int main(int argc, char** argv) {
  std::string body_file;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> const char* {
      return i + 1 < argc ? argv[++i] : "";
    };
    if (a == "--host") g_cfg.host = next();
    else if (a == "--port") g_cfg.port = atoi(next());
    else if (a == "--path") g_cfg.path = next();
    else if (a == "--method") g_cfg.method = next();
    else if (a == "--body") g_cfg.body = next();
    else if (a == "--body-file") body_file = next();
    else if (a == "--header") g_cfg.headers.push_back(next());
    else if (a == "--threads" || a == "-t") g_cfg.threads = atoi(next());
    else if (a == "--connections" || a == "-c")
      g_cfg.connections = atoi(next());
    else if (a == "--duration" || a == "-d") g_cfg.duration_s = atof(next());
    else if (a == "--label") g_cfg.label = next();
    else if (a == "--json") g_cfg.json_out = true;
  }
  if (!body_file.empty()) {
    FILE* f = fopen(body_file.c_str(), "r");
    if (!f) {
      fprintf(stderr, "cannot read %s\n", body_file.c_str());
      return 1;
    }
    g_cfg.body.clear();
    char buf[4096];
    size_t n;
    while ((n = fread(buf, 1, sizeof buf, f)) > 0) g_cfg.body.append(buf, n);
    fclose(f);
  }
  signal(SIGPIPE, SIG_IGN);

  // prebuilt request
  std::string request = g_cfg.method + " " + g_cfg.path +
                        " HTTP/1.1\r\nHost: " + g_cfg.host +
                        "\r\nContent-Type: application/json\r\n";
  for (auto& h : g_cfg.headers) request += h + "\r\n";
  request += "Content-Length: " + std::to_string(g_cfg.body.size()) +
             "\r\n\r\n" + g_cfg.body;

  int conns_per_thread = std::max(1, g_cfg.connections / g_cfg.threads);
  std::vector<std::thread> threads;
  std::vector<ThreadStats> stats(g_cfg.threads);
  std::vector<std::unique_ptr<EventLoop>> loops;
  for (int t = 0; t < g_cfg.threads; ++t)
    loops.push_back(std::make_unique<EventLoop>());

  int64_t t0 = now_us();
  for (int t = 0; t < g_cfg.threads; ++t) {
    threads.emplace_back([t, &stats, &loops, &request, conns_per_thread] {
      EventLoop& loop = *loops[t];
      std::vector<std::unique_ptr<BenchConn>> conns;
      for (int i = 0; i < conns_per_thread; ++i) {
        auto c = std::make_unique<BenchConn>();
        c->loop = &loop;
        c->stats = &stats[t];
        c->request = &request;
        c->open();
        conns.push_back(std::move(c));
      }
      loop.run();
    });
  }
  usleep((useconds_t)(g_cfg.duration_s * 1e6));
  g_stop.store(true);
  for (auto& l : loops) l->stop();
  for (auto& th : threads) th.join();
  double elapsed = (now_us() - t0) / 1e6;

  Histo total, ttfb;
  uint64_t non2xx = 0, socket_errors = 0;
  for (auto& s : stats) {
    total.merge(s.histo);
    ttfb.merge(s.ttfb);
    non2xx += s.non2xx;
    socket_errors += s.socket_errors;
  }
  double rps = (double)total.count / elapsed;
  if (g_cfg.json_out) {
    printf("{\"label\":\"%s\",\"rps\":%.2f,\"p50_ms\":%.3f,\"p75_ms\":%.3f,"
           "\"p90_ms\":%.3f,\"p95_ms\":%.3f,\"p99_ms\":%.3f,"
           "\"mean_ms\":%.3f,\"ttfb_p50_ms\":%.3f,"
           "\"ttfb_p99_ms\":%.3f,\"non2xx\":%llu,"
           "\"socket_errors\":%llu,"
           "\"requests\":%llu,\"duration_s\":%.2f}\n",
           g_cfg.label.c_str(), rps, total.percentile(0.50),
           total.percentile(0.75), total.percentile(0.90),
           total.percentile(0.95), total.percentile(0.99),
           total.count ? (double)total.sum_us / total.count / 1000.0 : 0.0,
           ttfb.percentile(0.50), ttfb.percentile(0.99),
           (unsigned long long)non2xx, (unsigned long long)socket_errors,
           (unsigned long long)total.count, elapsed);
  } else {
    printf("label,rps,p50_ms,p75_ms,p90_ms,p95_ms,p99_ms,"
           "ttfb_p50_ms,ttfb_p99_ms,non2xx,"
           "socket_errors,requests,duration_s\n");
    printf("%s,%.2f,%.3f,%.3f,%.3f,%.3f,%.3f,%.3f,%.3f,%llu,%llu,%llu,"
           "%.2f\n",
           g_cfg.label.c_str(), rps, total.percentile(0.50),
           total.percentile(0.75), total.percentile(0.90),
           total.percentile(0.95), total.percentile(0.99),
           ttfb.percentile(0.50), ttfb.percentile(0.99),
           (unsigned long long)non2xx, (unsigned long long)socket_errors,
           (unsigned long long)total.count, elapsed);
  }
  return 0;
}
