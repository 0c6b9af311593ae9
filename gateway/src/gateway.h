// llmlb-gw server core: worker event loops, client/upstream connections,
// the zero-copy SSE proxy pump, local admin API.
#pragma once

#include <atomic>
#include <deque>
#include <memory>
#include <thread>
#include <unordered_map>
#include <unordered_set>

#include "common.h"
#include "db.h"
#include "history.h"
#include "http.h"
#include "json.h"
#include "net.h"
#include "router.h"

namespace gw { class AnthropicStreamEncoder; }
#include "anthropic.h"

namespace gw {

struct GatewayConfig {
  std::string host = "0.0.0.0";
  int port = 32768;
  int workers = 0;                 // 0 = #cores
  std::string db_path = "llmlb.sqlite3";
  size_t body_limit = 20 * 1024 * 1024;   // api/mod.rs:58
  AdmissionConfig admission;
  int retention_days = 7;
  // audit rows older than this move to <db>.audit-archive.sqlite3
  // (reference bootstrap.rs:291-318); 0 disables archiving
  int audit_archive_days = 90;
  int health_interval_s = 30;
  double tps_ema_alpha = 0.2;
  double amdsmi_poll_interval_s = 2.0;
  int health_timeout_ms = 5000;
  int auto_sync_interval_s = 900;
  int consecutive_failures_for_offline = 2;
  bool require_auth = true;
  // persist request/response bodies (truncated to 16 KiB) in
  // request_history — the reference stores sanitized bodies
  // (api/openai_util.rs:137); off by default for throughput
  bool capture_bodies = false;
  std::string admin_token;          // LLMLB_ADMIN_TOKEN
  std::string log_level = "info";
  std::string log_dir;
  int log_retention_days = 7;
  // cloud providers (base_url overridable for tests; http only — this
  // build environment has no TLS stack and no egress)
  std::string openai_base = "";     // e.g. http://127.0.0.1:9001/v1
  std::string google_base = "";
  std::string anthropic_base = "";
  std::string openai_key, google_key, anthropic_key;
  // control plane (Python FastAPI) to which /dashboard + extended /api
  // routes are proxied; empty = disabled
  std::string control_plane_base = "";
};

class Gateway;
class Worker;
class ClientConn;
class UpstreamConn;

// metrics counters (Prometheus text at /api/metrics — cloud_metrics.rs
// analog, extended with gateway totals)
// broadcast-bus analog (events/mod.rs): bounded ring of recent gateway
// events, polled by /api/events and pushed onward by the control plane
struct EventRing {
  struct Event {
    int64_t ts_ms;
    std::string type;     // EndpointStatusChanged | TpsUpdated | ...
    std::string detail;   // small JSON fragment
  };
  std::mutex mu;
  std::deque<Event> ring;
  void emit(std::string type, std::string detail) {
    std::lock_guard<std::mutex> lk(mu);
    ring.push_back({wall_ms(), std::move(type), std::move(detail)});
    while (ring.size() > 1024) ring.pop_front();  // cap (events/mod.rs:13)
  }
  std::vector<Event> since(int64_t ts_ms) {
    std::lock_guard<std::mutex> lk(mu);
    std::vector<Event> out;
    for (auto& e : ring)
      if (e.ts_ms > ts_ms) out.push_back(e);
    return out;
  }
};

struct Metrics {
  std::atomic<long long> requests_total{0};
  std::atomic<long long> responses_2xx{0};
  std::atomic<long long> responses_4xx{0};
  std::atomic<long long> responses_5xx{0};
  std::atomic<long long> upstream_errors{0};
  std::atomic<long long> streamed_responses{0};
  std::atomic<long long> tokens_streamed{0};
  std::atomic<long long> cloud_requests{0};
  std::atomic<long long> active_requests{0};
  std::atomic<long long> admission_delays{0};
};

// auth store: api-key sha256 lookup (auth/middleware.rs analog)
// Per-key permission matrix mirroring the reference's 11 scopes
// (common/auth.rs:60-95: openai.inference, openai.models.read,
// endpoints.read, endpoints.manage, api_keys.manage, users.manage,
// invitations.manage, models.manage, registry.read, logs.read,
// metrics.read) plus "admin" (grants all) and the legacy "inference"
// (grants openai.inference + openai.models.read).  Stored perms are a
// comma-separated token list compared EXACTLY per token — substring
// grants ("badminton" ⊃ "admin") are rejected.
class AuthStore {
 public:
  explicit AuthStore(Db* db) : db_(db) {}
  void load();
  // returns true when authorized; perm = one scope name above.
  // client_is_local gates open mode (no keys configured): an exposed
  // gateway without its first key only serves loopback clients.
  bool check_bearer(std::string_view authorization, const char* perm,
                    bool client_is_local = false);
  bool check_key(std::string_view key, const char* perm,
                 bool client_is_local = false);
  std::string create_key(const std::string& name, const std::string& perms);
  bool revoke_key(const std::string& prefix_or_id);
  bool any_keys();
  void set_open_mode(bool open) { open_mode_ = open; }
  bool open_mode() const { return open_mode_; }
  // exact-token permission-set test (public for unit tests)
  static bool perms_grant(std::string_view perm_list, const char* perm);

 private:
  Db* db_;
  std::mutex mu_;
  std::unordered_map<std::string, std::string> hash_to_perms_;
  bool open_mode_ = false;  // no keys configured -> loopback-only allow
};

// ---------------------------------------------------------------------
// Proxy operation: one in-flight proxied request (client <-> upstream).
// ---------------------------------------------------------------------
struct ProxyOp {
  ClientConn* client = nullptr;       // owner
  UpstreamConn* upstream = nullptr;
  RequestLease lease;
  BodyReader body_reader;
  bool selected = false;      // came through TPS selection (retryable)
  int admission_delay_ms = 0; // held by delayed admission -> header
  SseTap tap;
  HttpResponse resp;          // views valid only at head-parse time
  std::string resp_content_type;  // copied at head time (views dangle)
  bool resp_head_seen = false;
  bool streaming = false;
  bool lease_released_at_head = false;
  int64_t start_us = 0;
  int64_t deadline_us = 0;   // inference timeout (swept by the worker)
  uint64_t timer_id = 0;     // unused on the fast path
  // bookkeeping for history / TPS
  std::string public_model;
  std::string upstream_model;
  std::string endpoint_id;
  ApiKind kind = ApiKind::kChat;
  std::string client_ip;
  std::string request_id;     // tracing (copied from the client conn)
  long long req_prompt_chars = 0;
  std::string captured_request;   // when cfg.capture_bodies
  bool is_cloud = false;
  std::string cloud_provider;
  // buffered (non-stream) response body (decoded payload)
  std::string buffered;
  std::shared_ptr<TpsState> tps;   // direct TPS state (no map lookup)
  // anthropic ingress: re-encode OpenAI SSE -> Anthropic SSE
  bool anthropic_reencode = false;
  std::unique_ptr<class AnthropicStreamEncoder> reenc;
};

// ---------------------------------------------------------------------
class UpstreamConn : public FdHandler {
 public:
  UpstreamConn(Worker* w, EndpointPtr ep, std::string host, int port);
  ~UpstreamConn() override;
  void on_event(uint32_t events) override;

  Worker* w;
  EndpointPtr ep;                 // may be null for cloud
  std::string pool_key;
  std::string host;
  int port;
  int fd = -1;
  bool connected = false;
  bool closed = false;
  Buffer in, out;
  ProxyOp* op = nullptr;          // active op (null when idle/pooled)
  uint32_t events = 0;
  bool reading_paused = false;

  bool start();                    // begin non-blocking connect
  void send_buffered();
  void want(uint32_t ev);
  void close_now();

 private:
  void handle_readable();
  void handle_writable();
};

// ---------------------------------------------------------------------
class ClientConn : public FdHandler {
 public:
  ClientConn(Worker* w, int fd, std::string peer_ip);
  ~ClientConn() override;
  void on_event(uint32_t events) override;

  Worker* w;
  int fd;
  std::string peer_ip;
  std::string request_id;   // X-Request-Id: client-supplied or generated
  // OWNED copies of the parsed head's method/path: the parsed
  // HttpRequest holds string_views INTO `in`, and any later socket
  // read may reallocate it while an async op is in flight — the
  // audit block in send_simple() read those views after the upstream
  // round trip (heap-use-after-free under pipelined load; found by a
  // 280 s ASAN soak at 100 conns)
  std::string cur_method, cur_path;
  Buffer in, out;
  HttpRequest req;
  bool have_head = false;
  bool want_close = false;
  bool closed = false;
  bool expect_ack_sent = false;
  std::unique_ptr<ProxyOp> op;
  uint32_t events = 0;
  bool response_open = false;  // streaming response in progress
  bool adm_delayed = false;    // delayed-admission hold already served
  int adm_delay_ms = 0;        // the hold length (response header)
  // one-shot connect-failure retry: the endpoint id to exclude on the
  // re-selection (non-empty also means "already retried once")
  std::string retry_exclude_ep;

  void flush_out();            // try write; arm EPOLLOUT if partial
  void close_now();
  void want(uint32_t ev);

  // response helpers
  void send_simple(int status, std::string_view content_type,
                   std::string_view body,
                   std::string_view extra_headers = {});
  void send_json_error(int status, std::string_view code,
                       std::string_view message);
  void finish_request_cycle();  // reset for next pipelined request

  void process_requests();

 private:
  void handle_readable();
  void dispatch();
};

// ---------------------------------------------------------------------
class Worker {
 public:
  Worker(Gateway* gw, int index);
  void start();
  void join();

  Gateway* gw;
  int index;
  EventLoop loop;
  std::thread thread;
  int listen_fd = -1;

  // keep-alive upstream pool: key -> idle conns
  std::unordered_map<std::string, std::vector<UpstreamConn*>> pool;
  UpstreamConn* acquire_upstream(const EndpointPtr& ep,
                                 const std::string& key,
                                 const std::string& host, int port);
  // per-worker record buffer: finalize_op appends lock-free (worker
  // thread), a 200 ms loop timer flushes to the HistoryWriter
  std::vector<RequestRecord> pending_records;
  std::vector<AuditRecord> pending_audits;
  void arm_record_flush();
  // request-id minting state (tracing)
  unsigned rng = 0;
  unsigned long long request_seq = 0;
  // ops with an armed inference timeout, swept every 250 ms
  std::unordered_set<ClientConn*> timed_ops;
  // conns parked by the delayed-admission arm (50-80% load -> 10-100ms
  // hold, balancer/mod.rs:2255-2270); membership validates the timer
  // callback (the conn may close while parked)
  std::unordered_set<ClientConn*> parked;
  void arm_op_sweep();
  void release_upstream(UpstreamConn* uc);   // pool for reuse
  void discard_upstream(UpstreamConn* uc);   // close

  struct Acceptor : FdHandler {
    Worker* w;
    void on_event(uint32_t events) override;
  } acceptor;
};

// ---------------------------------------------------------------------
class Gateway {
 public:
  explicit Gateway(GatewayConfig cfg);
  ~Gateway();
  bool init(std::string* err);
  void run();          // blocks until stop()
  void stop();         // async-signal-safe: just requests shutdown
  void shutdown_threads();

  GatewayConfig cfg;
  Db db;
  Registry registry;
  HistoryWriter history;
  AuthStore auth;
  Metrics metrics;
  EventRing events;
  std::atomic<bool> draining{false};   // inference gate (self-update)
  // cached cloud-provider model listings, merged into /v1/models
  // (openai.rs:451-467); refreshed by the health loop every 10 min
  std::mutex cloud_models_mu;
  struct CloudModel { std::string id, provider; };
  std::vector<CloudModel> cloud_models_cache;
  int64_t cloud_models_fetched_ms = 0;
  void refresh_cloud_models();
  std::atomic<long long> start_ms{0};

  std::vector<std::unique_ptr<Worker>> workers;

  // request dispatch (called on worker threads)
  void handle_request(ClientConn* c);

  // local (non-proxied) handlers — api_local.cpp
  void handle_local(ClientConn* c, const std::string& path,
                    std::string_view body);
  std::string render_models_json();
  std::string render_summary_json();
  std::string render_metrics_text();

  // proxy path — proxy.cpp
  void start_proxy(ClientConn* c, ApiKind kind, uint32_t required_caps,
                   const std::string& upstream_path);
  void start_cloud_proxy(ClientConn* c, ApiKind kind,
                         const std::string& provider,
                         const std::string& bare_model,
                         const std::string& upstream_path,
                         bool anthropic_native);
  void start_control_plane_proxy(ClientConn* c);
  void finalize_op(ClientConn* c, bool success, int status,
                   const std::string& error);
  // teardown for the connect-failure retry: releases the lease and
  // counters but KEEPS the request buffered (finalize_op would consume
  // it and finish the cycle)
  void abandon_op_for_retry(ClientConn* c);

  // health checking — health.cpp
  void health_loop();
  std::thread health_thread;
  std::atomic<bool> stopping{false};
  std::atomic<bool> shutdown_done_{false};

  // model sync — health.cpp
  bool sync_endpoint_models(const EndpointPtr& ep, std::string* err);
  // type detection probe chain — health.cpp
  std::string detect_endpoint_type(const std::string& host, int port,
                                   const std::string& prefix);
};

// blocking mini HTTP client used by the health checker / model sync /
// detection (not on the hot path)
struct MiniHttpResult {
  int status = 0;
  std::string body;
  std::string error;
  std::vector<std::pair<std::string, std::string>> headers;
  double elapsed_ms = 0;
  std::string header(std::string_view name) const {
    for (auto& [k, v] : headers)
      if (ieq(k, name)) return v;
    return "";
  }
};
MiniHttpResult mini_http_request(const std::string& host, int port,
                                 const std::string& method,
                                 const std::string& path,
                                 const std::string& body, int timeout_ms,
                                 const std::vector<std::string>& headers = {});

}  // namespace gw
