#include "gateway.h"

#include <arpa/inet.h>
#include <sys/resource.h>
#include <unistd.h>

namespace gw {

// ======================================================== AuthStore
void AuthStore::load() {
  std::lock_guard<std::mutex> lk(mu_);
  hash_to_perms_.clear();
  if (db_ && db_->ok()) {
    std::vector<DbRow> rows;
    db_->exec("SELECT key_hash, permissions FROM api_keys WHERE revoked=0",
              {}, &rows);
    for (auto& r : rows) hash_to_perms_[r[0].str()] = r[1].str();
  }
  const char* env_key = getenv("LLMLB_API_KEY");
  if (env_key && *env_key)
    hash_to_perms_[sha256_hex(env_key)] = "inference,admin";
  open_mode_ = hash_to_perms_.empty();
  if (open_mode_)
    LOG_WARN("no API keys configured: /v1/* runs in open mode for "
             "LOOPBACK clients only (create a key via POST /api/api-keys "
             "or set LLMLB_API_KEY to serve remote clients)");
}

// exact comma-separated token matching; "admin" grants everything, the
// legacy "inference" token covers the two openai.* read/infer scopes.
bool AuthStore::perms_grant(std::string_view perm_list, const char* perm) {
  std::string_view want(perm);
  size_t pos = 0;
  while (pos <= perm_list.size()) {
    size_t comma = perm_list.find(',', pos);
    std::string_view tok = perm_list.substr(
        pos, comma == std::string_view::npos ? std::string_view::npos
                                             : comma - pos);
    tok = trim(tok);
    if (!tok.empty()) {
      if (tok == want || tok == "admin") return true;
      if (tok == "inference" &&
          (want == "openai.inference" || want == "openai.models.read"))
        return true;
    }
    if (comma == std::string_view::npos) break;
    pos = comma + 1;
  }
  return false;
}

bool AuthStore::check_key(std::string_view key, const char* perm,
                          bool client_is_local) {
  if (open_mode_) return client_is_local;
  std::string h = sha256_hex(key);
  std::lock_guard<std::mutex> lk(mu_);
  auto it = hash_to_perms_.find(h);
  if (it == hash_to_perms_.end()) return false;
  return perms_grant(it->second, perm);
}

bool AuthStore::check_bearer(std::string_view authorization,
                             const char* perm, bool client_is_local) {
  if (open_mode_) return client_is_local;
  std::string_view key = authorization;
  if (starts_with(key, "Bearer ") || starts_with(key, "bearer "))
    key.remove_prefix(7);
  key = trim(key);
  if (key.empty()) return false;
  return check_key(key, perm, client_is_local);
}

std::string AuthStore::create_key(const std::string& name,
                                  const std::string& perms) {
  unsigned char raw[32];
  if (!secure_random_bytes(raw, sizeof raw)) {
    LOG_ERROR("secure_random_bytes failed; refusing to mint an API key");
    return "";
  }
  std::string key =
      "sk-" + b64_encode(std::string_view((char*)raw, sizeof raw), true,
                         false)
                  .substr(0, 43);
  std::string h = sha256_hex(key);
  {
    std::lock_guard<std::mutex> lk(mu_);
    hash_to_perms_[h] = perms;
    open_mode_ = false;
  }
  if (db_ && db_->ok())
    db_->exec("INSERT INTO api_keys (key_hash,name,prefix,permissions,"
              "created_at) VALUES (?,?,?,?,?)",
              {h, name, key.substr(0, 8), perms, iso8601_now()});
  return key;
}

bool AuthStore::revoke_key(const std::string& prefix_or_id) {
  if (db_ && db_->ok()) {
    db_->exec("UPDATE api_keys SET revoked=1 WHERE prefix=? OR id=?",
              {prefix_or_id, prefix_or_id});
  }
  load();
  return true;
}

bool AuthStore::any_keys() {
  std::lock_guard<std::mutex> lk(mu_);
  return !hash_to_perms_.empty();
}

// ======================================================== UpstreamConn
UpstreamConn::UpstreamConn(Worker* w_, EndpointPtr ep_, std::string host_,
                           int port_)
    : w(w_), ep(std::move(ep_)), host(std::move(host_)), port(port_) {}

UpstreamConn::~UpstreamConn() {
  if (fd >= 0) {
    w->loop.del(fd);
    ::close(fd);
  }
}

bool UpstreamConn::start() {
  fd = start_connect(host, port);
  if (fd < 0) return false;
  connected = false;
  events = EPOLLIN | EPOLLOUT;
  w->loop.add(fd, events, this);
  return true;
}

void UpstreamConn::want(uint32_t ev) {
  if (ev != events) {
    events = ev;
    w->loop.mod(fd, ev, this);
  }
}

void UpstreamConn::close_now() {
  if (closed) return;
  closed = true;
  // a conn closing while IDLE in the keep-alive pool (peer closed /
  // RST) must leave the pool NOW: deferred deletion runs on the next
  // loop turn, and acquire_upstream popping a freed pointer was a
  // use-after-free (caught by the r2 TSAN pass)
  if (!op && !pool_key.empty()) {
    auto it = w->pool.find(pool_key);
    if (it != w->pool.end()) {
      auto& v = it->second;
      v.erase(std::remove(v.begin(), v.end(), this), v.end());
    }
  }
  if (op) {
    // upstream died with an active op: surface to the client
    ProxyOp* o = op;
    op = nullptr;
    o->upstream = nullptr;
    if (o->client) {
      ClientConn* c = o->client;
      if (!o->resp_head_seen) {
        // TCP connect never completed on a TPS-selected endpoint:
        // retry ONCE against the next-best endpoint (the dead one
        // keeps the top TPS score until the health loop ejects it;
        // the request bytes are still buffered).  The reference
        // surfaces the 502 instead (openai.rs:1006-1079) — this is a
        // deliberate availability improvement: a killed backend costs
        // clients one extra connect, not an error, during the
        // eject window.
        if (!connected && o->selected && c->retry_exclude_ep.empty() &&
            !o->endpoint_id.empty()) {
          c->retry_exclude_ep = o->endpoint_id;
          LOG_WARN("connect to endpoint %s failed; retrying selection "
                   "without it", o->endpoint_id.c_str());
          c->w->gw->metrics.upstream_errors.fetch_add(1);
          c->w->gw->abandon_op_for_retry(c);
          c->process_requests();
        } else {
          c->w->gw->metrics.upstream_errors.fetch_add(1);
          c->send_json_error(502, "endpoint_upstream_error",
                             "upstream connection failed");
          c->w->gw->finalize_op(c, false, 502, "upstream_conn_failed");
        }
      } else if (o->streaming) {
        if (o->body_reader.eof_terminated()) {
          // EOF-terminated stream: this IS normal completion
          o->body_reader.finish_eof();
          c->w->gw->finalize_op(c, true, o->resp.status, "");
          c->want_close = true;  // framing was EOF-delimited
          c->flush_out();
        } else {
          // mid-stream failure: abort the client connection
          c->w->gw->finalize_op(c, false, o->resp.status,
                                "upstream_aborted");
          c->close_now();
        }
      } else {
        c->w->gw->metrics.upstream_errors.fetch_add(1);
        c->send_json_error(502, "endpoint_upstream_error",
                           "upstream closed before response completed");
        c->w->gw->finalize_op(c, false, 502, "upstream_eof");
      }
    }
  }
  w->loop.defer_delete([this] { delete this; });
}

void UpstreamConn::on_event(uint32_t ev) {
  if (closed) return;
  if (ev & (EPOLLERR | EPOLLHUP)) {
    // if there is pending readable data, drain it first
    if (!(ev & EPOLLIN)) {
      close_now();
      return;
    }
  }
  if (ev & EPOLLOUT) handle_writable();
  if (closed) return;
  if (ev & EPOLLIN) handle_readable();
}

void UpstreamConn::handle_writable() {
  if (!connected) {
    int err = 0;
    socklen_t len = sizeof err;
    getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &len);
    if (err != 0) {
      close_now();
      return;
    }
    connected = true;
  }
  send_buffered();
}

void UpstreamConn::send_buffered() {
  while (!out.empty()) {
    ssize_t n = ::send(fd, out.data(), out.size(), MSG_NOSIGNAL);
    if (n > 0) {
      out.consume((size_t)n);
    } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      want(EPOLLIN | EPOLLOUT);
      return;
    } else {
      close_now();
      return;
    }
  }
  want(EPOLLIN);
}

void UpstreamConn::handle_readable() {
  while (true) {
    char* p = in.reserve(65536);
    ssize_t n = ::recv(fd, p, 65536, 0);
    if (n > 0) {
      in.commit((size_t)n);
      if (op) {
        extern void proxy_on_upstream_data(UpstreamConn* uc);
        proxy_on_upstream_data(this);
        if (closed) return;
        if (reading_paused) return;  // client backpressure
      } else {
        // data on an idle pooled conn: protocol garbage; drop it
        close_now();
        return;
      }
      if ((size_t)n < 65536) return;
    } else if (n == 0) {
      close_now();
      return;
    } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
      return;
    } else {
      close_now();
      return;
    }
  }
}

// ======================================================== ClientConn
ClientConn::ClientConn(Worker* w_, int fd_, std::string ip)
    : w(w_), fd(fd_), peer_ip(std::move(ip)) {
  set_nonblocking(fd);
  set_nodelay(fd);
  events = EPOLLIN;
  w->loop.add(fd, events, this);
}

ClientConn::~ClientConn() {
  if (fd >= 0) {
    w->loop.del(fd);
    ::close(fd);
  }
}

void ClientConn::want(uint32_t ev) {
  if (ev != events) {
    events = ev;
    w->loop.mod(fd, ev, this);
  }
}

void ClientConn::close_now() {
  if (closed) return;
  closed = true;
  w->timed_ops.erase(this);
  w->parked.erase(this);
  if (op) {
    // client went away mid-op: tear down upstream, record failure
    if (op->timer_id) w->loop.cancel_timer(op->timer_id);
    if (op->upstream) {
      UpstreamConn* uc = op->upstream;
      uc->op = nullptr;
      op->upstream = nullptr;
      uc->close_now();
    }
    w->gw->metrics.active_requests.fetch_sub(1);
    op.reset();  // lease Drop records the failure
  }
  w->loop.defer_delete([this] { delete this; });
}

void ClientConn::on_event(uint32_t ev) {
  if (closed) return;
  if (ev & (EPOLLERR | EPOLLHUP)) {
    if (!(ev & EPOLLIN)) {
      close_now();
      return;
    }
  }
  if (ev & EPOLLOUT) flush_out();
  if (closed) return;
  if (ev & EPOLLIN) handle_readable();
}

void ClientConn::handle_readable() {
  while (true) {
    char* p = in.reserve(65536);
    ssize_t n = ::recv(fd, p, 65536, 0);
    if (n > 0) {
      in.commit((size_t)n);
      process_requests();
      if (closed) return;
      if ((size_t)n < 65536) return;
    } else if (n == 0) {
      close_now();
      return;
    } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
      return;
    } else {
      close_now();
      return;
    }
  }
}

void ClientConn::process_requests() {
  // one request at a time; further pipelined bytes wait in `in`.
  // NOTE: the parsed head holds string_views INTO `in`; any later read
  // may reallocate the buffer, so the head is re-parsed on every call
  // and views never outlive this function (parse cost ~100 ns).
  if (op || response_open) return;
  while (!closed) {
    int rc = parse_request_head(in.view(), req);
    if (rc == 0) return;
    if (rc < 0) {
      cur_method.clear();
      cur_path.clear();
      send_simple(400, "text/plain", "bad request");
      want_close = true;
      flush_out();
      return;
    }
    cur_method.assign(req.method);
    cur_path.assign(req.path);
    if (req.chunked) {
      send_json_error(411, "length_required",
                      "chunked request bodies are not supported");
      want_close = true;
      flush_out();
      return;
    }
    if (!expect_ack_sent && req.content_length > 0 &&
        ieq(req.header("expect"), "100-continue") &&
        in.size() < req.header_len + (size_t)req.content_length) {
      out.append("HTTP/1.1 100 Continue\r\n\r\n");
      expect_ack_sent = true;
      flush_out();
    }
    if ((size_t)req.content_length > w->gw->cfg.body_limit) {
      send_json_error(413, "payload_too_large", "body exceeds limit");
      want_close = true;
      flush_out();
      return;
    }
    size_t need = req.header_len + (size_t)req.content_length;
    if (in.size() < need) return;
    dispatch();
    if (op || response_open) return;  // async op in flight
    if (closed) return;
  }
}

void ClientConn::dispatch() {
  // request tracing: honour a client-supplied X-Request-Id, else mint
  // one; echoed on the response and forwarded upstream
  std::string_view rid = req.header("x-request-id");
  if ((adm_delayed || !retry_exclude_ep.empty()) && !request_id.empty()) {
    // re-dispatch (delayed-admission hold / connect retry): keep the id
  } else if (!rid.empty() && rid.size() <= 64) {
    request_id.assign(rid);
  } else {
    char buf[32];
    snprintf(buf, sizeof buf, "req_%08x%06llx", (unsigned)rand_r(&w->rng),
             (unsigned long long)(w->request_seq++ & 0xffffff));
    request_id = buf;
  }
  w->gw->handle_request(this);
  // a conn parked by the delayed-admission arm keeps its request
  // buffered; the timer re-enters process_requests() to re-parse and
  // re-dispatch it (head views cannot be kept across the hold: any
  // socket read may reallocate `in`)
  if (!op && !response_open && !w->parked.count(this))
    finish_request_cycle();
}

void ClientConn::finish_request_cycle() {
  size_t need = req.header_len + (size_t)req.content_length;
  in.consume(need);
  expect_ack_sent = false;
  adm_delayed = false;
  retry_exclude_ep.clear();
  req.headers.clear();  // keep vector capacity across requests
  req.header_len = 0;
  req.content_length = 0;
  req.chunked = false;
  if (want_close) {
    flush_out();
    if (out.empty()) close_now();
  }
}

void ClientConn::flush_out() {
  while (!out.empty()) {
    ssize_t n = ::send(fd, out.data(), out.size(), MSG_NOSIGNAL);
    if (n > 0) {
      out.consume((size_t)n);
    } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      want(EPOLLIN | EPOLLOUT);
      return;
    } else {
      close_now();
      return;
    }
  }
  // fully drained
  want(EPOLLIN);
  if (op && op->upstream && op->upstream->reading_paused) {
    // resume upstream reads (backpressure release)
    op->upstream->reading_paused = false;
    op->upstream->want(EPOLLIN);
  }
  if (want_close && !op && !response_open) close_now();
}

void ClientConn::send_simple(int status, std::string_view content_type,
                             std::string_view body,
                             std::string_view extra_headers) {
  // every-request audit trail (excl. health/metrics/static —
  // audit/middleware.rs:23-45), buffered per worker
  if (getenv("LLMLB_NO_AUDIT") == nullptr && !cur_path.empty() &&
      cur_path != "/health" && cur_path != "/api/health" &&
      cur_path != "/metrics" && cur_path != "/api/metrics" &&
      !starts_with(cur_path, "/dashboard")) {
    AuditRecord a;
    a.ts_ms = wall_ms();
    a.method = cur_method;
    a.path = cur_path;
    a.status = status;
    a.client_ip = peer_ip;
    w->pending_audits.push_back(std::move(a));
  }
  append_status_line(out, status, status_reason(status));
  char hdr[320];
  int n = snprintf(hdr, sizeof hdr,
                   "Content-Type: %.*s\r\nContent-Length: %zu\r\n",
                   (int)content_type.size(), content_type.data(),
                   body.size());
  out.append(hdr, n);
  if (!request_id.empty()) {
    out.append("X-Request-Id: ");
    out.append(request_id);
    out.append("\r\n");
  }
  if (!extra_headers.empty()) out.append(extra_headers);
  if (want_close) out.append("Connection: close\r\n");
  out.append("\r\n");
  out.append(body);
  // update metrics
  if (status < 400) w->gw->metrics.responses_2xx.fetch_add(1);
  else if (status < 500) w->gw->metrics.responses_4xx.fetch_add(1);
  else w->gw->metrics.responses_5xx.fetch_add(1);
  flush_out();
}

void ClientConn::send_json_error(int status, std::string_view code,
                                 std::string_view message) {
  // OpenAI error envelope (api/openai_util.rs:57)
  std::string body = "{\"error\":{\"message\":";
  Json::escape(message, body);
  body += ",\"type\":";
  Json::escape(status >= 500 ? "server_error" : "invalid_request_error",
               body);
  body += ",\"code\":";
  Json::escape(code, body);
  body += "}}";
  send_simple(status, "application/json", body);
}

// ======================================================== Worker
Worker::Worker(Gateway* gw_, int index_) : gw(gw_), index(index_) {
  acceptor.w = this;
  rng = (unsigned)(wall_ms() ^ (0x9e3779b9u * (unsigned)(index_ + 1)));
}

void Worker::Acceptor::on_event(uint32_t) {
  while (true) {
    sockaddr_in addr{};
    socklen_t alen = sizeof addr;
    int cfd = accept4(w->listen_fd, (sockaddr*)&addr, &alen,
                      SOCK_NONBLOCK | SOCK_CLOEXEC);
    if (cfd < 0) return;
    char ip[64];
    inet_ntop(AF_INET, &addr.sin_addr, ip, sizeof ip);
    new ClientConn(w, cfd, ip);  // owned by the loop; deleted on close
  }
}

void Worker::arm_record_flush() {
  loop.add_timer(200, [this] {
    if (!pending_records.empty()) gw->history.record_batch(pending_records);
    if (!pending_audits.empty()) gw->history.audit_batch(pending_audits);
    arm_record_flush();
  });
}

void Worker::arm_op_sweep() {
  loop.add_timer(250, [this] {
    int64_t now = now_us();
    // collect first: timeout handling mutates timed_ops
    std::vector<ClientConn*> expired;
    for (ClientConn* c : timed_ops)
      if (c->op && c->op->deadline_us && now > c->op->deadline_us)
        expired.push_back(c);
    for (ClientConn* c : expired) {
      ProxyOp* o = c->op.get();
      if (!o) continue;
      if (o->upstream) {
        UpstreamConn* uc = o->upstream;
        uc->op = nullptr;
        o->upstream = nullptr;
        uc->close_now();
      }
      if (!o->resp_head_seen) {
        c->send_json_error(504, "endpoint_timeout", "upstream timed out");
        gw->finalize_op(c, false, 504, "timeout");
      } else {
        gw->finalize_op(c, false, o->resp.status, "stream_timeout");
        c->close_now();
      }
    }
    arm_op_sweep();
  });
}

void Worker::start() {
  listen_fd = make_listener(gw->cfg.host, gw->cfg.port, true);
  if (listen_fd < 0) {
    LOG_ERROR("worker %d: cannot listen on %s:%d", index,
              gw->cfg.host.c_str(), gw->cfg.port);
    exit(1);
  }
  loop.add(listen_fd, EPOLLIN, &acceptor);
  arm_record_flush();
  arm_op_sweep();
  thread = std::thread([this] {
    char name[16];
    snprintf(name, sizeof name, "gw-worker-%d", index);
    pthread_setname_np(pthread_self(), name);
    loop.run();
  });
}

void Worker::join() {
  loop.stop();
  if (thread.joinable()) thread.join();
  // free idle pooled upstream connections: their deferred-delete path
  // runs on the (now stopped) event loop, so reclaim directly.  Only
  // still-open conns are deleted here — closed ones are owned by the
  // loop's deferred queue.
  for (auto& [key, conns] : pool) {
    for (UpstreamConn* uc : conns) {
      if (!uc->closed) {
        if (uc->fd >= 0) ::close(uc->fd);
        delete uc;
      }
    }
  }
  pool.clear();
}

UpstreamConn* Worker::acquire_upstream(const EndpointPtr& ep,
                                       const std::string& key,
                                       const std::string& host, int port) {
  auto it = pool.find(key);
  while (it != pool.end() && !it->second.empty()) {
    UpstreamConn* uc = it->second.back();
    it->second.pop_back();
    if (!uc->closed) return uc;
  }
  auto* uc = new UpstreamConn(this, ep, host, port);
  uc->pool_key = key;
  if (!uc->start()) {
    delete uc;
    return nullptr;
  }
  return uc;
}

void Worker::release_upstream(UpstreamConn* uc) {
  if (uc->closed) return;
  uc->op = nullptr;
  uc->in.clear();
  uc->out.clear();
  uc->reading_paused = false;
  uc->want(EPOLLIN);  // detect server-side close while pooled
  auto& vec = pool[uc->pool_key];
  if (vec.size() >= 32) {  // 32 idle conns/host (bootstrap.rs:95)
    uc->close_now();
    return;
  }
  vec.push_back(uc);
}

void Worker::discard_upstream(UpstreamConn* uc) { uc->close_now(); }

// ======================================================== Gateway
Gateway::Gateway(GatewayConfig cfg_)
    : cfg(std::move(cfg_)), registry(&db),
      history(&db, cfg.retention_days), auth(&db) {}

Gateway::~Gateway() { stop(); }

bool Gateway::init(std::string* err) {
  log_init(cfg.log_level, cfg.log_dir, cfg.log_retention_days);
  // raise RLIMIT_NOFILE (bootstrap.rs:374-422)
  rlimit rl{};
  if (getrlimit(RLIMIT_NOFILE, &rl) == 0 && rl.rlim_cur < 65536) {
    rl.rlim_cur = std::min<rlim_t>(65536, rl.rlim_max);
    setrlimit(RLIMIT_NOFILE, &rl);
  }
  if (!cfg.db_path.empty() && cfg.db_path != ":memory:none") {
    if (!db.open(cfg.db_path, err)) return false;
    if (cfg.audit_archive_days > 0)
      history.set_audit_archive(cfg.db_path + ".audit-archive.sqlite3",
                                cfg.audit_archive_days);
    if (!db.migrate(err)) return false;
  }
  registry.load_from_db();
  auth.load();
  history.start();
  // boot reseed (bootstrap.rs:127-159): 60-min live ring from
  // request_history, TPS EMAs from today's daily stats
  if (db.ok()) {
    std::vector<DbRow> rows;
    db.exec("SELECT ts_ms, endpoint_id, model, api_kind, status, "
            "duration_ms, prompt_tokens, completion_tokens, client_ip "
            "FROM request_history WHERE ts_ms > ? ORDER BY ts_ms",
            {DbParam((long long)(wall_ms() - 60 * 60 * 1000))}, &rows);
    for (auto& r : rows) {
      RequestRecord rec;
      rec.ts_ms = r[0].i;
      rec.endpoint_id = r[1].str();
      rec.model = r[2].str();
      rec.api_kind = r[3].str();
      rec.status = (int)r[4].i;
      rec.duration_ms = r[5].kind == DbValue::kReal ? r[5].d
                                              : (double)r[5].i;
      rec.prompt_tokens = r[6].i;
      rec.completion_tokens = r[7].i;
      rec.client_ip = r[8].str();
      history.ring().push(rec);
    }
    if (!rows.empty())
      LOG_INFO("reseeded %zu request-history ring entries", rows.size());
    registry.seed_tps_from_db();
  }
  start_ms = wall_ms();
  int nworkers = cfg.workers > 0
                     ? cfg.workers
                     : (int)std::thread::hardware_concurrency();
  if (nworkers < 1) nworkers = 1;
  for (int i = 0; i < nworkers; ++i)
    workers.push_back(std::make_unique<Worker>(this, i));
  return true;
}

void Gateway::run() {
  for (auto& w : workers) w->start();
  health_thread = std::thread([this] { health_loop(); });
  LOG_INFO("llmlb-gw listening on %s:%d with %d workers",
           cfg.host.c_str(), cfg.port, (int)workers.size());
  // main thread parks until stop() is requested (signal handlers only
  // set a flag — never join threads from signal context)
  while (!stopping.load()) usleep(50 * 1000);
  shutdown_threads();
}

void Gateway::stop() { stopping.store(true); }

void Gateway::shutdown_threads() {
  if (shutdown_done_.exchange(true)) return;
  stopping.store(true);
  for (auto& w : workers) w->loop.stop();
  if (health_thread.joinable()) health_thread.join();
  for (auto& w : workers) w->join();
  // drain the per-worker record buffers before stopping the writer —
  // the 200 ms flush timer may not have fired for the last requests,
  // and losing them breaks the boot-time history/TPS reseed
  for (auto& w : workers) {
    if (!w->pending_records.empty())
      history.record_batch(w->pending_records);
    if (!w->pending_audits.empty())
      history.audit_batch(w->pending_audits);
  }
  history.stop();
}

}  // namespace gw
