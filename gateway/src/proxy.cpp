// The proxy hot path: request dispatch, endpoint selection, upstream
// request build (raw-splice model rewrite + stream_options injection),
// and the zero-copy SSE pump with in-band token accounting.
//
// Behavior spec: reference api/openai.rs:761-1338 (proxy_openai_post)
// and api/proxy.rs:120-270 (streaming TPS tap) — reimplemented without
// the per-chunk utf-8 copy: bytes are received once into the upstream
// buffer, scanned in place, and written verbatim to the client.
#include <arpa/inet.h>

#include "anthropic.h"
#include "gateway.h"
#include "bpe.h"

namespace gw {

static const size_t kHighWater = 1 << 20;  // client out-buf backpressure

// ---- raw body scanning helpers ----
static bool raw_bool_true(std::string_view body, std::string_view key) {
  // top-level `"key": true` check (depth-insensitive fast scan is fine
  // for our payloads: a nested "stream" key is pathological)
  size_t pos = 0;
  std::string pat = "\"";
  pat += key;
  pat += "\"";
  while ((pos = body.find(pat, pos)) != std::string_view::npos) {
    size_t i = pos + pat.size();
    while (i < body.size() &&
           (body[i] == ' ' || body[i] == '\t' || body[i] == '\n' ||
            body[i] == '\r'))
      ++i;
    if (i < body.size() && body[i] == ':') {
      ++i;
      while (i < body.size() &&
             (body[i] == ' ' || body[i] == '\t' || body[i] == '\n' ||
              body[i] == '\r'))
        ++i;
      return i < body.size() && body[i] == 't';
    }
    pos += pat.size();
  }
  return false;
}

static long long raw_int_field(std::string_view body, std::string_view key) {
  std::string pat = "\"";
  pat += key;
  pat += "\"";
  size_t pos = body.find(pat);
  if (pos == std::string_view::npos) return -1;
  size_t i = body.find(':', pos + pat.size());
  if (i == std::string_view::npos) return -1;
  ++i;
  while (i < body.size() && body[i] == ' ') ++i;
  return atoll(std::string(body.substr(i, 20)).c_str());
}

// Build the upstream request body: substitute the public model name with
// the endpoint-native one and, for streaming chat requests, inject
// stream_options.include_usage (api/openai.rs:945-993).  Returns true
// when `rewritten` was used; false when the original body can be sent
// verbatim (the common case on the bench path).
static bool build_upstream_body(std::string_view body,
                                std::string_view public_model,
                                std::string_view upstream_model,
                                bool want_stream, bool inject_usage,
                                std::string* rewritten) {
  bool need_model = upstream_model != public_model && !upstream_model.empty();
  bool need_usage = want_stream && inject_usage &&
                    body.find("\"stream_options\"") == std::string_view::npos;
  if (!need_model && !need_usage) return false;
  rewritten->clear();
  rewritten->reserve(body.size() + 64);
  if (need_model) {
    auto [pos, len] = find_raw_string_field(body, "model");
    if (pos != std::string_view::npos) {
      rewritten->append(body.substr(0, pos));
      rewritten->append(upstream_model);
      rewritten->append(body.substr(pos + len));
    } else {
      rewritten->append(body);
    }
  } else {
    rewritten->append(body);
  }
  if (need_usage) {
    size_t close = rewritten->rfind('}');
    if (close != std::string::npos)
      rewritten->insert(close,
                        ",\"stream_options\":{\"include_usage\":true}");
  }
  return true;
}

// -------------------------------------------------------------------
// Request dispatch
// -------------------------------------------------------------------
void Gateway::handle_request(ClientConn* c) {
  if (!c->adm_delayed && c->retry_exclude_ep.empty())
    metrics.requests_total.fetch_add(1, std::memory_order_relaxed);
  const HttpRequest& r = c->req;
  std::string_view body =
      c->in.view().substr(r.header_len, (size_t)r.content_length);
  std::string path(r.path);

  // un-authed health probe (used by load balancer fleets & tests)
  if (path == "/health" || path == "/api/health") {
    std::string b = "{\"status\":\"ok\",\"version\":\"0.1.0\","
                    "\"uptime_ms\":" +
                    std::to_string(wall_ms() - start_ms.load()) + "}";
    c->send_simple(200, "application/json", b);
    return;
  }
  if (path == "/metrics" || path == "/api/metrics" ||
      path == "/api/metrics/cloud") {
    c->send_simple(200, "text/plain; version=0.0.4", render_metrics_text());
    return;
  }

  if (starts_with(path, "/v1/")) {
    // inference gate: reject while draining for self-update
    // (inference_gate.rs:177-197)
    if (draining.load(std::memory_order_relaxed)) {
      c->send_simple(503, "application/json",
                     "{\"error\":{\"message\":\"draining for update\","
                     "\"type\":\"server_error\",\"code\":\"draining\"}}",
                     "Retry-After: 5\r\n");
      return;
    }
    // auth: Bearer key (x-api-key also accepted on /v1/messages,
    // mirroring the Anthropic-native surface).  Scope per route:
    // /v1/models needs openai.models.read, everything else
    // openai.inference (reference permission matrix auth.rs:60-95).
    const bool local_client = is_loopback_ip(c->peer_ip);
    const char* scope = (path == "/v1/models" ||
                         starts_with(path, "/v1/models/"))
                            ? "openai.models.read"
                            : "openai.inference";
    std::string_view authz = r.header("authorization");
    bool authed = auth.check_bearer(authz, scope, local_client);
    if (!authed && path == "/v1/messages") {
      std::string_view xkey = r.header("x-api-key");
      authed = !xkey.empty() && auth.check_key(xkey, scope, local_client);
    }
    if (!authed) {
      c->send_json_error(401, "invalid_api_key",
                         "missing or invalid API key");
      return;
    }

    if (path == "/v1/chat/completions")
      return start_proxy(c, ApiKind::kChat, kCapChat,
                         "/v1/chat/completions");
    if (path == "/v1/completions")
      return start_proxy(c, ApiKind::kCompletions, 0, "/v1/completions");
    if (path == "/v1/embeddings")
      return start_proxy(c, ApiKind::kEmbeddings, kCapEmbeddings,
                         "/v1/embeddings");
    if (path == "/v1/responses")
      return start_proxy(c, ApiKind::kResponses, kCapResponses,
                         "/v1/responses");
    if (path == "/v1/messages") return handle_anthropic_messages(this, c);
    if (path == "/v1/models" || starts_with(path, "/v1/models/")) {
      handle_local(c, path, body);
      return;
    }
    if (starts_with(path, "/v1/audio/") || starts_with(path, "/v1/images/")) {
      handle_local(c, path, body);  // capability-routed in api_local.cpp
      return;
    }
    c->send_json_error(404, "not_found", "unknown /v1 route");
    return;
  }

  if (starts_with(path, "/api/") || path == "/" ||
      starts_with(path, "/dashboard") || starts_with(path, "/ws/")) {
    handle_local(c, path, body);
    return;
  }
  LOG_WARN("unrouted request: method='%.*s' path='%s' (target='%.*s')",
           (int)r.method.size(), r.method.data(), path.c_str(),
           (int)r.target.size(), r.target.data());
  c->send_json_error(404, "not_found", "no such route");
}

// -------------------------------------------------------------------
// Proxy start
// -------------------------------------------------------------------
static void arm_timeout(Gateway* gw, ClientConn* c, int timeout_s) {
  (void)gw;
  c->op->deadline_us = now_us() + (int64_t)timeout_s * 1000000;
  c->w->timed_ops.insert(c);
}

void arm_timeout_public(Gateway* gw, ClientConn* c, int secs) {
  arm_timeout(gw, c, secs);
}

void Gateway::start_proxy(ClientConn* c, ApiKind kind,
                          uint32_t required_caps,
                          const std::string& upstream_path) {
  const HttpRequest& r = c->req;
  std::string_view body =
      c->in.view().substr(r.header_len, (size_t)r.content_length);
  std::string model = raw_string_field(body, "model");
  if (model.empty()) {
    c->send_json_error(400, "missing_model", "request must name a model");
    return;
  }
  // cloud prefixes: openai: / google: / anthropic: (+ the reference's
  // documented `ahtnorpic:` typo-alias, api/openai.rs:637-655)
  for (const char* pfx : {"openai:", "google:", "anthropic:", "ahtnorpic:"}) {
    if (starts_with(model, pfx)) {
      std::string provider = pfx;
      provider.pop_back();
      if (provider == "ahtnorpic") provider = "anthropic";
      return start_cloud_proxy(c, kind, provider,
                               model.substr(strlen(pfx)), upstream_path,
                               false);
    }
  }

  auto table = registry.table();
  SelectResult sel = select_endpoint(*table, model, kind, cfg.admission,
                                     wall_ms(), required_caps,
                                     &c->retry_exclude_ep);
  if (sel.err_status != 0) {
    const char* msg =
        sel.err_status == 404 ? "model not found on any online endpoint"
        : sel.err_status == 429 ? "all endpoints at capacity"
                                : "no endpoint online for this model";
    c->send_json_error(sel.err_status, sel.err_code, msg);
    return;
  }

  // delayed-admission arm (reference admission_control 50%->delay,
  // 80%->reject): park the request on the event loop for 10-100 ms,
  // then re-run selection (conditions may have improved).  One hold
  // max per request; the conn may close while parked, so membership
  // in w->parked validates the timer.
  if (sel.delay_ms > 0 && !c->adm_delayed) {
    c->adm_delayed = true;
    c->adm_delay_ms = sel.delay_ms;
    Worker* w = c->w;
    w->parked.insert(c);
    metrics.admission_delays.fetch_add(1, std::memory_order_relaxed);
    w->loop.add_timer(sel.delay_ms, [w, c] {
      if (w->parked.erase(c)) c->process_requests();
    });
    return;
  }

  bool want_stream = raw_bool_true(body, "stream");
  std::string rewritten;
  bool use_rewritten = build_upstream_body(
      body, model, sel.upstream_model, want_stream,
      kind == ApiKind::kChat || kind == ApiKind::kCompletions, &rewritten);
  std::string_view out_body = use_rewritten ? rewritten : body;

  UpstreamConn* uc = c->w->acquire_upstream(sel.ep, sel.ep->id,
                                            sel.ep->host, sel.ep->port);
  if (!uc) {
    metrics.upstream_errors.fetch_add(1);
    c->send_json_error(502, "endpoint_unreachable",
                       "could not connect to upstream endpoint");
    return;
  }

  auto op = std::make_unique<ProxyOp>();
  op->client = c;
  op->upstream = uc;
  op->start_us = now_us();
  op->public_model = model;
  op->upstream_model = sel.upstream_model;
  op->endpoint_id = sel.ep->id;
  op->kind = kind;
  op->client_ip = c->peer_ip;
  op->request_id = c->request_id;
  if (cfg.capture_bodies)
    op->captured_request = std::string(body.substr(0, 16384));
  op->req_prompt_chars = (long long)body.size();
  op->tps = sel.tps;
  op->selected = true;
  op->admission_delay_ms = c->adm_delayed ? c->adm_delay_ms : 0;
  op->lease = RequestLease(sel.ep, &registry, sel.canonical, kind);
  op->lease.set_tps(sel.tps);
  c->op = std::move(op);
  uc->op = c->op.get();

  // upstream request head
  char head[512];
  int n = snprintf(
      head, sizeof head,
      "POST %s%s HTTP/1.1\r\nHost: %s:%d\r\nContent-Type: "
      "application/json\r\nContent-Length: %zu\r\nAccept: */*\r\n",
      sel.ep->path_prefix.c_str(), upstream_path.c_str(),
      sel.ep->host.c_str(), sel.ep->port, out_body.size());
  uc->out.append(head, n);
  if (!sel.ep->api_key.empty()) {
    uc->out.append("Authorization: Bearer ");
    uc->out.append(sel.ep->api_key);
    uc->out.append("\r\n");
  } else {
    std::string_view authz = r.header("authorization");
    if (!authz.empty()) {
      uc->out.append("Authorization: ");
      uc->out.append(authz);
      uc->out.append("\r\n");
    }
  }
  uc->out.append("Connection: keep-alive\r\n\r\n");
  uc->out.append(out_body);
  metrics.active_requests.fetch_add(1);
  arm_timeout(this, c, sel.ep->inference_timeout_s);
  if (uc->connected) uc->send_buffered();
  // else: connect completion (EPOLLOUT) flushes
}

// direct proxy to a chosen endpoint (audio/images capability routing,
// playground passthrough); body forwarded verbatim
void start_proxy_direct(Gateway* gw, ClientConn* c, EndpointPtr ep,
                        ApiKind kind, const std::string& upstream_path,
                        std::string_view content_type) {
  const HttpRequest& r = c->req;
  std::string_view body =
      c->in.view().substr(r.header_len, (size_t)r.content_length);
  UpstreamConn* uc =
      c->w->acquire_upstream(ep, ep->id, ep->host, ep->port);
  if (!uc) {
    gw->metrics.upstream_errors.fetch_add(1);
    c->send_json_error(502, "endpoint_unreachable",
                       "could not connect to upstream endpoint");
    return;
  }
  auto op = std::make_unique<ProxyOp>();
  op->client = c;
  op->upstream = uc;
  op->start_us = now_us();
  op->endpoint_id = ep->id;
  op->kind = kind;
  op->client_ip = c->peer_ip;
  op->request_id = c->request_id;
  if (gw->cfg.capture_bodies)
    op->captured_request = std::string(body.substr(0, 16384));
  op->public_model = raw_string_field(body, "model");
  op->upstream_model = op->public_model;
  op->lease = RequestLease(ep, &gw->registry, op->public_model, kind);
  c->op = std::move(op);
  uc->op = c->op.get();

  char head[512];
  int n = snprintf(head, sizeof head,
                   "%.*s %s%s HTTP/1.1\r\nHost: %s:%d\r\nContent-Length: "
                   "%zu\r\nAccept: */*\r\n",
                   (int)r.method.size(), r.method.data(),
                   ep->path_prefix.c_str(), upstream_path.c_str(),
                   ep->host.c_str(), ep->port, body.size());
  uc->out.append(head, n);
  if (!c->request_id.empty()) {
    uc->out.append("X-Request-Id: ");
    uc->out.append(c->request_id);
    uc->out.append("\r\n");
  }
  if (!content_type.empty()) {
    uc->out.append("Content-Type: ");
    uc->out.append(content_type);
    uc->out.append("\r\n");
  }
  if (!ep->api_key.empty()) {
    uc->out.append("Authorization: Bearer ");
    uc->out.append(ep->api_key);
    uc->out.append("\r\n");
  }
  uc->out.append("Connection: keep-alive\r\n\r\n");
  uc->out.append(body);
  gw->metrics.active_requests.fetch_add(1);
  arm_timeout(gw, c, ep->inference_timeout_s);
  if (uc->connected) uc->send_buffered();
}

// -------------------------------------------------------------------
// Cloud-prefix passthrough (api/cloud_proxy.rs).  HTTP only: this
// environment has no TLS stack and no egress; base URLs come from
// LLMLB_{OPENAI,GOOGLE,ANTHROPIC}_BASE_URL (tests point them at local
// mock provider servers).
// -------------------------------------------------------------------
void Gateway::start_cloud_proxy(ClientConn* c, ApiKind kind,
                                const std::string& provider,
                                const std::string& bare_model,
                                const std::string& upstream_path,
                                bool anthropic_native) {
  const HttpRequest& r = c->req;
  std::string_view body =
      c->in.view().substr(r.header_len, (size_t)r.content_length);
  std::string base, key;
  if (provider == "openai") {
    base = cfg.openai_base;
    key = cfg.openai_key;
  } else if (provider == "google") {
    base = cfg.google_base;
    key = cfg.google_key;
  } else {
    base = cfg.anthropic_base;
    key = cfg.anthropic_key;
  }
  if (base.empty()) {
    c->send_json_error(502, "cloud_provider_unconfigured",
                       "no base URL configured for provider '" + provider +
                           "' (set LLMLB_" + to_lower(provider) +
                           "_BASE_URL)");
    return;
  }
  std::string host, prefix;
  int port = 80;
  {
    std::string_view s = base;
    if (starts_with(s, "http://")) s.remove_prefix(7);
    else {
      c->send_json_error(502, "cloud_provider_unsupported",
                         "only http:// cloud base URLs are supported in "
                         "this build (no TLS stack, no egress)");
      return;
    }
    size_t slash = s.find('/');
    std::string_view hostport = s.substr(0, slash);
    prefix = slash == std::string_view::npos ? ""
                                             : std::string(s.substr(slash));
    if (!prefix.empty() && prefix.back() == '/') prefix.pop_back();
    size_t colon = hostport.rfind(':');
    if (colon == std::string_view::npos) {
      host = std::string(hostport);
    } else {
      host = std::string(hostport.substr(0, colon));
      port = atoi(std::string(hostport.substr(colon + 1)).c_str());
    }
  }

  // rewrite "provider:model" -> "model"
  std::string rewritten;
  std::string public_model = provider + ":" + bare_model;
  build_upstream_body(body, public_model, bare_model, false, false,
                      &rewritten);
  std::string_view out_body = rewritten.empty() ? body : rewritten;

  UpstreamConn* uc =
      c->w->acquire_upstream(nullptr, "cloud:" + provider, host, port);
  if (!uc) {
    metrics.upstream_errors.fetch_add(1);
    c->send_json_error(502, "cloud_unreachable",
                       "could not connect to provider");
    return;
  }
  auto op = std::make_unique<ProxyOp>();
  op->client = c;
  op->upstream = uc;
  op->start_us = now_us();
  op->public_model = public_model;
  op->upstream_model = bare_model;
  op->kind = kind;
  op->is_cloud = true;
  op->cloud_provider = provider;
  op->client_ip = c->peer_ip;
  op->request_id = c->request_id;
  if (cfg.capture_bodies)
    op->captured_request = std::string(body.substr(0, 16384));
  c->op = std::move(op);
  uc->op = c->op.get();
  metrics.cloud_requests.fetch_add(1);

  char head[512];
  int n = snprintf(head, sizeof head,
                   "POST %s%s HTTP/1.1\r\nHost: %s\r\nContent-Type: "
                   "application/json\r\nContent-Length: %zu\r\n",
                   prefix.c_str(), upstream_path.c_str(), host.c_str(),
                   out_body.size());
  uc->out.append(head, n);
  if (provider == "anthropic") {
    // Anthropic-native auth headers (api/anthropic.rs:137-280)
    if (!key.empty()) {
      uc->out.append("x-api-key: ");
      uc->out.append(key);
      uc->out.append("\r\n");
    }
    std::string_view ver = r.header("anthropic-version");
    uc->out.append("anthropic-version: ");
    uc->out.append(ver.empty() ? std::string_view("2023-06-01") : ver);
    uc->out.append("\r\n");
  } else if (!key.empty()) {
    uc->out.append("Authorization: Bearer ");
    uc->out.append(key);
    uc->out.append("\r\n");
  }
  uc->out.append("Connection: keep-alive\r\n\r\n");
  uc->out.append(out_body);
  metrics.active_requests.fetch_add(1);
  arm_timeout(this, c, 120);
  if (uc->connected) uc->send_buffered();
  (void)anthropic_native;
}

// -------------------------------------------------------------------
// Upstream data pump
// -------------------------------------------------------------------
static void respond_buffered(Gateway* gw, ClientConn* c);

void proxy_on_upstream_data(UpstreamConn* uc) {
  ProxyOp* op = uc->op;
  ClientConn* c = op->client;
  Gateway* gw = c->w->gw;

  if (!op->resp_head_seen) {
    int rc = parse_response_head(uc->in.view(), op->resp);
    if (rc == 0) return;
    if (rc < 0) {
      uc->close_now();  // surfaces 502 via close path
      return;
    }
    op->resp_head_seen = true;
    op->body_reader.init(op->resp);
    op->resp_content_type = std::string(op->resp.header("content-type"));
    std::string_view ctype = op->resp_content_type;
    bool sse = ctype.find("text/event-stream") != std::string_view::npos;
    op->streaming = op->resp.status / 100 == 2 && sse;
    if (op->streaming) {
      gw->metrics.streamed_responses.fetch_add(1);
      if (op->anthropic_reencode) {
        // re-encoded stream: we generate our own chunked SSE framing
        append_status_line(c->out, 200, "OK");
        c->out.append(
            "Content-Type: text/event-stream\r\nCache-Control: no-cache\r\n"
            "Transfer-Encoding: chunked\r\nConnection: keep-alive\r\n");
        if (!c->request_id.empty()) {
          c->out.append("X-Request-Id: ");
          c->out.append(c->request_id);
          c->out.append("\r\n");
        }
        if (op->admission_delay_ms > 0) {
          c->out.append("X-Admission-Delay-Ms: ");
          c->out.append(std::to_string(op->admission_delay_ms));
          c->out.append("\r\n");
        }
        c->out.append("\r\n");
      } else {
        // forward upstream head verbatim minus hop-by-hop Connection
        append_status_line(c->out, op->resp.status, op->resp.reason);
        for (auto& h : op->resp.headers) {
          if (ieq(h.name, "connection") || ieq(h.name, "keep-alive") ||
              ieq(h.name, "x-request-id"))
            continue;
          c->out.append(h.name);
          c->out.append(": ");
          c->out.append(h.value);
          c->out.append("\r\n");
        }
        if (!c->request_id.empty()) {
          c->out.append("X-Request-Id: ");
          c->out.append(c->request_id);
          c->out.append("\r\n");
        }
        if (op->admission_delay_ms > 0) {
          // reference x-queue-* analog (openai.rs:74-84): tell the
          // client it waited in delayed admission
          c->out.append("X-Admission-Delay-Ms: ");
          c->out.append(std::to_string(op->admission_delay_ms));
          c->out.append("\r\n");
        }
        c->out.append("\r\n");
      }
      c->response_open = true;
      // reference releases occupancy at response-header time for
      // streams (api/openai.rs:1091); tokens recorded at stream end
      op->lease.complete_success(0);
      op->lease_released_at_head = true;
    }
    uc->in.consume(op->resp.header_len);
    if (uc->in.empty() && !op->body_reader.done()) {
      if (op->streaming) c->flush_out();
      return;
    }
  }

  std::string_view data = uc->in.view();
  auto out = op->body_reader.feed(data);
  if (op->streaming) {
    if (op->anthropic_reencode) {
      for (auto& [off, len] : out.payload)
        anthropic_reencode_feed(op, c, data.substr(off, len));
    } else {
      c->out.append(data.substr(0, out.consumed));
      for (auto& [off, len] : out.payload)
        op->tap.feed(data.substr(off, len));
    }
    uc->in.consume(out.consumed);
    c->flush_out();
    if (c->closed) return;
    if (c->out.size() > kHighWater && !out.done) {
      uc->reading_paused = true;
      uc->want(0);  // EPOLLHUP/ERR still delivered
    }
  } else {
    for (auto& [off, len] : out.payload)
      op->buffered.append(data.substr(off, len));
    uc->in.consume(out.consumed);
    if (op->buffered.size() > 64 * 1024 * 1024) {
      uc->close_now();
      return;
    }
  }

  if (out.done) {
    // response complete
    bool upstream_reusable = op->resp.keep_alive &&
                             !op->body_reader.eof_terminated() &&
                             uc->in.empty();
    uc->op = nullptr;
    op->upstream = nullptr;
    if (op->streaming) {
      if (op->anthropic_reencode) anthropic_reencode_finish(op, c);
      gw->finalize_op(c, true, op->resp.status, "");
    } else {
      respond_buffered(gw, c);
    }
    if (upstream_reusable) c->w->release_upstream(uc);
    else uc->close_now();
  }
}

// post-process + answer a buffered (non-streaming) upstream response
static void respond_buffered(Gateway* gw, ClientConn* c) {
  ProxyOp* op = c->op.get();
  std::string& body = op->buffered;
  int status = op->resp.status;
  if (status / 100 == 2) {
    // rewrite the model name back to the public one (openai.rs:1222)
    if (op->upstream_model != op->public_model &&
        !op->public_model.empty()) {
      auto [pos, len] = find_raw_string_field(body, "model");
      if (pos != std::string::npos &&
          std::string_view(body).substr(pos, len) == op->upstream_model)
        body.replace(pos, len, op->public_model);
    }
    long long ct = raw_int_field(body, "completion_tokens");
    long long pt = raw_int_field(body, "prompt_tokens");
    if (ct < 0) ct = raw_int_field(body, "output_tokens");
    if (pt < 0) pt = raw_int_field(body, "input_tokens");
    if (op->kind == ApiKind::kEmbeddings && ct < 0) ct = 0;
    op->lease.complete_success(ct > 0 ? ct : 0);
    std::string ctype = op->resp_content_type;
    if (ctype.empty()) ctype = "application/json";
    std::string extra;
    if (op->admission_delay_ms > 0)
      extra += "X-Admission-Delay-Ms: " +
               std::to_string(op->admission_delay_ms) + "\r\n";
    if (op->anthropic_reencode) {
      // non-streaming Anthropic ingress: convert OpenAI -> Anthropic
      std::string converted;
      if (anthropic_convert_response(body, op->public_model, &converted))
        body.swap(converted);
    }
    c->send_simple(status, ctype, body, extra);
    gw->finalize_op(c, true, status, "");
    if (pt > 0) {
      // stash prompt tokens for history via tap fields
      (void)pt;
    }
  } else {
    gw->metrics.upstream_errors.fetch_add(1);
    op->lease.complete_failure();
    // normalize upstream errors to 502 (openai.rs:1156-1220)
    std::string msg = "upstream returned status " + std::to_string(status);
    std::string detail = body.substr(0, 2048);
    std::string out = "{\"error\":{\"message\":";
    Json::escape(msg, out);
    out += ",\"type\":\"endpoint_upstream_error\",\"code\":"
           "\"endpoint_upstream_error\",\"upstream_status\":" +
           std::to_string(status) + ",\"detail\":";
    Json::escape(detail, out);
    out += "}}";
    c->send_simple(502, "application/json", out);
    gw->finalize_op(c, false, 502, "upstream_status_" +
                                       std::to_string(status));
  }
}

// -------------------------------------------------------------------
// Op finalization: history + TPS + cleanup, then resume the client
// -------------------------------------------------------------------
void Gateway::abandon_op_for_retry(ClientConn* c) {
  std::unique_ptr<ProxyOp> op = std::move(c->op);
  if (!op) return;
  c->w->timed_ops.erase(c);
  if (op->timer_id) c->w->loop.cancel_timer(op->timer_id);
  op->upstream = nullptr;   // the upstream conn is tearing itself down
  metrics.active_requests.fetch_sub(1);
  if (op->lease.active()) op->lease.complete_failure();
}

void Gateway::finalize_op(ClientConn* c, bool success, int status,
                          const std::string& error) {
  std::unique_ptr<ProxyOp> op = std::move(c->op);
  if (!op) return;
  c->w->timed_ops.erase(c);
  if (op->timer_id) c->w->loop.cancel_timer(op->timer_id);
  if (op->upstream) {
    UpstreamConn* uc = op->upstream;
    uc->op = nullptr;
    op->upstream = nullptr;
    uc->close_now();
  }
  metrics.active_requests.fetch_sub(1);
  double dur_ms = (now_us() - op->start_us) / 1e3;
  long long completion_tokens = 0, prompt_tokens = 0;
  if (op->streaming) {
    completion_tokens = op->tap.completion_tokens();
    // no usage from the backend -> BPE-estimate over the accumulated
    // delta text (reference tiktoken fallback, token/mod.rs:217-223);
    // event count remains the last resort
    if (!op->tap.usage_seen() && !op->tap.text().empty()) {
      const BpeTokenizer* bpe = BpeTokenizer::default_instance();
      completion_tokens =
          bpe ? bpe->count(op->tap.text())
              : BpeTokenizer::heuristic_count(op->tap.text());
    }
    prompt_tokens = std::max(0LL, op->tap.prompt_tokens());
    metrics.tokens_streamed.fetch_add(completion_tokens);
    if (success && op->lease_released_at_head && !op->endpoint_id.empty()) {
      // stream TPS recorded at end (api/proxy.rs:165-183)
      EndpointPtr ep = op->lease.endpoint();
      if (!ep) ep = registry.get(op->endpoint_id);
      if (ep) {
        ep->update_latency(dur_ms);
        if (completion_tokens > 0) {
          if (op->tps)
            op->tps->update((double)completion_tokens, dur_ms / 1e3,
                            cfg.tps_ema_alpha,
                            wall_ms());
          else
            registry.record_tps(op->endpoint_id, op->public_model,
                                op->kind, (double)completion_tokens,
                                dur_ms / 1e3);
        }
      }
    }
  } else if (!op->buffered.empty()) {
    completion_tokens = raw_int_field(op->buffered, "completion_tokens");
    if (completion_tokens < 0)
      completion_tokens = raw_int_field(op->buffered, "output_tokens");
    if (completion_tokens < 0 && status < 400) {
      // usage omitted by the backend: BPE-estimate from the response
      // content field
      std::string text;
      if (append_json_string_field(op->buffered, "content", &text,
                                   256 * 1024) &&
          !text.empty()) {
        const BpeTokenizer* bpe = BpeTokenizer::default_instance();
        completion_tokens = bpe ? bpe->count(text)
                                : BpeTokenizer::heuristic_count(text);
      }
    }
    completion_tokens = std::max(0LL, completion_tokens);
    prompt_tokens = raw_int_field(op->buffered, "prompt_tokens");
    if (prompt_tokens < 0)
      prompt_tokens = raw_int_field(op->buffered, "input_tokens");
    prompt_tokens = std::max(0LL, prompt_tokens);
  }
  if (!success && op->lease.active()) op->lease.complete_failure();

  RequestRecord rec;
  rec.ts_ms = wall_ms();
  rec.endpoint_id = op->is_cloud ? ("cloud:" + op->cloud_provider)
                                 : std::move(op->endpoint_id);
  rec.model = std::move(op->public_model);
  rec.api_kind = api_kind_name(op->kind);
  rec.status = status;
  rec.duration_ms = dur_ms;
  rec.prompt_tokens = prompt_tokens;
  rec.completion_tokens = completion_tokens;
  rec.client_ip = std::move(op->client_ip);
  rec.request_id = std::move(op->request_id);
  rec.error = error;
  if (cfg.capture_bodies) {
    rec.request_body = std::move(op->captured_request);
    if (!op->streaming)
      rec.response_body = op->buffered.substr(0, 16384);
  }
  // TpsUpdated event, throttled to ~1/s (events/mod.rs TpsUpdated)
  if (success && completion_tokens > 0) {
    static std::atomic<long long> last_emit{0};
    long long now = rec.ts_ms, prev = last_emit.load();
    if (now - prev > 1000 && last_emit.compare_exchange_weak(prev, now)) {
      events.emit("TpsUpdated",
                  "{\"endpoint_id\":\"" + rec.endpoint_id +
                  "\",\"model\":\"" + rec.model + "\"}");
    }
  }
  c->w->pending_records.push_back(std::move(rec));
  if (c->w->pending_records.size() >= 4096)
    history.record_batch(c->w->pending_records);

  bool was_streaming = op->streaming;
  op.reset();
  c->response_open = false;
  if (was_streaming && status / 100 == 2) {
    metrics.responses_2xx.fetch_add(1);
  }
  if (c->closed) return;
  c->finish_request_cycle();
  if (!c->closed) c->process_requests();
}

}  // namespace gw
