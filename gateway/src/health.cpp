// Pull-based health checking, endpoint type detection and model sync —
// the reference's health/endpoint_checker.rs, detection/ and sync/
// subsystems, as one background thread with a blocking
// mini HTTP client (off the hot path).
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include "gateway.h"

namespace gw {

// telemetry.cpp
void telemetry_poll_once();
bool telemetry_get(int device_index, GpuInfo* out);

// ---------------------------------------------------------------- client
MiniHttpResult mini_http_request(const std::string& host, int port,
                                 const std::string& method,
                                 const std::string& path,
                                 const std::string& body, int timeout_ms,
                                 const std::vector<std::string>& headers) {
  MiniHttpResult res;
  int64_t t0 = now_us();
  int fd = start_connect(host, port);
  if (fd < 0) {
    res.error = "connect failed";
    return res;
  }
  auto wait_fd = [&](short ev) -> bool {
    pollfd p{fd, ev, 0};
    int64_t left = timeout_ms - (now_us() - t0) / 1000;
    if (left <= 0) return false;
    return poll(&p, 1, (int)left) > 0;
  };
  // wait for connect
  if (!wait_fd(POLLOUT)) {
    close(fd);
    res.error = "connect timeout";
    return res;
  }
  int err = 0;
  socklen_t elen = sizeof err;
  getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &elen);
  if (err != 0) {
    close(fd);
    res.error = "connect error";
    return res;
  }
  std::string req = method + " " + path + " HTTP/1.1\r\nHost: " + host +
                    "\r\nAccept: */*\r\n";
  for (auto& h : headers) req += h + "\r\n";
  if (!body.empty()) {
    req += "Content-Type: application/json\r\nContent-Length: " +
           std::to_string(body.size()) + "\r\n";
  }
  req += "Connection: close\r\n\r\n" + body;
  size_t off = 0;
  while (off < req.size()) {
    if (!wait_fd(POLLOUT)) {
      close(fd);
      res.error = "send timeout";
      return res;
    }
    ssize_t n = send(fd, req.data() + off, req.size() - off, MSG_NOSIGNAL);
    if (n <= 0) {
      if (errno == EAGAIN) continue;
      close(fd);
      res.error = "send failed";
      return res;
    }
    off += (size_t)n;
  }
  std::string raw;
  char buf[16384];
  HttpResponse resp;
  bool have_head = false;
  BodyReader bodyr;
  size_t body_start = 0;
  bool eof = false;
  while (true) {
    if (have_head && !bodyr.eof_terminated()) {
      // completion detectable without EOF (servers may keep the
      // connection open even though we sent Connection: close)
      BodyReader probe = bodyr;
      auto out = probe.feed(std::string_view(raw).substr(body_start));
      if (out.done) break;
    }
    if (!wait_fd(POLLIN)) {
      close(fd);
      if (!have_head) {
        res.error = "recv timeout";
        return res;
      }
      break;
    }
    ssize_t n = recv(fd, buf, sizeof buf, 0);
    if (n > 0) {
      raw.append(buf, (size_t)n);
      if (raw.size() > 16 * 1024 * 1024) break;
      if (!have_head) {
        int rc = parse_response_head(raw, resp);
        if (rc < 0) {
          close(fd);
          res.error = "bad response";
          return res;
        }
        if (rc == 1) {
          have_head = true;
          bodyr.init(resp);
          body_start = resp.header_len;
        }
      }
    } else if (n == 0) {
      eof = true;
      break;
    } else if (errno != EAGAIN) {
      break;
    }
  }
  close(fd);
  (void)eof;
  // re-parse now that `raw` is stable (views taken during the read
  // loop dangle whenever the string reallocated)
  resp = HttpResponse{};
  if (parse_response_head(raw, resp) != 1) {
    res.error = "bad response";
    return res;
  }
  res.status = resp.status;
  for (auto& h : resp.headers)
    res.headers.emplace_back(std::string(h.name), std::string(h.value));
  std::string_view rest = std::string_view(raw).substr(resp.header_len);
  if (resp.chunked) {
    ChunkedScanner cs;
    auto out = cs.scan(rest);
    for (auto& [o2, l] : out.payload) res.body.append(rest.substr(o2, l));
  } else {
    res.body = std::string(rest);
  }
  res.elapsed_ms = (now_us() - t0) / 1e3;
  return res;
}

// ---------------------------------------------------------------- detect
// Probe chain (reference detection/mod.rs:85-195 priority order):
// xllm -> lm_studio -> ollama -> vllm -> llama.cpp -> openai_compat
std::string Gateway::detect_endpoint_type(const std::string& host, int port,
                                          const std::string& prefix) {
  int t = cfg.health_timeout_ms;
  MiniHttpResult r = mini_http_request(host, port, "GET",
                                       prefix + "/api/system", "", t);
  if (r.status == 200 && r.body.find("xllm_version") != std::string::npos)
    return "xllm";
  r = mini_http_request(host, port, "GET", prefix + "/api/v1/models", "", t);
  if (r.status == 200 &&
      to_lower(r.header("server")).find("lm studio") != std::string::npos)
    return "lm_studio";
  r = mini_http_request(host, port, "GET", prefix + "/api/tags", "", t);
  if (r.status == 200 && r.body.find("\"models\"") != std::string::npos)
    return "ollama";
  r = mini_http_request(host, port, "GET", prefix + "/v1/models", "", t);
  if (r.status == 200) {
    std::string server = to_lower(r.header("server"));
    if (server.find("vllm") != std::string::npos) return "vllm";
    MiniHttpResult v = mini_http_request(host, port, "GET",
                                         prefix + "/v1/version", "", t);
    if (v.status == 200 &&
        v.body.find("llama.cpp") != std::string::npos)
      return "llama_cpp";
    if (r.body.find("\"data\"") != std::string::npos)
      return "openai_compat";
  }
  return "";  // unreachable / unsupported
}

// ---------------------------------------------------------------- mapping
// Static canonical (HF repo id) <-> engine alias table — the analog of
// the reference's BUILTIN_MAPPINGS (models/mapping.rs:44): any alias a
// backend reports resolves to the canonical id so every spelling routes.
struct AliasRow {
  const char* canonical;
  const char* alias;
};
static const AliasRow kBuiltinMappings[] = {
    {"meta-llama/Meta-Llama-3-8B-Instruct", "llama3:8b"},
    {"meta-llama/Meta-Llama-3-8B-Instruct", "llama-3-8b"},
    {"meta-llama/Meta-Llama-3-8B-Instruct", "Meta-Llama-3-8B-Instruct"},
    {"meta-llama/Meta-Llama-3-70B-Instruct", "llama3:70b"},
    {"meta-llama/Meta-Llama-3-70B-Instruct", "llama-3-70b"},
    {"openai/gpt-oss-20b", "gpt-oss:20b"},
    {"openai/whisper-large-v3", "whisper-large-v3"},
    {"openai/whisper-large-v3", "whisper-large-v3-synth"},
    {"BAAI/bge-large-en-v1.5", "bge-large"},
    {"mistralai/Mistral-7B-Instruct-v0.3", "mistral:7b"},
    {"Qwen/Qwen2.5-7B-Instruct", "qwen2.5:7b"},
};

std::string resolve_canonical_alias(const std::string& model_id) {
  for (const auto& row : kBuiltinMappings)
    if (model_id == row.alias) return row.canonical;
  return "";
}

// ---------------------------------------------------------------- sync
// GET /v1/models (OpenAI or Ollama shape) -> model list diff
// (reference sync/mod.rs:104, sync/parser.rs)
bool Gateway::sync_endpoint_models(const EndpointPtr& ep, std::string* err) {
  std::vector<std::string> hdrs;
  if (!ep->api_key.empty())
    hdrs.push_back("Authorization: Bearer " + ep->api_key);
  MiniHttpResult r;
  if (ep->type == "ollama") {
    r = mini_http_request(ep->host, ep->port, "GET",
                          ep->path_prefix + "/api/tags", "",
                          cfg.health_timeout_ms, hdrs);
  } else {
    r = mini_http_request(ep->host, ep->port, "GET",
                          ep->path_prefix + "/v1/models", "",
                          cfg.health_timeout_ms, hdrs);
  }
  if (r.status != 200) {
    if (err) *err = r.error.empty()
                        ? "status " + std::to_string(r.status)
                        : r.error;
    return false;
  }
  Json doc;
  if (!Json::parse(r.body, doc)) {
    if (err) *err = "invalid JSON from /v1/models";
    return false;
  }
  std::vector<EndpointModel> models;
  const Json* arr = doc.find("data");
  if (!arr || !arr->is_arr()) arr = doc.find("models");  // ollama shape
  if (arr && arr->is_arr()) {
    for (auto& m : arr->as_arr()) {
      EndpointModel em;
      em.id = m["id"].as_str();
      if (em.id.empty()) em.id = m["name"].as_str();  // ollama
      if (em.id.empty()) continue;
      // capability inference from the model name
      // (reference sync/capabilities.rs)
      std::string lower = to_lower(em.id);
      uint32_t caps = kCapChat | kCapCompletions | kCapResponses;
      if (lower.find("embed") != std::string::npos)
        caps = kCapEmbeddings;
      if (lower.find("whisper") != std::string::npos ||
          lower.find("asr") != std::string::npos)
        caps = kCapAudioTranscription;
      if (lower.find("tts") != std::string::npos ||
          lower.find("speech") != std::string::npos ||
          lower.find("vibevoice") != std::string::npos)
        caps = kCapAudioSpeech;
      if (lower.find("sd") == 0 ||
          lower.find("stable-diffusion") != std::string::npos ||
          lower.find("image") != std::string::npos)
        caps = kCapImageGen;
      // a backend that declares its own capabilities wins over the
      // name heuristic (our MI355X backend self-reports)
      if (const Json* cj = m.find("capabilities");
          cj && cj->is_str() && !cj->as_str().empty())
        caps = caps_from_csv(cj->as_str());
      em.caps = caps;
      const Json* mt = m.find("max_tokens");
      if (mt) em.max_tokens = mt->as_int(0);
      if (const Json* can = m.find("canonical_id"); can)
        em.canonical = can->as_str();
      if (em.canonical.empty())
        em.canonical = resolve_canonical_alias(em.id);
      if (em.canonical.empty()) em.canonical = em.id;
      models.push_back(std::move(em));
    }
  }
  registry.sync_models(ep->id, models);
  return true;
}

// ---------------------------------------------------------------- loop
// fetch {base}/models from each configured cloud provider and cache
// the ids prefixed for routing ("openai:gpt-4o" etc.); 10-min TTL like
// the reference's cloud model cache (api/cloud_models.rs)
void Gateway::refresh_cloud_models() {
  struct Prov { const char* name; const std::string* base;
                const std::string* key; };
  const Prov provs[] = {{"openai", &cfg.openai_base, &cfg.openai_key},
                        {"google", &cfg.google_base, &cfg.google_key},
                        {"anthropic", &cfg.anthropic_base,
                         &cfg.anthropic_key}};
  std::vector<CloudModel> fresh;
  for (auto& pv : provs) {
    if (pv.base->empty()) continue;
    // parse http://host:port/prefix
    std::string b = *pv.base;
    if (starts_with(b, "http://")) b = b.substr(7);
    size_t slash = b.find('/');
    std::string hostport = b.substr(0, slash);
    std::string prefix = slash == std::string::npos ? "" : b.substr(slash);
    size_t colon = hostport.find(':');
    std::string host = hostport.substr(0, colon);
    int port = colon == std::string::npos
                   ? 80
                   : atoi(hostport.c_str() + colon + 1);
    std::vector<std::string> hdrs;
    if (!pv.key->empty())
      hdrs.push_back("Authorization: Bearer " + *pv.key);
    MiniHttpResult r = mini_http_request(host, port, "GET",
                                         prefix + "/models", "",
                                         cfg.health_timeout_ms, hdrs);
    if (r.status != 200) continue;
    Json v;
    if (!Json::parse(r.body, v) || !v["data"].is_arr()) continue;
    for (auto& m : v["data"].as_arr()) {
      std::string id = m["id"].as_str();
      if (!id.empty())
        fresh.push_back({std::string(pv.name) + ":" + id, pv.name});
    }
  }
  std::lock_guard<std::mutex> lk(cloud_models_mu);
  cloud_models_cache = std::move(fresh);
  cloud_models_fetched_ms = wall_ms();
}

void Gateway::health_loop() {
  pthread_setname_np(pthread_self(), "gw-health");
  std::map<std::string, int64_t> last_sync_ms;
  int64_t last_telemetry_us = 0;
  int64_t last_cloud_ms = 0;
  while (!stopping.load()) {
    int64_t loop_start = now_us();
    if ((!cfg.openai_base.empty() || !cfg.google_base.empty() ||
         !cfg.anthropic_base.empty()) &&
        wall_ms() - last_cloud_ms > 10 * 60 * 1000) {
      refresh_cloud_models();
      last_cloud_ms = wall_ms();
    }
    // amdsmi poll (MI355X-native GPU telemetry -> admission control)
    if (now_us() - last_telemetry_us >
        (int64_t)(cfg.amdsmi_poll_interval_s * 1e6)) {
      telemetry_poll_once();
      last_telemetry_us = now_us();
    }
    for (auto& ep : registry.list()) {
      if (stopping.load()) return;
      // per-endpoint interval honoring (default 30 s)
      MiniHttpResult r = mini_http_request(
          ep->host, ep->port, "GET", ep->path_prefix + "/v1/models", "",
          cfg.health_timeout_ms);
      bool ok = r.status == 200;
      EndpointStatus cur = ep->status.load();
      if (ok) {
        ep->consecutive_failures.store(0);
        if (cur != EndpointStatus::kOnline) {
          // offline -> online: re-detect type + re-sync models
          // (endpoint_checker.rs:333-377)
          std::string dtype = detect_endpoint_type(ep->host, ep->port,
                                                   ep->path_prefix);
          if (!dtype.empty())
            registry.update_endpoint(ep->id, [&](Endpoint& e) {
              e.type = dtype;
            });
          std::string serr;
          sync_endpoint_models(ep, &serr);
          registry.set_status(ep->id, EndpointStatus::kOnline);
          events.emit("EndpointStatusChanged",
                      "{\"endpoint_id\":\"" + ep->id +
                      "\",\"status\":\"online\"}");
          LOG_INFO("endpoint %s (%s) online", ep->name.c_str(),
                   ep->base_url.c_str());
        } else {
          // throttled auto model sync (15 min default,
          // endpoint_checker.rs:426-512)
          int64_t now = wall_ms();
          int64_t& last = last_sync_ms[ep->id];
          if (now - last > (int64_t)cfg.auto_sync_interval_s * 1000) {
            last = now;
            std::string serr;
            sync_endpoint_models(ep, &serr);
          }
        }
        // attach local GPU telemetry when the endpoint is pinned to a
        // device (BASELINE: HBM occupancy straight from amdsmi)
        if (ep->device_index >= 0) {
          GpuInfo g;
          if (telemetry_get(ep->device_index, &g))
            registry.update_gpu_info(ep->id, g);
        }
        // backend-measured tokens/sec (hipEvent-fed) seeds the TPS
        // tracker until production samples exist — mirrors the
        // reference's startup TPS seeding (bootstrap.rs:143-159)
        {
          MiniHttpResult h = mini_http_request(
              ep->host, ep->port, "GET", ep->path_prefix + "/api/health",
              "", cfg.health_timeout_ms);
          if (h.status == 200) {
            Json doc;
            if (Json::parse(h.body, doc)) {
              double tps = doc["tokens_per_sec"].as_num(0);
              std::string hm = doc["model"].as_str();
              if (tps > 0 && !hm.empty()) {
                auto st = registry.tps_state(ep->id, hm, ApiKind::kChat);
                std::lock_guard<std::mutex> lk(st->mu);
                if (st->samples == 0) {   // never override production
                  st->ema = tps;
                  st->samples = 1;
                  st->updated_ms = wall_ms();
                }
              }
              // xLLM-style GPU fields from backends without a local
              // device binding (endpoint_checker.rs:514-557)
              if (ep->device_index < 0 &&
                  doc.find("gpu_memory_total_mb")) {
                GpuInfo g;
                g.hbm_total_gb =
                    doc["gpu_memory_total_mb"].as_num(0) / 1024.0;
                g.hbm_used_gb =
                    doc["gpu_memory_used_mb"].as_num(0) / 1024.0;
                g.device_index = -1;
                if (g.hbm_total_gb > 0)
                  registry.update_gpu_info(ep->id, g);
              }
            }
          }
        }
      } else {
        int fails = ep->consecutive_failures.fetch_add(1) + 1;
        if (cur == EndpointStatus::kPending) {
          registry.set_status(ep->id, EndpointStatus::kOffline);
        } else if (cur == EndpointStatus::kOnline) {
          registry.set_status(ep->id, EndpointStatus::kError);
          LOG_WARN("endpoint %s failed health check (1/%d)",
                   ep->name.c_str(), cfg.consecutive_failures_for_offline);
        } else if (cur == EndpointStatus::kError &&
                   fails >= cfg.consecutive_failures_for_offline) {
          registry.set_status(ep->id, EndpointStatus::kOffline);
          registry.clear_tps_for_endpoint(ep->id);  // eject + TPS reset
          events.emit("EndpointStatusChanged",
                      "{\"endpoint_id\":\"" + ep->id +
                      "\",\"status\":\"offline\"}");
          LOG_WARN("endpoint %s ejected (offline)", ep->name.c_str());
        }
      }
      // record health-check history
      if (db.ok()) {
        std::lock_guard<std::mutex> lk(db.mutex());
        db.exec("INSERT INTO health_checks (endpoint_id,ts_ms,ok,"
                "latency_ms,detail) VALUES (?,?,?,?,?)",
                {ep->id, (long long)wall_ms(), ok, r.elapsed_ms,
                 ok ? "" : r.error});
      }
    }
    // sleep until next tick (interval from config; responsive to stop)
    int64_t elapsed_ms = (now_us() - loop_start) / 1000;
    int64_t sleep_ms = (int64_t)cfg.health_interval_s * 1000 - elapsed_ms;
    // first pass after boot runs quickly to bring endpoints online
    static thread_local bool first = true;
    if (first) {
      first = false;
      sleep_ms = std::min<int64_t>(sleep_ms, 1000);
    }
    while (sleep_ms > 0 && !stopping.load()) {
      int64_t step = std::min<int64_t>(sleep_ms, 200);
      usleep((useconds_t)(step * 1000));
      sleep_ms -= step;
    }
  }
}

}  // namespace gw
