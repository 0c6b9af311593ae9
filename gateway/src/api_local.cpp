// Local (non-proxied) API handlers: /v1/models, endpoint admin CRUD,
// api-key management, dashboard summary/stats, request history, GPU
// telemetry view, audio/images capability routing, drain control.
#include "gateway.h"
#include "dashboard_html.h"

namespace gw {

void start_proxy_direct(Gateway* gw, ClientConn* c, EndpointPtr ep,
                        ApiKind kind, const std::string& upstream_path,
                        std::string_view content_type);

// telemetry.cpp
void telemetry_snapshot_json(std::string* out);

static Json endpoint_json(const EndpointPtr& ep,
                          Registry* reg = nullptr) {
  JsonObject o;
  o.emplace_back("id", Json(ep->id));
  o.emplace_back("name", Json(ep->name));
  o.emplace_back("base_url", Json(ep->base_url));
  o.emplace_back("endpoint_type", Json(ep->type));
  o.emplace_back("status", Json(status_name(ep->status.load())));
  o.emplace_back("device_index", Json((long long)ep->device_index));
  o.emplace_back("active_requests", Json((long long)ep->active.load()));
  o.emplace_back("total_requests",
                 Json((long long)ep->total_requests.load()));
  o.emplace_back("total_failures",
                 Json((long long)ep->total_failures.load()));
  o.emplace_back("latency_ema_ms", Json(ep->latency_ema_ms.load()));
  o.emplace_back("health_check_interval_secs",
                 Json((long long)ep->health_interval_s));
  o.emplace_back("inference_timeout_secs",
                 Json((long long)ep->inference_timeout_s));
  o.emplace_back("created_at", Json(ep->created_at));
  o.emplace_back("updated_at", Json(ep->updated_at));
  {
    std::lock_guard<std::mutex> lk(ep->gpu_mu);
    if (ep->gpu.updated_ms > 0) {
      JsonObject g;
      g.emplace_back("hbm_used_gb", Json(ep->gpu.hbm_used_gb));
      g.emplace_back("hbm_total_gb", Json(ep->gpu.hbm_total_gb));
      g.emplace_back("busy_pct", Json(ep->gpu.busy_pct));
      g.emplace_back("device_index",
                     Json((long long)ep->gpu.device_index));
      g.emplace_back("updated_ms", Json((long long)ep->gpu.updated_ms));
      o.emplace_back("gpu", Json(std::move(g)));
    }
  }
  JsonArray models;
  for (auto& m : ep->models) {
    JsonObject mo;
    mo.emplace_back("id", Json(m.id));
    mo.emplace_back("canonical_id", Json(m.canonical));
    mo.emplace_back("capabilities", Json(caps_to_csv(m.caps)));
    if (m.max_tokens) mo.emplace_back("max_tokens", Json(m.max_tokens));
    if (reg) {
      auto st = reg->tps_state(ep->id, m.canonical, ApiKind::kChat);
      std::lock_guard<std::mutex> lk(st->mu);
      if (st->samples > 0)
        mo.emplace_back("tokens_per_sec", Json(st->ema));
    }
    models.emplace_back(Json(std::move(mo)));
  }
  o.emplace_back("models", Json(std::move(models)));
  return Json(std::move(o));
}

std::string Gateway::render_models_json() {
  auto table = registry.table();
  JsonArray data;
  for (auto& key : table->model_keys_canonical) {
    auto it = table->by_model.find(key);
    if (it == table->by_model.end()) continue;
    const ModelRoute& route = it->second;
    JsonObject m;
    m.emplace_back("id", Json(key));
    m.emplace_back("object", Json("model"));
    m.emplace_back("created", Json((long long)(wall_ms() / 1000)));
    m.emplace_back("owned_by", Json("llmlb"));
    JsonArray eps;
    for (auto& ep : route.endpoints) eps.emplace_back(Json(ep->id));
    m.emplace_back("endpoint_ids", Json(std::move(eps)));
    m.emplace_back("capabilities", Json(caps_to_csv(route.caps)));
    if (route.max_tokens) m.emplace_back("max_tokens",
                                        Json(route.max_tokens));
    data.emplace_back(Json(std::move(m)));
  }
  {
    // cloud-provider models (cached; reference openai.rs:451-467) —
    // routable via their provider prefix, always "ready"
    std::lock_guard<std::mutex> lk(cloud_models_mu);
    for (auto& cm : cloud_models_cache) {
      JsonObject m;
      m.emplace_back("id", Json(cm.id));
      m.emplace_back("object", Json("model"));
      m.emplace_back("created", Json((long long)(wall_ms() / 1000)));
      m.emplace_back("owned_by", Json(cm.provider));
      m.emplace_back("endpoint_ids", Json(JsonArray{}));
      m.emplace_back("capabilities", Json("chat"));
      m.emplace_back("ready", Json(true));
      data.emplace_back(Json(std::move(m)));
    }
  }
  JsonObject o;
  o.emplace_back("object", Json("list"));
  o.emplace_back("data", Json(std::move(data)));
  return Json(std::move(o)).dump();
}

std::string Gateway::render_summary_json() {
  // SystemSummary analog (balancer/mod.rs:2470-2588)
  auto eps = registry.list();
  JsonArray earr;
  long long active = 0;
  for (auto& ep : eps) {
    active += ep->active.load();
    earr.emplace_back(endpoint_json(ep, &registry));
  }
  auto sum60 = history.ring().summarize(60 * 60 * 1000);
  JsonObject o;
  o.emplace_back("endpoints", Json(std::move(earr)));
  o.emplace_back("active_requests", Json(active));
  o.emplace_back("uptime_ms", Json((long long)(wall_ms() - start_ms)));
  JsonObject h;
  h.emplace_back("window_minutes", Json(60));
  h.emplace_back("requests", Json(sum60.requests));
  h.emplace_back("failures", Json(sum60.failures));
  h.emplace_back("prompt_tokens", Json(sum60.prompt_tokens));
  h.emplace_back("completion_tokens", Json(sum60.completion_tokens));
  h.emplace_back("avg_duration_ms", Json(sum60.avg_duration_ms));
  o.emplace_back("last_hour", Json(std::move(h)));
  return Json(std::move(o)).dump();
}

std::string Gateway::render_metrics_text() {
  char buf[2048];
  int n = snprintf(
      buf, sizeof buf,
      "# HELP llmlb_requests_total Total HTTP requests seen\n"
      "# TYPE llmlb_requests_total counter\n"
      "llmlb_requests_total %lld\n"
      "# TYPE llmlb_responses_total counter\n"
      "llmlb_responses_total{class=\"2xx\"} %lld\n"
      "llmlb_responses_total{class=\"4xx\"} %lld\n"
      "llmlb_responses_total{class=\"5xx\"} %lld\n"
      "# TYPE llmlb_upstream_errors_total counter\n"
      "llmlb_upstream_errors_total %lld\n"
      "# TYPE llmlb_streamed_responses_total counter\n"
      "llmlb_streamed_responses_total %lld\n"
      "# TYPE llmlb_tokens_streamed_total counter\n"
      "llmlb_tokens_streamed_total %lld\n"
      "# TYPE llmlb_cloud_requests_total counter\n"
      "llmlb_cloud_requests_total %lld\n"
      "# TYPE llmlb_active_requests gauge\n"
      "llmlb_active_requests %lld\n"
      "# TYPE llmlb_admission_delays_total counter\n"
      "llmlb_admission_delays_total %lld\n",
      metrics.requests_total.load(), metrics.responses_2xx.load(),
      metrics.responses_4xx.load(), metrics.responses_5xx.load(),
      metrics.upstream_errors.load(), metrics.streamed_responses.load(),
      metrics.tokens_streamed.load(), metrics.cloud_requests.load(),
      metrics.active_requests.load(), metrics.admission_delays.load());
  std::string out(buf, n);
  // per-endpoint gauges (ops dashboards scrape these to see the
  // balancer's view of each replica)
  out += "# TYPE llmlb_endpoint_active gauge\n"
         "# TYPE llmlb_endpoint_requests_total counter\n"
         "# TYPE llmlb_endpoint_failures_total counter\n"
         "# TYPE llmlb_endpoint_online gauge\n";
  for (auto& ep : registry.list()) {
    char line[512];
    int m = snprintf(
        line, sizeof line,
        "llmlb_endpoint_active{endpoint=\"%s\"} %lld\n"
        "llmlb_endpoint_requests_total{endpoint=\"%s\"} %lld\n"
        "llmlb_endpoint_failures_total{endpoint=\"%s\"} %lld\n"
        "llmlb_endpoint_online{endpoint=\"%s\"} %d\n",
        ep->name.c_str(), (long long)ep->active.load(),
        ep->name.c_str(), (long long)ep->total_requests.load(),
        ep->name.c_str(), (long long)ep->total_failures.load(),
        ep->name.c_str(),
        ep->status.load() == EndpointStatus::kOnline ? 1 : 0);
    out.append(line, m);
  }
  return out;
}

// capability-routed backend pick: FIRST online endpoint with the
// capability (reference api/audio.rs:160-190 — deliberately no TPS
// weighting on the audio/image paths)
// ModelStatEntry rows -> JSON (reference db/endpoint_daily_stats.rs:25-37):
// rows are (model, requests, failures, completion_tokens, duration_ms)
static Json model_stats_json(const std::vector<DbRow>& rows) {
  JsonArray arr;
  for (auto& row : rows) {
    long long req = row[1].i, fail = row[2].i;
    JsonObject o;
    o.emplace_back("model_id", Json(row[0].str()));
    o.emplace_back("total_requests", Json(req));
    o.emplace_back("successful_requests", Json(req - fail));
    o.emplace_back("failed_requests", Json(fail));
    o.emplace_back("total_output_tokens", Json((long long)row[3].i));
    o.emplace_back(
        "total_duration_ms",
        Json(row[4].kind == DbValue::kReal ? row[4].d : (double)row[4].i));
    arr.emplace_back(Json(std::move(o)));
  }
  return Json(std::move(arr));
}

static EndpointPtr pick_by_capability(Registry& reg, uint32_t cap) {
  for (auto& ep : reg.list()) {
    if (ep->status.load() != EndpointStatus::kOnline) continue;
    for (auto& m : ep->models)
      if (m.caps & cap) return ep;
  }
  return nullptr;
}

void Gateway::handle_local(ClientConn* c, const std::string& path,
                           std::string_view body) {
  const HttpRequest& r = c->req;
  std::string_view method = r.method;

  // ---- /v1/models -------------------------------------------------
  if (path == "/v1/models" && method == "GET") {
    c->send_simple(200, "application/json", render_models_json());
    return;
  }
  if (starts_with(path, "/v1/models/") && method == "GET") {
    std::string id = url_decode(path.substr(strlen("/v1/models/")));
    auto table = registry.table();
    auto it = table->by_model.find(id);
    if (it == table->by_model.end()) {
      c->send_json_error(404, "model_not_found", "no such model");
      return;
    }
    JsonObject m;
    m.emplace_back("id", Json(it->second.canonical));
    m.emplace_back("object", Json("model"));
    m.emplace_back("owned_by", Json("llmlb"));
    m.emplace_back("capabilities", Json(caps_to_csv(it->second.caps)));
    c->send_simple(200, "application/json", Json(std::move(m)).dump());
    return;
  }

  // ---- /v1/audio/* + /v1/images/* (capability-routed proxy) --------
  if (starts_with(path, "/v1/audio/") || starts_with(path, "/v1/images/")) {
    uint32_t cap;
    ApiKind kind;
    if (path == "/v1/audio/transcriptions") {
      cap = kCapAudioTranscription;
      kind = ApiKind::kAudio;
    } else if (path == "/v1/audio/speech") {
      cap = kCapAudioSpeech;
      kind = ApiKind::kAudio;
    } else if (starts_with(path, "/v1/images/")) {
      cap = kCapImageGen;
      kind = ApiKind::kImages;
    } else {
      c->send_json_error(404, "not_found", "unknown audio route");
      return;
    }
    EndpointPtr ep = pick_by_capability(registry, cap);
    if (!ep) {
      c->send_json_error(503, "no_capable_endpoint",
                         "no online endpoint supports this capability");
      return;
    }
    start_proxy_direct(this, c, ep, kind, path,
                       r.header("content-type"));
    return;
  }

  // ---- admin API: admin token, or a key holding the route's scope
  // (reference per-group middleware matrix, api/mod.rs:70-635) --------
  if (starts_with(path, "/api/")) {
    // route -> required scope (read vs manage split per method)
    const bool is_read = method == "GET" || method == "HEAD";
    const char* scope = "admin";
    if (starts_with(path, "/api/endpoints"))
      scope = is_read ? "endpoints.read" : "endpoints.manage";
    else if (starts_with(path, "/api/api-keys"))
      scope = "api_keys.manage";
    else if (starts_with(path, "/api/models/registry"))
      scope = "registry.read";
    else if (starts_with(path, "/api/models"))
      scope = is_read ? "registry.read" : "models.manage";
    else if (starts_with(path, "/api/dashboard") ||
             path == "/api/requests" || path == "/api/events" ||
             path == "/api/gpu")
      scope = "metrics.read";
    else if (starts_with(path, "/api/audit") ||
             starts_with(path, "/api/logs"))
      scope = "logs.read";
    std::string_view authz = r.header("authorization");
    bool admin_ok = false;
    if (!cfg.admin_token.empty()) {
      std::string_view key = authz;
      if (starts_with(key, "Bearer ")) key.remove_prefix(7);
      if (trim(key) == cfg.admin_token) admin_ok = true;
    }
    if (!admin_ok)
      admin_ok = auth.check_bearer(authz, scope,
                                   is_loopback_ip(c->peer_ip));
    if (!admin_ok) {
      c->send_json_error(401, "unauthorized", "admin credentials required");
      return;
    }

    // -- endpoints CRUD (api/endpoints.rs) --
    if (path == "/api/endpoints" && method == "GET") {
      JsonArray arr;
      for (auto& ep : registry.list())
        arr.emplace_back(endpoint_json(ep, &registry));
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (path == "/api/endpoints" && method == "POST") {
      Json in;
      if (!Json::parse(body, in) || !in.is_obj()) {
        c->send_json_error(400, "bad_json", "invalid JSON body");
        return;
      }
      std::string err;
      LOG_INFO("add_endpoint: start");
      EndpointPtr ep = registry.add_endpoint(
          in["name"].as_str(), in["base_url"].as_str(),
          in["endpoint_type"].as_str(), in["api_key"].as_str(), &err);
      if (!ep) {
        c->send_json_error(422, "invalid_endpoint", err);
        return;
      }
      if (in.find("device_index"))
        registry.update_endpoint(ep->id, [&](Endpoint& e) {
          e.device_index = (int)in["device_index"].as_int(-1);
        });
      // immediate async probe: health loop will pick it up on its next
      // tick; optionally detect type + sync now (blocking, admin path)
      LOG_INFO("add_endpoint: registered, detecting type");
      std::string dtype = detect_endpoint_type(ep->host, ep->port,
                                               ep->path_prefix);
      LOG_INFO("add_endpoint: detected '%s', syncing", dtype.c_str());
      if (!dtype.empty() && ep->type == "openai_compat")
        registry.update_endpoint(ep->id, [&](Endpoint& e) {
          e.type = dtype;
        });
      std::string serr;
      if (sync_endpoint_models(ep, &serr)) {
        registry.set_status(ep->id, EndpointStatus::kOnline);
      }
      LOG_INFO("add_endpoint: done (sync err='%s')", serr.c_str());
      c->send_simple(201, "application/json", endpoint_json(ep).dump());
      return;
    }
    if (starts_with(path, "/api/endpoints/")) {
      std::string rest = path.substr(strlen("/api/endpoints/"));
      size_t slash = rest.find('/');
      std::string id = url_decode(rest.substr(0, slash));
      std::string action = slash == std::string::npos
                               ? ""
                               : rest.substr(slash + 1);
      EndpointPtr ep = registry.get(id);
      if (!ep) {
        c->send_json_error(404, "not_found", "no such endpoint");
        return;
      }
      if (action.empty() && method == "GET") {
        c->send_simple(200, "application/json",
                       endpoint_json(ep, &registry).dump());
        return;
      }
      if (action.empty() && method == "DELETE") {
        registry.remove_endpoint(id);
        c->send_simple(200, "application/json", "{\"deleted\":true}");
        return;
      }
      if (action.empty() && (method == "PUT" || method == "PATCH")) {
        Json in;
        if (!Json::parse(body, in)) {
          c->send_json_error(400, "bad_json", "invalid JSON");
          return;
        }
        registry.update_endpoint(id, [&](Endpoint& e) {
          if (in.find("name")) e.name = in["name"].as_str();
          if (in.find("api_key")) e.api_key = in["api_key"].as_str();
          if (in.find("endpoint_type"))
            e.type = in["endpoint_type"].as_str();
          if (in.find("device_index"))
            e.device_index = (int)in["device_index"].as_int(-1);
          if (in.find("inference_timeout_secs"))
            e.inference_timeout_s =
                (int)in["inference_timeout_secs"].as_int(120);
        });
        c->send_simple(200, "application/json",
                       endpoint_json(registry.get(id)).dump());
        return;
      }
      if (action == "sync" && method == "POST") {
        std::string err;
        bool ok = sync_endpoint_models(ep, &err);
        if (ok) registry.set_status(id, EndpointStatus::kOnline);
        std::string bodyj = ok ? "{\"synced\":true}"
                               : "{\"synced\":false,\"error\":\"" + err +
                                     "\"}";
        c->send_simple(ok ? 200 : 502, "application/json", bodyj);
        return;
      }
      // -- per-endpoint request statistics (reference
      //    api/dashboard.rs:919-976, db/endpoint_daily_stats.rs) ----
      if (action == "today" && method == "GET") {
        char date[16];
        time_t t = time(nullptr);
        struct tm tmv;
        gmtime_r(&t, &tmv);
        strftime(date, sizeof date, "%Y-%m-%d", &tmv);
        std::vector<DbRow> rows;
        db.exec("SELECT COALESCE(SUM(requests),0),"
                "COALESCE(SUM(failures),0) FROM endpoint_daily_stats "
                "WHERE endpoint_id=? AND date=?",
                {id, date}, &rows);
        long long req = rows.empty() ? 0 : rows[0][0].i;
        long long fail = rows.empty() ? 0 : rows[0][1].i;
        JsonObject o;
        o.emplace_back("date", Json(std::string(date)));
        o.emplace_back("total_requests", Json(req));
        o.emplace_back("successful_requests", Json(req - fail));
        o.emplace_back("failed_requests", Json(fail));
        c->send_simple(200, "application/json",
                       Json(std::move(o)).dump());
        return;
      }
      if (action == "daily-stats" && method == "GET") {
        long long days = 7;
        std::string q(r.query);
        size_t p = q.find("days=");
        if (p != std::string::npos) days = atoll(q.c_str() + p + 5);
        if (days < 1) days = 1;
        if (days > 365) days = 365;
        std::vector<DbRow> rows;
        db.exec("SELECT date, SUM(requests), SUM(failures) FROM "
                "endpoint_daily_stats WHERE endpoint_id=? GROUP BY "
                "date ORDER BY date DESC LIMIT ?",
                {id, days}, &rows);
        JsonArray arr;
        for (auto& row : rows) {
          long long req = row[1].i, fail = row[2].i;
          JsonObject o;
          o.emplace_back("date", Json(row[0].str()));
          o.emplace_back("total_requests", Json(req));
          o.emplace_back("successful_requests", Json(req - fail));
          o.emplace_back("failed_requests", Json(fail));
          arr.emplace_back(Json(std::move(o)));
        }
        c->send_simple(200, "application/json",
                       Json(std::move(arr)).dump());
        return;
      }
      if (action == "model-stats" && method == "GET") {
        std::vector<DbRow> rows;
        db.exec("SELECT model, SUM(requests), SUM(failures), "
                "SUM(completion_tokens), SUM(total_duration_ms) FROM "
                "endpoint_daily_stats WHERE endpoint_id=? GROUP BY "
                "model ORDER BY SUM(requests) DESC",
                {id}, &rows);
        c->send_simple(200, "application/json",
                       model_stats_json(rows).dump());
        return;
      }
      if (action == "model-tps" && method == "GET") {
        // live per-model TPS EMAs for this endpoint enriched with
        // persisted totals (reference api/dashboard.rs:1206-1225;
        // source is Production — benchmark-sourced TPS lives under
        // /api/benchmarks/tps runs)
        auto list = registry.model_tps_list(id);
        JsonArray arr;
        for (auto& e : list) {
          std::vector<DbRow> rows;
          db.exec("SELECT COALESCE(SUM(completion_tokens),0),"
                  "COALESCE(SUM(total_duration_ms),0),"
                  "COALESCE(SUM(requests),0) FROM endpoint_daily_stats "
                  "WHERE endpoint_id=? AND model=?",
                  {id, e.model}, &rows);
          long long toks = rows.empty() ? 0 : rows[0][0].i;
          double dur = rows.empty()
                           ? 0.0
                           : (rows[0][1].kind == DbValue::kReal
                                  ? rows[0][1].d
                                  : (double)rows[0][1].i);
          long long reqs = rows.empty() ? 0 : rows[0][2].i;
          JsonObject o;
          o.emplace_back("model_id", Json(e.model));
          o.emplace_back("api_kind", Json(e.kind));
          o.emplace_back("source", Json(std::string("Production")));
          o.emplace_back("tps", Json(e.tps));
          o.emplace_back("request_count", Json(e.samples));
          o.emplace_back("total_output_tokens", Json(toks));
          o.emplace_back("average_duration_ms",
                         Json(reqs > 0 ? dur / reqs : 0.0));
          arr.emplace_back(Json(std::move(o)));
        }
        c->send_simple(200, "application/json",
                       Json(std::move(arr)).dump());
        return;
      }
      if (action == "chat" && method == "POST") {
        // playground: chat through THIS endpoint, bypassing selection
        // (reference api/endpoints.rs:1079 proxy_chat_completions)
        start_proxy_direct(this, c, ep, ApiKind::kChat,
                           "/v1/chat/completions",
                           r.header("content-type").empty()
                               ? std::string_view("application/json")
                               : r.header("content-type"));
        return;
      }
      // -- per-engine model adapters (reference metadata/*.rs,
      //    download/mod.rs, delete: types/endpoint.rs:164-166) -------
      if (starts_with(action, "models/") &&
          ends_with(action, "/metadata") && method == "GET") {
        std::string model = url_decode(action.substr(
            strlen("models/"),
            action.size() - strlen("models/") - strlen("/metadata")));
        MiniHttpResult res;
        if (ep->type == "ollama") {
          res = mini_http_request(ep->host, ep->port, "POST",
                                  ep->path_prefix + "/api/show",
                                  "{\"name\":\"" + model + "\"}",
                                  cfg.health_timeout_ms);
        } else if (ep->type == "lm_studio") {
          res = mini_http_request(ep->host, ep->port, "GET",
                                  ep->path_prefix + "/api/v1/models",
                                  "", cfg.health_timeout_ms);
        } else if (ep->type == "xllm") {
          res = mini_http_request(
              ep->host, ep->port, "GET",
              ep->path_prefix + "/api/models/" + model + "/metadata",
              "", cfg.health_timeout_ms);
        } else {
          c->send_json_error(501, "unsupported_engine",
                             "metadata adapter: ollama, lm_studio and "
                             "xllm endpoints only");
          return;
        }
        if (res.status != 200) {
          c->send_json_error(res.status == 404 ? 404 : 502,
                             "metadata_unavailable",
                             res.error.empty() ? res.body.substr(0, 200)
                                               : res.error);
          return;
        }
        c->send_simple(200, "application/json", res.body);
        return;
      }
      if (action == "models/download" && method == "POST") {
        Json in;
        Json::parse(body, in);
        std::string model = in["model"].as_str();
        if (model.empty()) model = in["name"].as_str();
        if (model.empty()) {
          c->send_json_error(400, "missing_model",
                             "body must name a model");
          return;
        }
        MiniHttpResult res;
        if (ep->type == "ollama") {
          res = mini_http_request(
              ep->host, ep->port, "POST", ep->path_prefix + "/api/pull",
              "{\"name\":\"" + model + "\",\"stream\":false}",
              ep->inference_timeout_s * 1000);
        } else if (ep->type == "xllm") {
          res = mini_http_request(
              ep->host, ep->port, "POST",
              ep->path_prefix + "/api/models/download",
              "{\"model\":\"" + model + "\"}",
              ep->inference_timeout_s * 1000);
        } else if (ep->type == "lm_studio") {
          res = mini_http_request(
              ep->host, ep->port, "POST",
              ep->path_prefix + "/api/v1/models/download",
              "{\"model\":\"" + model + "\"}",
              ep->inference_timeout_s * 1000);
        } else {
          c->send_json_error(501, "unsupported_engine",
                             "download adapter: ollama, lm_studio and "
                             "xllm endpoints only");
          return;
        }
        if (res.status != 200) {
          c->send_json_error(502, "download_failed",
                             res.error.empty() ? res.body.substr(0, 200)
                                               : res.error);
          return;
        }
        std::string serr;
        sync_endpoint_models(ep, &serr);   // pick up the new model
        c->send_simple(200, "application/json",
                       "{\"downloaded\":\"" + model + "\"}");
        return;
      }
      if (starts_with(action, "models/") && method == "DELETE") {
        std::string model =
            url_decode(action.substr(strlen("models/")));
        if (ep->type != "ollama") {
          // reference parity: model delete is an ollama-only
          // capability (types/endpoint.rs:164-166)
          c->send_json_error(501, "unsupported_engine",
                             "model delete: ollama endpoints only");
          return;
        }
        MiniHttpResult res = mini_http_request(
            ep->host, ep->port, "DELETE",
            ep->path_prefix + "/api/delete",
            "{\"name\":\"" + model + "\"}", cfg.health_timeout_ms);
        if (res.status != 200) {
          c->send_json_error(res.status == 404 ? 404 : 502,
                             "delete_failed",
                             res.error.empty() ? res.body.substr(0, 200)
                                               : res.error);
          return;
        }
        std::string serr;
        sync_endpoint_models(ep, &serr);
        c->send_simple(200, "application/json",
                       "{\"deleted\":\"" + model + "\"}");
        return;
      }
      if (action == "test" && method == "POST") {
        MiniHttpResult res = mini_http_request(
            ep->host, ep->port, "GET", ep->path_prefix + "/v1/models", "",
            cfg.health_timeout_ms);
        JsonObject o;
        o.emplace_back("reachable", Json(res.status > 0));
        o.emplace_back("status", Json((long long)res.status));
        o.emplace_back("latency_ms", Json(res.elapsed_ms));
        if (!res.error.empty()) o.emplace_back("error", Json(res.error));
        c->send_simple(200, "application/json", Json(std::move(o)).dump());
        return;
      }
      c->send_json_error(404, "not_found", "unknown endpoint action");
      return;
    }

    // -- api keys --
    if (path == "/api/api-keys" && method == "POST") {
      Json in;
      Json::parse(body, in);
      std::string name = in["name"].as_str();
      std::string perms = in["permissions"].as_str();
      if (perms.empty()) perms = "inference";
      std::string key = auth.create_key(name.empty() ? "key" : name, perms);
      if (key.empty()) {
        c->send_json_error(500, "entropy_unavailable",
                           "could not generate key material");
        return;
      }
      JsonObject o;
      o.emplace_back("api_key", Json(key));
      o.emplace_back("name", Json(name));
      o.emplace_back("permissions", Json(perms));
      c->send_simple(201, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (path == "/api/api-keys" && method == "GET") {
      std::vector<DbRow> rows;
      db.exec("SELECT id,name,prefix,permissions,created_at,revoked FROM "
              "api_keys", {}, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("id", Json((long long)row[0].i));
        o.emplace_back("name", Json(row[1].str()));
        o.emplace_back("prefix", Json(row[2].str()));
        o.emplace_back("permissions", Json(row[3].str()));
        o.emplace_back("created_at", Json(row[4].str()));
        o.emplace_back("revoked", Json(row[5].i != 0));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (starts_with(path, "/api/api-keys/") && method == "DELETE") {
      auth.revoke_key(path.substr(strlen("/api/api-keys/")));
      c->send_simple(200, "application/json", "{\"revoked\":true}");
      return;
    }

    // -- dashboard data --
    if (path == "/api/dashboard/summary" && method == "GET") {
      c->send_simple(200, "application/json", render_summary_json());
      return;
    }
    if (path == "/api/dashboard/overview" && method == "GET") {
      // one-call dashboard bootstrap (reference api/dashboard.rs:186-202
      // get_overview): summary + recent history + per-endpoint model
      // TPS, stamped with generation time
      int64_t t0 = now_us();
      Json summary;
      Json::parse(render_summary_json(), summary);
      JsonArray hist;
      for (auto& rec : history.ring().snapshot(100)) {
        JsonObject o;
        o.emplace_back("ts_ms", Json((long long)rec.ts_ms));
        o.emplace_back("endpoint_id", Json(rec.endpoint_id));
        o.emplace_back("model", Json(rec.model));
        o.emplace_back("status", Json((long long)rec.status));
        o.emplace_back("duration_ms", Json(rec.duration_ms));
        hist.emplace_back(Json(std::move(o)));
      }
      JsonArray tps;
      for (auto& ep : registry.list()) {
        for (auto& e : registry.model_tps_list(ep->id)) {
          JsonObject o;
          o.emplace_back("endpoint_id", Json(ep->id));
          o.emplace_back("model_id", Json(e.model));
          o.emplace_back("api_kind", Json(e.kind));
          o.emplace_back("tps", Json(e.tps));
          o.emplace_back("request_count", Json(e.samples));
          tps.emplace_back(Json(std::move(o)));
        }
      }
      JsonObject out;
      out.emplace_back("summary", std::move(summary));
      out.emplace_back("history", Json(std::move(hist)));
      out.emplace_back("endpoint_tps", Json(std::move(tps)));
      out.emplace_back("generated_at", Json(iso8601_now()));
      out.emplace_back("generation_time_ms",
                       Json((double)(now_us() - t0) / 1000.0));
      c->send_simple(200, "application/json", Json(std::move(out)).dump());
      return;
    }
    if (path == "/api/requests" && method == "GET") {
      // optional filters (reference api/dashboard.rs:661-689
      // list_request_responses / RequestHistoryQuery): model=, status=
      // (exact or "4xx"/"5xx" class), limit=
      std::string q(r.query);
      auto qget = [&](const char* k) -> std::string {
        std::string needle = std::string(k) + "=";
        size_t p = q.find(needle);
        if (p != 0 && (p == std::string::npos || q[p - 1] != '&'))
          return "";
        size_t v = p + needle.size();
        return url_decode(q.substr(v, q.find('&', v) - v));
      };
      std::string f_model = qget("model"), f_status = qget("status");
      long long limit = 500;
      if (!qget("limit").empty()) limit = atoll(qget("limit").c_str());
      if (limit < 1) limit = 1;
      if (limit > 500) limit = 500;
      auto recs = history.ring().snapshot(500);
      JsonArray arr;
      for (auto& rec : recs) {
        if (!f_model.empty() && rec.model != f_model) continue;
        if (!f_status.empty()) {
          if (f_status == "4xx") {
            if (rec.status < 400 || rec.status > 499) continue;
          } else if (f_status == "5xx") {
            if (rec.status < 500 || rec.status > 599) continue;
          } else if (rec.status != atoi(f_status.c_str())) {
            continue;
          }
        }
        JsonObject o;
        o.emplace_back("ts_ms", Json((long long)rec.ts_ms));
        o.emplace_back("endpoint_id", Json(rec.endpoint_id));
        o.emplace_back("model", Json(rec.model));
        o.emplace_back("api_kind", Json(rec.api_kind));
        o.emplace_back("status", Json((long long)rec.status));
        o.emplace_back("duration_ms", Json(rec.duration_ms));
        o.emplace_back("prompt_tokens", Json(rec.prompt_tokens));
        o.emplace_back("completion_tokens", Json(rec.completion_tokens));
        o.emplace_back("client_ip", Json(rec.client_ip));
        if (!rec.request_id.empty())
          o.emplace_back("request_id", Json(rec.request_id));
        if (!rec.error.empty()) o.emplace_back("error", Json(rec.error));
        arr.emplace_back(Json(std::move(o)));
      }
      // snapshot is oldest-first: a limit keeps the NEWEST records
      if ((long long)arr.size() > limit)
        arr.erase(arr.begin(), arr.end() - limit);
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (starts_with(path, "/api/requests/") && method == "GET") {
      // request-history detail incl. captured bodies (reference
      // api/dashboard.rs:692-705 get_request_response_detail);
      // accepts the numeric row id or the X-Request-Id value
      std::string rid = url_decode(path.substr(strlen("/api/requests/")));
      std::vector<DbRow> rows;
      db.exec("SELECT id,ts_ms,endpoint_id,model,api_kind,status,"
              "duration_ms,prompt_tokens,completion_tokens,client_ip,"
              "error,request_body,response_body,request_id FROM "
              "request_history WHERE id=? OR request_id=? LIMIT 1",
              {rid, rid}, &rows);
      if (rows.empty()) {
        c->send_json_error(404, "not_found", "no such request record");
        return;
      }
      auto& row = rows[0];
      JsonObject o;
      o.emplace_back("id", Json((long long)row[0].i));
      o.emplace_back("ts_ms", Json((long long)row[1].i));
      o.emplace_back("endpoint_id", Json(row[2].str()));
      o.emplace_back("model", Json(row[3].str()));
      o.emplace_back("api_kind", Json(row[4].str()));
      o.emplace_back("status", Json((long long)row[5].i));
      o.emplace_back("duration_ms",
                     Json(row[6].kind == DbValue::kReal ? row[6].d
                                                        : (double)row[6].i));
      o.emplace_back("prompt_tokens", Json((long long)row[7].i));
      o.emplace_back("completion_tokens", Json((long long)row[8].i));
      o.emplace_back("client_ip", Json(row[9].str()));
      if (!row[10].str().empty())
        o.emplace_back("error", Json(row[10].str()));
      o.emplace_back("request_body", Json(row[11].str()));
      o.emplace_back("response_body", Json(row[12].str()));
      if (!row[13].str().empty())
        o.emplace_back("request_id", Json(row[13].str()));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (path == "/api/dashboard/stats/tokens" && method == "GET") {
      // all-time token totals (reference api/dashboard.rs:216-225,
      // db/request_history.rs:946-953 TokenStatistics)
      std::vector<DbRow> rows;
      db.exec("SELECT COALESCE(SUM(prompt_tokens),0),"
              "COALESCE(SUM(completion_tokens),0) FROM "
              "endpoint_daily_stats",
              {}, &rows);
      long long in_t = rows.empty() ? 0 : rows[0][0].i;
      long long out_t = rows.empty() ? 0 : rows[0][1].i;
      JsonObject o;
      o.emplace_back("total_input_tokens", Json(in_t));
      o.emplace_back("total_output_tokens", Json(out_t));
      o.emplace_back("total_tokens", Json(in_t + out_t));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (path == "/api/dashboard/stats/tokens/daily" && method == "GET") {
      // daily token totals (reference api/dashboard.rs:257-309
      // DailyTokenStats, default 30 days)
      long long days = 30;
      std::string q(r.query);
      size_t p = q.find("days=");
      if (p != std::string::npos) days = atoll(q.c_str() + p + 5);
      if (days < 1) days = 1;
      if (days > 365) days = 365;
      std::vector<DbRow> rows;
      db.exec("SELECT date, SUM(prompt_tokens), SUM(completion_tokens),"
              "SUM(requests) FROM endpoint_daily_stats GROUP BY date "
              "ORDER BY date DESC LIMIT ?",
              {days}, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        long long in_t = row[1].i, out_t = row[2].i;
        JsonObject o;
        o.emplace_back("date", Json(row[0].str()));
        o.emplace_back("total_input_tokens", Json(in_t));
        o.emplace_back("total_output_tokens", Json(out_t));
        o.emplace_back("total_tokens", Json(in_t + out_t));
        o.emplace_back("request_count", Json((long long)row[3].i));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (starts_with(path, "/api/dashboard/metrics/") && method == "GET") {
      // per-endpoint health-metrics history (reference
      // api/dashboard.rs:205-211 get_node_metrics): the health loop's
      // persisted probe results, newest first
      std::string eid =
          url_decode(path.substr(strlen("/api/dashboard/metrics/")));
      std::vector<DbRow> rows;
      db.exec("SELECT ts_ms, ok, latency_ms, detail FROM health_checks "
              "WHERE endpoint_id=? ORDER BY ts_ms DESC LIMIT 500",
              {eid}, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("ts_ms", Json((long long)row[0].i));
        o.emplace_back("ok", Json(row[1].i != 0));
        o.emplace_back("latency_ms",
                       Json(row[2].kind == DbValue::kReal
                                ? row[2].d
                                : (double)row[2].i));
        std::string detail = row[3].str();
        Json dj;
        if (Json::parse(detail, dj))
          o.emplace_back("detail", std::move(dj));
        else if (!detail.empty())
          o.emplace_back("detail", Json(detail));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (path == "/api/dashboard/all-model-stats" && method == "GET") {
      // cross-endpoint per-model rollup (reference
      // api/dashboard.rs:969-976 get_all_model_stats)
      std::vector<DbRow> rows;
      db.exec("SELECT model, SUM(requests), SUM(failures), "
              "SUM(completion_tokens), SUM(total_duration_ms) FROM "
              "endpoint_daily_stats GROUP BY model ORDER BY "
              "SUM(requests) DESC",
              {}, &rows);
      c->send_simple(200, "application/json",
                     model_stats_json(rows).dump());
      return;
    }
    if (path == "/api/dashboard/daily-stats" && method == "GET") {
      std::vector<DbRow> rows;
      db.exec("SELECT date,endpoint_id,model,api_kind,requests,failures,"
              "prompt_tokens,completion_tokens,total_duration_ms FROM "
              "endpoint_daily_stats ORDER BY date DESC LIMIT 500",
              {}, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("date", Json(row[0].str()));
        o.emplace_back("endpoint_id", Json(row[1].str()));
        o.emplace_back("model", Json(row[2].str()));
        o.emplace_back("api_kind", Json(row[3].str()));
        o.emplace_back("requests", Json((long long)row[4].i));
        o.emplace_back("failures", Json((long long)row[5].i));
        o.emplace_back("prompt_tokens", Json((long long)row[6].i));
        o.emplace_back("completion_tokens", Json((long long)row[7].i));
        o.emplace_back("total_duration_ms",
                       Json(row[8].kind == DbValue::kReal ? row[8].d
                                                          : (double)row[8].i));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (path == "/api/events" && method == "GET") {
      int64_t since = 0;
      std::string q(r.query);
      size_t p = q.find("since=");
      if (p != std::string::npos)
        since = atoll(q.c_str() + p + 6);
      JsonArray arr;
      for (auto& e : events.since(since)) {
        JsonObject o;
        o.emplace_back("ts_ms", Json((long long)e.ts_ms));
        o.emplace_back("type", Json(e.type));
        Json detail;
        if (Json::parse(e.detail, detail))
          o.emplace_back("detail", std::move(detail));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (path == "/api/gpu" && method == "GET") {
      std::string out;
      telemetry_snapshot_json(&out);
      c->send_simple(200, "application/json", out);
      return;
    }
    // -- client analytics (reference api/dashboard.rs: rankings,
    //    heatmap, per-IP drilldown, monthly rollups) ------------------
    if (path == "/api/dashboard/clients" && method == "GET") {
      std::vector<DbRow> rows;
      db.exec("SELECT client_ip, COUNT(*) requests, "
              "SUM(completion_tokens) tokens, AVG(duration_ms) avg_ms, "
              "SUM(CASE WHEN status >= 400 THEN 1 ELSE 0 END) failures "
              "FROM request_history WHERE ts_ms > ? GROUP BY client_ip "
              "ORDER BY requests DESC LIMIT 50",
              {DbParam((long long)(wall_ms() - 7LL * 86400 * 1000))},
              &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("client_ip", Json(row[0].str()));
        o.emplace_back("requests", Json(row[1].i));
        o.emplace_back("tokens", Json(row[2].i));
        o.emplace_back("avg_ms", Json(row[3].kind == DbValue::kReal
                                          ? row[3].d
                                          : (double)row[3].i));
        o.emplace_back("failures", Json(row[4].i));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }
    if (path == "/api/dashboard/clients/heatmap" && method == "GET") {
      // weekday x hour 7x24 request counts over the last 7 days
      std::vector<DbRow> rows;
      db.exec("SELECT CAST(strftime('%w', ts_ms/1000, 'unixepoch') AS "
              "INTEGER) dow, CAST(strftime('%H', ts_ms/1000, "
              "'unixepoch') AS INTEGER) hour, COUNT(*) FROM "
              "request_history WHERE ts_ms > ? GROUP BY dow, hour",
              {DbParam((long long)(wall_ms() - 7LL * 86400 * 1000))},
              &rows);
      long long grid[7][24] = {};
      for (auto& row : rows) {
        int d = (int)row[0].i, h = (int)row[1].i;
        if (d >= 0 && d < 7 && h >= 0 && h < 24) grid[d][h] = row[2].i;
      }
      JsonArray outer;
      for (int d = 0; d < 7; ++d) {
        JsonArray inner;
        for (int h = 0; h < 24; ++h) inner.emplace_back(Json(grid[d][h]));
        outer.emplace_back(Json(std::move(inner)));
      }
      JsonObject o;
      o.emplace_back("days", Json(7));
      o.emplace_back("requests", Json(std::move(outer)));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (starts_with(path, "/api/dashboard/clients/") && method == "GET") {
      std::string ip = url_decode(
          path.substr(strlen("/api/dashboard/clients/")));
      long long cutoff = wall_ms() - 7LL * 86400 * 1000;
      std::vector<DbRow> tot, models, recent;
      db.exec("SELECT COUNT(*), SUM(prompt_tokens), "
              "SUM(completion_tokens), AVG(duration_ms), "
              "SUM(CASE WHEN status >= 400 THEN 1 ELSE 0 END) FROM "
              "request_history WHERE ts_ms > ? AND client_ip = ?",
              {DbParam(cutoff), DbParam(ip)}, &tot);
      if (tot.empty() || tot[0][0].i == 0) {
        c->send_json_error(404, "not_found",
                           "no requests from this client");
        return;
      }
      db.exec("SELECT model, COUNT(*), SUM(completion_tokens) FROM "
              "request_history WHERE ts_ms > ? AND client_ip = ? "
              "GROUP BY model ORDER BY 2 DESC",
              {DbParam(cutoff), DbParam(ip)}, &models);
      db.exec("SELECT ts_ms, model, api_kind, status, duration_ms, "
              "completion_tokens FROM request_history WHERE "
              "client_ip = ? ORDER BY ts_ms DESC LIMIT 50",
              {DbParam(ip)}, &recent);
      JsonObject o;
      o.emplace_back("client_ip", Json(ip));
      {
        JsonObject t;
        t.emplace_back("requests", Json(tot[0][0].i));
        t.emplace_back("prompt_tokens", Json(tot[0][1].i));
        t.emplace_back("completion_tokens", Json(tot[0][2].i));
        t.emplace_back("avg_ms", Json(tot[0][3].kind == DbValue::kReal
                                          ? tot[0][3].d
                                          : (double)tot[0][3].i));
        t.emplace_back("failures", Json(tot[0][4].i));
        o.emplace_back("totals", Json(std::move(t)));
      }
      {
        JsonArray arr;
        for (auto& m : models) {
          JsonObject mo;
          mo.emplace_back("model", Json(m[0].str()));
          mo.emplace_back("requests", Json(m[1].i));
          mo.emplace_back("tokens", Json(m[2].i));
          arr.emplace_back(Json(std::move(mo)));
        }
        o.emplace_back("models", Json(std::move(arr)));
      }
      {
        JsonArray arr;
        for (auto& r2 : recent) {
          JsonObject ro;
          ro.emplace_back("ts_ms", Json(r2[0].i));
          ro.emplace_back("model", Json(r2[1].str()));
          ro.emplace_back("api_kind", Json(r2[2].str()));
          ro.emplace_back("status", Json(r2[3].i));
          ro.emplace_back("duration_ms",
                          Json(r2[4].kind == DbValue::kReal
                                   ? r2[4].d
                                   : (double)r2[4].i));
          ro.emplace_back("completion_tokens", Json(r2[5].i));
          arr.emplace_back(Json(std::move(ro)));
        }
        o.emplace_back("recent", Json(std::move(arr)));
      }
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (path == "/api/dashboard/token-stats/monthly" && method == "GET") {
      std::vector<DbRow> rows;
      db.exec("SELECT substr(date, 1, 7) month, SUM(prompt_tokens), "
              "SUM(completion_tokens), SUM(requests), SUM(failures) "
              "FROM endpoint_daily_stats GROUP BY month ORDER BY "
              "month DESC LIMIT 24", {}, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("month", Json(row[0].str()));
        o.emplace_back("prompt_tokens", Json(row[1].i));
        o.emplace_back("completion_tokens", Json(row[2].i));
        o.emplace_back("requests", Json(row[3].i));
        o.emplace_back("failures", Json(row[4].i));
        arr.emplace_back(Json(std::move(o)));
      }
      c->send_simple(200, "application/json", Json(std::move(arr)).dump());
      return;
    }

    // -- audit query + full-text search (reference audit_log API +
    //    migrations/026 FTS5) -----------------------------------------
    if (path == "/api/audit" && method == "GET") {
      std::string q(r.query);
      auto param = [&](const char* key) -> std::string {
        std::string pat = std::string(key) + "=";
        size_t p = q.find(pat);
        if (p != 0 && (p == std::string::npos || q[p - 1] != '&'))
          return "";
        size_t v = p + pat.size();
        size_t e = q.find('&', v);
        return url_decode(q.substr(v, e == std::string::npos
                                          ? std::string::npos
                                          : e - v));
      };
      std::string sql =
          "SELECT id, ts_ms, method, path, status, client_ip, user, "
          "duration_ms, hash FROM audit_log WHERE 1=1";
      std::vector<DbParam> binds;
      if (!param("from").empty()) {
        sql += " AND ts_ms >= ?";
        binds.emplace_back((long long)atoll(param("from").c_str()));
      }
      if (!param("to").empty()) {
        sql += " AND ts_ms <= ?";
        binds.emplace_back((long long)atoll(param("to").c_str()));
      }
      if (!param("status").empty()) {
        sql += " AND status = ?";
        binds.emplace_back(atoi(param("status").c_str()));
      }
      if (!param("path").empty()) {
        sql += " AND path LIKE ?";
        binds.emplace_back(param("path") + "%");
      }
      if (!param("user").empty()) {
        sql += " AND user = ?";
        binds.emplace_back(param("user"));
      }
      std::string search = param("q");
      if (!search.empty()) {
        if (db.fts5()) {
          sql += " AND id IN (SELECT rowid FROM audit_fts WHERE "
                 "audit_fts MATCH ?)";
          binds.emplace_back(search);
        } else {  // libsqlite3 without FTS5: LIKE fallback
          sql += " AND (path LIKE ? OR method LIKE ? OR client_ip "
                 "LIKE ? OR user LIKE ?)";
          std::string like = "%" + search + "%";
          for (int i = 0; i < 4; ++i) binds.emplace_back(like);
        }
      }
      int limit = param("limit").empty() ? 100
                                         : atoi(param("limit").c_str());
      if (limit <= 0 || limit > 1000) limit = 100;
      int offset = param("offset").empty()
                       ? 0
                       : atoi(param("offset").c_str());
      sql += " ORDER BY id DESC LIMIT ? OFFSET ?";
      binds.emplace_back(limit);
      binds.emplace_back(offset);
      std::vector<DbRow> rows;
      db.exec(sql, binds, &rows);
      JsonArray arr;
      for (auto& row : rows) {
        JsonObject o;
        o.emplace_back("id", Json(row[0].i));
        o.emplace_back("ts_ms", Json(row[1].i));
        o.emplace_back("method", Json(row[2].str()));
        o.emplace_back("path", Json(row[3].str()));
        o.emplace_back("status", Json(row[4].i));
        o.emplace_back("client_ip", Json(row[5].str()));
        o.emplace_back("user", Json(row[6].str()));
        o.emplace_back("duration_ms",
                       Json(row[7].kind == DbValue::kReal
                                ? row[7].d
                                : (double)row[7].i));
        o.emplace_back("hash", Json(row[8].str()));
        arr.emplace_back(Json(std::move(o)));
      }
      JsonObject out;
      out.emplace_back("entries", Json(std::move(arr)));
      out.emplace_back("fts", Json(db.fts5()));
      c->send_simple(200, "application/json", Json(std::move(out)).dump());
      return;
    }
    if (path == "/api/audit/verify" && method == "GET") {
      std::string err;
      bool ok = history.verify_audit_chain(&err);
      JsonObject o;
      o.emplace_back("valid", Json(ok));
      if (!ok) o.emplace_back("error", Json(err));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    // -- drain gate control (update/self-update support) --
    if (path == "/api/system/drain" && method == "POST") {
      draining.store(true);
      c->send_simple(200, "application/json", "{\"draining\":true}");
      return;
    }
    if (path == "/api/system/resume" && method == "POST") {
      draining.store(false);
      c->send_simple(200, "application/json", "{\"draining\":false}");
      return;
    }
    // -- settings KV (reference db/settings.rs + dashboard settings
    //    pages) --------------------------------------------------------
    if (path == "/api/settings" && method == "GET") {
      std::vector<DbRow> rows;
      db.exec("SELECT key, value FROM settings", {}, &rows);
      JsonObject o;
      for (auto& row : rows)
        o.emplace_back(row[0].str(), Json(row[1].str()));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    if (path == "/api/settings" && (method == "PUT" || method == "POST")) {
      Json in;
      if (!Json::parse(body, in) || !in.is_obj()) {
        c->send_json_error(400, "bad_json", "expected a key->value object");
        return;
      }
      int nset = 0;
      for (auto& [key, val] : in.as_obj()) {
        if (key.size() > 128 || nset > 100) continue;
        db.exec("INSERT INTO settings (key, value) VALUES (?, ?) ON "
                "CONFLICT(key) DO UPDATE SET value=excluded.value",
                {DbParam(key), DbParam(val.is_str() ? val.as_str()
                                                    : val.dump())});
        ++nset;
      }
      c->send_simple(200, "application/json",
                     "{\"updated\":" + std::to_string(nset) + "}");
      return;
    }
    if (path == "/api/system/info" && method == "GET") {
      JsonObject o;
      o.emplace_back("version", Json("0.1.0"));
      o.emplace_back("workers", Json((long long)workers.size()));
      o.emplace_back("uptime_ms", Json((long long)(wall_ms() - start_ms)));
      o.emplace_back("draining", Json(draining.load()));
      c->send_simple(200, "application/json", Json(std::move(o)).dump());
      return;
    }
    // fall through to the Python control plane when configured
    if (!cfg.control_plane_base.empty()) {
      start_control_plane_proxy(c);
      return;
    }
    c->send_json_error(404, "not_found", "unknown /api route");
    return;
  }

  // embedded dashboard SPA (reference serves its React build from the
  // binary via include_dir, api/mod.rs:56,637-676; ours is a single
  // self-contained page generated into dashboard_html.h at build time).
  // The page itself is public; every API it calls requires the token.
  if (method == "GET" &&
      (path == "/" || path == "/dashboard" ||
       starts_with(path, "/dashboard/"))) {
    c->send_simple(200, "text/html; charset=utf-8",
                   std::string((const char*)kDashboardHtml,
                               kDashboardHtml_len));
    return;
  }

  // ws / anything else under the ops prefix: control plane
  if (!cfg.control_plane_base.empty()) {
    start_control_plane_proxy(c);
    return;
  }
  c->send_json_error(404, "not_found", "no such route");
}

void Gateway::start_control_plane_proxy(ClientConn* c) {
  // forward verbatim to the Python control plane (host:port http)
  std::string base = cfg.control_plane_base;
  std::string_view s = base;
  if (starts_with(s, "http://")) s.remove_prefix(7);
  size_t colon = s.rfind(':');
  std::string host(s.substr(0, colon));
  int port = colon == std::string_view::npos
                 ? 80
                 : atoi(std::string(s.substr(colon + 1)).c_str());
  auto ep = std::make_shared<Endpoint>();  // ephemeral descriptor
  ep->id = "control-plane";
  ep->host = host;
  ep->port = port;
  ep->inference_timeout_s = 30;
  ep->status = EndpointStatus::kOnline;
  start_proxy_direct(this, c, ep, ApiKind::kOther,
                     std::string(c->req.target),
                     c->req.header("content-type"));
}

}  // namespace gw
