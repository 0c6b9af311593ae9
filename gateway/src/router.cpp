#include "router.h"

#include <algorithm>

namespace gw {

const char* status_name(EndpointStatus s) {
  switch (s) {
    case EndpointStatus::kPending: return "pending";
    case EndpointStatus::kOnline: return "online";
    case EndpointStatus::kOffline: return "offline";
    case EndpointStatus::kError: return "error";
  }
  return "unknown";
}

EndpointStatus status_from(std::string_view s) {
  if (s == "online") return EndpointStatus::kOnline;
  if (s == "offline") return EndpointStatus::kOffline;
  if (s == "error") return EndpointStatus::kError;
  return EndpointStatus::kPending;
}

uint32_t caps_from_csv(std::string_view csv) {
  uint32_t caps = 0;
  for (auto& part : split(csv, ',')) {
    std::string p = to_lower(trim(part));
    if (p == "chat") caps |= kCapChat;
    else if (p == "embeddings" || p == "embedding") caps |= kCapEmbeddings;
    else if (p == "image_generation" || p == "image_gen")
      caps |= kCapImageGen;
    else if (p == "audio_transcription") caps |= kCapAudioTranscription;
    else if (p == "audio_speech" || p == "tts") caps |= kCapAudioSpeech;
    else if (p == "responses") caps |= kCapResponses;
    else if (p == "completions") caps |= kCapCompletions;
  }
  return caps;
}

std::string caps_to_csv(uint32_t caps) {
  std::string out;
  auto add = [&](const char* s) {
    if (!out.empty()) out += ',';
    out += s;
  };
  if (caps & kCapChat) add("chat");
  if (caps & kCapEmbeddings) add("embeddings");
  if (caps & kCapImageGen) add("image_generation");
  if (caps & kCapAudioTranscription) add("audio_transcription");
  if (caps & kCapAudioSpeech) add("audio_speech");
  if (caps & kCapResponses) add("responses");
  if (caps & kCapCompletions) add("completions");
  return out;
}

const char* api_kind_name(ApiKind k) {
  switch (k) {
    case ApiKind::kChat: return "chat_completions";
    case ApiKind::kCompletions: return "completions";
    case ApiKind::kEmbeddings: return "embeddings";
    case ApiKind::kResponses: return "responses";
    case ApiKind::kMessages: return "messages";
    case ApiKind::kAudio: return "audio";
    case ApiKind::kImages: return "images";
    case ApiKind::kOther: return "other";
  }
  return "other";
}

static bool parse_base_url(const std::string& url, std::string* host,
                           int* port, std::string* prefix) {
  std::string_view s = url;
  if (starts_with(s, "http://")) s.remove_prefix(7);
  else if (starts_with(s, "https://")) return false;  // no TLS upstream
  size_t slash = s.find('/');
  std::string_view hostport = s.substr(0, slash);
  *prefix = slash == std::string_view::npos ? "" : std::string(s.substr(slash));
  if (!prefix->empty() && prefix->back() == '/') prefix->pop_back();
  size_t colon = hostport.rfind(':');
  if (colon == std::string_view::npos) {
    *host = std::string(hostport);
    *port = 80;
  } else {
    *host = std::string(hostport.substr(0, colon));
    *port = atoi(std::string(hostport.substr(colon + 1)).c_str());
  }
  return !host->empty() && *port > 0 && *port < 65536;
}

bool Registry::load_from_db() {
  std::vector<DbRow> rows;
  if (!db_->exec(
          "SELECT id,name,base_url,endpoint_type,status,api_key,device_index,"
          "health_check_interval_secs,inference_timeout_secs,created_at,"
          "updated_at FROM endpoints", {}, &rows))
    return false;
  std::unique_lock<std::mutex> lk(mu_);
  for (auto& r : rows) {
    auto ep = std::make_shared<Endpoint>();
    ep->id = r[0].str();
    ep->name = r[1].str();
    ep->base_url = r[2].str();
    ep->type = r[3].str();
    ep->status = status_from(r[4].str());
    // previously-online endpoints start pending until re-checked
    if (ep->status == EndpointStatus::kOnline)
      ep->status = EndpointStatus::kPending;
    ep->api_key = r[5].str();
    ep->device_index = (int)r[6].i;
    ep->health_interval_s = (int)r[7].i;
    ep->inference_timeout_s = (int)r[8].i;
    ep->created_at = r[9].str();
    ep->updated_at = r[10].str();
    if (!parse_base_url(ep->base_url, &ep->host, &ep->port,
                        &ep->path_prefix)) {
      LOG_WARN("skipping endpoint %s: bad base_url %s", ep->id.c_str(),
               ep->base_url.c_str());
      continue;
    }
    endpoints_[ep->id] = ep;
  }
  std::vector<DbRow> mrows;
  db_->exec("SELECT endpoint_id,model_id,canonical_id,capabilities,"
            "max_tokens FROM endpoint_models", {}, &mrows);
  for (auto& r : mrows) {
    auto it = endpoints_.find(r[0].str());
    if (it == endpoints_.end()) continue;
    EndpointModel m;
    m.id = r[1].str();
    m.canonical = r[2].str().empty() ? m.id : r[2].str();
    m.caps = caps_from_csv(r[3].str());
    if (m.caps == 0) m.caps = kCapChat | kCapCompletions | kCapResponses;
    m.max_tokens = r[4].kind == DbValue::kInt ? r[4].i : 0;
    it->second->models.push_back(std::move(m));
  }
  lk.unlock();  // rebuild takes mu_ itself
  rebuild();
  return true;
}

EndpointPtr Registry::add_endpoint(const std::string& name,
                                   const std::string& base_url,
                                   const std::string& type,
                                   const std::string& api_key,
                                   std::string* err) {
  auto ep = std::make_shared<Endpoint>();
  ep->id = uuid4();
  ep->name = name.empty() ? base_url : name;
  ep->base_url = base_url;
  ep->type = type.empty() ? "openai_compat" : type;
  ep->api_key = api_key;
  ep->created_at = ep->updated_at = iso8601_now();
  if (!parse_base_url(base_url, &ep->host, &ep->port, &ep->path_prefix)) {
    if (err) *err = "invalid base_url (need http://host:port)";
    return nullptr;
  }
  {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& [id, existing] : endpoints_) {
      if (existing->base_url == base_url) {
        if (err) *err = "endpoint with this base_url already exists";
        return nullptr;
      }
    }
    endpoints_[ep->id] = ep;
  }
  if (db_ && db_->ok()) {
    db_->exec("INSERT INTO endpoints (id,name,base_url,endpoint_type,status,"
              "api_key,device_index,health_check_interval_secs,"
              "inference_timeout_secs,created_at,updated_at) VALUES "
              "(?,?,?,?,?,?,?,?,?,?,?)",
              {ep->id, ep->name, ep->base_url, ep->type, "pending",
               ep->api_key, (long long)ep->device_index,
               (long long)ep->health_interval_s,
               (long long)ep->inference_timeout_s, ep->created_at,
               ep->updated_at});
  }
  rebuild();
  return ep;
}

bool Registry::remove_endpoint(const std::string& id) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (!endpoints_.erase(id)) return false;
  }
  if (db_ && db_->ok()) {
    db_->exec("DELETE FROM endpoints WHERE id=?", {id});
    db_->exec("DELETE FROM endpoint_models WHERE endpoint_id=?", {id});
  }
  clear_tps_for_endpoint(id);
  rebuild();
  return true;
}

bool Registry::update_endpoint(const std::string& id,
                               const std::function<void(Endpoint&)>& fn) {
  EndpointPtr ep;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = endpoints_.find(id);
    if (it == endpoints_.end()) return false;
    ep = it->second;
    fn(*ep);
    ep->updated_at = iso8601_now();
  }
  if (db_ && db_->ok()) {
    db_->exec("UPDATE endpoints SET name=?,endpoint_type=?,api_key=?,"
              "device_index=?,status=?,updated_at=? WHERE id=?",
              {ep->name, ep->type, ep->api_key,
               (long long)ep->device_index,
               status_name(ep->status.load()), ep->updated_at, id});
  }
  rebuild();
  return true;
}

EndpointPtr Registry::get(const std::string& id) {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = endpoints_.find(id);
  return it == endpoints_.end() ? nullptr : it->second;
}

std::vector<EndpointPtr> Registry::list() {
  std::lock_guard<std::mutex> lk(mu_);
  std::vector<EndpointPtr> out;
  out.reserve(endpoints_.size());
  for (auto& [id, ep] : endpoints_) out.push_back(ep);
  return out;
}

void Registry::set_status(const std::string& id, EndpointStatus st) {
  EndpointPtr ep = get(id);
  if (!ep) return;
  EndpointStatus old = ep->status.exchange(st);
  if (old == st) return;
  if (db_ && db_->ok())
    db_->exec("UPDATE endpoints SET status=?,updated_at=? WHERE id=?",
              {status_name(st), iso8601_now(), id});
  rebuild();
}

void Registry::sync_models(const std::string& id,
                           const std::vector<EndpointModel>& models) {
  EndpointPtr ep = get(id);
  if (!ep) return;
  {
    std::lock_guard<std::mutex> lk(mu_);
    ep->models = models;
  }
  if (db_ && db_->ok()) {
    std::lock_guard<std::mutex> dblk(db_->mutex());
    db_->exec("DELETE FROM endpoint_models WHERE endpoint_id=?", {id});
    for (auto& m : models) {
      db_->exec("INSERT OR REPLACE INTO endpoint_models (endpoint_id,"
                "model_id,canonical_id,capabilities,max_tokens) VALUES "
                "(?,?,?,?,?)",
                {id, m.id, m.canonical, caps_to_csv(m.caps),
                 (long long)m.max_tokens});
    }
  }
  rebuild();
}

void Registry::update_gpu_info(const std::string& id, const GpuInfo& g) {
  EndpointPtr ep = get(id);
  if (!ep) return;
  std::lock_guard<std::mutex> lk(ep->gpu_mu);
  ep->gpu = g;
  ep->gpu.updated_ms = wall_ms();
}

void Registry::rebuild() {
  auto table = std::make_shared<RouteTable>();
  {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& [id, ep] : endpoints_) {
      table->all.push_back(ep);
      if (ep->status.load() != EndpointStatus::kOnline) continue;
      for (auto& m : ep->models) {
        // index under the native id AND the canonical id so any alias
        // routes (reference registry/endpoints.rs:16-54)
        for (const std::string& key : {m.id, m.canonical}) {
          if (key.empty()) continue;
          auto& route = table->by_model[key];
          if (route.model_key.empty()) {
            route.model_key = key;
            route.canonical = m.canonical;
            route.rr = std::make_shared<std::atomic<uint64_t>>(0);
          }
          // avoid duplicate endpoint entries per key
          bool dup = false;
          for (auto& e : route.endpoints)
            if (e->id == ep->id) dup = true;
          if (dup) continue;
          route.endpoints.push_back(ep);
          route.upstream_ids.push_back(m.id);
          route.tps.push_back(tps_state(ep->id, m.canonical,
                                        ApiKind::kChat));
          route.caps |= m.caps;
          route.max_tokens = std::max(route.max_tokens, m.max_tokens);
        }
      }
    }
    for (auto& [key, route] : table->by_model)
      if (key == route.canonical)
        table->model_keys_canonical.push_back(key);
    std::sort(table->model_keys_canonical.begin(),
              table->model_keys_canonical.end());
  }
  std::lock_guard<std::mutex> lk(table_mu_);
  table_ = table;
}

std::shared_ptr<TpsState> Registry::tps_state(const std::string& endpoint_id,
                                              const std::string& model,
                                              ApiKind kind) {
  std::string key = endpoint_id + "|" + model + "|" + api_kind_name(kind);
  std::lock_guard<std::mutex> lk(tps_mu_);
  auto& st = tps_[key];
  if (!st) st = std::make_shared<TpsState>();
  return st;
}

void Registry::record_tps(const std::string& endpoint_id,
                          const std::string& model, ApiKind kind,
                          double tokens, double duration_s) {
  tps_state(endpoint_id, model, kind)
      ->update(tokens, duration_s, 0.2, wall_ms());
}

double Registry::tps_score(const std::string& endpoint_id,
                           const std::string& model, ApiKind kind) {
  return tps_state(endpoint_id, model, kind)->score(wall_ms());
}

std::vector<Registry::ModelTps> Registry::model_tps_list(
    const std::string& endpoint_id) {
  std::vector<ModelTps> out;
  std::lock_guard<std::mutex> lk(tps_mu_);
  const std::string prefix = endpoint_id + "|";
  for (auto& [key, st] : tps_) {
    if (!starts_with(key, prefix)) continue;
    std::string rest = key.substr(prefix.size());   // "model|kind"
    size_t bar = rest.rfind('|');
    if (bar == std::string::npos) continue;
    std::lock_guard<std::mutex> slk(st->mu);
    if (st->samples == 0) continue;
    out.push_back({rest.substr(0, bar), rest.substr(bar + 1), st->ema,
                   st->samples});
  }
  return out;
}

void Registry::clear_tps_for_endpoint(const std::string& endpoint_id) {
  std::lock_guard<std::mutex> lk(tps_mu_);
  for (auto& [key, st] : tps_) {
    if (starts_with(key, endpoint_id + "|")) st->reset();
  }
}

// Boot-time TPS reseed from today's persisted daily stats
// (bootstrap.rs:143-159, balancer/mod.rs:2612-2634): one synthetic
// EMA sample per (endpoint, model, api_kind) so a restarted gateway
// is not score-blind until every backend re-reports.
void Registry::seed_tps_from_db() {
  if (!db_ || !db_->ok()) return;
  std::vector<DbRow> rows;
  db_->exec(
      "SELECT endpoint_id, model, api_kind, completion_tokens, "
      "total_duration_ms FROM endpoint_daily_stats WHERE "
      "date = date('now') AND requests > 0 AND total_duration_ms > 0",
      {}, &rows);
  int64_t now = wall_ms();
  int seeded = 0;
  for (auto& r : rows) {
    double tokens = (double)r[3].i;
    double dur_s = (r[4].kind == DbValue::kReal ? r[4].d
                                            : (double)r[4].i) /
               1000.0;
    if (tokens <= 0 || dur_s <= 0) continue;
    ApiKind kind = ApiKind::kOther;
    std::string kname = r[2].str();
    for (ApiKind k : {ApiKind::kChat, ApiKind::kCompletions,
                      ApiKind::kEmbeddings, ApiKind::kResponses,
                      ApiKind::kMessages, ApiKind::kAudio,
                      ApiKind::kImages})
      if (kname == api_kind_name(k)) kind = k;
    tps_state(r[0].str(), r[1].str(), kind)->seed(tokens / dur_s, now);
    ++seeded;
  }
  if (seeded)
    LOG_INFO("seeded %d TPS cells from today's daily stats", seeded);
}

SelectResult select_endpoint(const RouteTable& table,
                             const std::string& model, ApiKind kind,
                             const AdmissionConfig& adm, int64_t now_ms,
                             uint32_t required_caps,
                             const std::string* exclude) {
  SelectResult res;
  auto it = table.by_model.find(model);
  if (it == table.by_model.end()) {
    res.err_status = 404;
    res.err_code = "model_not_found";
    return res;
  }
  const ModelRoute& route = it->second;
  if (required_caps && !(route.caps & required_caps)) {
    res.err_status = 404;
    res.err_code = "capability_not_supported";
    return res;
  }
  // score each online candidate; admission-filter
  struct Cand {
    size_t idx;
    double score;
  };
  std::vector<Cand> cands;
  cands.reserve(route.endpoints.size());
  bool any_online = false, all_rejecting = true;
  for (size_t i = 0; i < route.endpoints.size(); ++i) {
    const EndpointPtr& ep = route.endpoints[i];
    if (ep->status.load(std::memory_order_relaxed) !=
        EndpointStatus::kOnline)
      continue;
    if (exclude && !exclude->empty() && ep->id == *exclude) continue;
    any_online = true;
    int active = ep->active.load(std::memory_order_relaxed);
    double frac = (double)active / adm.max_concurrent_per_endpoint;
    if (frac >= adm.reject_fraction) continue;  // 80 % -> reject this ep
    // MI355X HBM admission: refuse placement on a GPU already beyond
    // the HBM budget (fresh amdsmi sample required)
    {
      std::lock_guard<std::mutex> lk(ep->gpu_mu);
      if (ep->gpu.updated_ms > 0 && now_ms - ep->gpu.updated_ms < 30000 &&
          ep->gpu.hbm_total_gb > 0 &&
          ep->gpu.hbm_used_gb / ep->gpu.hbm_total_gb >
              adm.hbm_admission_fraction)
        continue;
    }
    all_rejecting = false;
    double score = adm.round_robin ? 0.0 : route.tps[i]->score(now_ms);
    cands.push_back({i, score});
  }
  if (cands.empty()) {
    res.err_status = any_online ? 429 : 503;
    res.err_code = any_online ? "capacity_exceeded" : "no_endpoint_online";
    (void)all_rejecting;
    return res;
  }
  // sort by TPS desc; RR among the top-score group (balancer/mod.rs:1960)
  std::sort(cands.begin(), cands.end(),
            [](const Cand& a, const Cand& b) { return a.score > b.score; });
  size_t top_end = 1;
  while (top_end < cands.size() &&
         cands[top_end].score >= cands[0].score * 0.999 - 1e-9)
    ++top_end;
  uint64_t rr = route.rr->fetch_add(1, std::memory_order_relaxed);
  const Cand& pick = cands[rr % top_end];
  res.ep = route.endpoints[pick.idx];
  {
    // delayed-admission arm: 50%..80% load -> 10..100 ms hold
    // (reference admission_control, balancer/mod.rs:2255-2270)
    double frac =
        (double)res.ep->active.load(std::memory_order_relaxed) /
        adm.max_concurrent_per_endpoint;
    if (frac >= adm.delay_fraction && adm.reject_fraction >
                                          adm.delay_fraction)
      res.delay_ms =
          10 + (int)((frac - adm.delay_fraction) /
                     (adm.reject_fraction - adm.delay_fraction) * 90.0);
  }
  res.upstream_model = route.upstream_ids[pick.idx];
  res.canonical = route.canonical;
  res.tps = route.tps[pick.idx];
  return res;
}

}  // namespace gw
