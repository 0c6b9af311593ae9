// Routing core: endpoint registry + TPS-weighted selection.
//
// Reimplements the behavior of the reference's balancer::LoadManager /
// registry::EndpointRegistry (SURVEY.md §2 components 5-8) natively:
//   * per-(endpoint, model, api_kind) tokens/sec EMA, alpha = 0.2
//     (balancer/types.rs:109), stale after 120 s (types.rs:20);
//   * selection sorts candidates by TPS score desc with round-robin
//     tie-break (balancer/mod.rs:1922-1985);
//   * RAII RequestLease: in-flight counters auto-complete as Error on
//     drop (balancer/lease.rs);
//   * admission control thresholds 50 % delay / 80 % reject
//     (balancer/mod.rs:2255-2270), extended MI355X-natively with
//     amdsmi HBM-occupancy gating (BASELINE.json: 288 GB HBM3E).
//
// Hot-path reads go through an immutable snapshot (shared_ptr swap);
// mutations (admin CRUD, health transitions, model sync) rebuild it.
#pragma once

#include <atomic>
#include <cmath>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "common.h"
#include "db.h"

namespace gw {

enum class EndpointStatus { kPending, kOnline, kOffline, kError };
const char* status_name(EndpointStatus s);
EndpointStatus status_from(std::string_view s);

// capability bits (reference types/endpoint.rs EndpointCapability)
enum Capability : uint32_t {
  kCapChat = 1,
  kCapEmbeddings = 2,
  kCapImageGen = 4,
  kCapAudioTranscription = 8,
  kCapAudioSpeech = 16,
  kCapResponses = 32,
  kCapCompletions = 64,
};
uint32_t caps_from_csv(std::string_view csv);
std::string caps_to_csv(uint32_t caps);

enum class ApiKind { kChat, kCompletions, kEmbeddings, kResponses,
                     kMessages, kAudio, kImages, kOther };
const char* api_kind_name(ApiKind k);

// per-(endpoint, model, kind) TPS EMA state
struct TpsState {
  std::mutex mu;
  double ema = 0.0;
  int64_t updated_ms = 0;
  double last_sample = 0.0;
  long long samples = 0;

  // boot-time seed from persisted daily stats (bootstrap.rs:143-159):
  // one synthetic sample so first selections after a restart are not
  // score-blind; the first live sample then EMA-blends over it
  void seed(double tps, int64_t now) {
    std::lock_guard<std::mutex> lk(mu);
    if (samples > 0) return;
    ema = tps;
    last_sample = tps;
    samples = 1;
    updated_ms = now;
  }

  void update(double tokens, double duration_s, double alpha, int64_t now) {
    if (duration_s <= 0.0) return;
    double tps = tokens / duration_s;
    std::lock_guard<std::mutex> lk(mu);
    ema = samples == 0 ? tps : alpha * tps + (1.0 - alpha) * ema;
    last_sample = tps;
    ++samples;
    updated_ms = now;
  }
  double score(int64_t now_ms, int64_t stale_ms = 120000) {
    std::lock_guard<std::mutex> lk(mu);
    if (samples == 0 || now_ms - updated_ms > stale_ms) return 0.0;
    return ema;
  }
  void reset() {
    std::lock_guard<std::mutex> lk(mu);
    ema = 0;
    samples = 0;
    updated_ms = 0;
  }
};

struct GpuInfo {
  double hbm_used_gb = 0, hbm_total_gb = 0;
  double busy_pct = 0;
  int device_index = -1;
  int64_t updated_ms = 0;
};

struct EndpointModel {
  std::string id;          // backend-native model id
  std::string canonical;   // canonical (HF-style) id, may equal id
  uint32_t caps = kCapChat | kCapCompletions | kCapResponses;
  long long max_tokens = 0;
};

struct Endpoint {
  std::string id;
  std::string name;
  std::string base_url;        // http://host:port
  std::string host;            // parsed
  int port = 80;
  std::string path_prefix;     // from base_url (usually empty)
  std::string type = "openai_compat";
  std::string api_key;         // bearer for upstream
  int device_index = -1;       // local GPU index for amdsmi telemetry
  int health_interval_s = 30;
  int inference_timeout_s = 120;
  std::string created_at, updated_at;

  // mutable runtime state
  std::atomic<EndpointStatus> status{EndpointStatus::kPending};
  std::atomic<int> active{0};
  std::atomic<long long> total_requests{0};
  std::atomic<long long> total_failures{0};
  std::atomic<int> consecutive_failures{0};
  std::atomic<double> latency_ema_ms{0.0};  // inference latency EMA α=0.2
  std::mutex gpu_mu;
  GpuInfo gpu;

  std::vector<EndpointModel> models;  // guarded by Registry::mu_

  void update_latency(double ms) {
    double cur = latency_ema_ms.load(std::memory_order_relaxed);
    double next = cur == 0.0 ? ms : 0.2 * ms + 0.8 * cur;
    latency_ema_ms.store(next, std::memory_order_relaxed);
  }
};
using EndpointPtr = std::shared_ptr<Endpoint>;

// immutable routing snapshot (rebuilt on registry change)
struct ModelRoute {
  std::string model_key;                       // lookup key (alias)
  std::string canonical;
  std::vector<EndpointPtr> endpoints;          // serving this model
  std::vector<std::shared_ptr<TpsState>> tps;  // parallel (chat kind)
  std::vector<std::string> upstream_ids;       // per-endpoint native id
  std::shared_ptr<std::atomic<uint64_t>> rr;   // round-robin cursor
  uint32_t caps = 0;
  long long max_tokens = 0;
};

struct RouteTable {
  std::unordered_map<std::string, ModelRoute> by_model;
  std::vector<EndpointPtr> all;
  // distinct canonical models (for /v1/models)
  std::vector<std::string> model_keys_canonical;
};

class EventBus;  // fwd (events.h)

class Registry {
 public:
  explicit Registry(Db* db) : db_(db) {}
  void set_event_bus(EventBus* bus) { bus_ = bus; }
  bool load_from_db();

  // CRUD (persists + rebuilds snapshot)
  EndpointPtr add_endpoint(const std::string& name,
                           const std::string& base_url,
                           const std::string& type,
                           const std::string& api_key, std::string* err);
  bool remove_endpoint(const std::string& id);
  bool update_endpoint(const std::string& id,
                       const std::function<void(Endpoint&)>& fn);
  EndpointPtr get(const std::string& id);
  std::vector<EndpointPtr> list();

  void set_status(const std::string& id, EndpointStatus st);
  // replace an endpoint's model list (sync); rebuilds the index
  void sync_models(const std::string& id,
                   const std::vector<EndpointModel>& models);
  void update_gpu_info(const std::string& id, const GpuInfo& g);

  std::shared_ptr<const RouteTable> table() const {
    std::lock_guard<std::mutex> lk(table_mu_);
    return table_;
  }

  // TPS tracking -------------------------------------------------------
  std::shared_ptr<TpsState> tps_state(const std::string& endpoint_id,
                                      const std::string& model,
                                      ApiKind kind);
  void record_tps(const std::string& endpoint_id, const std::string& model,
                  ApiKind kind, double tokens, double duration_s);
  void clear_tps_for_endpoint(const std::string& endpoint_id);
  // per-model TPS snapshot for one endpoint (dashboard.rs:1206-1225):
  // (model, kind, ema tokens/s, samples), live entries only
  struct ModelTps {
    std::string model, kind;
    double tps;
    long long samples;
  };
  std::vector<ModelTps> model_tps_list(const std::string& endpoint_id);
  // boot reseed from endpoint_daily_stats (today's rows)
  void seed_tps_from_db();
  double tps_score(const std::string& endpoint_id, const std::string& model,
                   ApiKind kind);

  void rebuild();  // rebuild snapshot from current state

 private:
  Db* db_;
  EventBus* bus_ = nullptr;
  mutable std::mutex mu_;        // endpoints map + models
  mutable std::mutex table_mu_;  // snapshot pointer
  std::map<std::string, EndpointPtr> endpoints_;
  std::shared_ptr<const RouteTable> table_ =
      std::make_shared<RouteTable>();
  std::mutex tps_mu_;
  std::map<std::string, std::shared_ptr<TpsState>> tps_;  // "ep|model|kind"
};

// ------------------------------------------------------------------ lease
// RAII in-flight accounting (reference balancer/lease.rs): if the lease
// is dropped without complete(), it auto-records a failure.
class RequestLease {
 public:
  RequestLease() = default;
  RequestLease(EndpointPtr ep, Registry* reg, std::string model,
               ApiKind kind)
      : ep_(std::move(ep)), reg_(reg), model_(std::move(model)),
        kind_(kind), done_(false) {
    start_us_ = now_us();
    ep_->active.fetch_add(1, std::memory_order_relaxed);
    ep_->total_requests.fetch_add(1, std::memory_order_relaxed);
  }
  RequestLease(const RequestLease&) = delete;
  RequestLease& operator=(const RequestLease&) = delete;
  RequestLease(RequestLease&& o) noexcept { *this = std::move(o); }
  RequestLease& operator=(RequestLease&& o) noexcept {
    release_if_needed();
    ep_ = std::move(o.ep_);
    reg_ = o.reg_;
    tps_ = std::move(o.tps_);
    model_ = std::move(o.model_);
    kind_ = o.kind_;
    start_us_ = o.start_us_;
    done_ = o.done_;
    o.ep_ = nullptr;
    o.done_ = true;
    return *this;
  }
  ~RequestLease() { release_if_needed(); }

  bool active() const { return ep_ && !done_; }
  EndpointPtr endpoint() const { return ep_; }
  int64_t start_us() const { return start_us_; }

  void set_tps(std::shared_ptr<TpsState> tps) { tps_ = std::move(tps); }
  void complete_success(long long completion_tokens) {
    if (!active()) return;
    done_ = true;
    ep_->active.fetch_sub(1, std::memory_order_relaxed);
    double dur_s = (now_us() - start_us_) / 1e6;
    ep_->update_latency(dur_s * 1000.0);
    ep_->consecutive_failures.store(0, std::memory_order_relaxed);
    if (completion_tokens > 0) {
      if (tps_)
        tps_->update((double)completion_tokens, dur_s, 0.2, wall_ms());
      else if (reg_)
        reg_->record_tps(ep_->id, model_, kind_, (double)completion_tokens,
                         dur_s);
    }
  }
  void complete_failure() {
    if (!active()) return;
    done_ = true;
    ep_->active.fetch_sub(1, std::memory_order_relaxed);
    ep_->total_failures.fetch_add(1, std::memory_order_relaxed);
  }

 private:
  void release_if_needed() {
    if (active()) complete_failure();  // Drop => Error (lease.rs:71)
  }
  EndpointPtr ep_;
  Registry* reg_ = nullptr;
  std::shared_ptr<TpsState> tps_;
  std::string model_;
  ApiKind kind_ = ApiKind::kChat;
  int64_t start_us_ = 0;
  bool done_ = true;
};

// ------------------------------------------------------------------ select
struct AdmissionConfig {
  int max_concurrent_per_endpoint = 100;   // LLMLB_QUEUE_MAX analog
  double delay_fraction = 0.5;
  double reject_fraction = 0.8;
  double hbm_admission_fraction = 0.92;    // of 288 GB HBM3E
  // LLMLB_LOAD_BALANCER_MODE=round_robin ignores TPS scores (pure RR
  // over admitting endpoints; reference config.rs load_balancer mode)
  bool round_robin = false;
};

struct SelectResult {
  EndpointPtr ep;
  std::string upstream_model;   // model id to send upstream
  std::string canonical;
  std::shared_ptr<TpsState> tps;  // the (endpoint, model, chat) EMA cell
  int err_status = 0;           // 0 = ok; else 404/429/503
  std::string err_code;
  // delayed admission (reference admission_control,
  // balancer/mod.rs:2255-2270): >=50% load on the picked endpoint ->
  // hold the request 10-100 ms before forwarding
  int delay_ms = 0;
};

// TPS-weighted selection with RR tie-break + admission control.
// exclude: endpoint id to skip (single retry after a connect failure
// — the dead endpoint keeps the best TPS score until the health loop
// ejects it, so without exclusion a retry would re-pick it)
SelectResult select_endpoint(const RouteTable& table,
                             const std::string& model, ApiKind kind,
                             const AdmissionConfig& adm, int64_t now_ms,
                             uint32_t required_caps = 0,
                             const std::string* exclude = nullptr);

}  // namespace gw
