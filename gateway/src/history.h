// Async request-history / daily-stats / audit writer: records are
// enqueued from the hot path (lock + push, no DB I/O) and flushed by a
// background thread in batches — the native analog of the reference's
// fire-and-forget tokio::spawn writes (api/proxy.rs:273-368) and the
// buffered audit writer (audit/writer.rs).
#pragma once

#include <condition_variable>
#include <deque>
#include <thread>

#include "common.h"
#include "db.h"
#include "sha256.h"

namespace gw {

struct RequestRecord {
  int64_t ts_ms = 0;
  std::string request_id;
  std::string request_body;    // captured only when LLMLB_CAPTURE_BODIES
  std::string response_body;
  std::string endpoint_id;
  std::string model;
  std::string api_kind;
  int status = 0;
  double duration_ms = 0;
  long long prompt_tokens = 0;
  long long completion_tokens = 0;
  std::string client_ip;
  std::string error;
};

struct AuditRecord {
  int64_t ts_ms = 0;
  std::string method;
  std::string path;
  int status = 0;
  std::string client_ip;
  std::string user;
  double duration_ms = 0;
};

// In-memory 60-minute ring for live dashboards (balancer/types.rs:22-24)
class HistoryRing {
 public:
  void push(const RequestRecord& r) {
    std::lock_guard<std::mutex> lk(mu_);
    ring_.push_back(r);
    int64_t cutoff = r.ts_ms - 60 * 60 * 1000;
    while (!ring_.empty() && ring_.front().ts_ms < cutoff)
      ring_.pop_front();
    if (ring_.size() > 100000) ring_.pop_front();
  }
  std::vector<RequestRecord> snapshot(size_t limit = 1000) {
    std::lock_guard<std::mutex> lk(mu_);
    size_t n = std::min(limit, ring_.size());
    return {ring_.end() - n, ring_.end()};
  }
  // aggregate over the ring
  struct Summary {
    long long requests = 0, failures = 0;
    long long prompt_tokens = 0, completion_tokens = 0;
    double avg_duration_ms = 0;
  };
  Summary summarize(int64_t window_ms) {
    std::lock_guard<std::mutex> lk(mu_);
    Summary s;
    int64_t cutoff = wall_ms() - window_ms;
    double total_dur = 0;
    for (auto& r : ring_) {
      if (r.ts_ms < cutoff) continue;
      ++s.requests;
      if (r.status >= 400 || r.status == 0) ++s.failures;
      s.prompt_tokens += r.prompt_tokens;
      s.completion_tokens += r.completion_tokens;
      total_dur += r.duration_ms;
    }
    if (s.requests) s.avg_duration_ms = total_dur / s.requests;
    return s;
  }

 private:
  std::mutex mu_;
  std::deque<RequestRecord> ring_;
};

class HistoryWriter {
 public:
  HistoryWriter(Db* db, int retention_days)
      : db_(db), retention_days_(retention_days) {}
  ~HistoryWriter() { stop(); }

  void start();
  void stop();

  void record(RequestRecord r) {
    ring_.push(r);
    {
      std::lock_guard<std::mutex> lk(mu_);
      queue_.push_back(std::move(r));
      if (queue_.size() > 100000) queue_.pop_front();  // bound memory
    }
    cv_.notify_one();
  }
  // batched variant used by the workers (one lock for N records)
  void record_batch(std::vector<RequestRecord>& rs) {
    if (rs.empty()) return;
    for (auto& r : rs) ring_.push(r);
    {
      std::lock_guard<std::mutex> lk(mu_);
      for (auto& r : rs) queue_.push_back(std::move(r));
      while (queue_.size() > 100000) queue_.pop_front();
    }
    cv_.notify_one();
    rs.clear();
  }
  void audit(AuditRecord r) {
    std::lock_guard<std::mutex> lk(mu_);
    audit_queue_.push_back(std::move(r));
    if (audit_queue_.size() > 100000) audit_queue_.pop_front();
    cv_.notify_one();
  }
  void audit_batch(std::vector<AuditRecord>& rs) {
    if (rs.empty()) return;
    {
      std::lock_guard<std::mutex> lk(mu_);
      for (auto& r : rs) audit_queue_.push_back(std::move(r));
      while (audit_queue_.size() > 100000) audit_queue_.pop_front();
    }
    cv_.notify_one();
    rs.clear();
  }

  HistoryRing& ring() { return ring_; }
  // verify the audit hash chain (reference audit/hash_chain.rs); when
  // old rows have been archived, verification restarts from the
  // anchored tip hash stored in settings ('audit_anchor')
  bool verify_audit_chain(std::string* err);

  // archive audit rows older than cutoff_ms into a second SQLite file
  // (reference bootstrap.rs:291-318: 90-day archive to a separate DB).
  // Returns the number of rows moved, -1 on error.  The hash of the
  // newest archived row is anchored in settings so the live chain
  // still verifies; the archive itself verifies from genesis.
  void set_audit_archive(const std::string& path, int days) {
    archive_path_ = path;
    archive_days_ = days;
  }
  int archive_audit_now(int64_t cutoff_ms, std::string* err = nullptr);

 private:
  void loop();
  void flush(std::vector<RequestRecord>& recs,
             std::vector<AuditRecord>& audits);

  Db* db_;
  int retention_days_;
  HistoryRing ring_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<RequestRecord> queue_;
  std::deque<AuditRecord> audit_queue_;
  std::thread thread_;
  bool stop_ = false;
  std::string last_audit_hash_;
  int64_t last_cleanup_ms_ = 0;
  std::string archive_path_;
  int archive_days_ = 90;
};

}  // namespace gw
