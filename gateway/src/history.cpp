#include "history.h"

namespace gw {

void HistoryWriter::start() {
  // seed the audit chain tip from the DB (genesis if empty)
  if (db_ && db_->ok()) {
    std::vector<DbRow> rows;
    db_->exec("SELECT hash FROM audit_log ORDER BY id DESC LIMIT 1", {},
              &rows);
    last_audit_hash_ = rows.empty() ? sha256_hex("genesis")
                                    : rows[0][0].str();
  }
  stop_ = false;
  thread_ = std::thread([this] { loop(); });
}

void HistoryWriter::stop() {
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (stop_) return;
    stop_ = true;
  }
  cv_.notify_one();
  if (thread_.joinable()) thread_.join();
}

void HistoryWriter::loop() {
  while (true) {
    std::vector<RequestRecord> recs;
    std::vector<AuditRecord> audits;
    {
      std::unique_lock<std::mutex> lk(mu_);
      // system_clock wait_until (NOT steady-clock wait_for): libstdc++
      // lowers wait_for to pthread_cond_clockwait, which this image's
      // libtsan does not intercept — TSAN then believes the writer
      // holds mu_ forever and cascades hundreds of false lock reports
      // over every worker (r2 sanitizer pass).  Semantics unchanged.
      cv_.wait_until(lk,
                     std::chrono::system_clock::now() +
                         std::chrono::milliseconds(500),
                     [this] {
                       return stop_ || queue_.size() > 512 ||
                              !audit_queue_.empty();
                     });
      recs.assign(queue_.begin(), queue_.end());
      queue_.clear();
      audits.assign(audit_queue_.begin(), audit_queue_.end());
      audit_queue_.clear();
      if (stop_ && recs.empty() && audits.empty()) return;
    }
    flush(recs, audits);
    // hourly retention cleanup (db/request_history.rs:1728)
    int64_t now = wall_ms();
    if (db_ && db_->ok() && now - last_cleanup_ms_ > 3600 * 1000) {
      last_cleanup_ms_ = now;
      int64_t cutoff = now - (int64_t)retention_days_ * 86400 * 1000;
      db_->exec("DELETE FROM request_history WHERE ts_ms < ?",
                {(long long)cutoff});
      db_->exec("DELETE FROM health_checks WHERE ts_ms < ?",
                {(long long)cutoff});
      if (!archive_path_.empty())
        archive_audit_now(now - (int64_t)archive_days_ * 86400 * 1000);
    }
  }
}

void HistoryWriter::flush(std::vector<RequestRecord>& recs,
                          std::vector<AuditRecord>& audits) {
  if (!db_ || !db_->ok()) return;
  std::lock_guard<std::mutex> dblk(db_->mutex());
  db_->exec("BEGIN", {});
  for (auto& r : recs) {
    db_->exec(
        "INSERT INTO request_history (ts_ms,endpoint_id,model,api_kind,"
        "status,duration_ms,prompt_tokens,completion_tokens,client_ip,"
        "error,request_body,response_body,request_id) VALUES "
        "(?,?,?,?,?,?,?,?,?,?,?,?,?)",
        {(long long)r.ts_ms, r.endpoint_id, r.model, r.api_kind,
         (long long)r.status, r.duration_ms, r.prompt_tokens,
         r.completion_tokens, r.client_ip, r.error, r.request_body,
         r.response_body, r.request_id});
    // daily stats upsert (db/endpoint_daily_stats.rs)
    char date[16];
    time_t t = (time_t)(r.ts_ms / 1000);
    struct tm tmv;
    gmtime_r(&t, &tmv);
    strftime(date, sizeof date, "%Y-%m-%d", &tmv);
    bool fail = r.status >= 400 || r.status == 0;
    db_->exec(
        "INSERT INTO endpoint_daily_stats (date,endpoint_id,model,api_kind,"
        "requests,failures,prompt_tokens,completion_tokens,"
        "total_duration_ms) VALUES (?,?,?,?,1,?,?,?,?) "
        "ON CONFLICT(date,endpoint_id,model,api_kind) DO UPDATE SET "
        "requests=requests+1, failures=failures+excluded.failures, "
        "prompt_tokens=prompt_tokens+excluded.prompt_tokens, "
        "completion_tokens=completion_tokens+excluded.completion_tokens, "
        "total_duration_ms=total_duration_ms+excluded.total_duration_ms",
        {date, r.endpoint_id, r.model, r.api_kind, fail ? 1LL : 0LL,
         r.prompt_tokens, r.completion_tokens, r.duration_ms});
  }
  for (auto& a : audits) {
    // tamper-evident hash chain: hash = sha256(prev_hash || record)
    char rec[512];
    snprintf(rec, sizeof rec, "%lld|%s|%s|%d|%s|%s|%.3f",
             (long long)a.ts_ms, a.method.c_str(), a.path.c_str(), a.status,
             a.client_ip.c_str(), a.user.c_str(), a.duration_ms);
    std::string h = sha256_hex(last_audit_hash_ + "|" + rec);
    db_->exec(
        "INSERT INTO audit_log (ts_ms,method,path,status,client_ip,user,"
        "duration_ms,prev_hash,hash) VALUES (?,?,?,?,?,?,?,?,?)",
        {(long long)a.ts_ms, a.method, a.path, (long long)a.status,
         a.client_ip, a.user, a.duration_ms, last_audit_hash_, h});
    last_audit_hash_ = h;
  }
  db_->exec("COMMIT", {});
}

int HistoryWriter::archive_audit_now(int64_t cutoff_ms,
                                     std::string* err) {
  if (!db_ || !db_->ok() || archive_path_.empty()) return 0;
  std::lock_guard<std::mutex> dblk(db_->mutex());
  if (!db_->exec("ATTACH ? AS audit_archive", {archive_path_})) {
    if (err) *err = "attach failed: " + archive_path_;
    return -1;
  }
  int moved = -1;
  do {
    if (!db_->exec(
            "CREATE TABLE IF NOT EXISTS audit_archive.audit_log ("
            "id INTEGER PRIMARY KEY, ts_ms INTEGER, method TEXT, "
            "path TEXT, status INTEGER, client_ip TEXT, user TEXT, "
            "duration_ms REAL, prev_hash TEXT, hash TEXT)", {}))
      break;
    db_->exec("BEGIN", {});
    std::vector<DbRow> rows;
    db_->exec("SELECT COUNT(*), MAX(id) FROM audit_log WHERE ts_ms < ?",
              {(long long)cutoff_ms}, &rows);
    long long n = rows.empty() ? 0 : rows[0][0].i;
    if (n == 0) {
      db_->exec("COMMIT", {});
      moved = 0;
      break;
    }
    long long maxid = rows[0][1].i;
    rows.clear();
    db_->exec("SELECT hash FROM audit_log WHERE id = ?", {maxid}, &rows);
    std::string tip = rows.empty() ? std::string() : rows[0][0].str();
    db_->exec("INSERT INTO audit_archive.audit_log "
              "SELECT id,ts_ms,method,path,status,client_ip,user,"
              "duration_ms,prev_hash,hash FROM main.audit_log "
              "WHERE ts_ms < ?", {(long long)cutoff_ms});
    db_->exec("DELETE FROM main.audit_log WHERE ts_ms < ?",
              {(long long)cutoff_ms});
    db_->exec("INSERT INTO settings (key,value) VALUES "
              "('audit_anchor',?) ON CONFLICT(key) DO UPDATE SET "
              "value=excluded.value", {tip});
    db_->exec("COMMIT", {});
    moved = (int)n;
  } while (false);
  db_->exec("DETACH audit_archive", {});
  return moved;
}

bool HistoryWriter::verify_audit_chain(std::string* err) {
  if (!db_ || !db_->ok()) return true;
  std::vector<DbRow> rows;
  std::lock_guard<std::mutex> dblk(db_->mutex());
  if (!db_->exec("SELECT ts_ms,method,path,status,client_ip,user,"
                 "duration_ms,prev_hash,hash FROM audit_log ORDER BY id",
                 {}, &rows))
    return false;
  std::string prev = sha256_hex("genesis");
  {
    std::vector<DbRow> anchor;
    db_->exec("SELECT value FROM settings WHERE key = 'audit_anchor'", {},
              &anchor);
    if (!anchor.empty() && !anchor[0][0].str().empty())
      prev = anchor[0][0].str();
  }
  for (auto& r : rows) {
    if (r[7].str() != prev) {
      if (err) *err = "chain broken: prev_hash mismatch";
      return false;
    }
    char rec[512];
    snprintf(rec, sizeof rec, "%lld|%s|%s|%d|%s|%s|%.3f", r[0].i,
             r[1].str().c_str(), r[2].str().c_str(), (int)r[3].i,
             r[4].str().c_str(), r[5].str().c_str(),
             r[6].kind == DbValue::kReal ? r[6].d : (double)r[6].i);
    std::string h = sha256_hex(prev + "|" + rec);
    if (h != r[8].str()) {
      if (err) *err = "chain broken: record hash mismatch";
      return false;
    }
    prev = h;
  }
  return true;
}

}  // namespace gw
