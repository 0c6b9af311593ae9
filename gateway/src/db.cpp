#include "db.h"

#include "common.h"

namespace gw {

bool Db::open(const std::string& path, std::string* err) {
  int rc = sqlite3_open_v2(
      path.c_str(), &db_,
      SQLITE_OPEN_READWRITE | SQLITE_OPEN_CREATE | SQLITE_OPEN_FULLMUTEX,
      nullptr);
  if (rc != SQLITE_OK) {
    if (err) *err = db_ ? sqlite3_errmsg(db_) : "open failed";
    if (db_) sqlite3_close_v2(db_);
    db_ = nullptr;
    return false;
  }
  sqlite3_busy_timeout(db_, 5000);
  char* em = nullptr;
  sqlite3_exec(db_, "PRAGMA journal_mode=WAL; PRAGMA synchronous=NORMAL;",
               nullptr, nullptr, &em);
  if (em) sqlite3_free(em);
  return true;
}

void Db::close() {
  if (db_) {
    sqlite3_close_v2(db_);
    db_ = nullptr;
  }
}

bool Db::exec(std::string_view sql, const std::vector<DbParam>& params,
              std::vector<DbRow>* out, std::string* err) {
  sqlite3_stmt* st = nullptr;
  int rc = sqlite3_prepare_v2(db_, sql.data(), (int)sql.size(), &st, nullptr);
  if (rc != SQLITE_OK) {
    if (err) *err = std::string("prepare: ") + sqlite3_errmsg(db_);
    LOG_ERROR("sqlite prepare failed: %s | sql=%.*s", sqlite3_errmsg(db_),
              (int)sql.size(), sql.data());
    return false;
  }
  for (size_t i = 0; i < params.size(); ++i) {
    const DbParam& p = params[i];
    int idx = (int)i + 1;
    switch (p.kind) {
      case DbParam::kNull: sqlite3_bind_null(st, idx); break;
      case DbParam::kInt: sqlite3_bind_int64(st, idx, p.i); break;
      case DbParam::kReal: sqlite3_bind_double(st, idx, p.d); break;
      case DbParam::kText:
        sqlite3_bind_text(st, idx, p.s.data(), (int)p.s.size(),
                          SQLITE_TRANSIENT);
        break;
    }
  }
  bool okflag = true;
  while (true) {
    rc = sqlite3_step(st);
    if (rc == SQLITE_ROW) {
      if (out) {
        int nc = sqlite3_column_count(st);
        DbRow row;
        row.reserve(nc);
        for (int c = 0; c < nc; ++c) {
          DbValue v;
          switch (sqlite3_column_type(st, c)) {
            case 1: v.kind = DbValue::kInt;
                    v.i = sqlite3_column_int64(st, c); break;
            case 2: v.kind = DbValue::kReal;
                    v.d = sqlite3_column_double(st, c); break;
            case SQLITE_NULL: v.kind = DbValue::kNull; break;
            default: {
              v.kind = DbValue::kText;
              const unsigned char* t = sqlite3_column_text(st, c);
              int n = sqlite3_column_bytes(st, c);
              if (t) v.s.assign((const char*)t, n);
              break;
            }
          }
          row.push_back(std::move(v));
        }
        out->push_back(std::move(row));
      }
      continue;
    }
    if (rc == SQLITE_DONE) break;
    if (err) *err = sqlite3_errmsg(db_);
    LOG_ERROR("sqlite step failed: %s", sqlite3_errmsg(db_));
    okflag = false;
    break;
  }
  sqlite3_finalize(st);
  return okflag;
}

long long Db::last_insert_rowid() { return sqlite3_last_insert_rowid(db_); }
int Db::changes() { return sqlite3_changes(db_); }

// ---------------------------------------------------------------------
// Schema.  One migration set (idempotent CREATE IF NOT EXISTS), covering
// the reference's table groups (SURVEY.md §2 component 27): endpoints,
// endpoint_models, registered models, request_history, daily stats,
// users, api_keys, invitations, audit_log, settings, health history.
// ---------------------------------------------------------------------
static const char* kSchema = R"SQL(
CREATE TABLE IF NOT EXISTS endpoints (
  id TEXT PRIMARY KEY,
  name TEXT NOT NULL,
  base_url TEXT NOT NULL UNIQUE,
  endpoint_type TEXT NOT NULL DEFAULT 'openai_compat',
  status TEXT NOT NULL DEFAULT 'pending',
  api_key TEXT DEFAULT '',
  device_index INTEGER NOT NULL DEFAULT -1,
  health_check_interval_secs INTEGER NOT NULL DEFAULT 30,
  inference_timeout_secs INTEGER NOT NULL DEFAULT 120,
  created_at TEXT NOT NULL,
  updated_at TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS endpoint_models (
  endpoint_id TEXT NOT NULL,
  model_id TEXT NOT NULL,
  canonical_id TEXT NOT NULL DEFAULT '',
  capabilities TEXT NOT NULL DEFAULT 'chat',
  max_tokens INTEGER,
  PRIMARY KEY (endpoint_id, model_id)
);
CREATE TABLE IF NOT EXISTS models (
  id TEXT PRIMARY KEY,
  name TEXT NOT NULL,
  source TEXT NOT NULL DEFAULT '',
  capabilities TEXT NOT NULL DEFAULT 'chat',
  size_bytes INTEGER,
  created_at TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS request_history (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  ts_ms INTEGER NOT NULL,
  endpoint_id TEXT,
  model TEXT,
  api_kind TEXT,
  status INTEGER,
  duration_ms REAL,
  prompt_tokens INTEGER,
  completion_tokens INTEGER,
  client_ip TEXT,
  error TEXT,
  request_body TEXT,
  response_body TEXT
);
CREATE INDEX IF NOT EXISTS idx_reqhist_ts ON request_history(ts_ms);
CREATE TABLE IF NOT EXISTS endpoint_daily_stats (
  date TEXT NOT NULL,
  endpoint_id TEXT NOT NULL,
  model TEXT NOT NULL,
  api_kind TEXT NOT NULL,
  requests INTEGER NOT NULL DEFAULT 0,
  failures INTEGER NOT NULL DEFAULT 0,
  prompt_tokens INTEGER NOT NULL DEFAULT 0,
  completion_tokens INTEGER NOT NULL DEFAULT 0,
  total_duration_ms REAL NOT NULL DEFAULT 0,
  PRIMARY KEY (date, endpoint_id, model, api_kind)
);
CREATE TABLE IF NOT EXISTS users (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  username TEXT NOT NULL UNIQUE,
  password_hash TEXT NOT NULL,
  role TEXT NOT NULL DEFAULT 'user',
  must_change_password INTEGER NOT NULL DEFAULT 0,
  created_at TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS api_keys (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  key_hash TEXT NOT NULL UNIQUE,
  name TEXT NOT NULL,
  prefix TEXT NOT NULL DEFAULT '',
  permissions TEXT NOT NULL DEFAULT 'inference',
  user_id INTEGER,
  created_at TEXT NOT NULL,
  last_used_at TEXT,
  revoked INTEGER NOT NULL DEFAULT 0
);
CREATE TABLE IF NOT EXISTS invitations (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  token TEXT NOT NULL UNIQUE,
  email TEXT DEFAULT '',
  role TEXT NOT NULL DEFAULT 'user',
  created_by INTEGER,
  created_at TEXT NOT NULL,
  expires_at TEXT,
  used_at TEXT
);
CREATE TABLE IF NOT EXISTS audit_log (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  ts_ms INTEGER NOT NULL,
  method TEXT,
  path TEXT,
  status INTEGER,
  client_ip TEXT,
  user TEXT,
  duration_ms REAL,
  prev_hash TEXT NOT NULL,
  hash TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS settings (
  key TEXT PRIMARY KEY,
  value TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS health_checks (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  endpoint_id TEXT NOT NULL,
  ts_ms INTEGER NOT NULL,
  ok INTEGER NOT NULL,
  latency_ms REAL,
  detail TEXT
);
CREATE INDEX IF NOT EXISTS idx_health_ts ON health_checks(ts_ms);
)SQL";

bool Db::migrate(std::string* err) {
  char* em = nullptr;
  int rc = sqlite3_exec(db_, kSchema, nullptr, nullptr, &em);
  if (rc != SQLITE_OK) {
    if (err && em) *err = em;
    if (em) sqlite3_free(em);
    return false;
  }
  // additive migrations (idempotent: ignore duplicate-column errors)
  sqlite3_exec(db_,
               "ALTER TABLE request_history ADD COLUMN request_id TEXT",
               nullptr, nullptr, nullptr);
  // audit full-text search (reference migrations/026): contentless-
  // delete FTS5 over audit_log kept in sync by triggers.  Some
  // libsqlite3 builds lack FTS5 — record availability and fall back
  // to LIKE queries in the search endpoint.
  int frc = sqlite3_exec(
      db_,
      "CREATE VIRTUAL TABLE IF NOT EXISTS audit_fts USING fts5("
      "method, path, client_ip, user, content='audit_log', "
      "content_rowid='id');"
      "CREATE TRIGGER IF NOT EXISTS audit_fts_ai AFTER INSERT ON "
      "audit_log BEGIN INSERT INTO audit_fts(rowid, method, path, "
      "client_ip, user) VALUES (new.id, new.method, new.path, "
      "new.client_ip, new.user); END;",
      nullptr, nullptr, nullptr);
  fts5_ = (frc == SQLITE_OK);
  return true;
}

}  // namespace gw
