// SQLite persistence for the gateway (endpoints, models, request
// history, daily stats, users, api keys, audit log) — the L2 layer of
// SURVEY.md §1, reimplemented natively.
//
// The build environment ships libsqlite3.so.0 but no sqlite3.h, so we
// declare the handful of stable public C-API entry points we use
// ourselves (the SQLite C API is a frozen, documented ABI).
#pragma once

#include <cstdint>
#include <functional>
#include <mutex>
#include <string>
#include <string_view>
#include <vector>

// ---- minimal sqlite3 ABI declarations (stable public API) ----
extern "C" {
typedef struct sqlite3 sqlite3;
typedef struct sqlite3_stmt sqlite3_stmt;
int sqlite3_open_v2(const char* filename, sqlite3** db, int flags,
                    const char* vfs);
int sqlite3_close_v2(sqlite3*);
int sqlite3_exec(sqlite3*, const char* sql,
                 int (*cb)(void*, int, char**, char**), void*, char** errmsg);
void sqlite3_free(void*);
int sqlite3_prepare_v2(sqlite3*, const char* sql, int n, sqlite3_stmt**,
                       const char** tail);
int sqlite3_step(sqlite3_stmt*);
int sqlite3_finalize(sqlite3_stmt*);
int sqlite3_reset(sqlite3_stmt*);
int sqlite3_bind_text(sqlite3_stmt*, int, const char*, int, void (*)(void*));
int sqlite3_bind_int64(sqlite3_stmt*, int, long long);
int sqlite3_bind_double(sqlite3_stmt*, int, double);
int sqlite3_bind_null(sqlite3_stmt*, int);
int sqlite3_column_count(sqlite3_stmt*);
const unsigned char* sqlite3_column_text(sqlite3_stmt*, int);
long long sqlite3_column_int64(sqlite3_stmt*, int);
double sqlite3_column_double(sqlite3_stmt*, int);
int sqlite3_column_type(sqlite3_stmt*, int);
int sqlite3_column_bytes(sqlite3_stmt*, int);
const char* sqlite3_errmsg(sqlite3*);
long long sqlite3_last_insert_rowid(sqlite3*);
int sqlite3_changes(sqlite3*);
int sqlite3_busy_timeout(sqlite3*, int ms);
}

#define SQLITE_OK 0
#define SQLITE_ROW 100
#define SQLITE_DONE 101
#define SQLITE_OPEN_READWRITE 0x00000002
#define SQLITE_OPEN_CREATE 0x00000004
#define SQLITE_OPEN_FULLMUTEX 0x00010000
#define SQLITE_TRANSIENT ((void (*)(void*))(intptr_t)(-1))
#define SQLITE_NULL 5

namespace gw {

// A cell value from a query result.
struct DbValue {
  enum Kind { kNull, kInt, kReal, kText } kind = kNull;
  long long i = 0;
  double d = 0;
  std::string s;
  std::string str() const { return kind == kText ? s : std::string(); }
};
using DbRow = std::vector<DbValue>;

// Bind parameter (variant-lite).
struct DbParam {
  enum Kind { kNull, kInt, kReal, kText } kind;
  long long i = 0;
  double d = 0;
  std::string s;
  DbParam() : kind(kNull) {}
  DbParam(std::nullptr_t) : kind(kNull) {}
  DbParam(int v) : kind(kInt), i(v) {}
  DbParam(long long v) : kind(kInt), i(v) {}
  DbParam(int64_t v, int) : kind(kInt), i(v) {}
  DbParam(unsigned long v) : kind(kInt), i((long long)v) {}
  DbParam(double v) : kind(kReal), d(v) {}
  DbParam(const char* v) : kind(kText), s(v) {}
  DbParam(std::string v) : kind(kText), s(std::move(v)) {}
  DbParam(std::string_view v) : kind(kText), s(v) {}
  DbParam(bool v) : kind(kInt), i(v ? 1 : 0) {}
};

class Db {
 public:
  Db() = default;
  ~Db() { close(); }
  bool open(const std::string& path, std::string* err = nullptr);
  void close();
  bool ok() const { return db_ != nullptr; }

  // Execute with positional params; rows (if any) via out.
  bool exec(std::string_view sql, const std::vector<DbParam>& params = {},
            std::vector<DbRow>* out = nullptr, std::string* err = nullptr);
  long long last_insert_rowid();
  int changes();
  bool fts5() const { return fts5_; }
  // run the schema migrations (idempotent)
  bool migrate(std::string* err = nullptr);

  std::mutex& mutex() { return mu_; }

 private:
  sqlite3* db_ = nullptr;
  bool fts5_ = false;
  std::mutex mu_;  // serialize writers; SQLite handles its own locking too
};

}  // namespace gw
