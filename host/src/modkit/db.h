// modkit-db equivalent: SQLite-backed tenant-scoped secure data access.
//
// Re-creates the reference's secure ORM semantics
// (libs/modkit-db/src/secure/mod.rs): every query goes through a
// SecureConn bound to an AccessScope; an EMPTY scope compiles to
// `WHERE 1=0` (deny-all default, secure/mod.rs:94-101), tenant scoping is
// never optional, and migrations run once under an advisory lock
// (modkit-db/src/{migration_runner,advisory_locks}.rs).
//
// SQLite's C ABI is stable; the image ships libsqlite3.so.0 without the
// header, so the handful of functions used are declared here directly.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <vector>

#include "../util/json.h"

extern "C" {
typedef struct sqlite3 sqlite3;
typedef struct sqlite3_stmt sqlite3_stmt;
int sqlite3_open(const char*, sqlite3**);
int sqlite3_close(sqlite3*);
int sqlite3_exec(sqlite3*, const char*, int (*)(void*, int, char**, char**),
                 void*, char**);
int sqlite3_prepare_v2(sqlite3*, const char*, int, sqlite3_stmt**,
                       const char**);
int sqlite3_step(sqlite3_stmt*);
int sqlite3_finalize(sqlite3_stmt*);
int sqlite3_bind_text(sqlite3_stmt*, int, const char*, int,
                      void (*)(void*));
int sqlite3_bind_double(sqlite3_stmt*, int, double);
int sqlite3_bind_int64(sqlite3_stmt*, int, long long);
int sqlite3_bind_null(sqlite3_stmt*, int);
int sqlite3_column_count(sqlite3_stmt*);
const char* sqlite3_column_name(sqlite3_stmt*, int);
int sqlite3_column_type(sqlite3_stmt*, int);
const unsigned char* sqlite3_column_text(sqlite3_stmt*, int);
double sqlite3_column_double(sqlite3_stmt*, int);
long long sqlite3_column_int64(sqlite3_stmt*, int);
const char* sqlite3_errmsg(sqlite3*);
long long sqlite3_last_insert_rowid(sqlite3*);
int sqlite3_changes(sqlite3*);
void sqlite3_free(void*);
}

namespace hs {

// Row-level access scope (modkit-security AccessScope algebra:
// src/access_scope.rs — for_tenant(s)/allow_all/deny_all; default DENY).
struct AccessScope {
  bool unrestricted = false;              // allow_all()
  std::vector<std::string> tenant_ids;    // empty + !unrestricted => deny
  std::optional<std::vector<std::string>> resource_ids;  // per-subject rows

  static AccessScope allow_all() { return {true, {}, std::nullopt}; }
  static AccessScope deny_all() { return {}; }
  static AccessScope for_tenant(const std::string& t) {
    return {false, {t}, std::nullopt};
  }
  static AccessScope for_tenants(std::vector<std::string> ts) {
    return {false, std::move(ts), std::nullopt};
  }
  bool is_deny_all() const { return !unrestricted && tenant_ids.empty(); }
};

struct DbValue {
  enum Kind { kNull, kText, kInt, kReal } kind = kNull;
  std::string text;
  long long i = 0;
  double d = 0;
  static DbValue S(std::string s) { return {kText, std::move(s), 0, 0}; }
  static DbValue I(long long v) { return {kInt, "", v, 0}; }
  static DbValue R(double v) { return {kReal, "", 0, v}; }
  static DbValue null() { return {}; }
};

using DbRow = std::map<std::string, Json>;

// One SQLite database (the reference's per-module Db from DbManager).
class Db {
 public:
  explicit Db(const std::string& path);   // ":memory:" or a file path
  ~Db();
  Db(const Db&) = delete;

  // schema migrations: applied once each, tracked in _migrations, under
  // a process-wide mutex + sqlite BEGIN IMMEDIATE (advisory-lock analog)
  void migrate(const std::string& module,
               const std::vector<std::pair<std::string, std::string>>& ms);

  void exec(const std::string& sql);
  std::vector<DbRow> query(const std::string& sql,
                           const std::vector<DbValue>& binds = {});
  long long last_insert_rowid();
  int changes();

  sqlite3* raw() { return db_; }
  std::mutex& mu() { return mu_; }

 private:
  sqlite3* db_ = nullptr;
  std::mutex mu_;
};

// PEP: every statement built here carries the scope's WHERE conditions.
// (reference: Entity::find().secure().scope_with(&scope))
class SecureConn {
 public:
  SecureConn(Db& db, AccessScope scope, std::string tenant_col = "tenant_id",
             std::string resource_col = "")
      : db_(db), scope_(std::move(scope)),
        tenant_col_(std::move(tenant_col)),
        resource_col_(std::move(resource_col)) {}

  struct Page {
    std::vector<DbRow> items;
    std::optional<std::string> next_cursor;
  };

  // multi-field signed ordering: (field, desc) list (modkit-odata
  // ODataOrderBy signed tokens, lib.rs:135)
  using OrderBy = std::vector<std::pair<std::string, bool>>;

  // SELECT with optional extra WHERE, deterministic (order keys, rowid)
  // keyset pagination and an opaque base64 cursor (modkit-odata CursorV1).
  Page select(const std::string& table, const std::string& extra_where,
              std::vector<DbValue> binds, const std::string& order_by,
              bool desc, int limit,
              const std::optional<std::string>& cursor);
  Page select(const std::string& table, const std::string& extra_where,
              std::vector<DbValue> binds, const OrderBy& order_by,
              int limit, const std::optional<std::string>& cursor);

  // INSERT: the tenant column is forced to the scope's single tenant.
  void insert(const std::string& table,
              const std::vector<std::pair<std::string, DbValue>>& cols);
  // UPDATE/DELETE: scope conditions always applied; returns changed rows.
  int update(const std::string& table,
             const std::vector<std::pair<std::string, DbValue>>& sets,
             const std::string& extra_where, std::vector<DbValue> binds);
  int remove(const std::string& table, const std::string& extra_where,
             std::vector<DbValue> binds);

  // the compiled scope condition ("1=0" when deny-all)
  std::string scope_sql(std::vector<DbValue>& binds) const;

 private:
  Db& db_;
  AccessScope scope_;
  std::string tenant_col_, resource_col_;
};

// Tiny OData-style $filter compiler: supports
//   <field> eq|ne|gt|ge|lt|le <literal>, contains/startswith/endswith,
//   and/or/not with parenthesised grouping (modkit-odata Expr AST,
//   /root/reference/libs/modkit-odata/src/lib.rs:23,:70).  Fields are
//   validated against an allow-list (x-odata-filter allowedFields).
// Returns SQL + binds; throws std::runtime_error on a bad filter.
std::string compile_odata_filter(const std::string& filter,
                                 const std::vector<std::string>& fields,
                                 std::vector<DbValue>& binds);

// "$orderby=a desc,b" / signed tokens "-a,b" → (field, desc) list,
// allow-listed; throws on unknown fields or malformed input.
std::vector<std::pair<std::string, bool>> parse_odata_orderby(
    const std::string& orderby, const std::vector<std::string>& fields);

}  // namespace hs
