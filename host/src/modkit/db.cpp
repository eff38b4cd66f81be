#include "db.h"

#include <cstring>
#include <stdexcept>

namespace hs {

static void bind_all(sqlite3* db, sqlite3_stmt* st,
                     const std::vector<DbValue>& binds) {
  for (size_t i = 0; i < binds.size(); ++i) {
    const DbValue& v = binds[i];
    int rc = 0;
    switch (v.kind) {
      case DbValue::kText:
        rc = sqlite3_bind_text(st, (int)i + 1, v.text.c_str(),
                               (int)v.text.size(), (void (*)(void*))-1);
        break;
      case DbValue::kInt:
        rc = sqlite3_bind_int64(st, (int)i + 1, v.i);
        break;
      case DbValue::kReal:
        rc = sqlite3_bind_double(st, (int)i + 1, v.d);
        break;
      case DbValue::kNull:
        rc = sqlite3_bind_null(st, (int)i + 1);
        break;
    }
    if (rc != 0)
      throw std::runtime_error(std::string("bind failed: ") +
                               sqlite3_errmsg(db));
  }
}

Db::Db(const std::string& path) {
  if (sqlite3_open(path.c_str(), &db_) != 0)
    throw std::runtime_error("sqlite open failed: " + path);
  exec("PRAGMA journal_mode=WAL; PRAGMA foreign_keys=ON;");
}

Db::~Db() {
  if (db_) sqlite3_close(db_);
}

void Db::exec(const std::string& sql) {
  char* err = nullptr;
  if (sqlite3_exec(db_, sql.c_str(), nullptr, nullptr, &err) != 0) {
    std::string msg = err ? err : "sqlite error";
    sqlite3_free(err);
    throw std::runtime_error(msg + " in: " + sql);
  }
}

std::vector<DbRow> Db::query(const std::string& sql,
                             const std::vector<DbValue>& binds) {
  sqlite3_stmt* st = nullptr;
  if (sqlite3_prepare_v2(db_, sql.c_str(), -1, &st, nullptr) != 0)
    throw std::runtime_error(std::string(sqlite3_errmsg(db_)) + " in: " +
                             sql);
  bind_all(db_, st, binds);
  std::vector<DbRow> rows;
  while (true) {
    int rc = sqlite3_step(st);
    if (rc == 101) break;           // SQLITE_DONE
    if (rc != 100) {                // SQLITE_ROW
      std::string msg = sqlite3_errmsg(db_);
      sqlite3_finalize(st);
      throw std::runtime_error(msg);
    }
    DbRow row;
    const int n = sqlite3_column_count(st);
    for (int c = 0; c < n; ++c) {
      const char* name = sqlite3_column_name(st, c);
      switch (sqlite3_column_type(st, c)) {
        case 1:   // INTEGER
          row[name] = Json((long)sqlite3_column_int64(st, c));
          break;
        case 2:   // FLOAT
          row[name] = Json(sqlite3_column_double(st, c));
          break;
        case 5:   // NULL
          row[name] = Json(nullptr);
          break;
        default: {
          const unsigned char* t = sqlite3_column_text(st, c);
          row[name] = Json(std::string(t ? (const char*)t : ""));
        }
      }
    }
    rows.push_back(std::move(row));
  }
  sqlite3_finalize(st);
  return rows;
}

long long Db::last_insert_rowid() { return sqlite3_last_insert_rowid(db_); }
int Db::changes() { return sqlite3_changes(db_); }

void Db::migrate(
    const std::string& module,
    const std::vector<std::pair<std::string, std::string>>& ms) {
  std::lock_guard<std::mutex> lk(mu_);
  exec("CREATE TABLE IF NOT EXISTS _migrations ("
       "module TEXT NOT NULL, name TEXT NOT NULL, applied_at TEXT,"
       "PRIMARY KEY (module, name))");
  // BEGIN IMMEDIATE takes the write lock up front — the single-node
  // analog of the reference's cross-process advisory lock
  exec("BEGIN IMMEDIATE");
  try {
    for (auto& [name, sql] : ms) {
      auto done = query(
          "SELECT 1 FROM _migrations WHERE module=?1 AND name=?2",
          {DbValue::S(module), DbValue::S(name)});
      if (!done.empty()) continue;
      exec(sql);
      sqlite3_stmt* st = nullptr;
      sqlite3_prepare_v2(db_,
                         "INSERT INTO _migrations VALUES (?1, ?2, "
                         "datetime('now'))",
                         -1, &st, nullptr);
      bind_all(db_, st, {DbValue::S(module), DbValue::S(name)});
      sqlite3_step(st);
      sqlite3_finalize(st);
    }
    exec("COMMIT");
  } catch (...) {
    exec("ROLLBACK");
    throw;
  }
}

// --------------------------------------------------------------- secure

std::string SecureConn::scope_sql(std::vector<DbValue>& binds) const {
  if (scope_.unrestricted) return "1=1";
  if (scope_.is_deny_all()) return "1=0";   // deny-all default
  std::string sql = tenant_col_ + " IN (";
  for (size_t i = 0; i < scope_.tenant_ids.size(); ++i) {
    if (i) sql += ",";
    sql += "?";
    binds.push_back(DbValue::S(scope_.tenant_ids[i]));
  }
  sql += ")";
  if (scope_.resource_ids && !resource_col_.empty()) {
    if (scope_.resource_ids->empty()) return "1=0";
    sql += " AND " + resource_col_ + " IN (";
    for (size_t i = 0; i < scope_.resource_ids->size(); ++i) {
      if (i) sql += ",";
      sql += "?";
      binds.push_back(DbValue::S((*scope_.resource_ids)[i]));
    }
    sql += ")";
  }
  return sql;
}

// opaque cursor: "v1|<order_value>|<rowid>" base64url-ish (no padding)
static std::string cur_encode(const std::string& s) {
  static const char* tbl =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";
  std::string out;
  const unsigned char* d = (const unsigned char*)s.data();
  for (size_t i = 0; i < s.size(); i += 3) {
    unsigned v = d[i] << 16;
    if (i + 1 < s.size()) v |= d[i + 1] << 8;
    if (i + 2 < s.size()) v |= d[i + 2];
    out.push_back(tbl[(v >> 18) & 63]);
    out.push_back(tbl[(v >> 12) & 63]);
    if (i + 1 < s.size()) out.push_back(tbl[(v >> 6) & 63]);
    if (i + 2 < s.size()) out.push_back(tbl[v & 63]);
  }
  return out;
}

static std::optional<std::string> cur_decode(const std::string& b) {
  auto val = [](char c) -> int {
    if (c >= 'A' && c <= 'Z') return c - 'A';
    if (c >= 'a' && c <= 'z') return c - 'a' + 26;
    if (c >= '0' && c <= '9') return c - '0' + 52;
    if (c == '-') return 62;
    if (c == '_') return 63;
    return -1;
  };
  std::string out;
  unsigned acc = 0;
  int bits = 0;
  for (char c : b) {
    int v = val(c);
    if (v < 0) return std::nullopt;
    acc = (acc << 6) | (unsigned)v;
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      out.push_back((char)((acc >> bits) & 0xff));
    }
  }
  return out;
}

SecureConn::Page SecureConn::select(
    const std::string& table, const std::string& extra_where,
    std::vector<DbValue> binds0, const std::string& order_by, bool desc,
    int limit, const std::optional<std::string>& cursor) {
  std::vector<DbValue> binds;
  std::string where = scope_sql(binds);
  for (auto& b : binds0) binds.push_back(b);
  if (!extra_where.empty()) where += " AND (" + extra_where + ")";
  const std::string ord = order_by.empty() ? "rowid" : order_by;
  const char* cmp = desc ? "<" : ">";
  if (cursor && !cursor->empty()) {
    auto raw = cur_decode(*cursor);
    if (!raw) throw std::runtime_error("bad cursor");
    // v1|<ord value>|<rowid>
    size_t p1 = raw->find('|'), p2 = raw->rfind('|');
    if (p1 == std::string::npos || p2 <= p1 ||
        raw->substr(0, p1) != "v1")
      throw std::runtime_error("bad cursor");
    const std::string ov = raw->substr(p1 + 1, p2 - p1 - 1);
    const std::string rid = raw->substr(p2 + 1);
    where += " AND (" + ord + " " + cmp + " ?";
    binds.push_back(DbValue::S(ov));
    where += " OR (" + ord + " = ? AND rowid " + cmp + " ?))";
    binds.push_back(DbValue::S(ov));
    binds.push_back(DbValue::I(atoll(rid.c_str())));
  }
  std::string sql = "SELECT rowid AS _rid, * FROM " + table + " WHERE " +
                    where + " ORDER BY " + ord + (desc ? " DESC" : "") +
                    ", rowid" + (desc ? " DESC" : "") + " LIMIT " +
                    std::to_string(limit + 1);
  std::lock_guard<std::mutex> lk(db_.mu());
  auto rows = db_.query(sql, binds);
  Page page;
  const bool more = (int)rows.size() > limit;
  if (more) rows.resize(limit);
  if (more && !rows.empty()) {
    const DbRow& last = rows.back();
    std::string ov;
    auto it = last.find(ord);
    if (it != last.end()) {
      ov = it->second.is_string() ? it->second.as_string()
                                  : it->second.dump();
    }
    long long rid = (long long)last.at("_rid").as_int(0);
    page.next_cursor =
        cur_encode("v1|" + ov + "|" + std::to_string(rid));
  }
  for (auto& r : rows) r.erase("_rid");
  page.items = std::move(rows);
  return page;
}

void SecureConn::insert(
    const std::string& table,
    const std::vector<std::pair<std::string, DbValue>>& cols) {
  // writes need an unambiguous tenant: a single-tenant scope (the tenant
  // column is FORCED from it), or allow_all WITH an explicit tenant col
  if (scope_.is_deny_all())
    throw std::runtime_error("insert under deny-all scope");
  if (!scope_.unrestricted && scope_.tenant_ids.size() != 1)
    throw std::runtime_error("insert requires a single-tenant scope");
  bool caller_tenant = false;
  std::string names, marks;
  std::vector<DbValue> binds;
  for (auto& [n, v] : cols) {
    if (n == tenant_col_) {
      if (!scope_.unrestricted) continue;   // forced from scope below
      caller_tenant = true;
    }
    if (!names.empty()) { names += ","; marks += ","; }
    names += n;
    marks += "?";
    binds.push_back(v);
  }
  if (!scope_.unrestricted) {
    names += std::string(names.empty() ? "" : ",") + tenant_col_;
    marks += std::string(marks.empty() ? "" : ",") + "?";
    binds.push_back(DbValue::S(scope_.tenant_ids[0]));
  } else if (!caller_tenant) {
    throw std::runtime_error(
        "unrestricted insert must name the tenant column explicitly");
  }
  std::string sql = "INSERT INTO " + table + " (" + names + ") VALUES (" +
                    marks + ")";
  std::lock_guard<std::mutex> lk(db_.mu());
  db_.query(sql, binds);
}

int SecureConn::update(
    const std::string& table,
    const std::vector<std::pair<std::string, DbValue>>& sets,
    const std::string& extra_where, std::vector<DbValue> binds0) {
  std::string set_sql;
  std::vector<DbValue> binds;
  for (auto& [n, v] : sets) {
    if (!set_sql.empty()) set_sql += ",";
    set_sql += n + "=?";
    binds.push_back(v);
  }
  std::string where = scope_sql(binds);
  for (auto& b : binds0) binds.push_back(b);
  if (!extra_where.empty()) where += " AND (" + extra_where + ")";
  std::lock_guard<std::mutex> lk(db_.mu());
  db_.query("UPDATE " + table + " SET " + set_sql + " WHERE " + where,
            binds);
  return db_.changes();
}

int SecureConn::remove(const std::string& table,
                       const std::string& extra_where,
                       std::vector<DbValue> binds0) {
  std::vector<DbValue> binds;
  std::string where = scope_sql(binds);
  for (auto& b : binds0) binds.push_back(b);
  if (!extra_where.empty()) where += " AND (" + extra_where + ")";
  std::lock_guard<std::mutex> lk(db_.mu());
  db_.query("DELETE FROM " + table + " WHERE " + where, binds);
  return db_.changes();
}

// ------------------------------------------------------- $filter compiler

static void skip_ws(const std::string& s, size_t& i) {
  while (i < s.size() && s[i] == ' ') ++i;
}

static std::string take_token(const std::string& s, size_t& i) {
  skip_ws(s, i);
  size_t start = i;
  while (i < s.size() && (isalnum((unsigned char)s[i]) || s[i] == '_'))
    ++i;
  return s.substr(start, i - start);
}

std::string compile_odata_filter(const std::string& filter,
                                 const std::vector<std::string>& fields,
                                 std::vector<DbValue>& binds) {
  auto allowed = [&](const std::string& f) {
    for (auto& x : fields)
      if (x == f) return true;
    throw std::runtime_error("field not filterable: " + f);
  };
  auto parse_literal = [&](const std::string& s, size_t& i) -> DbValue {
    skip_ws(s, i);
    if (i < s.size() && s[i] == '\'') {
      std::string v;
      ++i;
      while (i < s.size()) {
        if (s[i] == '\'') {
          if (i + 1 < s.size() && s[i + 1] == '\'') { v += '\''; i += 2; }
          else { ++i; break; }
        } else v += s[i++];
      }
      return DbValue::S(v);
    }
    size_t start = i;
    while (i < s.size() && s[i] != ' ' && s[i] != ')') ++i;
    std::string v = s.substr(start, i - start);
    if (v == "true") return DbValue::I(1);
    if (v == "false") return DbValue::I(0);
    if (v == "null") return DbValue::null();
    return DbValue::R(atof(v.c_str()));
  };

  std::string sql;
  size_t i = 0;
  while (i < filter.size()) {
    skip_ws(filter, i);
    if (i >= filter.size()) break;
    if (!sql.empty()) {
      std::string conj = take_token(filter, i);
      if (conj != "and")
        throw std::runtime_error("only 'and' is supported in $filter");
      sql += " AND ";
    }
    skip_ws(filter, i);
    if (filter.compare(i, 9, "contains(") == 0) {
      i += 9;
      std::string f = take_token(filter, i);
      allowed(f);
      skip_ws(filter, i);
      if (i >= filter.size() || filter[i] != ',')
        throw std::runtime_error("bad contains()");
      ++i;
      DbValue v = parse_literal(filter, i);
      skip_ws(filter, i);
      if (i >= filter.size() || filter[i] != ')')
        throw std::runtime_error("bad contains()");
      ++i;
      sql += f + " LIKE ?";
      binds.push_back(DbValue::S("%" + v.text + "%"));
      continue;
    }
    std::string f = take_token(filter, i);
    if (f.empty()) throw std::runtime_error("bad $filter");
    allowed(f);
    std::string op = take_token(filter, i);
    const char* sqlop = nullptr;
    if (op == "eq") sqlop = "=";
    else if (op == "ne") sqlop = "!=";
    else if (op == "gt") sqlop = ">";
    else if (op == "ge") sqlop = ">=";
    else if (op == "lt") sqlop = "<";
    else if (op == "le") sqlop = "<=";
    else throw std::runtime_error("bad operator: " + op);
    DbValue v = parse_literal(filter, i);
    if (v.kind == DbValue::kNull) {
      sql += f + (op == "eq" ? " IS NULL" : " IS NOT NULL");
    } else {
      sql += f + std::string(sqlop) + "?";
      binds.push_back(v);
    }
  }
  return sql.empty() ? "1=1" : sql;
}

}  // namespace hs
