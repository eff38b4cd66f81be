#include "db.h"

#include <cstring>
#include <functional>
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
  OrderBy ob;
  if (!order_by.empty()) ob.emplace_back(order_by, desc);
  return select(table, extra_where, std::move(binds0), ob, limit, cursor);
}

SecureConn::Page SecureConn::select(
    const std::string& table, const std::string& extra_where,
    std::vector<DbValue> binds0, const OrderBy& order_by, int limit,
    const std::optional<std::string>& cursor) {
  std::vector<DbValue> binds;
  std::string where = scope_sql(binds);
  for (auto& b : binds0) binds.push_back(b);
  if (!extra_where.empty()) where += " AND (" + extra_where + ")";

  // deterministic total order: user keys then rowid tie-break (rowid
  // direction follows the first key so a single-key page reads like v1)
  OrderBy keys = order_by;
  const bool rdesc = keys.empty() ? false : keys[0].second;
  keys.emplace_back("rowid", rdesc);

  if (cursor && !cursor->empty()) {
    // CursorV2: "v2|<len>:<val>...<len>:<rowid>" base64url — one value
    // per order key + the rowid (modkit-odata CursorV1 semantics:
    // opaque, keyset, order-bound — lib.rs:353)
    auto raw = cur_decode(*cursor);
    if (!raw || raw->compare(0, 3, "v2|") != 0)
      throw std::runtime_error("bad cursor");
    std::vector<std::string> vals;
    size_t i = 3;
    while (i < raw->size()) {
      size_t colon = raw->find(':', i);
      if (colon == std::string::npos) throw std::runtime_error("bad cursor");
      long len = atol(raw->substr(i, colon - i).c_str());
      if (len < 0 || colon + 1 + (size_t)len > raw->size())
        throw std::runtime_error("bad cursor");
      vals.push_back(raw->substr(colon + 1, (size_t)len));
      i = colon + 1 + (size_t)len;
    }
    if (vals.size() != keys.size())
      throw std::runtime_error("cursor does not match $orderby");
    // lexicographic keyset predicate with per-key direction:
    // k0 cmp0 v0 OR (k0 = v0 AND (k1 cmp1 v1 OR (... rowid cmpN vN)))
    std::function<std::string(size_t)> pred = [&](size_t k) {
      const auto& [col, d] = keys[k];
      const char* cmp = d ? "<" : ">";
      DbValue v = col == "rowid"
                      ? DbValue::I(atoll(vals[k].c_str()))
                      : DbValue::S(vals[k]);
      if (k + 1 == keys.size()) {
        binds.push_back(v);
        return col + " " + cmp + " ?";
      }
      std::string out = col + " " + cmp + " ?";
      binds.push_back(v);
      out += " OR (" + col + " = ? AND (";
      binds.push_back(v);
      out += pred(k + 1) + "))";
      return out;
    };
    where += " AND (" + pred(0) + ")";
  }

  std::string ord_sql;
  for (auto& [col, d] : keys) {
    if (!ord_sql.empty()) ord_sql += ", ";
    ord_sql += col + (d ? " DESC" : "");
  }
  std::string sql = "SELECT rowid AS _rid, * FROM " + table + " WHERE " +
                    where + " ORDER BY " + ord_sql + " LIMIT " +
                    std::to_string(limit + 1);
  std::lock_guard<std::mutex> lk(db_.mu());
  auto rows = db_.query(sql, binds);
  Page page;
  const bool more = (int)rows.size() > limit;
  if (more) rows.resize(limit);
  if (more && !rows.empty()) {
    const DbRow& last = rows.back();
    std::string payload = "v2|";
    for (auto& [col, d] : keys) {
      std::string ov;
      if (col == "rowid") {
        ov = std::to_string((long long)last.at("_rid").as_int(0));
      } else {
        auto it = last.find(col);
        if (it != last.end())
          ov = it->second.is_string() ? it->second.as_string()
                                      : it->second.dump();
      }
      payload += std::to_string(ov.size()) + ":" + ov;
    }
    page.next_cursor = cur_encode(payload);
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
  // Recursive-descent $filter compiler (modkit-odata Expr AST parity,
  // /root/reference/libs/modkit-odata/src/lib.rs:23,:70):
  //   expr    := and_expr ('or' and_expr)*
  //   and_expr:= unary ('and' unary)*
  //   unary   := 'not' unary | '(' expr ')' | primary
  //   primary := contains|startswith|endswith '(' field ',' lit ')'
  //            | field (eq|ne|gt|ge|lt|le) literal
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
    while (i < s.size() && s[i] != ' ' && s[i] != ')' && s[i] != ',') ++i;
    std::string v = s.substr(start, i - start);
    if (v.empty()) throw std::runtime_error("bad literal");
    if (v == "true") return DbValue::I(1);
    if (v == "false") return DbValue::I(0);
    if (v == "null") return DbValue::null();
    return DbValue::R(atof(v.c_str()));
  };

  // recursion depth bound: adversarial '((((...' must not overflow the
  // stack (the reference fuzzes exactly this surface, fuzz_odata_filter)
  int depth = 0;
  std::function<std::string(size_t&)> parse_or;
  auto parse_func = [&](size_t& i, const char* name, size_t nlen,
                        const char* pre, const char* post) -> std::string {
    i += nlen;
    std::string f = take_token(filter, i);
    allowed(f);
    skip_ws(filter, i);
    if (i >= filter.size() || filter[i] != ',')
      throw std::runtime_error(std::string("bad ") + name + "()");
    ++i;
    DbValue v = parse_literal(filter, i);
    skip_ws(filter, i);
    if (i >= filter.size() || filter[i] != ')')
      throw std::runtime_error(std::string("bad ") + name + "()");
    ++i;
    // escape LIKE wildcards in the needle so user data matches literally
    std::string needle;
    for (char c : v.text) {
      if (c == '%' || c == '_' || c == '\\') needle += '\\';
      needle += c;
    }
    binds.push_back(DbValue::S(pre + needle + post));
    return f + " LIKE ? ESCAPE '\\'";
  };
  std::function<std::string(size_t&)> parse_unary =
      [&](size_t& i) -> std::string {
    skip_ws(filter, i);
    if (++depth > 32) throw std::runtime_error("$filter too deeply nested");
    std::string out;
    if (i < filter.size() && filter[i] == '(') {
      ++i;
      out = "(" + parse_or(i) + ")";
      skip_ws(filter, i);
      if (i >= filter.size() || filter[i] != ')')
        throw std::runtime_error("unbalanced parentheses");
      ++i;
    } else if (filter.compare(i, 4, "not ") == 0 ||
               filter.compare(i, 4, "not(") == 0) {
      i += 3;
      out = "NOT (" + parse_unary(i) + ")";
    } else if (filter.compare(i, 9, "contains(") == 0) {
      out = parse_func(i, "contains", 9, "%", "%");
    } else if (filter.compare(i, 11, "startswith(") == 0) {
      out = parse_func(i, "startswith", 11, "", "%");
    } else if (filter.compare(i, 9, "endswith(") == 0) {
      out = parse_func(i, "endswith", 9, "%", "");
    } else {
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
        if (op != "eq" && op != "ne")
          throw std::runtime_error("null only supports eq/ne");
        out = f + (op == "eq" ? " IS NULL" : " IS NOT NULL");
      } else {
        out = f + std::string(sqlop) + "?";
        binds.push_back(v);
      }
    }
    --depth;
    return out;
  };
  auto parse_and = [&](size_t& i) -> std::string {
    std::string out = parse_unary(i);
    while (true) {
      size_t save = i;
      skip_ws(filter, i);
      size_t t0 = i;
      std::string tok = take_token(filter, i);
      if (tok == "and") {
        out += " AND " + parse_unary(i);
      } else {
        i = (tok.empty() ? i : t0);
        if (tok.empty()) i = save;
        break;
      }
    }
    return out;
  };
  parse_or = [&](size_t& i) -> std::string {
    std::string out = parse_and(i);
    while (true) {
      size_t save = i;
      skip_ws(filter, i);
      size_t t0 = i;
      std::string tok = take_token(filter, i);
      if (tok == "or") {
        out += " OR " + parse_and(i);
      } else {
        i = (tok.empty() ? i : t0);
        if (tok.empty()) i = save;
        break;
      }
    }
    return out;
  };

  size_t i = 0;
  skip_ws(filter, i);
  if (i >= filter.size()) return "1=1";
  std::string sql = parse_or(i);
  skip_ws(filter, i);
  if (i < filter.size())
    throw std::runtime_error("trailing input in $filter at offset " +
                             std::to_string(i));
  return sql;
}

std::vector<std::pair<std::string, bool>> parse_odata_orderby(
    const std::string& orderby, const std::vector<std::string>& fields) {
  // "$orderby=a desc,b asc" (OData) and the reference's signed-token
  // encoding "-a,b" (modkit-odata lib.rs:135) both parse to the same
  // (field, desc) list; fields are allow-listed like $filter fields.
  auto allowed = [&](const std::string& f) {
    for (auto& x : fields)
      if (x == f) return true;
    throw std::runtime_error("field not orderable: " + f);
  };
  std::vector<std::pair<std::string, bool>> out;
  size_t i = 0;
  while (i < orderby.size()) {
    skip_ws(orderby, i);
    bool desc = false;
    if (i < orderby.size() && (orderby[i] == '-' || orderby[i] == '+')) {
      desc = orderby[i] == '-';
      ++i;
    }
    std::string f = take_token(orderby, i);
    if (f.empty()) throw std::runtime_error("bad $orderby");
    allowed(f);
    skip_ws(orderby, i);
    size_t save = i;
    std::string dir = take_token(orderby, i);
    if (dir == "desc") desc = true;
    else if (dir == "asc") desc = false;
    else i = save;
    out.emplace_back(f, desc);
    skip_ws(orderby, i);
    if (i < orderby.size()) {
      if (orderby[i] != ',')
        throw std::runtime_error("bad $orderby separator");
      ++i;
    }
    if (out.size() > 8) throw std::runtime_error("$orderby too long");
  }
  if (out.empty()) throw std::runtime_error("bad $orderby");
  return out;
}

}  // namespace hs
