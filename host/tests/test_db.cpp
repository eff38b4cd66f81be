// C++ unit tests for the modkit platform layer (the reference's colocated
// `#[test]` layer): secure ORM scoping semantics, cursor
// pagination, $filter compilation, migrations, JWT validation.
// Assert-based; built+run by `make test-bin` via tests/test_host_unit.py.
#include <cassert>
#include <cstdio>
#include <ctime>

#include "../src/modkit/auth.h"
#include "../src/modkit/db.h"

using namespace hs;

static int checks = 0;
#define CHECK(x)                                                          \
  do {                                                                    \
    if (!(x)) {                                                           \
      fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #x);        \
      return 1;                                                           \
    }                                                                     \
    ++checks;                                                             \
  } while (0)

int test_secure_scoping() {
  Db db(":memory:");
  db.migrate("t", {{"0001",
                    "CREATE TABLE things (id TEXT, tenant_id TEXT,"
                    " owner TEXT, v INTEGER)"}});
  // migrations are idempotent
  db.migrate("t", {{"0001", "CREATE TABLE things_should_not_exist (x)"}});
  CHECK(db.query("SELECT name FROM sqlite_master WHERE name="
                 "'things_should_not_exist'").empty());

  SecureConn a(db, AccessScope::for_tenant("A"));
  SecureConn b(db, AccessScope::for_tenant("B"));
  for (int i = 0; i < 3; ++i)
    a.insert("things", {{"id", DbValue::S("a" + std::to_string(i))},
                        {"owner", DbValue::S("ua")},
                        {"v", DbValue::I(i)}});
  b.insert("things", {{"id", DbValue::S("b0")},
                      {"owner", DbValue::S("ub")},
                      {"v", DbValue::I(9)}});

  // tenant scoping
  CHECK(a.select("things", "", {}, "id", false, 10, std::nullopt)
            .items.size() == 3);
  CHECK(b.select("things", "", {}, "id", false, 10, std::nullopt)
            .items.size() == 1);
  // deny-all default: EMPTY scope sees nothing (secure/mod.rs:94-101)
  SecureConn deny(db, AccessScope::deny_all());
  CHECK(deny.select("things", "", {}, "id", false, 10, std::nullopt)
            .items.empty());
  // allow_all sees everything
  SecureConn root(db, AccessScope::allow_all());
  CHECK(root.select("things", "", {}, "id", false, 10, std::nullopt)
            .items.size() == 4);
  // resource scoping
  AccessScope ra = AccessScope::for_tenant("A");
  ra.resource_ids = std::vector<std::string>{"nobody"};
  SecureConn ares(db, ra, "tenant_id", "owner");
  CHECK(ares.select("things", "", {}, "id", false, 10, std::nullopt)
            .items.empty());
  // scoped update/delete cannot cross tenants
  CHECK(b.update("things", {{"v", DbValue::I(100)}}, "", {}) == 1);
  CHECK(a.select("things", "v=100", {}, "id", false, 10, std::nullopt)
            .items.empty());
  CHECK(b.remove("things", "id=?", {DbValue::S("a0")}) == 0);
  // insert requires a single-tenant scope
  bool threw = false;
  try { root.insert("things", {{"id", DbValue::S("x")}}); }
  catch (...) { threw = true; }
  CHECK(threw);
  return 0;
}

int test_cursor_pagination() {
  Db db(":memory:");
  db.exec("CREATE TABLE r (tenant_id TEXT, k TEXT, i INTEGER)");
  SecureConn c(db, AccessScope::for_tenant("T"));
  // duplicate order keys exercise the (k = ? AND rowid > ?) tiebreak
  for (int i = 0; i < 10; ++i)
    c.insert("r", {{"k", DbValue::S(i < 5 ? "dup" : "k" +
                                    std::to_string(i))},
                   {"i", DbValue::I(i)}});
  std::vector<long> seen;
  std::optional<std::string> cur;
  int rounds = 0;
  while (rounds++ < 20) {
    auto page = c.select("r", "", {}, "k", false, 3, cur);
    for (auto& row : page.items)
      seen.push_back(row.at("i").as_int());
    if (!page.next_cursor) break;
    cur = page.next_cursor;
  }
  CHECK(seen.size() == 10);
  for (int i = 0; i < 5; ++i) CHECK(seen[i] == i);       // dup block stable
  // bad cursor is a clean error, not UB
  bool threw = false;
  try { c.select("r", "", {}, "k", false, 3, std::string("@@bad@@")); }
  catch (...) { threw = true; }
  CHECK(threw);

  // multi-field signed ordering: k DESC then i ASC, cursor-stable
  SecureConn::OrderBy ob{{"k", true}, {"i", false}};
  std::vector<long> seen2;
  cur.reset();
  rounds = 0;
  while (rounds++ < 20) {
    auto page = c.select("r", "", {}, ob, 3, cur);
    for (auto& row : page.items)
      seen2.push_back(row.at("i").as_int());
    if (!page.next_cursor) break;
    cur = page.next_cursor;
  }
  CHECK(seen2.size() == 10);
  // k DESC puts k9..k5 first (9,8,7,6,5), then the dup block i ASC
  std::vector<long> want{9, 8, 7, 6, 5, 0, 1, 2, 3, 4};
  for (int i = 0; i < 10; ++i) CHECK(seen2[i] == want[i]);
  // a cursor from one ordering cannot be replayed under another
  auto p1 = c.select("r", "", {}, ob, 3, std::nullopt);
  threw = false;
  try { c.select("r", "", {}, "k", false, 3, p1.next_cursor); }
  catch (...) { threw = true; }
  CHECK(threw);
  return 0;
}

int test_filter_compile() {
  std::vector<DbValue> binds;
  std::string sql = compile_odata_filter(
      "k eq 'x' and i gt 3 and contains(k,'y')", {"k", "i"}, binds);
  CHECK(sql.find("k=?") != std::string::npos);
  CHECK(sql.find("i>?") != std::string::npos);
  CHECK(sql.find("k LIKE ?") != std::string::npos);
  CHECK(binds.size() == 3);
  bool threw = false;
  try {
    binds.clear();
    compile_odata_filter("secret eq 'x'", {"k"}, binds);  // not allowed
  } catch (...) { threw = true; }
  CHECK(threw);
  // or / not / grouping (modkit-odata Expr parity, lib.rs:23)
  binds.clear();
  sql = compile_odata_filter(
      "(k eq 'a' or k eq 'b') and not (i lt 2)", {"k", "i"}, binds);
  CHECK(sql.find(" OR ") != std::string::npos);
  CHECK(sql.find("NOT (") != std::string::npos);
  CHECK(sql.find("(k=? OR k=?)") != std::string::npos);
  CHECK(binds.size() == 3);
  // startswith/endswith; LIKE wildcards in the needle are escaped
  binds.clear();
  sql = compile_odata_filter("startswith(k,'a%_b')", {"k"}, binds);
  CHECK(sql.find("LIKE ? ESCAPE") != std::string::npos);
  CHECK(binds[0].text == "a\\%\\_b%");
  binds.clear();
  sql = compile_odata_filter("endswith(k,'z')", {"k"}, binds);
  CHECK(binds[0].text == "%z");
  // unbalanced parens / trailing garbage / deep nesting all throw
  for (const char* bad :
       {"(k eq 'a'", "k eq 'a' k eq 'b'", "k eq 'a')",
        "((((((((((((((((((((((((((((((((((k eq 'a'"}) {
    threw = false;
    try {
      binds.clear();
      compile_odata_filter(bad, {"k"}, binds);
    } catch (...) { threw = true; }
    CHECK(threw);
  }
  // $orderby: "a desc,b" and signed tokens "-a,+b"; allow-listed
  auto ob = parse_odata_orderby("k desc, i", {"k", "i"});
  CHECK(ob.size() == 2 && ob[0].first == "k" && ob[0].second &&
        ob[1].first == "i" && !ob[1].second);
  auto ob2 = parse_odata_orderby("-k,+i", {"k", "i"});
  CHECK(ob2 == ob);
  threw = false;
  try { parse_odata_orderby("secret", {"k"}); }
  catch (...) { threw = true; }
  CHECK(threw);
  // quoting: an embedded quote stays a literal (no injection)
  binds.clear();
  sql = compile_odata_filter("k eq 'a''; DROP TABLE r--'", {"k"}, binds);
  CHECK(binds.size() == 1 && binds[0].text.find("DROP") != std::string::npos);
  CHECK(sql.find("DROP") == std::string::npos);
  return 0;
}

int test_jwt() {
  JwtValidator v;
  v.hs256_secret = "unit-secret";
  v.issuer = "unit";
  Json claims = Json::object();
  claims["sub"] = "u1";
  claims["iss"] = "unit";
  claims["tid"] = "tenant-9";
  claims["exp"] = (long)(time(nullptr) + 300);
  claims["scope"] = "a b";
  std::string tok = v.sign_hs256(claims);
  std::string err;
  auto sc = v.validate(tok, &err);
  CHECK(sc.has_value());
  CHECK(sc->subject_id == "u1" && sc->tenant_id == "tenant-9");
  CHECK(sc->scopes.size() == 2 && sc->scopes[0] == "a");
  // tampered payload rejected
  std::string bad = tok;
  bad[bad.find('.') + 2] ^= 1;
  CHECK(!v.validate(bad, &err).has_value());
  // expired rejected
  claims["exp"] = (long)(time(nullptr) - 3600);
  CHECK(!v.validate(v.sign_hs256(claims), &err).has_value());
  CHECK(err == "token expired");
  // wrong issuer rejected
  claims["exp"] = (long)(time(nullptr) + 300);
  claims["iss"] = "evil";
  CHECK(!v.validate(v.sign_hs256(claims), &err).has_value());
  // alg=none style (unsigned) rejected
  CHECK(!v.validate("eyJhbGciOiJub25lIn0.e30.", &err).has_value());
  return 0;
}

int main() {
  if (test_secure_scoping()) return 1;
  if (test_cursor_pagination()) return 1;
  if (test_filter_compile()) return 1;
  if (test_jwt()) return 1;
  printf("ok: %d checks\n", checks);
  return 0;
}
