#include "users_info.h"

#include <sys/stat.h>

#include <ctime>

#include "system_modules.h"

namespace hs {

namespace {

std::string now_ts() {
  char ts[32];
  time_t now = time(nullptr);
  strftime(ts, sizeof ts, "%Y-%m-%d %H:%M:%S", gmtime(&now));
  return ts;
}

std::string gen_id(const char* prefix) {
  static std::atomic<uint64_t> ctr{1};
  return std::string(prefix) + "-" +
         std::to_string((unsigned long)time(nullptr)) + "-" +
         std::to_string(ctr.fetch_add(1));
}

class LocalUsersInfoClient : public UsersInfoClient {
 public:
  explicit LocalUsersInfoClient(UsersInfoModule* m) : m_(m) {}
  std::optional<Json> get_user(const std::string& tenant,
                               const std::string& id) override {
    SecureConn conn(*m_->db_, AccessScope::for_tenant(tenant));
    auto page = conn.select("users", "id=?", {DbValue::S(id)}, "id", false,
                            1, std::nullopt);
    if (page.items.empty()) return std::nullopt;
    Json u = Json::object();
    for (auto& [k, v] : page.items[0]) u[k] = v;
    return u;
  }
  long count_users(const std::string& tenant) override {
    SecureConn conn(*m_->db_, AccessScope::for_tenant(tenant));
    auto page = conn.select("users", "", {}, "id", false, 100000,
                            std::nullopt);
    return (long)page.items.size();
  }

 private:
  UsersInfoModule* m_;
};

}  // namespace

void UsersInfoModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  std::string file =
      ctx.full_config.path("modules.users-info.database.file").as_string("");
  if (file.empty()) {
    std::string home = ctx.home_dir;
    if (!home.empty() && home[0] == '~') {
      const char* h = getenv("HOME");
      home = std::string(h ? h : "/tmp") + home.substr(1);
    }
    mkdir(home.c_str(), 0755);
    file = home + "/users-info.db";
  }
  db_ = std::make_unique<Db>(file);
  // 4 migrations like the reference example (users/cities + indexes)
  db_->migrate("users-info", {
      {"0001_create_users",
       "CREATE TABLE users (id TEXT PRIMARY KEY, tenant_id TEXT NOT NULL,"
       " email TEXT NOT NULL, display_name TEXT, city_id TEXT,"
       " created_at TEXT NOT NULL)"},
      {"0002_create_cities",
       "CREATE TABLE cities (id TEXT PRIMARY KEY, tenant_id TEXT NOT NULL,"
       " name TEXT NOT NULL)"},
      {"0003_users_email_idx",
       "CREATE UNIQUE INDEX users_email ON users (tenant_id, email)"},
      {"0004_users_city_idx",
       "CREATE INDEX users_city ON users (city_id)"},
  });
  ctx.hub->register_client<UsersInfoClient>(
      "users-info", std::make_shared<LocalUsersInfoClient>(this));
}

AccessScope UsersInfoModule::scope_for(const SecurityContext& sec,
                                       const std::string& action) {
  auto pdp = hub_->get<AuthzResolverClient>("authz-resolver");
  if (!pdp) return AccessScope::deny_all();
  EvaluationRequest er;
  er.subject = sec;
  er.action = action;
  er.resource = "users-info:users";
  er.tenant_id = sec.tenant_id;
  EvaluationResponse r = pdp->evaluate(er);
  if (!r.allow)
    throw Problem{403, "Forbidden", "about:blank",
                  r.deny_reason.empty() ? "access denied" : r.deny_reason,
                  "pdp_deny"};
  return r.tenant_scope.empty() ? AccessScope::for_tenant(sec.tenant_id)
                                : AccessScope::for_tenants(r.tenant_scope);
}

static SecurityContext sec_of(HttpRequest& rq) {
  return SecurityContext::from_json(rq.extensions.at("security"));
}

static Json row_json(const DbRow& r) {
  Json o = Json::object();
  for (auto& [k, v] : r)
    if (k != "tenant_id") o[k] = v;
  return o;
}

void UsersInfoModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  const std::vector<std::string> filterable = {"email", "display_name",
                                               "city_id", "created_at"};
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/users-info/v1/users";
    op.operation_id = "create_user";
    op.summary = "Create a user";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      const std::string email = body.at("email").as_string();
      if (email.empty() || email.find('@') == std::string::npos)
        throw Problem{400, "Bad Request", "about:blank",
                      "valid 'email' is required", "validation_error"};
      SecureConn conn(*db_, scope_for(sec, "write"));
      Json u = Json::object();
      u["id"] = gen_id("user");
      u["email"] = email;
      u["display_name"] = body.at("display_name").as_string("");
      u["city_id"] = body.at("city_id").as_string("");
      u["created_at"] = now_ts();
      try {
        conn.insert("users",
                    {{"id", DbValue::S(u.at("id").as_string())},
                     {"email", DbValue::S(email)},
                     {"display_name",
                      DbValue::S(u.at("display_name").as_string())},
                     {"city_id", DbValue::S(u.at("city_id").as_string())},
                     {"created_at",
                      DbValue::S(u.at("created_at").as_string())}});
      } catch (const std::exception& e) {
        throw Problem{409, "Conflict", "about:blank",
                      "email already exists in tenant", "conflict"};
      }
      Json ev = Json::object();
      ev["type"] = "user.created";
      ev["user"] = u;
      events_.publish(ev);
      w.respond(201, "application/json", u.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/users-info/v1/users";
    op.operation_id = "list_users";
    op.summary = "List users ($filter/$top/cursor, Page envelope)";
    op.odata_filter_fields = filterable;
    op.authenticated = true;
    op.tags = {"users-info"};
    rest.register_op(op, [this, filterable](HttpRequest& rq,
                                            ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "read"));
      std::vector<DbValue> binds;
      std::string where;
      auto fit = rq.query.find("$filter");
      if (fit != rq.query.end()) {
        try { where = compile_odata_filter(fit->second, filterable, binds); }
        catch (const std::exception& e) {
          throw Problem{400, "Bad Request", "about:blank", e.what(),
                        "validation_error"};
        }
      }
      int top = 50;
      auto tit = rq.query.find("$top");
      if (tit != rq.query.end())
        top = std::max(1, std::min(1000, atoi(tit->second.c_str())));
      std::optional<std::string> cursor;
      auto cit = rq.query.find("cursor");
      if (cit != rq.query.end()) cursor = cit->second;
      // $orderby: multi-field signed ordering over the same allow-list
      SecureConn::OrderBy ob{{"email", false}};
      auto oit = rq.query.find("$orderby");
      SecureConn::Page page;
      try {
        if (oit != rq.query.end())
          ob = parse_odata_orderby(oit->second, filterable);
        page = conn.select("users", where, binds, ob, top, cursor);
      } catch (const std::exception& e) {
        throw Problem{400, "Bad Request", "about:blank", e.what(),
                      "validation_error"};
      }
      Json items = Json::array();
      for (auto& r : page.items) items.push_back(row_json(r));
      Json pi = Json::object();
      pi["limit"] = (long)top;
      if (page.next_cursor) pi["next_cursor"] = *page.next_cursor;
      Json out = Json::object();
      out["items"] = items;
      out["page_info"] = pi;
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/users-info/v1/users/events";
    op.operation_id = "user_events";
    op.summary = "SSE stream of user lifecycle events";
    op.authenticated = true;
    op.sse = true;
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      sec_of(rq);          // authn enforced by gateway; subscribe:
      events_.serve(w, 2000);
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/users-info/v1/users/{id}";
    op.operation_id = "get_user";
    op.summary = "Get a user";
    op.authenticated = true;
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "read"));
      auto page = conn.select("users", "id=?",
                              {DbValue::S(rq.path_params.at("id"))}, "id",
                              false, 1, std::nullopt);
      if (page.items.empty()) throw Problem::not_found("no such user");
      w.respond(200, "application/json", row_json(page.items[0]).dump());
    });
  }
  {
    OperationSpec op;
    op.method = "DELETE";
    op.path = "/users-info/v1/users/{id}";
    op.operation_id = "delete_user";
    op.summary = "Delete a user";
    op.authenticated = true;
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "write"));
      const std::string id = rq.path_params.at("id");
      int n = conn.remove("users", "id=?", {DbValue::S(id)});
      if (!n) throw Problem::not_found("no such user");
      Json ev = Json::object();
      ev["type"] = "user.deleted";
      ev["id"] = id;
      events_.publish(ev);
      w.respond(204, "application/json", "");
    });
  }
  // cities: tenant-scoped reference data
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/users-info/v1/cities";
    op.operation_id = "create_city";
    op.summary = "Create a city";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      if (body.at("name").as_string().empty())
        throw Problem::bad_request("'name' is required");
      SecureConn conn(*db_, scope_for(sec, "write"));
      const std::string id = gen_id("city");
      conn.insert("cities", {{"id", DbValue::S(id)},
                             {"name",
                              DbValue::S(body.at("name").as_string())}});
      Json out = Json::object();
      out["id"] = id;
      out["name"] = body.at("name");
      w.respond(201, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/users-info/v1/cities";
    op.operation_id = "list_cities";
    op.summary = "List cities";
    op.authenticated = true;
    op.tags = {"users-info"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "read"));
      auto page = conn.select("cities", "", {}, "name", false, 1000,
                              std::nullopt);
      Json items = Json::array();
      for (auto& r : page.items) items.push_back(row_json(r));
      Json out = Json::object();
      out["items"] = items;
      w.respond(200, "application/json", out.dump());
    });
  }
}

}  // namespace hs
