#include "user_settings.h"

#include <sys/stat.h>
#include <ctime>

#include "system_modules.h"

namespace hs {

void UserSettingsModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  std::string file = ctx.full_config
                         .path("modules.simple-user-settings.database.file")
                         .as_string("");
  if (file.empty()) {
    // default under home_dir (DbManager per-module database resolution)
    std::string home = ctx.home_dir;
    if (!home.empty() && home[0] == '~') {
      const char* h = getenv("HOME");
      home = std::string(h ? h : "/tmp") + home.substr(1);
    }
    mkdir(home.c_str(), 0755);
    file = home + "/simple-user-settings.db";
  }
  db_ = std::make_unique<Db>(file);
  db_->migrate("simple-user-settings", {
      {"0001_create_settings",
       "CREATE TABLE settings ("
       "  tenant_id TEXT NOT NULL,"
       "  subject_id TEXT NOT NULL,"
       "  key TEXT NOT NULL,"
       "  value TEXT NOT NULL,"
       "  updated_at TEXT NOT NULL DEFAULT (datetime('now')),"
       "  UNIQUE (tenant_id, subject_id, key))"},
  });
}

// PDP -> AccessScope (authz-resolver evaluate; deny => Problem 403;
// constraints become the row scope — SURVEY.md §5.9 PDP/PEP model)
AccessScope UserSettingsModule::scope_for(const SecurityContext& sec,
                                          const std::string& action) {
  auto pdp = hub_->get<AuthzResolverClient>("authz-resolver");
  if (!pdp) return AccessScope::deny_all();
  EvaluationRequest er;
  er.subject = sec;
  er.action = action;
  er.resource = "simple-user-settings:settings";
  er.tenant_id = sec.tenant_id;
  EvaluationResponse r = pdp->evaluate(er);
  if (!r.allow)
    throw Problem{403, "Forbidden", "about:blank",
                  r.deny_reason.empty() ? "access denied" : r.deny_reason,
                  "pdp_deny"};
  AccessScope sc = r.tenant_scope.empty()
                       ? AccessScope::for_tenant(sec.tenant_id)
                       : AccessScope::for_tenants(r.tenant_scope);
  sc.resource_ids = std::vector<std::string>{sec.subject_id};
  return sc;
}

static SecurityContext sec_of(HttpRequest& rq) {
  return SecurityContext::from_json(rq.extensions.at("security"));
}

void UserSettingsModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  const std::vector<std::string> filterable = {"key", "updated_at"};

  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/simple-user-settings/v1/settings";
    op.operation_id = "list_settings";
    op.summary = "List settings ($filter, $top, cursor)";
    op.odata_filter_fields = filterable;
    op.authenticated = true;
    op.tags = {"simple-user-settings"};
    rest.register_op(op, [this, filterable](HttpRequest& rq,
                                            ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "read"), "tenant_id",
                      "subject_id");
      std::vector<DbValue> binds;
      std::string where;
      auto fit = rq.query.find("$filter");
      if (fit != rq.query.end()) {
        try {
          where = compile_odata_filter(fit->second, filterable, binds);
        } catch (const std::exception& e) {
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
      SecureConn::OrderBy ob{{"key", false}};
      auto oit = rq.query.find("$orderby");
      SecureConn::Page page;
      try {
        if (oit != rq.query.end())
          ob = parse_odata_orderby(oit->second, filterable);
        page = conn.select("settings", where, binds, ob, top, cursor);
      } catch (const std::exception& e) {
        throw Problem{400, "Bad Request", "about:blank", e.what(),
                      "validation_error"};
      }
      Json items = Json::array();
      for (auto& r : page.items) {
        Json it = Json::object();
        it["key"] = r["key"];
        try { it["value"] = Json::parse(r["value"].as_string()); }
        catch (...) { it["value"] = r["value"]; }
        it["updated_at"] = r["updated_at"];
        items.push_back(it);
      }
      // Page envelope (modkit-odata page.rs:5-16)
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
    op.path = "/simple-user-settings/v1/settings/{key}";
    op.operation_id = "get_setting";
    op.summary = "Get one setting";
    op.authenticated = true;
    op.tags = {"simple-user-settings"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "read"), "tenant_id",
                      "subject_id");
      auto page = conn.select("settings", "key=?",
                              {DbValue::S(rq.path_params.at("key"))},
                              "key", false, 1, std::nullopt);
      if (page.items.empty()) throw Problem::not_found("no such setting");
      Json out = Json::object();
      out["key"] = page.items[0]["key"];
      try {
        out["value"] = Json::parse(page.items[0]["value"].as_string());
      } catch (...) { out["value"] = page.items[0]["value"]; }
      out["updated_at"] = page.items[0]["updated_at"];
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "PUT";
    op.path = "/simple-user-settings/v1/settings/{key}";
    op.operation_id = "put_setting";
    op.summary = "Create/update a setting";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"simple-user-settings"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      if (!body.contains("value"))
        throw Problem{400, "Bad Request", "about:blank",
                      "'value' is required", "validation_error"};
      const std::string key = rq.path_params.at("key");
      SecureConn conn(*db_, scope_for(sec, "write"), "tenant_id",
                      "subject_id");
      const std::string vjson = body.at("value").dump();
      char ts[32];
      time_t now = time(nullptr);
      strftime(ts, sizeof ts, "%Y-%m-%d %H:%M:%S", gmtime(&now));
      int changed = conn.update(
          "settings",
          {{"value", DbValue::S(vjson)},
           {"updated_at", DbValue::S(ts)}},
          "key=?", {DbValue::S(key)});
      if (!changed) {
        conn.insert("settings", {{"subject_id", DbValue::S(sec.subject_id)},
                                 {"key", DbValue::S(key)},
                                 {"value", DbValue::S(vjson)}});
      }
      Json out = Json::object();
      out["key"] = key;
      out["value"] = body.at("value");
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "DELETE";
    op.path = "/simple-user-settings/v1/settings/{key}";
    op.operation_id = "delete_setting";
    op.summary = "Delete a setting";
    op.authenticated = true;
    op.tags = {"simple-user-settings"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = sec_of(rq);
      SecureConn conn(*db_, scope_for(sec, "write"), "tenant_id",
                      "subject_id");
      int n = conn.remove("settings", "key=?",
                          {DbValue::S(rq.path_params.at("key"))});
      if (!n) throw Problem::not_found("no such setting");
      w.respond(204, "application/json", "");
    });
  }
}

}  // namespace hs
