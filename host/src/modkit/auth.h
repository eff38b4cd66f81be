// modkit-auth equivalent: JWT validation (HS256 + RS256 via OpenSSL).
//
// The reference's modkit-auth library validates bearer JWTs through an
// AuthDispatcher (key providers + claims validation,
// libs/modkit-auth/src/{dispatcher,validation}.rs).  There is no network
// in this deployment, so keys come from config (shared HS256 secret or a
// PEM-encoded RS256 public key) instead of a JWKS fetch; claims
// validation (exp/nbf with leeway, optional iss/aud) matches the
// reference semantics.  sub -> subject_id, <tenant_claim> -> tenant_id,
// scope/scp -> scopes.
#pragma once

#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>

#include "modkit.h"

namespace hs {

// JWKS endpoint fetch + cache with key rotation (reference:
// libs/modkit-auth/src/providers/jwks.rs).  Keys are resolved by `kid`;
// an unknown kid triggers ONE re-fetch (rotation) with a cool-down so a
// flood of bad tokens cannot hammer the IdP.  `discovery_url` points at
// an OIDC discovery document (/.well-known/openid-configuration,
// libs/modkit-auth/src/oauth2/discovery.rs) and resolves jwks_uri + the
// expected issuer.  Plain HTTP (in-cluster IdP / TLS-terminating
// sidecar — this host's outbound client has no TLS).
class JwksCache {
 public:
  std::string jwks_uri;            // direct JWKS endpoint, or
  std::string discovery_url;       // resolve jwks_uri + issuer via OIDC
  std::string tls_ca_file;         // CA bundle for https IdPs
  int ttl_s = 300;                 // full-refresh interval
  int rotate_cooldown_s = 2;       // min gap between miss-driven fetches

  bool configured() const {
    return !jwks_uri.empty() || !discovery_url.empty();
  }
  // PEM public key for `kid`; refreshes on stale cache or unknown kid
  std::optional<std::string> key_for(const std::string& kid,
                                     std::string* err = nullptr);
  // issuer from the discovery document ("" until discovered)
  std::string discovered_issuer();

 private:
  bool refresh(std::string* err);
  std::mutex mu_;
  std::map<std::string, std::string> keys_;   // kid -> PEM
  std::string issuer_;
  double fetched_at_ = 0, last_attempt_ = 0;
};

struct JwtValidator {
  std::string hs256_secret;       // enables HS256 when non-empty
  std::string rs256_public_pem;   // enables RS256 when non-empty
  std::string issuer;             // checked when non-empty
  std::string audience;           // checked when non-empty
  std::string tenant_claim = "tid";
  int leeway_s = 30;
  std::shared_ptr<JwksCache> jwks;  // enables RS256-by-kid when set

  bool configured() const {
    return !hs256_secret.empty() || !rs256_public_pem.empty() ||
           (jwks && jwks->configured());
  }
  // nullopt + err set on any failure (never throws)
  std::optional<SecurityContext> validate(const std::string& token,
                                          std::string* err = nullptr) const;

  // test helper: mint an HS256 token from a claims object
  std::string sign_hs256(const Json& claims) const;
};

}  // namespace hs
