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

#include <optional>
#include <string>

#include "modkit.h"

namespace hs {

struct JwtValidator {
  std::string hs256_secret;       // enables HS256 when non-empty
  std::string rs256_public_pem;   // enables RS256 when non-empty
  std::string issuer;             // checked when non-empty
  std::string audience;           // checked when non-empty
  std::string tenant_claim = "tid";
  int leeway_s = 30;

  bool configured() const {
    return !hs256_secret.empty() || !rs256_public_pem.empty();
  }
  // nullopt + err set on any failure (never throws)
  std::optional<SecurityContext> validate(const std::string& token,
                                          std::string* err = nullptr) const;

  // test helper: mint an HS256 token from a claims object
  std::string sign_hs256(const Json& claims) const;
};

}  // namespace hs
