#include "auth.h"

#include <unistd.h>

#include <openssl/bio.h>
#include <openssl/bn.h>
#include <openssl/evp.h>
#include <openssl/hmac.h>
#include <openssl/pem.h>
#include <openssl/rsa.h>

#include <chrono>
#include <ctime>
#include <vector>

#include "../http/client.h"

namespace hs {

namespace {

int b64url_val(char c) {
  if (c >= 'A' && c <= 'Z') return c - 'A';
  if (c >= 'a' && c <= 'z') return c - 'a' + 26;
  if (c >= '0' && c <= '9') return c - '0' + 52;
  if (c == '-' || c == '+') return 62;
  if (c == '_' || c == '/') return 63;
  return -1;
}

std::optional<std::string> b64url_decode(const std::string& s) {
  std::string out;
  unsigned acc = 0;
  int bits = 0;
  for (char c : s) {
    if (c == '=') break;
    int v = b64url_val(c);
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

std::string b64url_encode(const unsigned char* d, size_t n) {
  static const char* tbl =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";
  std::string out;
  for (size_t i = 0; i < n; i += 3) {
    unsigned v = d[i] << 16;
    if (i + 1 < n) v |= d[i + 1] << 8;
    if (i + 2 < n) v |= d[i + 2];
    out.push_back(tbl[(v >> 18) & 63]);
    out.push_back(tbl[(v >> 12) & 63]);
    if (i + 1 < n) out.push_back(tbl[(v >> 6) & 63]);
    if (i + 2 < n) out.push_back(tbl[v & 63]);
  }
  return out;
}

bool hmac_sha256(const std::string& key, const std::string& msg,
                 unsigned char out[32]) {
  unsigned int len = 32;
  return HMAC(EVP_sha256(), key.data(), (int)key.size(),
              (const unsigned char*)msg.data(), msg.size(), out,
              &len) != nullptr;
}

bool rs256_verify(const std::string& pem, const std::string& msg,
                  const std::string& sig) {
  BIO* bio = BIO_new_mem_buf(pem.data(), (int)pem.size());
  if (!bio) return false;
  EVP_PKEY* pkey = PEM_read_bio_PUBKEY(bio, nullptr, nullptr, nullptr);
  BIO_free(bio);
  if (!pkey) return false;
  EVP_MD_CTX* ctx = EVP_MD_CTX_new();
  bool ok = false;
  if (ctx &&
      EVP_DigestVerifyInit(ctx, nullptr, EVP_sha256(), nullptr, pkey) == 1 &&
      EVP_DigestVerify(ctx, (const unsigned char*)sig.data(), sig.size(),
                       (const unsigned char*)msg.data(), msg.size()) == 1)
    ok = true;
  if (ctx) EVP_MD_CTX_free(ctx);
  EVP_PKEY_free(pkey);
  return ok;
}

bool const_eq(const std::string& a, const std::string& b) {
  if (a.size() != b.size()) return false;
  unsigned char d = 0;
  for (size_t i = 0; i < a.size(); ++i)
    d |= (unsigned char)(a[i] ^ b[i]);
  return d == 0;
}

}  // namespace

// ------------------------------------------------------------ JwksCache

namespace {

// parse http(s)://host[:port]/path
bool parse_http_url(const std::string& url, std::string* host, int* port,
                    std::string* path, bool* https) {
  *https = url.rfind("https://", 0) == 0;
  if (!*https && url.rfind("http://", 0) != 0) return false;
  std::string rest = url.substr(*https ? 8 : 7);
  size_t slash = rest.find('/');
  std::string hp = slash == std::string::npos ? rest : rest.substr(0, slash);
  *path = slash == std::string::npos ? "/" : rest.substr(slash);
  size_t colon = hp.find(':');
  if (colon == std::string::npos) {
    *host = hp;
    *port = *https ? 443 : 80;
  } else {
    *host = hp.substr(0, colon);
    *port = atoi(hp.c_str() + colon + 1);
  }
  return !host->empty() && *port > 0;
}

// JWK (kty=RSA, base64url n/e) -> PEM SubjectPublicKeyInfo
#pragma GCC diagnostic push
#pragma GCC diagnostic ignored "-Wdeprecated-declarations"
std::string jwk_rsa_to_pem(const std::string& n_b64,
                           const std::string& e_b64) {
  auto n_raw = b64url_decode(n_b64);
  auto e_raw = b64url_decode(e_b64);
  if (!n_raw || !e_raw || n_raw->empty() || e_raw->empty()) return "";
  BIGNUM* n = BN_bin2bn((const unsigned char*)n_raw->data(),
                        (int)n_raw->size(), nullptr);
  BIGNUM* e = BN_bin2bn((const unsigned char*)e_raw->data(),
                        (int)e_raw->size(), nullptr);
  RSA* rsa = RSA_new();
  if (!n || !e || !rsa || RSA_set0_key(rsa, n, e, nullptr) != 1) {
    if (rsa) RSA_free(rsa); else { BN_free(n); BN_free(e); }
    return "";
  }
  EVP_PKEY* pkey = EVP_PKEY_new();
  EVP_PKEY_assign_RSA(pkey, rsa);      // pkey owns rsa (and n/e)
  BIO* bio = BIO_new(BIO_s_mem());
  std::string pem;
  if (PEM_write_bio_PUBKEY(bio, pkey) == 1) {
    char* data = nullptr;
    long len = BIO_get_mem_data(bio, &data);
    pem.assign(data, (size_t)len);
  }
  BIO_free(bio);
  EVP_PKEY_free(pkey);
  return pem;
}
#pragma GCC diagnostic pop

std::optional<Json> http_get_json(const std::string& url,
                                  std::string* err,
                                  const std::string& ca_file = "") {
  std::string host, path;
  int port = 0;
  bool https = false;
  if (!parse_http_url(url, &host, &port, &path, &https)) {
    if (err) *err = "bad url: " + url;
    return std::nullopt;
  }
  TlsOpts tls;
  tls.enable = https;
  tls.ca_file = ca_file;
  // idempotent GET with bounded retry/backoff (the reference's
  // modkit-http retry layer, src/layers/retry.rs): transient connect
  // failures and 5xx from the IdP retry twice before surfacing
  std::optional<ClientResponse> resp;
  for (int attempt = 0; attempt < 3; ++attempt) {
    if (attempt) usleep(100000u << (attempt - 1));   // 100ms, 200ms
    resp = http_request(host, port, "GET", path,
                        {{"accept", "application/json"}}, "", 5000,
                        nullptr, nullptr, &tls);
    if (resp && resp->status < 500) break;
  }
  if (!resp || resp->status != 200) {
    if (err)
      *err = "fetch failed: " + url + " status " +
             std::to_string(resp ? resp->status : 0);
    return std::nullopt;
  }
  try {
    return Json::parse(resp->body);
  } catch (...) {
    if (err) *err = "bad JSON from " + url;
    return std::nullopt;
  }
}

double mono_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

}  // namespace

bool JwksCache::refresh(std::string* err) {
  // caller holds mu_
  last_attempt_ = mono_s();
  if (jwks_uri.empty()) {
    auto disc = http_get_json(discovery_url, err, tls_ca_file);
    if (!disc) return false;
    jwks_uri = disc->at("jwks_uri").as_string();
    issuer_ = disc->at("issuer").as_string();
    if (jwks_uri.empty()) {
      if (err) *err = "discovery document has no jwks_uri";
      return false;
    }
  }
  auto jwks = http_get_json(jwks_uri, err, tls_ca_file);
  if (!jwks) return false;
  const Json& keys = jwks->at("keys");
  if (!keys.is_array()) {
    if (err) *err = "JWKS has no keys[]";
    return false;
  }
  std::map<std::string, std::string> fresh;
  for (auto& k : keys.arr()) {
    if (k.at("kty").as_string() != "RSA") continue;
    const std::string use = k.at("use").as_string("sig");
    if (use != "sig") continue;
    const std::string alg = k.at("alg").as_string("RS256");
    if (alg != "RS256") continue;
    const std::string kid = k.at("kid").as_string();
    std::string pem = jwk_rsa_to_pem(k.at("n").as_string(),
                                     k.at("e").as_string());
    if (!kid.empty() && !pem.empty()) fresh[kid] = pem;
  }
  keys_ = std::move(fresh);
  fetched_at_ = mono_s();
  return true;
}

std::optional<std::string> JwksCache::key_for(const std::string& kid,
                                              std::string* err) {
  std::lock_guard<std::mutex> lk(mu_);
  const double now = mono_s();
  auto it = keys_.find(kid);
  if (it != keys_.end() && now - fetched_at_ < ttl_s) return it->second;
  // stale hit, or miss (possible key rotation): re-fetch with cool-down
  if (now - last_attempt_ >= rotate_cooldown_s || fetched_at_ == 0) {
    std::string ferr;
    if (!refresh(&ferr)) {
      // keep serving a stale hit if the IdP is briefly unreachable
      if (it != keys_.end()) return it->second;
      if (err) *err = ferr;
      return std::nullopt;
    }
  }
  it = keys_.find(kid);
  if (it != keys_.end()) return it->second;
  if (err) *err = "unknown kid: " + kid;
  return std::nullopt;
}

std::string JwksCache::discovered_issuer() {
  std::lock_guard<std::mutex> lk(mu_);
  if (issuer_.empty() && !discovery_url.empty() &&
      mono_s() - last_attempt_ >= rotate_cooldown_s) {
    std::string err;
    refresh(&err);
  }
  return issuer_;
}

std::optional<SecurityContext> JwtValidator::validate(
    const std::string& token, std::string* err) const {
  auto fail = [&](const char* m) {
    if (err) *err = m;
    return std::nullopt;
  };
  size_t d1 = token.find('.');
  size_t d2 = token.rfind('.');
  if (d1 == std::string::npos || d2 == d1) return fail("malformed token");
  const std::string signed_part = token.substr(0, d2);
  auto hdr_raw = b64url_decode(token.substr(0, d1));
  auto pl_raw = b64url_decode(token.substr(d1 + 1, d2 - d1 - 1));
  auto sig_raw = b64url_decode(token.substr(d2 + 1));
  if (!hdr_raw || !pl_raw || !sig_raw) return fail("bad base64url");
  Json hdr, pl;
  try {
    hdr = Json::parse(*hdr_raw);
    pl = Json::parse(*pl_raw);
  } catch (...) {
    return fail("bad JSON");
  }
  const std::string alg = hdr.at("alg").as_string();
  if (alg == "HS256") {
    if (hs256_secret.empty()) return fail("HS256 not configured");
    unsigned char mac[32];
    if (!hmac_sha256(hs256_secret, signed_part, mac))
      return fail("hmac failed");
    if (!const_eq(std::string((char*)mac, 32), *sig_raw))
      return fail("bad signature");
  } else if (alg == "RS256") {
    std::string pem = rs256_public_pem;
    const std::string kid = hdr.at("kid").as_string();
    if (jwks && jwks->configured() && !kid.empty()) {
      std::string jerr;
      auto k = jwks->key_for(kid, &jerr);
      if (!k) {
        if (err) *err = "jwks: " + jerr;
        return std::nullopt;
      }
      pem = *k;
    }
    if (pem.empty()) return fail("RS256 not configured");
    if (!rs256_verify(pem, signed_part, *sig_raw))
      return fail("bad signature");
  } else {
    return fail("unsupported alg");   // incl. alg=none — always rejected
  }
  const long now = (long)time(nullptr);
  if (pl.contains("exp") && now > pl.at("exp").as_int(0) + leeway_s)
    return fail("token expired");
  if (pl.contains("nbf") && now + leeway_s < pl.at("nbf").as_int(0))
    return fail("token not yet valid");
  std::string want_iss = issuer;
  if (want_iss.empty() && jwks) want_iss = jwks->discovered_issuer();
  if (!want_iss.empty() && pl.at("iss").as_string() != want_iss)
    return fail("wrong issuer");
  if (!audience.empty()) {
    const Json& aud = pl.at("aud");
    bool ok = aud.as_string() == audience;
    if (!ok && aud.is_array())
      for (auto& a : aud.arr())
        if (a.as_string() == audience) ok = true;
    if (!ok) return fail("wrong audience");
  }
  SecurityContext c;
  c.subject_id = pl.at("sub").as_string();
  if (c.subject_id.empty()) return fail("missing sub");
  c.tenant_id = pl.at(tenant_claim).as_string(kDefaultTenantId);
  c.subject_type = pl.at("typ").as_string("user");
  // scopes: RFC8693-style space-separated "scope" or array "scp"
  const std::string sc = pl.at("scope").as_string();
  size_t p = 0;
  while (p < sc.size()) {
    size_t q = sc.find(' ', p);
    if (q == std::string::npos) q = sc.size();
    if (q > p) c.scopes.push_back(sc.substr(p, q - p));
    p = q + 1;
  }
  if (pl.at("scp").is_array())
    for (auto& x : pl.at("scp").arr()) c.scopes.push_back(x.as_string());
  return c;
}

std::string JwtValidator::sign_hs256(const Json& claims) const {
  const std::string hdr = R"({"alg":"HS256","typ":"JWT"})";
  std::string part =
      b64url_encode((const unsigned char*)hdr.data(), hdr.size()) + "." +
      [&] {
        std::string p = claims.dump();
        return b64url_encode((const unsigned char*)p.data(), p.size());
      }();
  unsigned char mac[32];
  hmac_sha256(hs256_secret, part, mac);
  return part + "." + b64url_encode(mac, 32);
}

}  // namespace hs
