// brpc_amd: client/server authentication.
// Parity: reference brpc/authenticator.h — the client generates a
// credential that rides RpcMeta.authentication_data (baidu_std field 7);
// the server verifies it once per connection and caches an AuthContext on
// the Socket, readable from Controller::auth_context().
// Delta vs reference: the credential is attached to EVERY request when
// auth is configured (not only the first on a connection) — wire-compatible
// (the field is optional) and immune to the concurrent-first-call race; the
// server still verifies only once per connection.
#pragma once

#include <string>

#include "base/endpoint.h"

namespace bam {

// Result of a successful verification, cached per connection.
class AuthContext {
 public:
  const std::string& user() const { return user_; }
  void set_user(const std::string& v) { user_ = v; }
  const std::string& group() const { return group_; }
  void set_group(const std::string& v) { group_ = v; }
  const std::string& roles() const { return roles_; }
  void set_roles(const std::string& v) { roles_ = v; }
  const std::string& starter() const { return starter_; }
  void set_starter(const std::string& v) { starter_ = v; }
  bool is_service() const { return is_service_; }
  void set_is_service(bool v) { is_service_ = v; }

 private:
  bool is_service_ = false;
  std::string user_;
  std::string group_;
  std::string roles_;
  std::string starter_;
};

class Authenticator {
 public:
  virtual ~Authenticator() = default;

  // Client side: fill `auth_str` with the credential to send. 0 = success.
  virtual int GenerateCredential(std::string* auth_str) const = 0;

  // Server side: verify `auth_str` from `client_addr`; may fill `out_ctx`.
  // 0 = accepted; nonzero rejects the request with ERPCAUTH.
  virtual int VerifyCredential(const std::string& auth_str, const EndPoint& client_addr,
                               AuthContext* out_ctx) const = 0;

  // Extra error text returned to rejected clients.
  virtual std::string GetUnauthorizedErrorText() const { return ""; }
};

// Built-in shared-secret authenticator ("user\0password" credential).
// NOTE: bound into Python as a ready-made C++ object because user-defined
// Python authenticators would have to run on protocol fibers, which never
// take the GIL by design (see usercode_pool.h).
class PasswordAuthenticator : public Authenticator {
 public:
  PasswordAuthenticator(std::string user, std::string password)
      : user_(std::move(user)), password_(std::move(password)) {}

  int GenerateCredential(std::string* auth_str) const override {
    auth_str->assign(user_);
    auth_str->push_back('\0');
    auth_str->append(password_);
    return 0;
  }

  int VerifyCredential(const std::string& auth_str, const EndPoint&,
                       AuthContext* out_ctx) const override {
    size_t sep = auth_str.find('\0');
    if (sep == std::string::npos) return -1;
    if (auth_str.compare(0, sep, user_) != 0 || auth_str.compare(sep + 1, std::string::npos,
                                                                 password_) != 0) {
      return -1;
    }
    out_ctx->set_user(user_);
    return 0;
  }

  std::string GetUnauthorizedErrorText() const override { return "bad user/password"; }

 private:
  std::string user_;
  std::string password_;
};

}  // namespace bam
