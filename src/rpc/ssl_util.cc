// brpc_amd: TLS utilities over OpenSSL (see ssl_util.h).
#include "rpc/ssl_util.h"

#include <errno.h>
#include <openssl/bio.h>
#include <openssl/ec.h>
#include <openssl/err.h>
#include <openssl/evp.h>
#include <openssl/pem.h>
#include <openssl/ssl.h>
#include <openssl/x509.h>

#include <mutex>

namespace bam {
namespace ssl {

namespace {

thread_local char tls_err[256];

void capture_err(const char* what) {
  unsigned long e = ERR_get_error();
  char buf[160] = {0};
  if (e != 0) ERR_error_string_n(e, buf, sizeof(buf));
  snprintf(tls_err, sizeof(tls_err), "%s: %s", what, buf[0] ? buf : "unknown");
  ERR_clear_error();
}

void init_openssl() {
  static std::once_flag once;
  std::call_once(once, [] {
    SSL_library_init();
    SSL_load_error_strings();
  });
}

bool looks_like_pem(const std::string& s) { return s.rfind("-----BEGIN", 0) == 0; }

bool use_cert(SSL_CTX* ctx, const std::string& cert, const std::string& key) {
  if (looks_like_pem(cert)) {
    BIO* bio = BIO_new_mem_buf(cert.data(), (int)cert.size());
    X509* x = PEM_read_bio_X509(bio, nullptr, nullptr, nullptr);
    BIO_free(bio);
    if (x == nullptr || SSL_CTX_use_certificate(ctx, x) != 1) {
      if (x != nullptr) X509_free(x);
      capture_err("use_certificate");
      return false;
    }
    X509_free(x);
  } else if (SSL_CTX_use_certificate_chain_file(ctx, cert.c_str()) != 1) {
    capture_err("use_certificate_chain_file");
    return false;
  }
  if (looks_like_pem(key)) {
    BIO* bio = BIO_new_mem_buf(key.data(), (int)key.size());
    EVP_PKEY* pk = PEM_read_bio_PrivateKey(bio, nullptr, nullptr, nullptr);
    BIO_free(bio);
    if (pk == nullptr || SSL_CTX_use_PrivateKey(ctx, pk) != 1) {
      if (pk != nullptr) EVP_PKEY_free(pk);
      capture_err("use_private_key");
      return false;
    }
    EVP_PKEY_free(pk);
  } else if (SSL_CTX_use_PrivateKey_file(ctx, key.c_str(), SSL_FILETYPE_PEM) != 1) {
    capture_err("use_private_key_file");
    return false;
  }
  if (SSL_CTX_check_private_key(ctx) != 1) {
    capture_err("check_private_key");
    return false;
  }
  return true;
}

}  // namespace

const char* LastError() { return tls_err; }

void* NewServerCtx(const std::string& cert, const std::string& key) {
  init_openssl();
  SSL_CTX* ctx = SSL_CTX_new(TLS_server_method());
  if (ctx == nullptr) {
    capture_err("SSL_CTX_new");
    return nullptr;
  }
  SSL_CTX_set_mode(ctx, SSL_MODE_ENABLE_PARTIAL_WRITE | SSL_MODE_ACCEPT_MOVING_WRITE_BUFFER);
  if (!use_cert(ctx, cert, key)) {
    SSL_CTX_free(ctx);
    return nullptr;
  }
  return ctx;
}

void* NewClientCtx() {
  init_openssl();
  SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
  if (ctx == nullptr) {
    capture_err("SSL_CTX_new");
    return nullptr;
  }
  SSL_CTX_set_mode(ctx, SSL_MODE_ENABLE_PARTIAL_WRITE | SSL_MODE_ACCEPT_MOVING_WRITE_BUFFER);
  SSL_CTX_set_verify(ctx, SSL_VERIFY_NONE, nullptr);
  return ctx;
}

void* NewSsl(void* ctx, int fd, bool client) {
  SSL* s = SSL_new((SSL_CTX*)ctx);
  if (s == nullptr) {
    capture_err("SSL_new");
    return nullptr;
  }
  if (SSL_set_fd(s, fd) != 1) {
    capture_err("SSL_set_fd");
    SSL_free(s);
    return nullptr;
  }
  if (client) {
    SSL_set_connect_state(s);
  } else {
    SSL_set_accept_state(s);
  }
  return s;
}

void FreeSsl(void* ssl) {
  if (ssl != nullptr) SSL_free((SSL*)ssl);
}

int HandshakeStep(void* vssl) {
  SSL* s = (SSL*)vssl;
  ERR_clear_error();
  int rc = SSL_do_handshake(s);
  if (rc == 1) return 1;
  int err = SSL_get_error(s, rc);
  if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) return 0;
  capture_err("SSL_do_handshake");
  return -1;
}

ssize_t Write(void* vssl, const void* data, size_t n) {
  SSL* s = (SSL*)vssl;
  ERR_clear_error();
  int rc = SSL_write(s, data, (int)n);
  if (rc > 0) return rc;
  int err = SSL_get_error(s, rc);
  if (err == SSL_ERROR_WANT_WRITE || err == SSL_ERROR_WANT_READ) {
    errno = EAGAIN;
    return -1;
  }
  capture_err("SSL_write");
  if (errno == 0) errno = EPIPE;
  return -1;
}

ssize_t Read(void* vssl, void* out, size_t n) {
  SSL* s = (SSL*)vssl;
  ERR_clear_error();
  int rc = SSL_read(s, out, (int)n);
  if (rc > 0) return rc;
  int err = SSL_get_error(s, rc);
  if (err == SSL_ERROR_ZERO_RETURN) return 0;  // clean TLS shutdown
  if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) {
    errno = EAGAIN;
    return -1;
  }
  if (err == SSL_ERROR_SYSCALL && rc == 0) return 0;  // peer closed without notify
  capture_err("SSL_read");
  if (errno == 0) errno = ECONNRESET;
  return -1;
}

int GenerateSelfSignedCert(std::string* cert_pem, std::string* key_pem, const std::string& cn) {
  init_openssl();
  EC_KEY* ec = EC_KEY_new_by_curve_name(NID_X9_62_prime256v1);
  if (ec == nullptr || EC_KEY_generate_key(ec) != 1) {
    capture_err("EC_KEY_generate_key");
    if (ec != nullptr) EC_KEY_free(ec);
    return -1;
  }
  EVP_PKEY* pk = EVP_PKEY_new();
  EVP_PKEY_assign_EC_KEY(pk, ec);  // pk owns ec now
  X509* x = X509_new();
  ASN1_INTEGER_set(X509_get_serialNumber(x), (long)1);
  X509_gmtime_adj(X509_get_notBefore(x), 0);
  X509_gmtime_adj(X509_get_notAfter(x), 365L * 24 * 3600);
  X509_set_pubkey(x, pk);
  X509_NAME* name = X509_get_subject_name(x);
  X509_NAME_add_entry_by_txt(name, "CN", MBSTRING_ASC, (const unsigned char*)cn.c_str(), -1,
                             -1, 0);
  X509_set_issuer_name(x, name);
  int rc = -1;
  if (X509_sign(x, pk, EVP_sha256()) != 0) {
    BIO* cb = BIO_new(BIO_s_mem());
    BIO* kb = BIO_new(BIO_s_mem());
    if (PEM_write_bio_X509(cb, x) == 1 && PEM_write_bio_PrivateKey(kb, pk, nullptr, nullptr, 0,
                                                                   nullptr, nullptr) == 1) {
      char* p = nullptr;
      long n = BIO_get_mem_data(cb, &p);
      cert_pem->assign(p, (size_t)n);
      n = BIO_get_mem_data(kb, &p);
      key_pem->assign(p, (size_t)n);
      rc = 0;
    } else {
      capture_err("PEM_write");
    }
    BIO_free(cb);
    BIO_free(kb);
  } else {
    capture_err("X509_sign");
  }
  X509_free(x);
  EVP_PKEY_free(pk);
  return rc;
}

}  // namespace ssl
}  // namespace bam
