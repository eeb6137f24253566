// brpc_amd: TLS support utilities.
// Parity: reference brpc SSL support (ChannelSSLOptions/ServerSSLOptions,
// socket.cpp SSL read/write paths). Implemented on OpenSSL (the
// toolchain's headers; linked with rpath so the matching libssl travels
// with the interpreter environment). void* is used for SSL_CTX*/SSL* so
// socket.h stays OpenSSL-free.
#pragma once

#include <string>

namespace bam {
namespace ssl {

// Server context from PEM data or file paths (auto-detected: content
// starting with "-----BEGIN" is treated as in-memory PEM). nullptr on error.
void* NewServerCtx(const std::string& cert_pem_or_file, const std::string& key_pem_or_file);

// Client context; certificate verification is OFF by default (parity with
// the reference's default ChannelSSLOptions).
void* NewClientCtx();

// Wraps fd into a new SSL handle bound to ctx. client=true -> connect state.
void* NewSsl(void* ctx, int fd, bool client);
void FreeSsl(void* ssl);

// Drives the handshake one step (non-blocking fd).
// Returns 1 done, 0 wants more IO (again later), -1 fatal.
int HandshakeStep(void* ssl);

// Returns >0 bytes moved; 0 clean shutdown (read only); -1 with errno
// EAGAIN when the transport blocks, other errno on fatal error.
ssize_t Write(void* ssl, const void* data, size_t n);
ssize_t Read(void* ssl, void* out, size_t n);

// Self-signed EC P-256 certificate for tests/examples. 0 on success.
int GenerateSelfSignedCert(std::string* cert_pem, std::string* key_pem,
                           const std::string& cn = "localhost");

const char* LastError();  // thread-local textual OpenSSL error

}  // namespace ssl
}  // namespace bam
