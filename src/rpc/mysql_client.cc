// brpc_amd: MySQL client implementation (see mysql_client.h).
#include "rpc/mysql_client.h"

#include <errno.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <openssl/bio.h>
#include <openssl/evp.h>
#include <openssl/pem.h>
#include <openssl/rsa.h>
#include <openssl/sha.h>

#include "base/codecs.h"
#include "base/endpoint.h"

#include <vector>

namespace bam {

namespace {

// capability flags (protocol constants)
constexpr uint32_t kClientLongPassword = 0x1;
constexpr uint32_t kClientProtocol41 = 0x200;
constexpr uint32_t kClientSecureConnection = 0x8000;
constexpr uint32_t kClientPluginAuth = 0x80000;
constexpr uint32_t kClientConnectWithDb = 0x8;

constexpr uint8_t kComQuit = 0x01;
constexpr uint8_t kComInitDb = 0x02;
constexpr uint8_t kComQuery = 0x03;
constexpr uint8_t kComPing = 0x0e;

// mysql_native_password: SHA1(pwd) XOR SHA1(salt + SHA1(SHA1(pwd)))
std::string native_scramble(const std::string& password, const std::string& salt) {
  if (password.empty()) return "";
  std::string h1 = SHA1Hash(password);
  std::string h2 = SHA1Hash(h1);
  std::string h3 = SHA1Hash(salt + h2);
  std::string out(h1.size(), '\0');
  for (size_t i = 0; i < h1.size(); ++i) out[i] = h1[i] ^ h3[i];
  return out;
}

// caching_sha2_password fast path:
// XOR(SHA256(pwd), SHA256(SHA256(SHA256(pwd)) || nonce))
std::string sha256s(const std::string& in) {
  unsigned char d[SHA256_DIGEST_LENGTH];
  SHA256((const unsigned char*)in.data(), in.size(), d);
  return std::string((const char*)d, sizeof(d));
}
std::string caching_sha2_scramble(const std::string& password, const std::string& nonce) {
  if (password.empty()) return "";
  std::string h1 = sha256s(password);
  std::string h2 = sha256s(sha256s(h1) + nonce);
  std::string out(h1.size(), '\0');
  for (size_t i = 0; i < h1.size(); ++i) out[i] = h1[i] ^ h2[i];
  return out;
}
std::string scramble_for(const std::string& plugin, const std::string& password,
                         const std::string& nonce) {
  if (plugin == "caching_sha2_password") return caching_sha2_scramble(password, nonce);
  return native_scramble(password, nonce);
}

// length-encoded integer; returns bytes consumed, 0 on error, -1 for NULL.
int lenc_int(const char* p, size_t n, uint64_t* v) {
  if (n == 0) return 0;
  uint8_t first = (uint8_t)p[0];
  if (first < 0xfb) {
    *v = first;
    return 1;
  }
  if (first == 0xfb) return -1;  // NULL
  if (first == 0xfc) {
    if (n < 3) return 0;
    *v = (uint8_t)p[1] | ((uint64_t)(uint8_t)p[2] << 8);
    return 3;
  }
  if (first == 0xfd) {
    if (n < 4) return 0;
    *v = (uint8_t)p[1] | ((uint64_t)(uint8_t)p[2] << 8) | ((uint64_t)(uint8_t)p[3] << 16);
    return 4;
  }
  if (n < 9) return 0;
  uint64_t x = 0;
  memcpy(&x, p + 1, 8);
  *v = x;
  return 9;
}

// length-encoded string; consumed bytes or 0; null=true for NULL value.
int lenc_str(const char* p, size_t n, std::string* out, bool* null) {
  uint64_t len;
  int k = lenc_int(p, n, &len);
  if (k == -1) {
    *null = true;
    return 1;
  }
  if (k == 0 || n < (size_t)k + len) return 0;
  *null = false;
  out->assign(p + k, len);
  return k + (int)len;
}

}  // namespace

MysqlClient::~MysqlClient() { Close(); }

void MysqlClient::Close() {
  if (fd_ >= 0) {
    std::string quit(1, (char)kComQuit);
    write_packet(quit, 0);
    ::close(fd_);
    fd_ = -1;
  }
}

int MysqlClient::read_n(void* buf, size_t n) {
  char* p = (char*)buf;
  size_t got = 0;
  while (got < n) {
    struct pollfd pfd{fd_, POLLIN, 0};
    int pr = ::poll(&pfd, 1, timeout_ms_);
    if (pr <= 0) return -1;
    ssize_t r = ::recv(fd_, p + got, n - got, 0);
    if (r <= 0) return -1;
    got += (size_t)r;
  }
  return 0;
}

int MysqlClient::read_packet(std::string* payload, uint8_t* seq) {
  uint8_t head[4];
  if (read_n(head, 4) != 0) return -1;
  uint32_t len = head[0] | ((uint32_t)head[1] << 8) | ((uint32_t)head[2] << 16);
  *seq = head[3];
  payload->resize(len);
  if (len != 0 && read_n(&(*payload)[0], len) != 0) return -1;
  return 0;
}

int MysqlClient::write_packet(const std::string& payload, uint8_t seq) {
  uint8_t head[4] = {(uint8_t)(payload.size() & 0xff), (uint8_t)((payload.size() >> 8) & 0xff),
                     (uint8_t)((payload.size() >> 16) & 0xff), seq};
  std::string buf((const char*)head, 4);
  buf += payload;
  size_t off = 0;
  while (off < buf.size()) {
    ssize_t w = ::send(fd_, buf.data() + off, buf.size() - off, MSG_NOSIGNAL);
    if (w <= 0) {
      if (errno == EAGAIN || errno == EINTR) continue;
      return -1;
    }
    off += (size_t)w;
  }
  return 0;
}

int MysqlClient::Connect(const std::string& host, int port, const std::string& user,
                         const std::string& password, const std::string& db,
                         int timeout_ms) {
  Close();
  timeout_ms_ = timeout_ms;
  EndPoint ep;
  if (hostname2endpoint(host.c_str(), port, &ep) != 0) return -1;
  fd_ = ::socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd_ < 0) return -1;
  struct sockaddr_in sa;
  memset(&sa, 0, sizeof(sa));
  sa.sin_family = AF_INET;
  sa.sin_addr = ep.ip;
  sa.sin_port = htons((uint16_t)ep.port);
  if (::connect(fd_, (struct sockaddr*)&sa, sizeof(sa)) != 0) {
    Close();
    return -1;
  }
  int one = 1;
  setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));

  // ---- HandshakeV10 ----
  std::string hs;
  uint8_t seq;
  if (read_packet(&hs, &seq) != 0 || hs.size() < 33 || (uint8_t)hs[0] != 10) {
    Close();
    return -1;
  }
  size_t pos = 1;
  size_t z = hs.find('\0', pos);
  if (z == std::string::npos) {
    Close();
    return -1;
  }
  server_version_ = hs.substr(pos, z - pos);
  pos = z + 1;
  if (hs.size() < pos + 4 + 8 + 1 + 2) {
    Close();
    return -1;
  }
  pos += 4;  // thread id
  std::string salt = hs.substr(pos, 8);
  pos += 8 + 1;          // auth-plugin-data-part-1 + filler
  pos += 2;              // capability_flags_1
  if (hs.size() >= pos + 1 + 2 + 2 + 1 + 10) {
    pos += 1 + 2 + 2;    // charset + status + capability_flags_2
    uint8_t auth_len = (uint8_t)hs[pos];
    pos += 1 + 10;       // auth data len + reserved
    // auth-plugin-data-part-2: max(13, auth_len - 8), includes trailing '\0'
    size_t part2 = auth_len > 8 ? (size_t)auth_len - 8 : 13;
    if (part2 < 13) part2 = 13;
    if (hs.size() >= pos + part2) {
      salt += hs.substr(pos, part2 - 1);  // drop the trailing '\0'
    }
  }

  // server's auth plugin name trails the salt (if CLIENT_PLUGIN_AUTH)
  std::string server_plugin = "mysql_native_password";
  {
    size_t z2 = hs.find('\0', hs.size() > 64 ? hs.size() - 64 : 0);
    // simplest robust read: last NUL-terminated token of the packet
    size_t last_nul = hs.find_last_of('\0');
    if (last_nul != std::string::npos && last_nul + 1 <= hs.size()) {
      size_t prev = hs.find_last_of('\0', last_nul - 1);
      std::string tail = hs.substr(prev + 1, last_nul - prev - 1);
      if (tail == "caching_sha2_password" || tail == "mysql_native_password")
        server_plugin = tail;
    }
    (void)z2;
  }

  // ---- HandshakeResponse41 ----
  uint32_t caps = kClientLongPassword | kClientProtocol41 | kClientSecureConnection |
                  kClientPluginAuth;
  if (!db.empty()) caps |= kClientConnectWithDb;
  std::string resp;
  resp.append((const char*)&caps, 4);
  uint32_t max_packet = 16 << 20;
  resp.append((const char*)&max_packet, 4);
  resp.push_back(33);  // utf8_general_ci
  resp.append(23, '\0');
  resp.append(user);
  resp.push_back('\0');
  std::string scramble = scramble_for(server_plugin, password, salt);
  resp.push_back((char)scramble.size());
  resp.append(scramble);
  if (!db.empty()) {
    resp.append(db);
    resp.push_back('\0');
  }
  resp.append(server_plugin);
  resp.push_back('\0');
  if (write_packet(resp, (uint8_t)(seq + 1)) != 0) {
    Close();
    return -1;
  }

  // ---- auth continuation: OK / ERR / AuthSwitch (0xFE) / AuthMoreData (0x01)
  std::string current_nonce = salt;
  bool sent_pubkey_request = false;
  for (int hop = 0; hop < 6; ++hop) {
    std::string fin;
    if (read_packet(&fin, &seq) != 0 || fin.empty()) {
      Close();
      return -1;
    }
    uint8_t tag = (uint8_t)fin[0];
    if (tag == 0x00) return 0;  // OK
    if (tag == 0xff) {          // ERR
      int code = fin.size() >= 3 ? ((uint8_t)fin[1] | ((int)(uint8_t)fin[2] << 8)) : -1;
      Close();
      return code;
    }
    if (tag == 0xfe) {  // AuthSwitchRequest: plugin\0 nonce
      size_t z3 = fin.find('\0', 1);
      if (z3 == std::string::npos) {
        Close();
        return -1;
      }
      std::string plugin = fin.substr(1, z3 - 1);
      std::string nonce = fin.substr(z3 + 1);
      while (!nonce.empty() && nonce.back() == '\0') nonce.pop_back();
      current_nonce = nonce;
      std::string sc = scramble_for(plugin, password, nonce);
      if (write_packet(sc, (uint8_t)(seq + 1)) != 0) {
        Close();
        return -1;
      }
      continue;
    }
    if (tag == 0x01) {  // AuthMoreData (caching_sha2)
      if (fin.size() >= 2 && (uint8_t)fin[1] == 0x03) continue;  // fast auth ok -> OK next
      if (fin.size() >= 2 && (uint8_t)fin[1] == 0x04 && !sent_pubkey_request) {
        // FULL auth over plain TCP (parity: reference policy/mysql full
        // caching_sha2): request the server's RSA public key (0x02), then
        // send RSA-OAEP(password||NUL XOR nonce).
        sent_pubkey_request = true;
        if (write_packet(std::string(1, '\x02'), (uint8_t)(seq + 1)) != 0) {
          Close();
          return -1;
        }
        continue;
      }
      if (sent_pubkey_request && fin.size() > 1) {
        // AuthMoreData carrying the PEM public key.
        std::string pem = fin.substr(1);
        std::string plain = password;
        plain.push_back('\0');
        for (size_t i = 0; i < plain.size(); ++i)
          plain[i] = (char)(plain[i] ^ current_nonce[i % current_nonce.size()]);
        std::string enc;
        {
          BIO* bio = BIO_new_mem_buf(pem.data(), (int)pem.size());
          EVP_PKEY* pkey = bio != nullptr ? PEM_read_bio_PUBKEY(bio, nullptr, nullptr, nullptr)
                                          : nullptr;
          if (bio != nullptr) BIO_free(bio);
          EVP_PKEY_CTX* ctx = pkey != nullptr ? EVP_PKEY_CTX_new(pkey, nullptr) : nullptr;
          bool ok = ctx != nullptr && EVP_PKEY_encrypt_init(ctx) > 0 &&
                    EVP_PKEY_CTX_set_rsa_padding(ctx, RSA_PKCS1_OAEP_PADDING) > 0;
          size_t outlen = 0;
          if (ok)
            ok = EVP_PKEY_encrypt(ctx, nullptr, &outlen, (const uint8_t*)plain.data(),
                                  plain.size()) > 0;
          if (ok) {
            enc.resize(outlen);
            ok = EVP_PKEY_encrypt(ctx, (uint8_t*)&enc[0], &outlen,
                                  (const uint8_t*)plain.data(), plain.size()) > 0;
            enc.resize(outlen);
          }
          if (ctx != nullptr) EVP_PKEY_CTX_free(ctx);
          if (pkey != nullptr) EVP_PKEY_free(pkey);
          if (!ok) {
            Close();
            return -1;
          }
        }
        if (write_packet(enc, (uint8_t)(seq + 1)) != 0) {
          Close();
          return -1;
        }
        continue;
      }
      Close();
      return -1;
    }
    Close();
    return -1;
  }
  Close();
  return -1;
}

int MysqlClient::Ping() {
  if (fd_ < 0) return -1;
  if (write_packet(std::string(1, (char)kComPing), 0) != 0) return -1;
  std::string p;
  uint8_t seq;
  if (read_packet(&p, &seq) != 0 || p.empty()) return -1;
  return (uint8_t)p[0] == 0x00 ? 0 : -1;
}

// ---- prepared statements (binary protocol) ----

int64_t MysqlClient::Prepare(const std::string& sql, int* param_count) {
  if (fd_ < 0) return -1;
  std::string cmd(1, (char)0x16);  // COM_STMT_PREPARE
  cmd += sql;
  if (write_packet(cmd, 0) != 0) return -1;
  std::string p;
  uint8_t seq;
  if (read_packet(&p, &seq) != 0 || p.empty()) return -1;
  if ((uint8_t)p[0] == 0xff)
    return p.size() >= 3 ? ((uint8_t)p[1] | ((int64_t)(uint8_t)p[2] << 8)) : -1;
  if ((uint8_t)p[0] != 0x00 || p.size() < 12) return -1;
  // PREPARE_OK: statement_id u32, num_columns u16, num_params u16
  uint32_t stmt_id;
  memcpy(&stmt_id, p.data() + 1, 4);
  uint16_t ncols, nparams;
  memcpy(&ncols, p.data() + 5, 2);
  memcpy(&nparams, p.data() + 7, 2);
  if (param_count != nullptr) *param_count = nparams;
  // drain param + column definition packets (+ possible EOFs)
  for (int i = 0; i < nparams; ++i)
    if (read_packet(&p, &seq) != 0) return -1;
  if (nparams > 0 && (read_packet(&p, &seq) != 0)) return -1;  // EOF
  for (int i = 0; i < ncols; ++i)
    if (read_packet(&p, &seq) != 0) return -1;
  if (ncols > 0 && (read_packet(&p, &seq) != 0)) return -1;  // EOF
  return (int64_t)stmt_id;
}

int MysqlClient::ExecutePrepared(int64_t stmt_id, const std::vector<std::string>& params,
                                 MysqlResult* out) {
  *out = MysqlResult();
  if (fd_ < 0) return -1;
  std::string cmd(1, (char)0x17);  // COM_STMT_EXECUTE
  uint32_t sid = (uint32_t)stmt_id;
  cmd.append((const char*)&sid, 4);
  cmd.push_back(0);                         // CURSOR_TYPE_NO_CURSOR
  uint32_t iter = 1;
  cmd.append((const char*)&iter, 4);
  if (!params.empty()) {
    std::string null_bitmap((params.size() + 7) / 8, '\0');
    cmd += null_bitmap;
    cmd.push_back(1);  // new_params_bound
    for (size_t i = 0; i < params.size(); ++i) {
      cmd.push_back((char)0xfe);  // MYSQL_TYPE_STRING
      cmd.push_back(0);
    }
    for (const std::string& v : params) {
      // length-encoded string
      if (v.size() < 0xfb) {
        cmd.push_back((char)v.size());
      } else {
        cmd.push_back((char)0xfc);
        uint16_t l = (uint16_t)v.size();
        cmd.append((const char*)&l, 2);
      }
      cmd += v;
    }
  }
  if (write_packet(cmd, 0) != 0) return -1;
  std::string p;
  uint8_t seq;
  if (read_packet(&p, &seq) != 0 || p.empty()) return -1;
  uint8_t first = (uint8_t)p[0];
  if (first == 0xff) {
    out->error_code = p.size() >= 3 ? ((uint8_t)p[1] | ((int)(uint8_t)p[2] << 8)) : -1;
    out->error_message = p.substr(p.size() > 9 && p[3] == '#' ? 9 : 3);
    return out->error_code;
  }
  if (first == 0x00) {
    out->ok = true;
    size_t pos = 1;
    uint64_t v = 0;
    int k = lenc_int(p.data() + pos, p.size() - pos, &v);
    if (k > 0) {
      out->affected_rows = v;
      pos += (size_t)k;
      if (lenc_int(p.data() + pos, p.size() - pos, &v) > 0) out->last_insert_id = v;
    }
    return 0;
  }
  // binary resultset: column count, defs, EOF, rows (0x00-headed), EOF
  uint64_t ncols = 0;
  if (lenc_int(p.data(), p.size(), &ncols) <= 0 || ncols == 0 || ncols > 4096) return -1;
  std::vector<uint8_t> col_types;
  for (uint64_t i = 0; i < ncols; ++i) {
    if (read_packet(&p, &seq) != 0) return -1;
    // column def: catalog..org_name lenc strings, then 0x0c fixed block
    size_t pos = 0;
    std::string field;
    bool nul;
    for (int f = 0; f < 6; ++f) {
      int k = lenc_str(p.data() + pos, p.size() - pos, &field, &nul);
      if (k <= 0) break;
      if (f == 4) out->columns.push_back(field);
      pos += (size_t)k;
    }
    // fixed block: filler(1) charset(2) length(4) type(1) ...
    col_types.push_back(pos + 8 < p.size() ? (uint8_t)p[pos + 7] : 0xfe);
  }
  if (read_packet(&p, &seq) != 0) return -1;  // EOF after defs
  for (;;) {
    if (read_packet(&p, &seq) != 0) return -1;
    if (!p.empty() && (uint8_t)p[0] == 0xfe && p.size() < 9) break;  // EOF
    if (p.empty() || (uint8_t)p[0] != 0x00) return -1;
    // binary row: 0x00 header + null bitmap (offset 2) + values
    size_t nb = (ncols + 9) / 8;
    if (p.size() < 1 + nb) return -1;
    const uint8_t* bitmap = (const uint8_t*)p.data() + 1;
    size_t pos = 1 + nb;
    std::vector<std::string> row;
    for (uint64_t c = 0; c < ncols; ++c) {
      if (bitmap[(c + 2) / 8] & (1 << ((c + 2) % 8))) {
        row.push_back("");
        continue;
      }
      uint8_t t = col_types[c];
      char buf[32];
      if (t == 0x01) {  // TINY
        row.push_back(std::to_string((int)(int8_t)p[pos]));
        pos += 1;
      } else if (t == 0x02) {  // SHORT
        int16_t v16;
        memcpy(&v16, p.data() + pos, 2);
        row.push_back(std::to_string(v16));
        pos += 2;
      } else if (t == 0x03) {  // LONG
        int32_t v32;
        memcpy(&v32, p.data() + pos, 4);
        row.push_back(std::to_string(v32));
        pos += 4;
      } else if (t == 0x08) {  // LONGLONG
        int64_t v64;
        memcpy(&v64, p.data() + pos, 8);
        row.push_back(std::to_string((long long)v64));
        pos += 8;
      } else if (t == 0x04 || t == 0x05) {  // FLOAT/DOUBLE
        double d = 0;
        if (t == 0x04) {
          float f;
          memcpy(&f, p.data() + pos, 4);
          d = f;
          pos += 4;
        } else {
          memcpy(&d, p.data() + pos, 8);
          pos += 8;
        }
        snprintf(buf, sizeof(buf), "%g", d);
        row.push_back(buf);
      } else {  // string-ish: length-encoded bytes
        std::string v;
        bool nul2;
        int k = lenc_str(p.data() + pos, p.size() - pos, &v, &nul2);
        if (k <= 0) return -1;
        row.push_back(v);
        pos += (size_t)k;
      }
    }
    out->rows.push_back(std::move(row));
  }
  out->ok = true;
  return 0;
}

void MysqlClient::CloseStatement(int64_t stmt_id) {
  if (fd_ < 0) return;
  std::string cmd(1, (char)0x19);  // COM_STMT_CLOSE (no response)
  uint32_t sid = (uint32_t)stmt_id;
  cmd.append((const char*)&sid, 4);
  write_packet(cmd, 0);
}

int MysqlClient::Query(const std::string& sql, MysqlResult* out) {
  *out = MysqlResult();
  if (fd_ < 0) {
    out->error_code = -1;
    out->error_message = "not connected";
    return -1;
  }
  std::string cmd(1, (char)kComQuery);
  cmd += sql;
  if (write_packet(cmd, 0) != 0) {
    out->error_code = -1;
    out->error_message = "write failed";
    return -1;
  }
  std::string p;
  uint8_t seq;
  if (read_packet(&p, &seq) != 0 || p.empty()) {
    out->error_code = -1;
    out->error_message = "read failed";
    return -1;
  }
  uint8_t first = (uint8_t)p[0];
  if (first == 0xff) {  // ERR: code u16, sql-state marker'#'+5, message
    out->error_code = p.size() >= 3 ? ((uint8_t)p[1] | ((int)(uint8_t)p[2] << 8)) : -1;
    size_t mpos = 3;
    if (p.size() > 3 && p[3] == '#') mpos = 9;
    out->error_message = p.substr(mpos);
    return out->error_code;
  }
  if (first == 0x00) {  // OK: affected_rows lenc, last_insert_id lenc
    out->ok = true;
    size_t pos = 1;
    uint64_t v = 0;
    int k = lenc_int(p.data() + pos, p.size() - pos, &v);
    if (k > 0) {
      out->affected_rows = v;
      pos += k;
      k = lenc_int(p.data() + pos, p.size() - pos, &v);
      if (k > 0) out->last_insert_id = v;
    }
    return 0;
  }
  // resultset: first packet = column count (lenc)
  uint64_t ncols = 0;
  if (lenc_int(p.data(), p.size(), &ncols) <= 0 || ncols == 0 || ncols > 4096) {
    out->error_code = -1;
    out->error_message = "bad column count";
    return -1;
  }
  // column definitions: catalog, schema, table, org_table, name, org_name...
  for (uint64_t i = 0; i < ncols; ++i) {
    if (read_packet(&p, &seq) != 0) return -1;
    size_t pos = 0;
    std::string field;
    bool null;
    for (int f = 0; f < 5; ++f) {  // 5th lenc-string = column name
      int k = lenc_str(p.data() + pos, p.size() - pos, &field, &null);
      if (k <= 0) {
        field.clear();
        break;
      }
      pos += (size_t)k;
    }
    out->columns.push_back(field);
  }
  // optional EOF (pre-DEPRECATE_EOF servers)
  if (read_packet(&p, &seq) != 0) return -1;
  bool was_eof = !p.empty() && (uint8_t)p[0] == 0xfe && p.size() < 9;
  for (;;) {
    if (!was_eof) {
      // p already holds a row (or terminator)
      was_eof = true;  // only skip the pre-read once
    } else {
      if (read_packet(&p, &seq) != 0) return -1;
    }
    if (!p.empty() && (uint8_t)p[0] == 0xfe && p.size() < 9) break;  // EOF
    if (!p.empty() && (uint8_t)p[0] == 0xff) {                      // mid-stream ERR
      out->error_code = p.size() >= 3 ? ((uint8_t)p[1] | ((int)(uint8_t)p[2] << 8)) : -1;
      out->error_message = p.substr(p.size() > 9 && p[3] == '#' ? 9 : 3);
      return out->error_code;
    }
    std::vector<std::string> row;
    size_t pos = 0;
    for (uint64_t i = 0; i < ncols; ++i) {
      std::string v;
      bool null = false;
      int k = lenc_str(p.data() + pos, p.size() - pos, &v, &null);
      if (k <= 0) break;
      pos += (size_t)k;
      row.push_back(null ? "" : v);
    }
    out->rows.push_back(std::move(row));
  }
  out->ok = true;
  return 0;
}

}  // namespace bam
