// brpc_amd: MySQL client protocol.
// Parity: reference policy/mysql/ (mysql.cpp, mysql_auth_handshake.cpp,
// mysql_auth_scramble.cpp — 4k loc, client side) — clean-room subset:
//  * packet framing: [payload_len u24 LE][sequence u8][payload]
//  * HandshakeV10 parse + HandshakeResponse41 with mysql_native_password
//    scramble: SHA1(pwd) XOR SHA1(salt + SHA1(SHA1(pwd)))
//  * COM_QUERY with OK / ERR / resultset (column defs + text rows with
//    length-encoded values, EOF or OK-terminated), COM_PING, COM_INIT_DB
//  * caching_sha2_password fast path (XOR(SHA256(pwd),
//    SHA256(SHA256(SHA256(pwd)) + nonce))) and AuthSwitchRequest loop
//  * prepared statements: COM_STMT_PREPARE / EXECUTE / CLOSE with binary
//    result rows (common column types decoded, others hex-dumped)
// Deltas vs reference: caching_sha2 full-auth (RSA password exchange on
// cache miss over plaintext) is not implemented — use the fast path or
// TLS; transactions work as plain queries (BEGIN/COMMIT/ROLLBACK).
#pragma once

#include <stdint.h>

#include <string>
#include <vector>

namespace bam {

struct MysqlResult {
  bool ok = false;
  uint64_t affected_rows = 0;
  uint64_t last_insert_id = 0;
  int error_code = 0;          // server error (ERR packet) or -1 on transport
  std::string error_message;
  std::vector<std::string> columns;
  std::vector<std::vector<std::string>> rows;  // text protocol values ("" for NULL)
};

class MysqlClient {
 public:
  ~MysqlClient();

  // Connects + authenticates. user/password per mysql_native_password;
  // db optional. Returns 0, or -1 (transport) / server error code.
  int Connect(const std::string& host, int port, const std::string& user,
              const std::string& password, const std::string& db = "",
              int timeout_ms = 3000);

  // Runs one statement. Fills *out (resultset or OK info). Returns 0 on
  // success (including statement-level success), server errno otherwise.
  int Query(const std::string& sql, MysqlResult* out);

  int Ping();

  // Prepared statements (COM_STMT_PREPARE / COM_STMT_EXECUTE, binary
  // protocol). Params are sent as MYSQL_TYPE_STRING; result rows come
  // back through the binary row format and are surfaced as text in
  // MysqlResult (ints/strings; other column types are hex-dumped).
  // Prepare returns a statement id (>0) or -1 / server errno.
  int64_t Prepare(const std::string& sql, int* param_count = nullptr);
  int ExecutePrepared(int64_t stmt_id, const std::vector<std::string>& params,
                      MysqlResult* out);
  void CloseStatement(int64_t stmt_id);

  void Close();
  bool connected() const { return fd_ >= 0; }
  const std::string& server_version() const { return server_version_; }

 private:
  int read_packet(std::string* payload, uint8_t* seq);
  int write_packet(const std::string& payload, uint8_t seq);
  int read_n(void* buf, size_t n);

  int fd_ = -1;
  int timeout_ms_ = 3000;
  std::string server_version_;
};

}  // namespace bam
