// brpc_amd: minimal glog-style streaming logger.
// Capability parity: reference butil/logging.h (LOG/CHECK macros, severity,
// VLOG). Clean-room implementation, no Chromium code.
#pragma once

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <sstream>
#include <string>

namespace bam {

enum LogSeverity { LOG_TRACE = -1, LOG_DEBUG = 0, LOG_INFO = 1, LOG_WARNING = 2, LOG_ERROR = 3, LOG_FATAL = 4 };

// Global minimum severity actually emitted (runtime-settable, like
// reference butil/logging.h min_log_level).
extern std::atomic<int> g_min_log_level;

typedef void (*LogSinkFn)(int severity, const char* file, int line, const std::string& msg);
// Replace the default stderr sink (used by /flags-style runtime config and tests).
void set_log_sink(LogSinkFn fn);

class LogMessage {
 public:
  LogMessage(const char* file, int line, int severity) : file_(file), line_(line), severity_(severity) {}
  ~LogMessage();
  std::ostream& stream() { return stream_; }

 private:
  const char* file_;
  int line_;
  int severity_;
  std::ostringstream stream_;
};

// Swallows the stream when the log level is disabled.
class LogMessageVoidify {
 public:
  LogMessageVoidify() {}
  void operator&(std::ostream&) {}
};

}  // namespace bam

#define BAM_LOG_IS_ON(sev) (::bam::LOG_##sev >= ::bam::g_min_log_level.load(std::memory_order_relaxed))

#define LOG(sev)                  \
  !BAM_LOG_IS_ON(sev) ? (void)0 : \
    ::bam::LogMessageVoidify() & ::bam::LogMessage(__FILE__, __LINE__, ::bam::LOG_##sev).stream()

#define LOG_IF(sev, cond) \
  (!BAM_LOG_IS_ON(sev) || !(cond)) ? (void)0 : ::bam::LogMessageVoidify() & ::bam::LogMessage(__FILE__, __LINE__, ::bam::LOG_##sev).stream()

#define CHECK(cond) \
  (cond) ? (void)0 : ::bam::LogMessageVoidify() & ::bam::LogMessage(__FILE__, __LINE__, ::bam::LOG_FATAL).stream() << "Check failed: " #cond " "

#define CHECK_EQ(a, b) CHECK((a) == (b))
#define CHECK_NE(a, b) CHECK((a) != (b))
#define CHECK_LT(a, b) CHECK((a) < (b))
#define CHECK_LE(a, b) CHECK((a) <= (b))
#define CHECK_GT(a, b) CHECK((a) > (b))
#define CHECK_GE(a, b) CHECK((a) >= (b))

#define PLOG(sev) LOG(sev) << "[errno=" << errno << " " << strerror(errno) << "] "
