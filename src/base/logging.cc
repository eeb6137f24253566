#include "base/logging.h"

#include <time.h>
#include <unistd.h>

namespace bam {

std::atomic<int> g_min_log_level{LOG_INFO};

static std::atomic<LogSinkFn> g_sink{nullptr};

void set_log_sink(LogSinkFn fn) { g_sink.store(fn, std::memory_order_release); }

LogMessage::~LogMessage() {
  std::string msg = stream_.str();
  LogSinkFn sink = g_sink.load(std::memory_order_acquire);
  if (sink != nullptr) {
    sink(severity_, file_, line_, msg);
  } else {
    static const char kSevChar[] = {'T', 'D', 'I', 'W', 'E', 'F'};
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    struct tm tm_buf;
    localtime_r(&ts.tv_sec, &tm_buf);
    const char* base = strrchr(file_, '/');
    base = base ? base + 1 : file_;
    fprintf(stderr, "%c%02d%02d %02d:%02d:%02d.%06ld %5d %s:%d] %s\n",
            kSevChar[severity_ + 1], tm_buf.tm_mon + 1, tm_buf.tm_mday, tm_buf.tm_hour,
            tm_buf.tm_min, tm_buf.tm_sec, ts.tv_nsec / 1000, (int)getpid(), base, line_, msg.c_str());
  }
  if (severity_ >= LOG_FATAL) {
    fflush(stderr);
    abort();
  }
}

}  // namespace bam
