// Minimal leveled logger (reference: horovod/common/logging.{cc,h} —
// HOROVOD_LOG_LEVEL + timestamp toggle).
#pragma once

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>

namespace hvd {

enum class LogLevel : int { TRACE = 0, DEBUG = 1, INFO = 2, WARNING = 3,
                            ERROR = 4, FATAL = 5 };

inline LogLevel MinLogLevel() {
  static LogLevel lvl = [] {
    const char* e = std::getenv("HOROVOD_LOG_LEVEL");
    if (!e) return LogLevel::WARNING;
    if (!strcasecmp(e, "trace")) return LogLevel::TRACE;
    if (!strcasecmp(e, "debug")) return LogLevel::DEBUG;
    if (!strcasecmp(e, "info")) return LogLevel::INFO;
    if (!strcasecmp(e, "warning")) return LogLevel::WARNING;
    if (!strcasecmp(e, "error")) return LogLevel::ERROR;
    if (!strcasecmp(e, "fatal")) return LogLevel::FATAL;
    return LogLevel::WARNING;
  }();
  return lvl;
}

inline bool LogTimestamps() {
  static bool ts = [] {
    const char* e = std::getenv("HOROVOD_LOG_HIDE_TIME");
    return e == nullptr || e[0] == '0';
  }();
  return ts;
}

// HVD_LOG(INFO, "comm init rank %d", rank);
#define HVD_LOG(level, ...)                                                  \
  do {                                                                       \
    if ((int)::hvd::LogLevel::level >= (int)::hvd::MinLogLevel()) {          \
      if (::hvd::LogTimestamps()) {                                          \
        char tbuf_[32];                                                      \
        std::time_t t_ = std::time(nullptr);                                 \
        std::strftime(tbuf_, sizeof(tbuf_), "%H:%M:%S",                      \
                      std::localtime(&t_));                                  \
        std::fprintf(stderr, "[horovod_amd %s %s] ", tbuf_, #level);         \
      } else {                                                               \
        std::fprintf(stderr, "[horovod_amd %s] ", #level);                   \
      }                                                                      \
      std::fprintf(stderr, __VA_ARGS__);                                     \
      std::fprintf(stderr, "\n");                                            \
    }                                                                        \
  } while (0)

}  // namespace hvd
