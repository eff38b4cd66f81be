// Structured leveled logging with per-section levels, console + rotating
// file sink — the reference's tracing-subscriber setup re-created natively
// (reference libs/modkit/src/bootstrap/host/logging.rs; config section
// `logging.default.{console_level,file,file_level,max_size_mb,max_backups}`).
#pragma once

#include <cstdarg>
#include <mutex>
#include <string>

namespace hs {

enum class LogLevel { kTrace, kDebug, kInfo, kWarn, kError, kOff };

LogLevel parse_level(const std::string& s, LogLevel dflt = LogLevel::kInfo);

class Logger {
 public:
  static Logger& get();

  void configure(LogLevel console, const std::string& file_path,
                 LogLevel file_level, long max_size_mb, int max_backups);
  void log(LogLevel lvl, const char* section, const char* fmt, ...)
      __attribute__((format(printf, 4, 5)));

 private:
  void write_line(LogLevel lvl, const char* section, const std::string& msg);
  void rotate_if_needed();

  std::mutex mu_;
  LogLevel console_ = LogLevel::kInfo;
  LogLevel file_level_ = LogLevel::kOff;
  std::string file_path_;
  long max_size_ = 0;
  int max_backups_ = 0;
  int fd_ = -1;
};

#define HS_LOG(lvl, section, ...) \
  ::hs::Logger::get().log(::hs::LogLevel::lvl, section, __VA_ARGS__)
#define LOG_INFO(section, ...) HS_LOG(kInfo, section, __VA_ARGS__)
#define LOG_WARN(section, ...) HS_LOG(kWarn, section, __VA_ARGS__)
#define LOG_ERROR(section, ...) HS_LOG(kError, section, __VA_ARGS__)
#define LOG_DEBUG(section, ...) HS_LOG(kDebug, section, __VA_ARGS__)

}  // namespace hs
