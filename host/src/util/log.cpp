#include "log.h"

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <ctime>

namespace hs {

LogLevel parse_level(const std::string& s, LogLevel dflt) {
  if (s == "trace") return LogLevel::kTrace;
  if (s == "debug") return LogLevel::kDebug;
  if (s == "info") return LogLevel::kInfo;
  if (s == "warn" || s == "warning") return LogLevel::kWarn;
  if (s == "error") return LogLevel::kError;
  if (s == "off" || s == "none") return LogLevel::kOff;
  return dflt;
}

static const char* level_name(LogLevel l) {
  switch (l) {
    case LogLevel::kTrace: return "TRACE";
    case LogLevel::kDebug: return "DEBUG";
    case LogLevel::kInfo: return "INFO";
    case LogLevel::kWarn: return "WARN";
    case LogLevel::kError: return "ERROR";
    default: return "OFF";
  }
}

Logger& Logger::get() {
  static Logger inst;
  return inst;
}

void Logger::configure(LogLevel console, const std::string& file_path,
                       LogLevel file_level, long max_size_mb,
                       int max_backups) {
  std::lock_guard<std::mutex> lk(mu_);
  console_ = console;
  file_level_ = file_path.empty() ? LogLevel::kOff : file_level;
  file_path_ = file_path;
  max_size_ = max_size_mb * 1024 * 1024;
  max_backups_ = max_backups;
  if (fd_ >= 0) { close(fd_); fd_ = -1; }
  if (!file_path_.empty()) {
    // create parent dir (one level)
    auto slash = file_path_.rfind('/');
    if (slash != std::string::npos)
      mkdir(file_path_.substr(0, slash).c_str(), 0755);
    fd_ = open(file_path_.c_str(), O_CREAT | O_WRONLY | O_APPEND, 0644);
  }
}

void Logger::rotate_if_needed() {
  if (fd_ < 0 || max_size_ <= 0) return;
  struct stat st;
  if (fstat(fd_, &st) != 0 || st.st_size < max_size_) return;
  close(fd_);
  for (int i = max_backups_ - 1; i >= 1; --i) {
    std::string from = file_path_ + "." + std::to_string(i);
    std::string to = file_path_ + "." + std::to_string(i + 1);
    rename(from.c_str(), to.c_str());
  }
  if (max_backups_ > 0)
    rename(file_path_.c_str(), (file_path_ + ".1").c_str());
  fd_ = open(file_path_.c_str(), O_CREAT | O_WRONLY | O_TRUNC, 0644);
}

void Logger::log(LogLevel lvl, const char* section, const char* fmt, ...) {
  if (lvl < console_ && lvl < file_level_) return;
  char buf[2048];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof buf, fmt, ap);
  va_end(ap);
  write_line(lvl, section, buf);
}

void Logger::write_line(LogLevel lvl, const char* section,
                        const std::string& msg) {
  char ts[64];
  struct timespec now;
  clock_gettime(CLOCK_REALTIME, &now);
  struct tm tmv;
  gmtime_r(&now.tv_sec, &tmv);
  snprintf(ts, sizeof ts, "%04d-%02d-%02dT%02d:%02d:%02d.%03ldZ",
           tmv.tm_year + 1900, tmv.tm_mon + 1, tmv.tm_mday, tmv.tm_hour,
           tmv.tm_min, tmv.tm_sec, now.tv_nsec / 1000000);
  char line[2304];
  int n = snprintf(line, sizeof line, "%s %-5s %s: %s\n", ts,
                   level_name(lvl), section, msg.c_str());
  std::lock_guard<std::mutex> lk(mu_);
  if (lvl >= console_) fwrite(line, 1, size_t(n), stderr);
  if (fd_ >= 0 && lvl >= file_level_) {
    rotate_if_needed();
    [[maybe_unused]] ssize_t w = write(fd_, line, size_t(n));
  }
}

}  // namespace hs
