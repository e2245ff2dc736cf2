// Flags, logging, metrics (§5.5, §5.6).
// Flat typed flags: --key=val CLI over NEZHA_* env defaults.
#pragma once

#include <cstdarg>
#include <map>
#include <set>
#include <sstream>
#include <string>
#include <vector>

#include "src/core/common.h"

namespace nz {

// ----------------------------------------------------------------- flags ---
class Flags {
 public:
  // Parse argv (--key=val / --key val); earlier NEZHA_<KEY> env is the default.
  void parse(int argc, const char** argv);
  void set(const std::string& k, const std::string& v) { kv_[k] = v; }
  std::string get_str(const std::string& k, const std::string& dflt) const;
  int64_t get_int(const std::string& k, int64_t dflt) const;
  double get_float(const std::string& k, double dflt) const;
  bool get_bool(const std::string& k, bool dflt) const;
  std::string dump_json() const;  // resolved config, embedded in metrics
  // Warn (rank-tagged) about parsed --flags no get_* ever consulted — a
  // typo'd flag would otherwise be SILENTLY ignored (e.g. --ckpt-out vs
  // --save). Call after all get_* lookups.
  void warn_unused() const;

 private:
  std::map<std::string, std::string> kv_;
  mutable std::set<std::string> used_;
  std::string resolve(const std::string& k) const;  // CLI > env > ""
};

// ------------------------------------------------------------------- log ---
// Rank-tagged stderr logger; level via NEZHA_LOG (debug|info|warn|error).
enum class LogLevel { DEBUG = 0, INFO = 1, WARN = 2, ERROR = 3 };
void log_set_rank(int rank);
void log_msg(LogLevel lvl, const char* fmt, ...);
#define NZ_LOG_INFO(...) ::nz::log_msg(::nz::LogLevel::INFO, __VA_ARGS__)
#define NZ_LOG_WARN(...) ::nz::log_msg(::nz::LogLevel::WARN, __VA_ARGS__)
#define NZ_LOG_ERROR(...) ::nz::log_msg(::nz::LogLevel::ERROR, __VA_ARGS__)
#define NZ_LOG_DEBUG(...) ::nz::log_msg(::nz::LogLevel::DEBUG, __VA_ARGS__)

// --------------------------------------------------------------- metrics ---
// Per-step JSONL writer (rank 0): step, loss, samples/sec, ms splits.
class MetricsWriter {
 public:
  explicit MetricsWriter(const std::string& path);  // empty path -> disabled
  ~MetricsWriter();
  void write(const std::map<std::string, double>& nums,
             const std::map<std::string, std::string>& strs = {});

 private:
  FILE* f_ = nullptr;
};

// Wall-clock seconds (monotonic).
double now_sec();

}  // namespace nz
