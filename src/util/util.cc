#include "src/util/util.h"

#include <algorithm>
#include <cctype>
#include <chrono>
#include <cstdio>

namespace nz {

// ----------------------------------------------------------------- flags ---
void Flags::parse(int argc, const char** argv) {
  for (int i = 0; i < argc; ++i) {
    std::string a = argv[i];
    if (a.rfind("--", 0) != 0) continue;
    a = a.substr(2);
    auto eq = a.find('=');
    if (eq != std::string::npos) {
      kv_[a.substr(0, eq)] = a.substr(eq + 1);
    } else if (i + 1 < argc && argv[i + 1][0] != '-') {
      kv_[a] = argv[++i];
    } else {
      kv_[a] = "1";  // bare flag => true
    }
  }
}

std::string Flags::resolve(const std::string& k) const {
  used_.insert(k);
  auto it = kv_.find(k);
  if (it != kv_.end()) return it->second;
  std::string env = "NEZHA_";
  for (char c : k) env += char(c == '-' ? '_' : std::toupper((unsigned char)c));
  const char* v = getenv(env.c_str());
  return v ? v : "";
}

std::string Flags::get_str(const std::string& k, const std::string& dflt) const {
  auto v = resolve(k);
  return v.empty() ? dflt : v;
}
int64_t Flags::get_int(const std::string& k, int64_t dflt) const {
  auto v = resolve(k);
  return v.empty() ? dflt : strtoll(v.c_str(), nullptr, 10);
}
double Flags::get_float(const std::string& k, double dflt) const {
  auto v = resolve(k);
  return v.empty() ? dflt : strtod(v.c_str(), nullptr);
}
bool Flags::get_bool(const std::string& k, bool dflt) const {
  auto v = resolve(k);
  if (v.empty()) return dflt;
  return v == "1" || v == "true" || v == "yes";
}
std::string Flags::dump_json() const {
  std::ostringstream os;
  os << "{";
  bool first = true;
  for (auto& [k, v] : kv_) {
    if (!first) os << ",";
    first = false;
    os << "\"" << k << "\":\"" << v << "\"";
  }
  os << "}";
  return os.str();
}

void Flags::warn_unused() const {
  for (auto& [k, v] : kv_) {
    if (!used_.count(k))
      log_msg(LogLevel::WARN, "flag --%s=%s was never consulted (typo?)",
              k.c_str(), v.c_str());
  }
}

// ------------------------------------------------------------------- log ---
static int g_rank = -1;
static LogLevel g_level = [] {
  const char* v = getenv("NEZHA_LOG");
  if (!v) return LogLevel::INFO;
  std::string s = v;
  if (s == "debug") return LogLevel::DEBUG;
  if (s == "warn") return LogLevel::WARN;
  if (s == "error") return LogLevel::ERROR;
  return LogLevel::INFO;
}();

void log_set_rank(int rank) { g_rank = rank; }

void log_msg(LogLevel lvl, const char* fmt, ...) {
  if ((int)lvl < (int)g_level) return;
  const char* names[] = {"DEBUG", "INFO", "WARN", "ERROR"};
  char buf[4096];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  if (g_rank >= 0)
    fprintf(stderr, "[nezha %s r%d] %s\n", names[(int)lvl], g_rank, buf);
  else
    fprintf(stderr, "[nezha %s] %s\n", names[(int)lvl], buf);
}

// --------------------------------------------------------------- metrics ---
MetricsWriter::MetricsWriter(const std::string& path) {
  if (!path.empty()) f_ = fopen(path.c_str(), "a");
}
MetricsWriter::~MetricsWriter() {
  if (f_) fclose(f_);
}
void MetricsWriter::write(const std::map<std::string, double>& nums,
                          const std::map<std::string, std::string>& strs) {
  if (!f_) return;
  fprintf(f_, "{");
  bool first = true;
  for (auto& [k, v] : nums) {
    fprintf(f_, "%s\"%s\":%.9g", first ? "" : ",", k.c_str(), v);
    first = false;
  }
  for (auto& [k, v] : strs) {
    // values that are themselves JSON objects embed as-is; everything else
    // is quoted with escaping (an unescaped embed once produced malformed
    // JSONL when the CLI passed its config dump as a plain string)
    if (!v.empty() && v.front() == '{' && v.back() == '}') {
      fprintf(f_, "%s\"%s\":%s", first ? "" : ",", k.c_str(), v.c_str());
    } else {
      fprintf(f_, "%s\"%s\":\"", first ? "" : ",", k.c_str());
      for (char c : v) {
        if (c == '"' || c == '\\') fputc('\\', f_);
        if ((unsigned char)c >= 0x20) fputc(c, f_);
      }
      fprintf(f_, "\"");
    }
    first = false;
  }
  fprintf(f_, "}\n");
  fflush(f_);
}

double now_sec() {
  using namespace std::chrono;
  return duration<double>(steady_clock::now().time_since_epoch()).count();
}

}  // namespace nz
