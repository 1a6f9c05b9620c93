// strutil.hpp — small string helpers shared by the pruner and the exporter.
#pragma once

#include <algorithm>
#include <cctype>
#include <chrono>
#include <cstdint>
#include <cstdio>
#include <ctime>
#include <string>
#include <vector>

namespace strutil {

inline std::string lower(std::string s) {
  std::transform(s.begin(), s.end(), s.begin(),
                 [](unsigned char c) { return static_cast<char>(std::tolower(c)); });
  return s;
}

inline std::string trim(const std::string& s) {
  size_t b = s.find_first_not_of(" \t\r\n");
  if (b == std::string::npos) return "";
  size_t e = s.find_last_not_of(" \t\r\n");
  return s.substr(b, e - b + 1);
}

inline std::vector<std::string> split(const std::string& s, char sep) {
  std::vector<std::string> out;
  size_t start = 0;
  while (true) {
    size_t pos = s.find(sep, start);
    if (pos == std::string::npos) {
      out.push_back(s.substr(start));
      break;
    }
    out.push_back(s.substr(start, pos - start));
    start = pos + 1;
  }
  return out;
}

inline bool starts_with(const std::string& s, const std::string& p) {
  return s.size() >= p.size() && s.compare(0, p.size(), p) == 0;
}

inline bool ends_with(const std::string& s, const std::string& p) {
  return s.size() >= p.size() && s.compare(s.size() - p.size(), p.size(), p) == 0;
}

inline size_t count_occurrences(const std::string& haystack, const std::string& needle) {
  if (needle.empty()) return 0;
  size_t n = 0, pos = 0;
  while ((pos = haystack.find(needle, pos)) != std::string::npos) {
    n++;
    pos += needle.size();
  }
  return n;
}

// Percent-encode for use in URL path segments / query values.
inline std::string url_encode(const std::string& s) {
  static const char* hex = "0123456789ABCDEF";
  std::string out;
  out.reserve(s.size() * 3);
  for (unsigned char c : s) {
    if (std::isalnum(c) || c == '-' || c == '_' || c == '.' || c == '~') {
      out += static_cast<char>(c);
    } else {
      out += '%';
      out += hex[c >> 4];
      out += hex[c & 0xF];
    }
  }
  return out;
}

// RFC3339 UTC timestamp, second precision: 2026-01-02T03:04:05Z
inline std::string rfc3339_now() {
  auto now = std::chrono::system_clock::now();
  std::time_t t = std::chrono::system_clock::to_time_t(now);
  std::tm tm{};
  gmtime_r(&t, &tm);
  char buf[40];
  std::snprintf(buf, sizeof buf, "%04d-%02d-%02dT%02d:%02d:%02dZ", tm.tm_year + 1900,
                tm.tm_mon + 1, tm.tm_mday, tm.tm_hour, tm.tm_min, tm.tm_sec);
  return buf;
}

// RFC3339 UTC with microseconds (K8s MicroTime): 2026-01-02T03:04:05.123456Z
inline std::string rfc3339_micro_now() {
  auto now = std::chrono::system_clock::now();
  std::time_t t = std::chrono::system_clock::to_time_t(now);
  auto us = std::chrono::duration_cast<std::chrono::microseconds>(now.time_since_epoch()).count() %
            1000000;
  std::tm tm{};
  gmtime_r(&t, &tm);
  char buf[48];
  std::snprintf(buf, sizeof buf, "%04d-%02d-%02dT%02d:%02d:%02d.%06ldZ", tm.tm_year + 1900,
                tm.tm_mon + 1, tm.tm_mday, tm.tm_hour, tm.tm_min, tm.tm_sec,
                static_cast<long>(us));
  return buf;
}

// Parse an RFC3339 timestamp (K8s creationTimestamp shape: 2026-01-02T03:04:05Z,
// optional fractional seconds, optional ±hh:mm offset) into unix seconds.
// Returns false on malformed input. Hand-rolled: this runs once per
// candidate pod in the decision hot loop and sscanf costs ~1-2 µs a call.
inline bool parse_rfc3339(const std::string& s, double* out) {
  const char* p = s.c_str();
  auto digits = [&](int n, long* v) {
    long acc = 0;
    for (int i = 0; i < n; i++) {
      if (*p < '0' || *p > '9') return false;
      acc = acc * 10 + (*p++ - '0');
    }
    *v = acc;
    return true;
  };
  auto expect = [&](char c) { return *p == c ? (++p, true) : false; };
  long y, mo, d, h, mi, sec;
  if (!digits(4, &y) || !expect('-') || !digits(2, &mo) || !expect('-') || !digits(2, &d))
    return false;
  if (*p != 'T' && *p != 't' && *p != ' ') return false;
  p++;
  if (!digits(2, &h) || !expect(':') || !digits(2, &mi) || !expect(':') || !digits(2, &sec))
    return false;
  double frac = 0.0;
  if (*p == '.') {
    p++;
    double scale = 0.1;
    if (*p < '0' || *p > '9') return false;
    while (*p >= '0' && *p <= '9') {
      frac += (*p++ - '0') * scale;
      scale *= 0.1;
    }
  }
  // days since epoch (civil-from-days inverse; Howard Hinnant's algorithm)
  long yy = y - (mo <= 2 ? 1 : 0);
  long era = (yy >= 0 ? yy : yy - 399) / 400;
  unsigned yoe = static_cast<unsigned>(yy - era * 400);
  unsigned doy = static_cast<unsigned>((153 * (mo + (mo > 2 ? -3 : 9)) + 2) / 5 + d - 1);
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  long days = era * 146097 + static_cast<long>(doe) - 719468;
  double base = static_cast<double>(days) * 86400.0 + h * 3600.0 + mi * 60.0 +
                static_cast<double>(sec) + frac;
  if (*p == 'Z' || *p == 'z') {
    p++;
  } else if (*p == '+' || *p == '-') {
    char sign = *p++;
    long oh, om;
    if (!digits(2, &oh) || !expect(':') || !digits(2, &om)) return false;
    long off = oh * 3600 + om * 60;
    base += sign == '+' ? -off : off;
  }
  if (*p != '\0') return false;
  *out = base;
  return true;
}

// Random 32-hex-char id (uuid4 "simple" form) from /dev/urandom with a
// rand_r fallback; used for Event names: gpuscaler-<id>.
std::string uuid4_simple();

}  // namespace strutil
