// querytest.cpp — debug CLI for raw PromQL queries.
//
// Equivalent of the reference's second binary (SURVEY.md §2.1 "querytest
// debug tool"; reference gpu-pruner/src/bin/querytest.rs): arg1 = raw PromQL,
// arg2 = Prometheus URL. Prints vector/matrix results as an aligned table on
// stdout and writes output.csv. Scalar results are unsupported, matching the
// reference.
#include <cstdio>
#include <ctime>
#include <fstream>
#include <vector>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "prom.hpp"

namespace {

std::string ts_to_string(double ts) {
  std::time_t t = static_cast<std::time_t>(ts);
  std::tm tm{};
  if (!gmtime_r(&t, &tm)) return "invalid timestamp";
  char buf[40];
  std::snprintf(buf, sizeof buf, "%04d-%02d-%02d %02d:%02d:%02d UTC", tm.tm_year + 1900,
                tm.tm_mon + 1, tm.tm_mday, tm.tm_hour, tm.tm_min, tm.tm_sec);
  return buf;
}

std::string sample_value(const jsn::Value& pair) {
  const jsn::Value& v = pair[1];
  return v.is_string() ? v.as_string() : v.dump();
}

void add_row(std::vector<std::vector<std::string>>& rows, const jsn::Value& metric,
             const jsn::Value& sample_pair) {
  std::vector<std::string> row;
  if (metric.is_object())
    for (const auto& [k, v] : metric.obj()) row.push_back(v.as_string_or(v.dump()));
  row.push_back(ts_to_string(sample_pair[0].as_double()));
  row.push_back(sample_value(sample_pair));
  rows.push_back(std::move(row));
}

std::string csv_escape(const std::string& s) {
  if (s.find_first_of(",\"\n") == std::string::npos) return s;
  std::string out = "\"";
  for (char c : s) {
    if (c == '"') out += "\"\"";
    else out += c;
  }
  out += "\"";
  return out;
}

}  // namespace

int main(int argc, char** argv) {
  logx::init(logx::Format::Default);
  if (argc < 3) {
    std::fprintf(stderr, "usage: querytest '<promql>' <prometheus-url>\n");
    return 2;
  }
  std::string query = argv[1];
  std::string url = argv[2];
  LOGI("querytest", "Prometheus URL: " + url);
  LOGI("querytest", "Query: " + query);

  try {
    std::string token = pruner::get_prometheus_token();
    // Same hard-coded local CA convention as the reference tool
    // (querytest.rs:16: certs/prometheus.crt) when the file exists.
    std::optional<std::string> cert;
    if (std::ifstream("certs/prometheus.crt").good()) cert = "certs/prometheus.crt";
    pruner::PromClient client(url, token, pruner::TlsModeOpt::Verify, cert);
    jsn::Value data = client.query(query);

    std::string result_type = data.get("resultType").as_string();
    const jsn::Value& result = data.get("result");
    std::vector<std::vector<std::string>> rows;

    if (result_type == "vector") {
      for (const auto& series : result.arr())
        add_row(rows, series.get("metric"), series.get("value"));
    } else if (result_type == "matrix") {
      for (const auto& series : result.arr())
        for (const auto& sample : series.get("values").arr())
          add_row(rows, series.get("metric"), sample);
    } else {
      LOGE("querytest", "Scalar data not supported");
      return 1;
    }

    // aligned table on stdout
    std::vector<size_t> widths;
    for (const auto& row : rows) {
      if (widths.size() < row.size()) widths.resize(row.size(), 0);
      for (size_t i = 0; i < row.size(); i++) widths[i] = std::max(widths[i], row[i].size());
    }
    for (const auto& row : rows) {
      std::string line = "|";
      for (size_t i = 0; i < row.size(); i++) {
        line += " " + row[i] + std::string(widths[i] - row[i].size(), ' ') + " |";
      }
      std::puts(line.c_str());
    }

    std::ofstream csv("output.csv");
    for (const auto& row : rows) {
      for (size_t i = 0; i < row.size(); i++) {
        if (i) csv << ",";
        csv << csv_escape(row[i]);
      }
      csv << "\n";
    }
    return 0;
  } catch (const std::exception& e) {
    LOGE("querytest", e.what());
    return 1;
  }
}
