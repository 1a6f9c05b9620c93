// core_py.cpp — _pruner_core: Python bindings over the native pruner core.
//
// Exposes the pure-logic layers (query builder, resource flags, scale-target
// model, event generation, series parsing, CLI parsing) for the pytest suite
// and bench harness. The daemon binaries do NOT go through Python — these
// bindings exist so the unit-test surface of the reference
// (gpu-pruner/src/lib.rs:578-998, main.rs:572-740) can be ported to pytest
// against the very same native code the binaries link.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../pruner/config.hpp"
#include "../pruner/engine.hpp"
#include "../pruner/k8s.hpp"
#include "../pruner/prom.hpp"
#include "../pruner/promql.hpp"
#include "../pruner/resources.hpp"
#include "../common/http.hpp"
#include "../common/miniyaml.hpp"
#include "../pruner/informer.hpp"
#include "../pruner/leader.hpp"
#include "../pruner/otlp.hpp"
#include "../pruner/synthbench.hpp"

namespace py = pybind11;
using namespace pruner;

namespace {

QueryArgs query_args_from_json(const std::string& json_args) {
  jsn::Value v = jsn::parse(json_args);
  QueryArgs qa;
  if (v.contains("duration")) qa.duration_min = v.get("duration").as_int(30);
  if (v.get("namespace").is_string()) qa.namespace_re = v.get("namespace").as_string();
  if (v.get("model_name").is_string()) qa.model_name_re = v.get("model_name").as_string();
  if (v.get("power_threshold").is_num())
    qa.power_threshold_w = v.get("power_threshold").as_double();
  qa.honor_labels = v.get("honor_labels").as_bool(false);
  return qa;
}

Kind kind_from_str(const std::string& s) {
  if (s == "Deployment") return Kind::Deployment;
  if (s == "ReplicaSet") return Kind::ReplicaSet;
  if (s == "StatefulSet") return Kind::StatefulSet;
  if (s == "InferenceService") return Kind::InferenceService;
  if (s == "Notebook") return Kind::Notebook;
  throw py::value_error("unknown kind: " + s);
}

Config config_from_json(const std::string& json_cfg) {
  jsn::Value v = jsn::parse(json_cfg);
  Config c;
  if (v.contains("duration")) c.duration_min = v.get("duration").as_int(30);
  if (v.contains("grace_period")) c.grace_period_s = v.get("grace_period").as_int(300);
  if (v.get("namespace").is_string()) c.namespace_ = v.get("namespace").as_string();
  if (v.get("model_name").is_string()) c.model_name = v.get("model_name").as_string();
  if (v.get("power_threshold").is_num())
    c.power_threshold = v.get("power_threshold").as_double();
  c.honor_labels = v.get("honor_labels").as_bool(false);
  if (v.contains("max_concurrency"))
    c.max_concurrency = static_cast<int>(v.get("max_concurrency").as_int(32));
  if (v.get("run_mode").as_string_or("dry-run") == "scale-down")
    c.run_mode = RunMode::ScaleDown;
  c.enabled_resources = v.get("enabled_resources").as_string_or("drsin");
  std::string strat = v.get("eval_strategy").as_string_or("auto");
  c.eval_strategy = strat == "get"     ? EvalStrategy::PerPodGet
                    : strat == "list"  ? EvalStrategy::NamespaceList
                    : strat == "watch" ? EvalStrategy::Watch
                                       : EvalStrategy::Auto;
  if (v.get("prometheus_url").is_string())
    c.prometheus_url = v.get("prometheus_url").as_string();
  return c;
}

}  // namespace

PYBIND11_MODULE(_pruner_core, m) {
  m.doc() = "native core of the MI355X gpu-pruner (query builder, scaling model, engine)";

  logx::init(logx::Format::Default);

  // ---- query builder ----
  m.def("render_query", [](const std::string& json_args) {
    return build_idle_query(query_args_from_json(json_args));
  },
      "Build the idle-GPU PromQL query from a JSON args object "
      "(duration, namespace, model_name, power_threshold, honor_labels)");

  // ---- resource flags ----
  m.def("get_enabled_resources",
        [](const std::string& s) { return static_cast<int>(get_enabled_resources(s)); });
  m.attr("RK_DEPLOYMENT") = static_cast<int>(RK_DEPLOYMENT);
  m.attr("RK_REPLICA_SET") = static_cast<int>(RK_REPLICA_SET);
  m.attr("RK_STATEFUL_SET") = static_cast<int>(RK_STATEFUL_SET);
  m.attr("RK_INFERENCE_SERVICE") = static_cast<int>(RK_INFERENCE_SERVICE);
  m.attr("RK_NOTEBOOK") = static_cast<int>(RK_NOTEBOOK);

  // ---- scale-target model ----
  py::class_<ScaleKind>(m, "ScaleKind")
      .def(py::init([](const std::string& kind, const std::string& obj_json) {
             return ScaleKind{kind_from_str(kind), jsn::parse(obj_json)};
           }),
           py::arg("kind"), py::arg("object_json"))
      .def_property_readonly("kind", [](const ScaleKind& sk) { return sk.kind_str(); })
      .def_property_readonly("name", [](const ScaleKind& sk) { return sk.name(); })
      .def_property_readonly("namespace",
                             [](const ScaleKind& sk) -> py::object {
                               auto ns = sk.ns();
                               return ns ? py::cast(*ns) : py::none();
                             })
      .def_property_readonly("uid",
                             [](const ScaleKind& sk) -> py::object {
                               auto u = sk.uid();
                               return u ? py::cast(*u) : py::none();
                             })
      .def_property_readonly("api_version",
                             [](const ScaleKind& sk) { return sk.api_version(); })
      .def_property_readonly("resource_version",
                             [](const ScaleKind& sk) -> py::object {
                               auto rv = sk.resource_version();
                               return rv ? py::cast(*rv) : py::none();
                             })
      .def_property_readonly("resource_kind",
                             [](const ScaleKind& sk) { return static_cast<int>(kind_flag(sk.kind)); })
      .def("object_json", [](const ScaleKind& sk) { return sk.object.dump(); })
      .def("__eq__", [](const ScaleKind& a, const ScaleKind& b) { return a == b; },
           py::is_operator())
      .def("__hash__", [](const ScaleKind& sk) { return sk.hash(); })
      .def("__repr__", [](const ScaleKind& sk) {
        return "<ScaleKind " + sk.kind_str() + " " + sk.ns().value_or("") + ":" + sk.name() + ">";
      });

  m.def("generate_scale_event",
        [](const ScaleKind& sk) { return generate_scale_event(sk).dump(); });

  // ---- series parsing ----
  m.def("parse_pod_metric", [](const std::string& series_json) {
    PodMetricData pmd = parse_pod_metric(jsn::parse(series_json));
    py::dict d;
    d["name"] = pmd.name;
    d["namespace"] = pmd.ns;
    d["container"] = pmd.container;
    d["node_type"] = pmd.node_type;
    d["gpu_model"] = pmd.gpu_model;
    d["value"] = pmd.value;
    return d;
  });

  py::register_exception<PodConvertError>(m, "PodConvertError");

  // ---- CLI ----
  m.def("parse_cli", [](const std::vector<std::string>& args) {
    CliResult r = parse_cli(args);
    py::dict d;
    d["help"] = r.show_help;
    d["error"] = r.error ? py::cast(*r.error) : py::object(py::none());
    py::dict c;
    c["duration"] = r.config.duration_min;
    c["daemon_mode"] = r.config.daemon_mode;
    c["enabled_resources"] = r.config.enabled_resources;
    c["check_interval"] = r.config.check_interval_s;
    c["namespace"] =
        r.config.namespace_ ? py::cast(*r.config.namespace_) : py::object(py::none());
    c["grace_period"] = r.config.grace_period_s;
    c["model_name"] =
        r.config.model_name ? py::cast(*r.config.model_name) : py::object(py::none());
    c["power_threshold"] = r.config.power_threshold ? py::cast(*r.config.power_threshold)
                                                    : py::object(py::none());
    c["honor_labels"] = r.config.honor_labels;
    c["run_mode"] = r.config.run_mode == RunMode::ScaleDown ? "scale-down" : "dry-run";
    c["prometheus_url"] = r.config.prometheus_url;
    c["prometheus_tls_mode"] =
        r.config.prometheus_tls_mode == TlsModeOpt::Skip ? "skip" : "verify";
    c["log_format"] = r.config.log_format == LogFormatOpt::Json      ? "json"
                      : r.config.log_format == LogFormatOpt::Pretty ? "pretty"
                                                                     : "default";
    c["max_concurrency"] = r.config.max_concurrency;
    c["queue_capacity"] = r.config.queue_capacity;
    c["leader_elect"] = r.config.leader_elect;
    c["leader_lease_duration_s"] = r.config.leader_lease_duration_s;
    c["leader_renew_period_s"] = r.config.leader_renew_period_s;
    d["config"] = c;
    return d;
  });

  // ---- engine (against a live endpoint: fake fixtures or a real cluster) ----
  // Used by unit tests and bench.py; the daemon binary has its own main loop.
  m.def("find_root_object",
        [](const std::string& pod_json) -> py::object {
          KubeClient kube(KubeConfig::resolve());
          auto sk = find_root_object(kube, jsn::parse(pod_json));
          if (!sk) return py::none();
          return py::cast(*sk);
        },
        "Owner-reference walk using the env-configured apiserver");

  m.def("scale", [](const ScaleKind& sk) {
    KubeClient kube(KubeConfig::resolve());
    scale(kube, sk);
  });

  m.def("evaluate_candidates",
        [](const std::string& result_vector_json, const std::string& cfg_json) {
          KubeClient kube(KubeConfig::resolve());
          Config cfg = config_from_json(cfg_json);
          QueryOutcome out;
          std::vector<ScaleKind> roots =
              evaluate_candidates(kube, jsn::parse(result_vector_json), cfg, &out);
          py::gil_scoped_acquire gil;
          py::dict d;
          d["num_series"] = out.num_series;
          d["num_unique_pods"] = out.num_unique_pods;
          d["shutdown_events"] = out.shutdown_events;
          py::list lst;
          for (auto& sk : roots) lst.append(py::cast(sk));
          d["roots"] = lst;
          return d;
        },
        py::call_guard<py::gil_scoped_release>(),
        "Full decision pass over a parsed Prometheus vector result");

  m.def("run_tick",
        [](const std::string& cfg_json) {
          Config cfg = config_from_json(cfg_json);
          auto t0 = std::chrono::steady_clock::now();
          auto prom = build_prom_client(cfg);
          KubeClient kube(KubeConfig::resolve());
          std::string query = build_idle_query(cfg.query_args());
          jsn::Value result = prom->query_vector(query);
          auto t1 = std::chrono::steady_clock::now();
          QueryOutcome out;
          std::vector<ScaleKind> roots = evaluate_candidates(kube, result, cfg, &out);
          auto t2 = std::chrono::steady_clock::now();
          size_t scaled = 0;
          if (cfg.run_mode == RunMode::ScaleDown) {
            uint8_t enabled = get_enabled_resources(cfg.enabled_resources);
            scaled = scale_all(kube, roots, enabled, cfg.max_concurrency);
          }
          auto t3 = std::chrono::steady_clock::now();
          auto ms = [](auto a, auto b) {
            return std::chrono::duration<double, std::milli>(b - a).count();
          };
          py::gil_scoped_acquire gil;
          py::dict d;
          d["num_series"] = out.num_series;
          d["num_unique_pods"] = out.num_unique_pods;
          d["shutdown_events"] = out.shutdown_events;
          d["scaled"] = scaled;
          py::dict phases;
          phases["query_ms"] = ms(t0, t1);
          phases["evaluate_ms"] = ms(t1, t2);
          phases["actuate_ms"] = ms(t2, t3);
          d["phase_ms"] = phases;
          return d;
        },
        py::call_guard<py::gil_scoped_release>(),
        "One full decision tick: Prometheus query + evaluation + (in "
        "scale-down mode) inline actuation of every selected root");

  m.def("get_prometheus_token", [] { return get_prometheus_token(); });

  m.def("http_stream_lines",
        [](const std::string& url) {
          auto parsed = http::Url::parse(url);
          if (!parsed) throw std::runtime_error("bad url");
          std::vector<std::string> lines;
          {
            py::gil_scoped_release nogil;
            http::Client client(*parsed, http::ClientOptions{});
            http::Request r;
            r.path = parsed->path;
            auto stream = client.open_stream(r);
            std::string line;
            while (stream->read_line(&line)) lines.push_back(line);
          }
          py::list out;
          for (auto& l : lines) out.append(l);
          return out;
        },
        "Open a streaming GET and return every decoded body line "
        "(BodyStream chunk-decoder test surface)");

  // Lease-based leader elector (k8s config from the environment) — lets
  // tests drive single acquire/renew attempts deterministically.
  py::class_<LeaderElector>(m, "LeaderElector")
      .def(py::init([](const std::string& ns, const std::string& lease_name,
                       const std::string& identity, int lease_duration_s,
                       int renew_period_s) {
             return new LeaderElector(KubeConfig::resolve(), ns, lease_name, identity,
                                      lease_duration_s, renew_period_s);
           }),
           py::arg("namespace"), py::arg("lease_name"), py::arg("identity"),
           py::arg("lease_duration_s") = 15, py::arg("renew_period_s") = 5)
      .def("try_acquire_or_renew", &LeaderElector::try_acquire_or_renew,
           py::call_guard<py::gil_scoped_release>())
      .def("start", &LeaderElector::start)
      .def("stop", &LeaderElector::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("is_leader", &LeaderElector::is_leader);

  m.def("informers_reset",
        [] {
          py::gil_scoped_release nogil;
          InformerRegistry::global().stop_all();
        },
        "Stop and drop all persistent watch informers (test isolation)");

  m.def("resolve_kube_config", [] {
    KubeConfig cfg = KubeConfig::resolve();
    py::dict d;
    d["url"] = cfg.url;
    d["token"] = cfg.token ? py::cast(*cfg.token) : py::object(py::none());
    d["token_file"] =
        cfg.token_file ? py::cast(*cfg.token_file) : py::object(py::none());
    d["ca_file"] = cfg.ca_file ? py::cast(*cfg.ca_file) : py::object(py::none());
    d["ca_data"] = cfg.ca_data ? py::cast(*cfg.ca_data) : py::object(py::none());
    d["client_cert_data"] =
        cfg.client_cert_data ? py::cast(*cfg.client_cert_data) : py::object(py::none());
    d["client_key_data"] =
        cfg.client_key_data ? py::cast(*cfg.client_key_data) : py::object(py::none());
    d["skip_tls"] = cfg.skip_tls;
    d["default_namespace"] = cfg.default_namespace;
    d["exec_command"] =
        cfg.exec ? py::cast(cfg.exec->command) : py::object(py::none());
    return d;
  });

  // ---- native synthetic backend (benchmark harness) ----
  py::class_<SyntheticBackend>(m, "SyntheticBackend")
      .def(py::init([](int n_pods, int pods_per_parent, int gpus_per_pod, int latency_us,
                       const std::string& model_name) {
             SynthOptions o;
             o.n_pods = n_pods;
             o.pods_per_parent = pods_per_parent;
             o.gpus_per_pod = gpus_per_pod;
             o.latency_us = latency_us;
             if (!model_name.empty()) o.model_name = model_name;
             return new SyntheticBackend(o);
           }),
           py::arg("n_pods") = 1000, py::arg("pods_per_parent") = 2,
           py::arg("gpus_per_pod") = 1, py::arg("latency_us") = 0,
           py::arg("model_name") = "")
      .def("start", &SyntheticBackend::start)
      .def("stop", &SyntheticBackend::stop)
      .def_property_readonly("prom_url", &SyntheticBackend::prom_url)
      .def_property_readonly("k8s_url", &SyntheticBackend::k8s_url)
      .def("set_series_value", &SyntheticBackend::set_series_value)
      .def_property_readonly("events_posted", &SyntheticBackend::events_posted)
      .def_property_readonly("scale_patches", &SyntheticBackend::scale_patches)
      .def_property_readonly("watch_streams", &SyntheticBackend::watch_streams)
      .def_property_readonly("requests_served", &SyntheticBackend::requests_served)
      .def_property_readonly("expected_parents", &SyntheticBackend::expected_parents);

  // OTLP lifecycle (the binaries call these in main(); bench.py/config-5
  // runs need them from Python)
  m.def("otlp_init", [](const std::string& service) { otlp::init(service); },
        py::arg("service") = "gpu-pruner");
  m.def("otlp_shutdown", [] { otlp::shutdown(); },
        py::call_guard<py::gil_scoped_release>());
  m.def("otlp_enabled", [] { return otlp::enabled(); });

  // test helper: raw GET through the native HTTP client (used to pin
  // chunked / close-delimited / keep-alive decoding against fixture servers)
  m.def("_http_get",
        [](const std::string& url) {
          http::Response resp;
          {
            py::gil_scoped_release nogil;
            resp = http::fetch(url);
          }
          py::dict d;
          d["status"] = resp.status;
          d["body"] = py::bytes(resp.body);
          py::dict headers;
          for (const auto& [k, v] : resp.headers) headers[py::str(k)] = v;
          d["headers"] = headers;
          return d;
        });

  // test helper: the kubeconfig YAML-subset reader (common/miniyaml.hpp)
  m.def("_yaml_to_json", [](const std::string& src) {
    return miniyaml::parse(src).dump();
  });
  py::register_exception<miniyaml::Error>(m, "YamlError");

  m.def("counters_snapshot", [] {
    py::dict d;
    for (const auto& [k, v] : logx::counters_snapshot()) d[py::str(k)] = v;
    return d;
  });
  m.def("counters_reset", [] { logx::counters_reset_for_test(); });
}
