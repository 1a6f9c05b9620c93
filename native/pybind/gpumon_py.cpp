// gpumon_py.cpp — _gpumon: Python bindings over the mi355-exporter internals.
//
// Gives pytest and bench.py direct access to the ROCm sampler, the pod
// attribution chain, and the metrics renderer — the same native objects the
// mi355-exporter binary runs. GPU-marked tests exercise Sampler on real
// gfx950; the attribution/rendering paths are CPU-testable via the
// GPU_EXPORTER_*_ROOT / POD_MAP_FILE overrides.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../common/grpc_client.hpp"
#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../exporter/attrib.hpp"
#include "../exporter/podresources.hpp"
#include "../exporter/registry.hpp"
#include "../exporter/sampler.hpp"

namespace py = pybind11;
using namespace exporter;

namespace {

py::dict sample_to_dict(const DeviceSample& d) {
  py::dict s;
  s["index"] = d.index;
  s["model_name"] = d.model_name;
  s["unique_id"] = d.unique_id;
  s["pci_bdf"] = d.pci_bdf;
  s["drm_render_minor"] = d.drm_render_minor;
  s["kfd_gpu_id"] = d.kfd_gpu_id;
  s["busy_percent"] = d.busy_percent;
  s["gr_engine_active"] = d.gr_engine_active;
  s["mem_busy_percent"] = d.mem_busy_percent;
  s["power_w"] = d.power_w;
  s["vram_used_b"] = d.vram_used_b;
  s["vram_total_b"] = d.vram_total_b;
  s["temp_edge_c"] = d.temp_edge_c;
  s["gfx_clock_mhz"] = d.gfx_clock_mhz;
  s["energy_j"] = d.energy_j;
  s["metrics_table_ok"] = d.metrics_table_ok;
  s["gfx_activity_acc"] = d.gfx_activity_acc;
  s["firmware_timestamp"] = d.firmware_timestamp;
  s["read_ok"] = d.read_ok;
  s["healthy"] = d.healthy;
  s["staleness_s"] = d.staleness_s;
  s["xgmi_link_width"] = d.xgmi_link_width;
  s["xgmi_link_speed"] = d.xgmi_link_speed;
  s["xgmi_read_kb"] = d.xgmi_read_kb;
  s["xgmi_write_kb"] = d.xgmi_write_kb;
  return s;
}

DeviceSample sample_from_json(const jsn::Value& v) {
  DeviceSample d;
  d.index = static_cast<uint32_t>(v.get("index").as_int(0));
  d.model_name = v.get("model_name").as_string_or("AMD Instinct MI355X");
  d.unique_id = v.get("unique_id").as_string_or("");
  d.drm_render_minor = static_cast<uint32_t>(v.get("drm_render_minor").as_int(128));
  d.kfd_gpu_id = static_cast<uint64_t>(v.get("kfd_gpu_id").as_int(0));
  d.pci_bdf = v.get("pci_bdf").as_string_or("");
  d.busy_percent = v.get("busy_percent").as_double(0);
  d.gr_engine_active = v.get("gr_engine_active").as_double(0);
  d.mem_busy_percent = v.get("mem_busy_percent").as_double(0);
  d.power_w = v.get("power_w").as_double(0);
  d.vram_used_b = v.get("vram_used_b").as_double(0);
  d.vram_total_b = v.get("vram_total_b").as_double(0);
  d.temp_edge_c = v.get("temp_edge_c").as_double(0);
  d.gfx_clock_mhz = v.get("gfx_clock_mhz").as_double(0);
  d.healthy = v.get("healthy").as_bool(true);
  d.staleness_s = v.get("staleness_s").as_double(0);
  return d;
}

}  // namespace

PYBIND11_MODULE(_gpumon, m) {
  m.doc() = "native mi355-exporter internals: ROCm sampler, attribution, registry";

  logx::init(logx::Format::Default);

  py::register_exception<SamplerError>(m, "SamplerError");
  py::register_exception<grpcx::GrpcError>(m, "GrpcError");

  py::class_<Sampler>(m, "Sampler")
      .def(py::init<int, double, double>(), py::arg("poll_interval_ms") = 1000,
           py::arg("idle_epsilon") = 0.005, py::arg("window_s") = 30.0)
      .def("init", &Sampler::init,
           "Initialize rocm_smi and enumerate devices (raises SamplerError "
           "without an AMD GPU)")
      .def("start", &Sampler::start)
      .def("stop", &Sampler::stop)
      .def("poll_once", &Sampler::poll_once, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("device_count", &Sampler::device_count)
      .def("snapshot", [](Sampler& s) {
        auto samples = s.snapshot();
        py::list out;
        for (const auto& d : samples) out.append(sample_to_dict(d));
        return out;
      });

  // Pure sliding-window integrator — CPU-unit-testable core of the sampler's
  // scrape-idempotent activity ratio (no rsmi involved).
  py::class_<ActivityWindow>(m, "ActivityWindow")
      .def(py::init<>())
      .def("add", &ActivityWindow::add, py::arg("t"), py::arg("ratio"),
           py::arg("known") = true)
      .def("ratio",
           [](const ActivityWindow& w, double now, double window_s) {
             double known = 0.0;
             double r = w.ratio(now, window_s, &known);
             return py::make_tuple(r, known);
           },
           py::arg("now"), py::arg("window_s"),
           "-> (time-weighted ratio over known segments, known seconds)")
      .def("set_retention", &ActivityWindow::set_retention)
      .def_property_readonly("size", &ActivityWindow::size);

  m.def("list_pod_resources",
        [](const std::string& socket_path, int timeout_ms) {
          std::vector<PodResourcesEntry> entries;
          {
            py::gil_scoped_release nogil;
            entries = list_pod_resources(socket_path, timeout_ms);
          }
          py::list out;
          for (const auto& e : entries) {
            py::dict d;
            d["pod"] = e.pod;
            d["namespace"] = e.ns;
            d["container"] = e.container;
            py::list devs;
            for (const auto& cd : e.devices) {
              py::dict dd;
              dd["resource_name"] = cd.resource_name;
              dd["device_ids"] = cd.device_ids;
              devs.append(dd);
            }
            d["devices"] = devs;
            out.append(d);
          }
          return out;
        },
        py::arg("socket_path"), py::arg("timeout_ms") = 5000,
        "Unary v1.PodResourcesLister/List over the kubelet unix socket "
        "(hand-rolled h2c + protobuf, no grpc library)");

  py::register_exception<PodResourcesError>(m, "PodResourcesError");

  m.def("grpc_unary_call",
        [](const std::string& host, uint16_t port, const std::string& method_path,
           py::bytes request_msg, int timeout_ms) -> py::bytes {
          std::string req = request_msg;
          std::string resp;
          {
            py::gil_scoped_release nogil;
            grpcx::Target t;
            t.host = host;
            t.port = port;
            t.authority = host + ":" + std::to_string(port);
            resp = grpcx::unary_call(t, method_path, req, timeout_ms);
          }
          return py::bytes(resp);
        },
        py::arg("host"), py::arg("port"), py::arg("method_path"),
        py::arg("request_msg"), py::arg("timeout_ms") = 5000,
        "Raw unary gRPC over h2c (native/common/grpc_client.cpp test surface)");

  m.def("hpack_decode",
        [](py::bytes block) {
          std::string b = block;
          grpcx::HpackDecoder d;
          py::list out;
          for (const auto& [k, v] : d.decode_block(b))
            out.append(py::make_tuple(k, v));
          return out;
        },
        py::arg("block"),
        "Decode one HPACK header block (response-side decoder test surface)");

  m.def("hpack_decode_blocks",
        [](const std::vector<py::bytes>& blocks) {
          grpcx::HpackDecoder d;  // dynamic table persists across blocks
          py::list out;
          for (const auto& blk : blocks) {
            std::string b = blk;
            py::list hs;
            for (const auto& [k, v] : d.decode_block(b))
              hs.append(py::make_tuple(k, v));
            out.append(hs);
          }
          return out;
        },
        py::arg("blocks"),
        "Decode successive header blocks sharing one dynamic table "
        "(RFC 7541 Appendix C multi-request examples)");

  m.def("huffman_decode",
        [](py::bytes data) {
          std::string d = data;
          return py::bytes(grpcx::huffman_decode(
              reinterpret_cast<const uint8_t*>(d.data()), d.size()));
        },
        py::arg("data"), "RFC 7541 Appendix B Huffman string decode");

  m.def("decode_list_response", [](py::bytes payload) {
    std::string data = payload;
    auto entries = decode_list_response(data);
    py::list out;
    for (const auto& e : entries) {
      py::dict d;
      d["pod"] = e.pod;
      d["namespace"] = e.ns;
      d["container"] = e.container;
      py::list devs;
      for (const auto& cd : e.devices) {
        py::dict dd;
        dd["resource_name"] = cd.resource_name;
        dd["device_ids"] = cd.device_ids;
        devs.append(dd);
      }
      d["devices"] = devs;
      out.append(d);
    }
    return out;
  });

  m.def("pod_uid_from_cgroup", [](const std::string& text) -> py::object {
    auto uid = pod_uid_from_cgroup(text);
    return uid ? py::cast(*uid) : py::none();
  });

  m.def("kfd_gpu_pids", [] {
    py::dict out;
    for (const auto& [gpu_id, pids] : kfd_gpu_pids()) out[py::cast(gpu_id)] = pids;
    return out;
  });

  py::class_<Attributor>(m, "Attributor")
      .def(py::init<>())
      .def("resolve",
           [](Attributor& a, const std::vector<std::pair<uint32_t, uint64_t>>& idx_kfd) {
             // release the GIL during the native work: the apiserver this
             // talks to may be an in-process Python fixture
             std::map<uint32_t, PodAttribution> resolved;
             {
               py::gil_scoped_release nogil;
               resolved = a.resolve(idx_kfd);
             }
             py::dict out;
             for (const auto& [idx, attr] : resolved) {
               py::dict v;
               v["pod"] = attr.pod;
               v["namespace"] = attr.ns;
               v["container"] = attr.container;
               out[py::cast(idx)] = v;
             }
             return out;
           })
      .def("resolve_full",
           [](Attributor& a, const std::string& samples_json) {
             jsn::Value sv = jsn::parse(samples_json);
             std::vector<DeviceSample> devices;
             for (const auto& v : sv.arr()) devices.push_back(sample_from_json(v));
             std::map<uint32_t, PodAttribution> resolved;
             {
               py::gil_scoped_release nogil;
               resolved = a.resolve_full(devices);
             }
             py::dict out;
             for (const auto& [idx, attr] : resolved) {
               py::dict v;
               v["pod"] = attr.pod;
               v["namespace"] = attr.ns;
               v["container"] = attr.container;
               out[py::cast(idx)] = v;
             }
             return out;
           },
           "Allocation-first attribution: PodResources socket, then KFD fallback")
      .def("lookup_uid", [](Attributor& a, const std::string& uid) -> py::object {
        std::optional<PodAttribution> attr;
        {
          py::gil_scoped_release nogil;
          attr = a.lookup_uid(uid);
        }
        if (!attr) return py::none();
        py::dict v;
        v["pod"] = attr->pod;
        v["namespace"] = attr->ns;
        v["container"] = attr->container;
        return v;
      });

  // Render synthetic samples (JSON list) — lets the exposition format be
  // pinned by CPU tests.
  m.def("render_metrics",
        [](const std::string& samples_json, const std::string& attribs_json,
           const std::string& hostname, const std::string& node_type) {
          jsn::Value sv = jsn::parse(samples_json);
          std::vector<DeviceSample> samples;
          for (const auto& v : sv.arr()) samples.push_back(sample_from_json(v));
          std::map<uint32_t, PodAttribution> attribs;
          if (!attribs_json.empty()) {
            jsn::Value av = jsn::parse(attribs_json);
            for (const auto& [k, v] : av.obj()) {
              attribs[static_cast<uint32_t>(std::stoul(k))] = PodAttribution{
                  v.get("pod").as_string(), v.get("namespace").as_string(),
                  v.get("container").as_string_or("")};
            }
          }
          RenderOptions opts;
          opts.hostname = hostname;
          if (!node_type.empty()) opts.const_labels.emplace_back("node_type", node_type);
          return render_metrics(samples, attribs, opts);
        },
        py::arg("samples_json"), py::arg("attribs_json") = "",
        py::arg("hostname") = "test-node", py::arg("node_type") = "");
}
