// pybind11 bindings: expose the native dispatcher core (matching,
// resolution, scheduler) so the pytest suite drives the EXACT C++ code the
// server binary runs — mirroring how the reference unit-tests its scheduler
// and matching in-process (reference src/dispatcher.rs:942-984,
// src/control.rs:1385-1459).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "config.h"
#include "core.h"
#include "json.h"
#include "matching.h"
#include "scheduler.h"

namespace py = pybind11;
using namespace omq;

namespace {

// Test harness around AppState + schedule_once with mock backends.
class Harness {
public:
    Harness() { st_.settings.stuck_timeout_s = 0; }  // off by default

    size_t add_backend(const std::string& url, bool online,
                       const std::string& api,
                       std::vector<std::string> available,
                       std::vector<std::string> loaded, int max_conc) {
        std::lock_guard<std::mutex> g(st_.backends_mu);
        BackendStatus b;
        b.url = url;
        b.is_online = online;
        b.api_type = api == "ollama"   ? ApiType::Ollama
                     : api == "openai" ? ApiType::OpenAi
                     : api == "both"   ? ApiType::Both
                                       : ApiType::Unknown;
        b.available_models = std::move(available);
        b.loaded_models = std::move(loaded);
        b.max_concurrency = max_conc;
        st_.backends.push_back(std::move(b));
        return st_.backends.size() - 1;
    }

    void set_online(size_t i, bool online) {
        std::lock_guard<std::mutex> g(st_.backends_mu);
        st_.backends.at(i).is_online = online;
    }
    void set_control_op(size_t i, bool active) {
        std::lock_guard<std::mutex> g(st_.control_mu);
        if (active)
            st_.control_ops[i] = {ControlAction::Load, "x", "x", now_ms()};
        else
            st_.control_ops.erase(i);
    }
    void set_processed(const std::string& user, int64_t n) {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        st_.users[user].processed = n;
    }
    void set_vip(const std::string& u) { st_.vip_user = u; }
    void set_boost(const std::string& u) { st_.boost_user = u; }
    void set_stuck_timeout(int64_t s) { st_.settings.stuck_timeout_s = s; }
    void set_debug_log(bool on) { st_.log.set_debug(on); }

    void enqueue(const std::string& user, const std::string& model,
                 const std::string& path) {
        Task t;
        t.method = "POST";
        t.path = path;
        t.user_id = user;
        t.requested_model = model;
        t.queued_at_ms = now_ms();
        std::lock_guard<std::mutex> g(st_.queues_mu);
        st_.users[user].queue.push_back(std::move(t));
    }
    void enqueue_aged(const std::string& user, const std::string& model,
                      const std::string& path, int64_t age_ms) {
        Task t;
        t.method = "POST";
        t.path = path;
        t.user_id = user;
        t.requested_model = model;
        t.queued_at_ms = now_ms() - age_ms;
        std::lock_guard<std::mutex> g(st_.queues_mu);
        st_.users[user].queue.push_back(std::move(t));
    }

    // returns (user, backend_idx, model) or None
    py::object schedule() {
        Dispatch d;
        if (!schedule_once(st_, &d)) return py::none();
        inflight_.push_back(d);
        return py::make_tuple(d.user, d.backend_idx, d.task.requested_model);
    }

    void finish(size_t inflight_idx, bool ok) {
        finish_dispatch(st_, inflight_.at(inflight_idx), ok,
                        ok ? "ok" : "error");
    }
    size_t inflight_count() const { return inflight_.size(); }

    int queue_len(const std::string& user) {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        auto it = st_.users.find(user);
        return it == st_.users.end() ? 0 : (int)it->second.queue.size();
    }
    int64_t processed(const std::string& user) {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        return st_.users[user].processed;
    }
    int64_t dropped(const std::string& user) {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        return st_.users[user].dropped;
    }
    int active_requests(size_t i) {
        std::lock_guard<std::mutex> g(st_.backends_mu);
        return st_.backends.at(i).active_requests;
    }
    std::vector<std::pair<std::string, std::string>> log_events() {
        std::vector<std::pair<std::string, std::string>> out;
        for (auto& e : st_.log.snapshot()) out.emplace_back(e.kind, e.text);
        return out;
    }

private:
    AppState st_;
    std::vector<Dispatch> inflight_;
};

}  // namespace

PYBIND11_MODULE(_dispatch, m) {
    m.doc() = "ollamamq_amd native dispatcher core (C++)";

    m.def("smart_model_match_one", &smart_model_match_one);
    m.def("smart_model_match", &smart_model_match);
    m.def("fuzzy_model_match", &fuzzy_model_match);
    m.def("model_routable", &model_routable);
    m.def("resolve_model_name",
          [](const std::string& req, std::vector<std::string> avail,
             std::map<std::string, std::string> native) -> py::object {
              auto r = resolve_model_name(req, avail, native);
              if (!r) return py::none();
              return py::str(*r);
          },
          py::arg("requested"), py::arg("available"),
          py::arg("native_display") = std::map<std::string, std::string>{});

    m.def("candidate_order", &candidate_order);

    m.def("load_config", [](const std::string& path) -> py::object {
        AppConfig cfg;
        std::string err;
        if (!load_config(path, &cfg, &err))
            throw std::runtime_error(err);
        py::dict d;
        d["backends"] = cfg.backends;
        py::dict st;
        st["port"] = cfg.settings.port;
        st["host"] = cfg.settings.host;
        st["timeout"] = cfg.settings.timeout_s;
        st["load_keep_alive"] = cfg.settings.load_keep_alive_s;
        st["allow_all_routes"] = cfg.settings.allow_all_routes;
        st["stuck_timeout"] = cfg.settings.stuck_timeout_s;
        d["settings"] = st;
        py::list models;
        for (const auto& mdl : cfg.models) {
            py::dict e;
            e["name"] = mdl.name;
            e["identifier"] = mdl.identifier;
            e["max_ctx"] = mdl.max_ctx;
            e["keep_alive"] = mdl.keep_alive;
            e["max_concurrent_requests"] = mdl.max_concurrent_requests;
            e["backends"] = mdl.backends;
            models.append(e);
        }
        d["models"] = models;
        return d;
    });
    m.def("normalize_backend_url", &normalize_backend_url);

    m.def("json_roundtrip", [](const std::string& s) -> py::object {
        auto j = Json::parse(s);
        if (!j) return py::none();
        return py::str(j->dump());
    });
    m.def("json_get_model", [](const std::string& body) {
        auto j = Json::parse(body);
        return j ? j->get_str("model") : std::string();
    });

    py::class_<Harness>(m, "Harness")
        .def(py::init<>())
        .def("add_backend", &Harness::add_backend, py::arg("url"),
             py::arg("online") = true, py::arg("api") = "ollama",
             py::arg("available") = std::vector<std::string>{},
             py::arg("loaded") = std::vector<std::string>{},
             py::arg("max_conc") = 1)
        .def("set_online", &Harness::set_online)
        .def("set_control_op", &Harness::set_control_op)
        .def("set_processed", &Harness::set_processed)
        .def("set_vip", &Harness::set_vip)
        .def("set_boost", &Harness::set_boost)
        .def("set_stuck_timeout", &Harness::set_stuck_timeout)
        .def("set_debug_log", &Harness::set_debug_log)
        .def("enqueue", &Harness::enqueue, py::arg("user"),
             py::arg("model") = "", py::arg("path") = "/api/chat")
        .def("enqueue_aged", &Harness::enqueue_aged)
        .def("schedule", &Harness::schedule)
        .def("finish", &Harness::finish)
        .def("inflight_count", &Harness::inflight_count)
        .def("queue_len", &Harness::queue_len)
        .def("processed", &Harness::processed)
        .def("dropped", &Harness::dropped)
        .def("active_requests", &Harness::active_requests)
        .def("log_events", &Harness::log_events);
}
