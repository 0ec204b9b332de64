#include "control.h"

#include <algorithm>

#include <sstream>

#include <thread>

#include "matching.h"
#include "scheduler.h"

namespace omq {

void apply_probe(BackendStatus& b, const ProbeResult& p) {
    // merge (reference apply_probe, src/control.rs:340-352)
    b.is_online = p.online;
    if (p.api_type != ApiType::Unknown) b.api_type = p.api_type;
    b.available_models = p.available_models;
    b.loaded_models = p.loaded_models;
    b.loaded_ctx = p.loaded_ctx;
    if (p.lmstudio) b.lmstudio = true;
    if (!p.native_display.empty()) b.native_display = p.native_display;
    if (!p.native_instance.empty()) b.native_instance = p.native_instance;
    for (const auto& e : p.bad_endpoints) b.known_bad_endpoints.insert(e);
    for (const auto& e : p.good_endpoints) b.known_bad_endpoints.erase(e);
}

void probe_all(AppState& st, bool full_reprobe) {
    // snapshot (idx, impl, was_online, skip) without holding locks during IO
    struct Item {
        size_t idx;
        std::shared_ptr<Backend> impl;
        bool was_online;
        std::set<std::string> skip;
    };
    std::vector<Item> items;
    {
        std::lock_guard<std::mutex> g(st.backends_mu);
        for (size_t i = 0; i < st.backends.size(); i++) {
            Item it;
            it.idx = i;
            it.impl = st.impls[i];
            it.was_online = st.backends[i].is_online;
            // offline backends get a clean full probe every tick, and a
            // skip set that would drop BOTH primary probes is cleared
            // entirely (reference dispatcher.rs:427-439)
            if (!full_reprobe && it.was_online) {
                it.skip = st.backends[i].known_bad_endpoints;
                if (it.skip.count("/api/tags") &&
                    it.skip.count("/v1/models"))
                    it.skip.clear();
            }
            items.push_back(std::move(it));
        }
    }
    bool changed = false;
    for (auto& it : items) {
        ProbeResult p = it.impl->probe(it.skip);
        std::lock_guard<std::mutex> g(st.backends_mu);
        if (it.idx >= st.backends.size()) continue;
        auto& b = st.backends[it.idx];
        if (!it.was_online && p.online) {
            // recovery: clear learned memory (dispatcher.rs:449-453)
            b.known_bad_endpoints.clear();
        }
        const bool delta = b.is_online != p.online ||
                           b.api_type != p.api_type ||
                           b.available_models != p.available_models ||
                           b.loaded_models != p.loaded_models;
        apply_probe(b, p);
        changed |= delta;
    }
    if (changed) st.notify();
}

static Json control_err(const std::string& msg) {
    Json j = Json::object();
    j.set("error", Json::string(msg));
    return j;
}

ControlOutcome start_model_control(AppState& st, const ControlRequest& req) {
    std::string canonical;
    std::shared_ptr<Backend> impl;
    BackendStatus snap;
    {
        // lock order: control before backends (control.rs:963-965)
        std::scoped_lock lk(st.control_mu, st.backends_mu);
        if (req.backend_idx >= st.backends.size())
            return {404, control_err("backend not found")};
        auto& b = st.backends[req.backend_idx];
        if (st.control_ops.count(req.backend_idx))
            return {409, control_err("another model operation is already "
                                     "running on this backend")};
        if (!b.is_online)
            return {400, control_err("backend is offline")};
        if (b.active_requests > 0)
            return {409, control_err("backend is busy serving requests")};
        impl = st.impls[req.backend_idx];
        if (!impl->supports_control(b))
            return {400, control_err("backend has no model-control API")};
        auto resolved = resolve_model_name(req.model, b.available_models,
                                           b.native_display);
        if (!resolved) {
            if (req.action == ControlAction::Unload) {
                // allow unloading a loaded-but-not-listed model by exact name
                for (const auto& l : b.loaded_models)
                    if (l == req.model) resolved = req.model;
            }
            if (!resolved)
                return {404, control_err("model not found on this backend: " +
                                         req.model)};
        }
        canonical = *resolved;
        if (req.action == ControlAction::Unload) {
            bool loaded = false;
            for (const auto& l : b.loaded_models)
                if (smart_model_match_one(canonical, l) ||
                    l == canonical)
                    loaded = true;
            if (!loaded)
                return {400, control_err("model is not loaded: " + canonical)};
        }
        st.control_ops[req.backend_idx] =
            ControlOp{req.action, canonical, req.model, now_ms()};
        snap = b;
    }
    st.log.push("CTL", std::string(req.action == ControlAction::Load
                                       ? "load "
                                       : "unload ") +
                           canonical + " on b" +
                           std::to_string(req.backend_idx) +
                           (req.identifier.empty()
                                ? ""
                                : " [id: " + req.identifier + "]"));
    st.notify();  // backend is now scheduler-busy

    const ControlRequest r = req;
    const std::string model = canonical;
    std::thread([&st, r, model, impl, snap] {
        std::string err;
        if (r.action == ControlAction::Load)
            err = impl->load_model(model, r.num_ctx,
                                   r.keep_alive ? r.keep_alive
                                                : st.settings.load_keep_alive_s,
                                   snap);
        else
            err = impl->unload_model(model, snap);
        {
            std::scoped_lock lk(st.control_mu);
            st.control_ops.erase(r.backend_idx);
            st.control_history.push_back(ControlResult{
                r.action, model, r.backend_idx, err.empty(), err, now_ms()});
            while (st.control_history.size() > 20)
                st.control_history.pop_front();
        }
        st.log.push("CTL", (err.empty() ? "done: " : "failed: ") +
                               model + (err.empty() ? "" : " (" + err + ")"));
        // immediate post-op re-probe (control.rs:1137-1146)
        ProbeResult p = impl->probe({});
        {
            std::lock_guard<std::mutex> g(st.backends_mu);
            if (r.backend_idx < st.backends.size())
                apply_probe(st.backends[r.backend_idx], p);
        }
        st.notify();
    }).detach();

    Json body = Json::object();
    body.set("status", Json::string("accepted"));
    body.set("action", Json::string(req.action == ControlAction::Load
                                        ? "load"
                                        : "unload"));
    // reference echoes the backend URL, not the index (control.rs:1283)
    body.set("backend", Json::string(snap.url));
    body.set("backend_index", Json::number((double)req.backend_idx));
    body.set("model", Json::string(canonical));
    if (!req.identifier.empty())
        body.set("identifier", Json::string(req.identifier));
    if (req.num_ctx > 0) body.set("num_ctx", Json::number((double)req.num_ctx));
    if (req.action == ControlAction::Load)
        body.set("keep_alive",
                 Json::number((double)(req.keep_alive
                                           ? req.keep_alive
                                           : st.settings.load_keep_alive_s)));
    return {202, std::move(body)};
}

// ------------------------------------------------------------- admin API
ControlOutcome admin_stats(AppState& st) {
    Json out = Json::object();
    out.set("uptime_s", Json::number(
        st.started_ms ? (now_ms() - st.started_ms) / 1000.0 : 0));
    int64_t processed = 0, dropped = 0, queued = 0, processing = 0;
    Json users = Json::array();
    {
        std::lock_guard<std::mutex> g(st.queues_mu);
        for (const auto& [name, us] : st.users) {
            processed += us.processed;
            dropped += us.dropped;
            queued += (int64_t)us.queue.size();
            processing += us.processing;
            Json u = Json::object();
            u.set("user", Json::string(name));
            u.set("queued", Json::number((double)us.queue.size()));
            u.set("processing", Json::number((double)us.processing));
            u.set("processed", Json::number((double)us.processed));
            u.set("dropped", Json::number((double)us.dropped));
            users.arr.push_back(std::move(u));
        }
    }
    out.set("processed", Json::number((double)processed));
    out.set("dropped", Json::number((double)dropped));
    out.set("queued", Json::number((double)queued));
    out.set("processing", Json::number((double)processing));
    out.set("users", std::move(users));
    {
        std::lock_guard<std::mutex> g(st.backends_mu);
        Json bs = Json::array();
        for (const auto& b : st.backends) {
            Json e = Json::object();
            e.set("url", Json::string(b.url));
            e.set("online", Json::boolean(b.is_online));
            e.set("active_requests", Json::number(b.active_requests));
            e.set("processed_count", Json::number((double)b.processed_count));
            bs.arr.push_back(std::move(e));
        }
        out.set("backends", std::move(bs));
    }
    {
        std::lock_guard<std::mutex> g(st.waits_mu);
        std::vector<int64_t> w(st.wait_samples_ms.begin(),
                               st.wait_samples_ms.end());
        std::sort(w.begin(), w.end());
        Json qw = Json::object();
        auto pct = [&](double p) -> double {
            if (w.empty()) return 0;
            size_t i = (size_t)(p * (w.size() - 1));
            return (double)w[i];
        };
        qw.set("samples", Json::number((double)w.size()));
        qw.set("p50_ms", Json::number(pct(0.50)));
        qw.set("p90_ms", Json::number(pct(0.90)));
        qw.set("p99_ms", Json::number(pct(0.99)));
        out.set("queue_wait", std::move(qw));
    }
    return {200, std::move(out)};
}

ControlOutcome admin_models_state(AppState& st) {
    // bare JSON array, sorted model lists, operation with requested +
    // elapsed_secs — exact reference shape (control.rs:1320-1350)
    Json arr = Json::array();
    std::scoped_lock lk(st.control_mu, st.backends_mu);
    for (size_t i = 0; i < st.backends.size(); i++) {
        const auto& b = st.backends[i];
        Json e = Json::object();
        e.set("index", Json::number((double)i));
        e.set("url", Json::string(b.url));
        e.set("online", Json::boolean(b.is_online));
        e.set("api", Json::string(api_type_name(b.api_type)));
        e.set("lmstudio", Json::boolean(b.lmstudio));
        e.set("active_requests", Json::number(b.active_requests));
        auto sorted_list = [](std::vector<std::string> v) {
            std::sort(v.begin(), v.end());
            Json a = Json::array();
            for (auto& m : v) a.arr.push_back(Json::string(m));
            return a;
        };
        e.set("available_models", sorted_list(b.available_models));
        e.set("loaded_models", sorted_list(b.loaded_models));
        auto op = st.control_ops.find(i);
        if (op != st.control_ops.end()) {
            Json o = Json::object();
            o.set("action", Json::string(op->second.action ==
                                                 ControlAction::Load
                                             ? "load"
                                             : "unload"));
            o.set("model", Json::string(op->second.model));
            o.set("requested", Json::string(op->second.requested));
            o.set("elapsed_secs",
                  Json::number((double)((now_ms() -
                                         op->second.started_ms) / 1000)));
            e.set("operation", std::move(o));
        } else {
            e.set("operation", Json::null());
        }
        arr.arr.push_back(std::move(e));
    }
    return {200, std::move(arr)};
}

// backend selector: index | "any" | URL substring (control.rs:1179-1240)
static std::optional<size_t> parse_backend_selector(
    AppState& st, const Json& body, const std::string& model,
    ControlAction action, std::string* err) {
    std::string sel;
    const Json* bsel = body.find("backend");
    if (bsel) {
        if (bsel->is_num()) {
            const size_t idx = (size_t)bsel->num;
            std::lock_guard<std::mutex> g(st.backends_mu);
            if (idx >= st.backends.size()) {
                *err = "backend index out of range";
                return std::nullopt;
            }
            return idx;
        }
        sel = bsel->str;
        // trim (reference control.rs:1192)
        const auto b0 = sel.find_first_not_of(" \t");
        if (b0 == std::string::npos) sel.clear();
        else sel = sel.substr(b0, sel.find_last_not_of(" \t") - b0 + 1);
    }
    // numeric STRING is an index too (reference control.rs:1210-1211);
    // without this, "2" would substring-match any URL containing a 2
    if (!sel.empty() &&
        sel.find_first_not_of("0123456789") == std::string::npos) {
        const size_t idx = (size_t)strtoull(sel.c_str(), nullptr, 10);
        std::lock_guard<std::mutex> g(st.backends_mu);
        if (idx >= st.backends.size()) {
            *err = "backend index out of range";
            return std::nullopt;
        }
        return idx;
    }
    std::scoped_lock lk(st.control_mu, st.backends_mu);
    if (sel.empty() || lower(sel) == "any") {
        // first online, idle, control-capable backend with the model
        // resolvable (load) or loaded (unload)
        for (size_t i = 0; i < st.backends.size(); i++) {
            const auto& b = st.backends[i];
            if (!b.is_online || b.active_requests > 0 ||
                st.control_ops.count(i))
                continue;
            if (!st.impls[i]->supports_control(b)) continue;
            if (action == ControlAction::Load) {
                if (resolve_model_name(model, b.available_models,
                                       b.native_display))
                    return i;
            } else {
                for (const auto& l : b.loaded_models)
                    if (smart_model_match_one(model, l)) return i;
            }
        }
        *err = "no suitable backend found for model " + model;
        return std::nullopt;
    }
    // URL substring, case-insensitive, first match (control.rs:1213-1219)
    const std::string sl = lower(sel);
    for (size_t i = 0; i < st.backends.size(); i++) {
        const std::string u = lower(st.backends[i].url);
        if (u == sl || u.find(sl) != std::string::npos) return i;
    }
    *err = "no backend matches selector " + sel;
    return std::nullopt;
}

static ControlOutcome admin_control(AppState& st, const std::string& body,
                                    ControlAction action) {
    auto j = Json::parse(body);
    if (!j) return {400, control_err("invalid JSON body")};
    const std::string model = j->get_str("model");
    if (model.empty() ||
        model.find_first_not_of(" \t\r\n") == std::string::npos)
        return {400, control_err("missing model")};
    std::string sel_err;
    auto idx = parse_backend_selector(st, *j, model, action, &sel_err);
    if (!idx) return {404, control_err(sel_err)};
    ControlRequest req;
    req.action = action;
    req.model = model;
    req.backend_idx = *idx;
    req.num_ctx = (int64_t)j->get_num("num_ctx", 0);
    req.keep_alive = (int64_t)j->get_num("keep_alive", 0);
    req.identifier = j->get_str("identifier");
    // apply per-model config defaults (max_ctx/keep_alive) when present
    {
        std::lock_guard<std::mutex> g(st.models_mu);
        for (const auto& mc : st.model_config)
            if (smart_model_match_one(model, mc.name) ||
                mc.identifier == model) {
                if (req.num_ctx == 0) req.num_ctx = mc.max_ctx;
                if (req.keep_alive == 0) req.keep_alive = mc.keep_alive;
            }
    }
    return start_model_control(st, req);
}

ControlOutcome admin_model_load(AppState& st, const std::string& body) {
    return admin_control(st, body, ControlAction::Load);
}
ControlOutcome admin_model_unload(AppState& st, const std::string& body) {
    return admin_control(st, body, ControlAction::Unload);
}

// -------------------------------------------------- declarative config
void apply_model_config(AppState& st) {
    std::vector<ModelConfigEntry> cfg;
    {
        std::lock_guard<std::mutex> g(st.models_mu);
        cfg = st.model_config;
    }
    if (cfg.empty()) return;
    // group entries by backend index (via selector list), then apply
    // sequentially per backend on one thread per backend
    std::map<size_t, std::vector<ModelConfigEntry>> per_backend;
    {
        std::lock_guard<std::mutex> g(st.backends_mu);
        for (const auto& e : cfg) {
            if (e.backends.empty()) {
                // every backend that can resolve it gets a chance: first one
                for (size_t i = 0; i < st.backends.size(); i++)
                    if (resolve_model_name(e.name,
                                           st.backends[i].available_models,
                                           st.backends[i].native_display)) {
                        per_backend[i].push_back(e);
                        break;
                    }
            } else {
                for (const auto& sel : e.backends) {
                    const std::string sl = lower(sel);
                    for (size_t i = 0; i < st.backends.size(); i++) {
                        if (lower(st.backends[i].url).find(sl) !=
                                std::string::npos ||
                            sel == std::to_string(i)) {
                            per_backend[i].push_back(e);
                            break;
                        }
                    }
                }
            }
        }
    }
    for (auto& [idx, entries] : per_backend) {
        std::thread([&st, idx = idx, entries = entries] {
            for (const auto& e : entries) {
                // live re-probe (control.rs:599-720): skip when resident at
                // the right ctx; reload on mismatch
                std::shared_ptr<Backend> impl;
                {
                    std::lock_guard<std::mutex> g(st.backends_mu);
                    if (idx >= st.impls.size()) return;
                    impl = st.impls[idx];
                }
                ProbeResult p = impl->probe({});
                {
                    std::lock_guard<std::mutex> g(st.backends_mu);
                    apply_probe(st.backends[idx], p);
                }
                bool resident = false;
                int64_t resident_ctx = 0;
                std::string canonical = e.name;
                {
                    std::lock_guard<std::mutex> g(st.backends_mu);
                    const auto& b = st.backends[idx];
                    auto res = resolve_model_name(e.name, b.available_models,
                                                  b.native_display);
                    if (res) canonical = *res;
                    for (const auto& l : b.loaded_models)
                        if (smart_model_match_one(canonical, l)) {
                            resident = true;
                            auto c = b.loaded_ctx.find(l);
                            if (c != b.loaded_ctx.end())
                                resident_ctx = c->second;
                        }
                }
                const bool ctx_mismatch =
                    resident && e.max_ctx > 0 && resident_ctx > 0 &&
                    resident_ctx != e.max_ctx;
                if (resident && !ctx_mismatch) {
                    st.log.push("CTL", "config: " + canonical +
                                           " already resident on b" +
                                           std::to_string(idx) + ", skip");
                    continue;
                }
                if (ctx_mismatch) {
                    ControlRequest ur{ControlAction::Unload, canonical, idx,
                                      0, 0};
                    start_model_control(st, ur);
                    // settle (reference waits 1 s between unload and load)
                    for (int i = 0; i < 100; i++) {
                        std::this_thread::sleep_for(
                            std::chrono::milliseconds(100));
                        std::lock_guard<std::mutex> g(st.control_mu);
                        if (!st.control_ops.count(idx)) break;
                    }
                    std::this_thread::sleep_for(std::chrono::seconds(1));
                }
                ControlRequest lr{ControlAction::Load, canonical, idx,
                                  e.max_ctx, e.keep_alive, e.identifier};
                start_model_control(st, lr);
                // wait for completion before next entry on this backend
                for (int i = 0; i < 6000; i++) {
                    std::this_thread::sleep_for(
                        std::chrono::milliseconds(100));
                    std::lock_guard<std::mutex> g(st.control_mu);
                    if (!st.control_ops.count(idx)) break;
                }
            }
        }).detach();
    }
}

}  // namespace omq

namespace omq {

// ------------------------------------------------------- /metrics (Prom)
// Prometheus text exposition of the same counters admin_stats reports.
std::string metrics_text(AppState& st) {
    std::ostringstream o;
    o << "# HELP ollamamq_uptime_seconds Dispatcher uptime.\n"
         "# TYPE ollamamq_uptime_seconds gauge\n"
      << "ollamamq_uptime_seconds "
      << (st.started_ms ? (now_ms() - st.started_ms) / 1000.0 : 0) << "\n";
    int64_t processed = 0, dropped = 0, queued = 0, processing = 0;
    {
        std::lock_guard<std::mutex> g(st.queues_mu);
        for (const auto& [name, us] : st.users) {
            processed += us.processed;
            dropped += us.dropped;
            queued += (int64_t)us.queue.size();
            processing += us.processing;
        }
    }
    o << "# TYPE ollamamq_requests_processed_total counter\n"
      << "ollamamq_requests_processed_total " << processed << "\n"
      << "# TYPE ollamamq_requests_dropped_total counter\n"
      << "ollamamq_requests_dropped_total " << dropped << "\n"
      << "# TYPE ollamamq_requests_queued gauge\n"
      << "ollamamq_requests_queued " << queued << "\n"
      << "# TYPE ollamamq_requests_processing gauge\n"
      << "ollamamq_requests_processing " << processing << "\n";
    {
        std::lock_guard<std::mutex> g(st.backends_mu);
        o << "# TYPE ollamamq_backend_online gauge\n"
             "# TYPE ollamamq_backend_active_requests gauge\n"
             "# TYPE ollamamq_backend_processed_total counter\n";
        for (const auto& b : st.backends) {
            o << "ollamamq_backend_online{url=\"" << b.url << "\"} "
              << (b.is_online ? 1 : 0) << "\n"
              << "ollamamq_backend_active_requests{url=\"" << b.url
              << "\"} " << b.active_requests << "\n"
              << "ollamamq_backend_processed_total{url=\"" << b.url
              << "\"} " << b.processed_count << "\n";
        }
    }
    {
        std::lock_guard<std::mutex> g(st.waits_mu);
        std::vector<int64_t> w(st.wait_samples_ms.begin(),
                               st.wait_samples_ms.end());
        std::sort(w.begin(), w.end());
        auto pct = [&](double p) -> double {
            if (w.empty()) return 0;
            return (double)w[(size_t)(p * (w.size() - 1))];
        };
        o << "# TYPE ollamamq_queue_wait_ms summary\n"
          << "ollamamq_queue_wait_ms{quantile=\"0.5\"} " << pct(0.50) << "\n"
          << "ollamamq_queue_wait_ms{quantile=\"0.9\"} " << pct(0.90) << "\n"
          << "ollamamq_queue_wait_ms{quantile=\"0.99\"} " << pct(0.99)
          << "\n"
          << "ollamamq_queue_wait_ms_count " << w.size() << "\n";
    }
    return o.str();
}

}  // namespace omq
