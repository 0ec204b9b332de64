#include "server.h"

#include <algorithm>
#include <cstring>

#include "config.h"
#include "control.h"
#include "http_backend.h"
#include "json.h"
#include "scheduler.h"

namespace omq {

// the proxied API surface (reference src/main.rs:264-291)
static const char* kProxied[] = {
    "/", "/api/generate", "/api/chat", "/api/embed", "/api/embeddings",
    "/api/tags", "/api/show", "/api/create", "/api/copy", "/api/delete",
    "/api/pull", "/api/push", "/api/ps", "/api/version",
    "/v1/chat/completions", "/v1/completions", "/v1/embeddings",
    "/v1/models",
};

bool is_proxied_route(const std::string& path) {
    for (const char* p : kProxied)
        if (path == p) return true;
    if (path.rfind("/api/blobs/", 0) == 0) return true;      // /api/blobs/{d}
    if (path.rfind("/v1/models/", 0) == 0) return true;      // /v1/models/{m}
    return false;
}

Server::Server(std::string cfg_path) : config_path(std::move(cfg_path)) {}

Server::~Server() { stop(); }

void Server::add_http_backend(const std::string& url) {
    const std::string u = normalize_backend_url(url);
    add_backend(std::make_shared<HttpBackend>(
                    u, st_.settings.timeout_s,
                    st_.settings.load_keep_alive_s),
                u);
}

void Server::add_backend(std::shared_ptr<Backend> impl,
                         const std::string& name) {
    std::lock_guard<std::mutex> g(st_.backends_mu);
    BackendStatus b;
    b.url = name;
    st_.backends.push_back(std::move(b));
    st_.impls.push_back(std::move(impl));
}

int Server::port() const { return http_ ? http_->port() : 0; }

bool Server::start(std::string* err) {
    st_.started_ms = now_ms();
    st_.load_blocked();
    http_ = std::make_unique<HttpServer>(
        st_.settings.host, st_.settings.port,
        [this](const HttpRequest& r, HttpConn& c) { handle(r, c); });
    if (!http_->start(err)) return false;
    started_ = true;
    worker_ = std::thread([this] { run_worker(); });
    health_ = std::thread([this] { health_loop(); });
    return true;
}

void Server::stop() {
    if (!started_.exchange(false)) {
        if (http_) http_->stop();
        return;
    }
    st_.shutting_down = true;
    st_.notify();
    if (http_) http_->stop();
    if (worker_.joinable()) worker_.join();
    if (health_.joinable()) health_.join();
}

// ------------------------------------------------------------------- auth
static bool const_time_eq(const std::string& a, const std::string& b) {
    // constant-time comparison (reference main.rs:79-89)
    unsigned char acc = a.size() == b.size() ? 0 : 1;
    for (size_t i = 0; i < a.size(); i++)
        acc |= (unsigned char)(a[i] ^ b[b.empty() ? 0 : i % b.size()]);
    return acc == 0;
}

bool Server::auth_ok(const HttpRequest& req) const {
    const std::string& key = st_.settings.api_key;
    if (key.empty()) return true;
    const std::string xk = req.header("X-API-Key");
    if (!xk.empty() && const_time_eq(xk, key)) return true;
    std::string auth = req.header("Authorization");
    if (auth.size() > 7) {
        std::string scheme = auth.substr(0, 7);
        std::transform(scheme.begin(), scheme.end(), scheme.begin(),
                       ::tolower);
        if (scheme == "bearer " && const_time_eq(auth.substr(7), key))
            return true;
    }
    return false;
}

// ----------------------------------------------------------------- router
void Server::handle(const HttpRequest& req, HttpConn& conn) {
    // /health outside the auth layer (reference main.rs:305-307)
    if (req.path == "/health") {
        conn.send(200, {{"Content-Type", "text/plain"}}, "OK");
        return;
    }
    if (!auth_ok(req)) {
        conn.send(401,
                  {{"Content-Type", "application/json"},
                   {"WWW-Authenticate", "Bearer"}},
                  "{\"error\":\"unauthorized\"}");
        return;
    }
    if (req.path == "/admin/stats" && req.method == "GET") {
        auto out = admin_stats(st_);
        conn.send(out.http_status, {}, out.body.dump());
        return;
    }
    if (req.path == "/metrics" && req.method == "GET") {
        // Prometheus exposition (beyond the reference: SURVEY §5 calls for
        // a first-class metrics surface for the tokens/queue-wait metric)
        conn.send(200,
                  {{"Content-Type",
                    "text/plain; version=0.0.4; charset=utf-8"}},
                  metrics_text(st_));
        return;
    }
    if (req.path == "/admin/models" && req.method == "GET") {
        auto out = admin_models_state(st_);
        conn.send(out.http_status, {}, out.body.dump());
        return;
    }
    if (req.path == "/admin/models/load" && req.method == "POST") {
        auto out = admin_model_load(st_, req.body);
        conn.send(out.http_status, {}, out.body.dump());
        return;
    }
    if (req.path == "/admin/models/unload" && req.method == "POST") {
        auto out = admin_model_unload(st_, req.body);
        conn.send(out.http_status, {}, out.body.dump());
        return;
    }
    if (is_proxied_route(req.path) || st_.settings.allow_all_routes) {
        proxy_handler(req, conn);
        return;
    }
    conn.send(404, {}, "{\"error\":\"not found\"}");
}

// ------------------------------------------------------- ingress + stream
// reference proxy_handler (src/dispatcher.rs:841-940)
void Server::proxy_handler(const HttpRequest& req, HttpConn& conn) {
    std::string user = req.header("X-User-ID");
    if (user.empty()) user = "anonymous";

    {
        // IP check first, distinct bodies (reference dispatcher.rs:857-865)
        std::lock_guard<std::mutex> g(st_.blocked_mu);
        if (st_.blocked_ips.count(req.client_ip)) {
            conn.send(403, {}, "{\"error\":\"IP blocked\"}");
            return;
        }
        if (st_.blocked_users.count(user)) {
            conn.send(403, {}, "{\"error\":\"User blocked\"}");
            return;
        }
    }
    {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        st_.user_ips[user] = req.client_ip;
    }

    Task t;
    t.method = req.method;
    t.path = req.path;
    t.query = req.query;
    for (const auto& [k, v] : req.headers) {
        std::string lk = k;
        std::transform(lk.begin(), lk.end(), lk.begin(), ::tolower);
        if (lk == "host" || lk == "content-length" ||
            lk == "transfer-encoding" || lk == "connection")
            continue;  // Host stripped (reference dispatcher.rs:873-874)
        t.headers.emplace_back(k, v);
    }
    t.body = req.body;
    t.user_id = user;
    if (!req.body.empty()) {
        auto j = Json::parse(req.body);
        if (j) t.requested_model = j->get_str("model");
    }
    t.resp = std::make_shared<ResponseChannel>();
    t.queued_at_ms = now_ms();
    auto resp = t.resp;

    {
        std::lock_guard<std::mutex> g(st_.queues_mu);
        st_.users[user].queue.push_back(std::move(t));
    }
    st_.log.push("IN", user + " " + req.method + " " + req.path);
    st_.notify();

    // wait for the first part, then stream (reference dispatcher.rs:916-939)
    const int timeout_ms = (int)(st_.settings.timeout_s * 1000);
    if (!resp->wait_started(timeout_ms)) {
        resp->mark_client_gone();
        conn.send(500, {}, "{\"error\":\"Worker failed to respond\"}");
        return;
    }
    int status;
    std::vector<std::pair<std::string, std::string>> headers;
    {
        std::lock_guard<std::mutex> g(resp->mu);
        status = resp->status ? resp->status : 500;
        headers = resp->headers;
    }
    if (!conn.begin_stream(status, headers)) {
        resp->mark_client_gone();
        return;
    }
    std::string chunk;
    while (resp->next_chunk(&chunk, timeout_ms)) {
        if (!conn.write_chunk(chunk)) {
            // client disconnected: propagate so the executor stops
            resp->mark_client_gone();
            return;
        }
    }
    conn.end_stream();
}

// ------------------------------------------------------------- scheduler
void Server::run_worker() {
    while (!st_.shutting_down) {
        Dispatch d;
        while (schedule_once(st_, &d)) {
            // late block re-check (reference dispatcher.rs:718-735)
            bool blocked;
            {
                std::lock_guard<std::mutex> g(st_.blocked_mu);
                blocked = st_.blocked_users.count(d.user) > 0;
            }
            std::shared_ptr<Backend> impl;
            {
                std::lock_guard<std::mutex> g(st_.backends_mu);
                impl = st_.impls[d.backend_idx];
            }
            auto dd = std::make_shared<Dispatch>(std::move(d));
            if (blocked || (dd->task.resp && dd->task.resp->client_gone)) {
                if (dd->task.resp) dd->task.resp->finish();
                finish_dispatch(st_, *dd, false,
                                blocked ? "blocked" : "client gone");
                continue;
            }
            std::thread([this, dd, impl] {
                const int status = impl->execute(dd->task);
                const bool ok = status >= 200 && status < 500;
                finish_dispatch(st_, *dd, ok,
                                ok ? "ok" : "backend error");
            }).detach();
        }
        st_.wait_work(500);
    }
}

// ------------------------------------------------------------ health loop
// reference src/dispatcher.rs:403-492: 10 s probes, full reprobe every 6th
void Server::health_loop() {
    int cycle = 0;
    while (!st_.shutting_down) {
        probe_all(st_, cycle % 6 == 0);   // full reprobe every 6th (~60 s)
        cycle++;
        const int step = 50;
        for (int waited = 0;
             waited < st_.settings.probe_interval_ms && !st_.shutting_down;
             waited += step)
            std::this_thread::sleep_for(std::chrono::milliseconds(step));
    }
}

void reload_model_config(AppState& st, const std::string& config_path) {
    AppConfig cfg;
    std::string err;
    if (!load_config(config_path, &cfg, &err)) {
        st.log.push("CTL", "config reload failed: " + err);
        return;
    }
    {
        std::lock_guard<std::mutex> g(st.models_mu);
        st.model_config = cfg.models;
    }
    apply_model_config(st);
}

}  // namespace omq
