// WorkerBackend: an in-process MI355X GPU worker as a scheduler backend.
//
// This replaces the reference's process/network boundary (an HTTP hop to an
// external Ollama server, reference src/dispatcher.rs:742) with a direct
// submit to a framework-owned per-GPU engine process over a unix domain
// socket: one engine process per GPU (the torch.distributed/RCCL process
// model), newline-framed JSON control messages, raw byte streaming for
// token output, cancellation by socket close (the worker aborts the
// sequence and frees its KV when the dispatcher hangs up — the reference
// merely dropped bytes, SURVEY.md §7 hard-part 4).
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <cstring>

#include "control.h"
#include "json.h"
#include "server.h"
#include "tui.h"

namespace omq {

namespace {

int uds_connect(const std::string& path, int timeout_ms) {
    int fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) return -1;
    timeval tv{timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
    setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (connect(fd, (sockaddr*)&addr, sizeof addr) != 0) {
        ::close(fd);
        return -1;
    }
    return fd;
}

bool send_all(int fd, const std::string& s) {
    const char* p = s.data();
    size_t n = s.size();
    while (n) {
        ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
        if (w <= 0) return false;
        p += w;
        n -= (size_t)w;
    }
    return true;
}

// Buffered line reader: the old one-byte-per-recv() header read cost
// ~60 syscalls per request on the token hot path's front door (ADVICE
// r01); this reads in 16 KiB gulps and hands any overshoot back to the
// caller via rest() so the byte stream after the header is preserved.
class LineReader {
public:
    explicit LineReader(int fd) : fd_(fd) {}

    bool line(std::string& out) {
        out.clear();
        while (true) {
            auto nl = buf_.find('\n', scan_);
            if (nl != std::string::npos) {
                out = buf_.substr(0, nl);
                buf_.erase(0, nl + 1);
                scan_ = 0;
                return true;
            }
            scan_ = buf_.size();
            if (buf_.size() > (1 << 20)) return false;
            char chunk[16384];
            ssize_t r = ::recv(fd_, chunk, sizeof chunk, 0);
            if (r <= 0) return false;
            buf_.append(chunk, (size_t)r);
        }
    }

    // bytes read past the last newline (start of the body stream)
    std::string take_rest() { return std::move(buf_); }

private:
    int fd_;
    std::string buf_;
    size_t scan_ = 0;
};

bool recv_line(int fd, std::string& line) {
    LineReader lr(fd);
    return lr.line(line);   // control/probe paths: nothing follows
}

class WorkerBackend : public Backend {
public:
    WorkerBackend(std::string sock, int64_t timeout_s)
        : sock_(std::move(sock)), timeout_s_(timeout_s) {}

    ProbeResult probe(const std::set<std::string>&) override {
        ProbeResult r;
        int fd = uds_connect(sock_, 2000);
        if (fd < 0) return r;
        Json req = Json::object();
        req.set("cmd", Json::string("probe"));
        if (!send_all(fd, req.dump() + "\n")) {
            ::close(fd);
            return r;
        }
        std::string line;
        if (recv_line(fd, line)) {
            auto j = Json::parse(line);
            if (j) {
                r.online = j->get_bool("online", true);
                r.api_type = ApiType::Both;  // workers speak Ollama + OpenAI
                if (const Json* ms = j->find("models"))
                    for (const auto& m : ms->arr)
                        if (m.is_str()) r.available_models.push_back(m.str);
                if (const Json* ls = j->find("loaded"))
                    for (const auto& m : ls->arr)
                        if (m.is_str()) r.loaded_models.push_back(m.str);
                if (const Json* cx = j->find("ctx"))
                    for (const auto& [k, v] : cx->obj)
                        if (v.is_num()) r.loaded_ctx[k] = (int64_t)v.num;
                // clamp: a buggy worker reporting 0/negative would make
                // this backend permanently unschedulable
                max_conc_ =
                    std::max(1, (int)j->get_num("max_concurrency", 1));
            }
        }
        ::close(fd);
        return r;
    }

    int execute(const Task& task) override {
        auto resp = task.resp;
        int fd = uds_connect(sock_, 5000);
        if (fd >= 0) {
            // the 5 s connect-time SO_RCVTIMEO must NOT govern the token
            // stream: admission behind a full batch or a long prefill can
            // legally pause it longer.  Use the configured request
            // timeout (reference -t semantics, default 300 s).
            timeval tv{(time_t)(timeout_s_ > 0 ? timeout_s_ : 300), 0};
            setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
        }
        if (fd < 0) {
            if (resp) {
                resp->send_status(502,
                                  {{"Content-Type", "application/json"}});
                resp->send_chunk("{\"error\":\"worker unavailable\"}");
                resp->finish();
            }
            return -1;
        }
        Json req = Json::object();
        req.set("cmd", Json::string("request"));
        req.set("method", Json::string(task.method));
        req.set("path", Json::string(task.path));
        req.set("user", Json::string(task.user_id));
        req.set("body", Json::string(task.body));
        bool ok = send_all(fd, req.dump() + "\n");
        int status = -1;
        std::string line;
        LineReader lr(fd);
        if (ok && lr.line(line)) {
            auto j = Json::parse(line);
            if (j) {
                status = (int)j->get_num("status", 200);
                if (status < 100 || status > 599) status = 502;
                std::string ct = j->get_str("content_type",
                                            "application/x-ndjson");
                if (resp)
                    resp->send_status(status, {{"Content-Type", ct}});
                bool client_ok = true;
                std::string rest = lr.take_rest();
                if (!rest.empty() && resp)
                    client_ok = resp->send_chunk(rest);
                // stream raw bytes until worker closes
                char buf[65536];
                while (client_ok) {
                    ssize_t r = ::recv(fd, buf, sizeof buf, 0);
                    if (r <= 0) break;
                    if (resp && !resp->send_chunk(std::string(buf, r))) {
                        // client gone: hang up so the worker cancels
                        break;
                    }
                }
            }
        }
        if (status < 0 && resp) {
            resp->send_status(502, {{"Content-Type", "application/json"}});
            resp->send_chunk("{\"error\":\"worker failed\"}");
        }
        if (resp) resp->finish();
        ::close(fd);
        return status;
    }

    std::string load_model(const std::string& model, int64_t num_ctx,
                           int64_t keep_alive,
                           const BackendStatus&) override {
        Json req = Json::object();
        req.set("cmd", Json::string("load"));
        req.set("model", Json::string(model));
        if (num_ctx > 0) req.set("num_ctx", Json::number((double)num_ctx));
        req.set("keep_alive", Json::number((double)keep_alive));
        return control(req);
    }

    std::string unload_model(const std::string& model,
                             const BackendStatus&) override {
        Json req = Json::object();
        req.set("cmd", Json::string("unload"));
        req.set("model", Json::string(model));
        return control(req);
    }

    bool supports_control(const BackendStatus&) const override {
        return true;
    }

    int max_concurrency() const { return max_conc_; }

private:
    std::string control(const Json& req) {
        int fd = uds_connect(sock_, 5000);
        if (fd < 0) return "worker unavailable";
        std::string err = "worker did not answer";
        if (send_all(fd, req.dump() + "\n")) {
            std::string line;
            // loads allocate ~16 GB in HBM: allow minutes
            timeval tv{600, 0};
            setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
            if (recv_line(fd, line)) {
                auto j = Json::parse(line);
                if (j) err = j->get_bool("ok") ? "" : j->get_str(
                                 "error", "worker error");
            }
        }
        ::close(fd);
        return err;
    }

    std::string sock_;
    int64_t timeout_s_;
    int max_conc_ = 1;
};

}  // namespace

void add_worker_backend(Server& server, const std::string& spec) {
    std::string path = spec;
    int max_conc = 0;
    auto q = spec.find('?');
    if (q != std::string::npos) {
        path = spec.substr(0, q);
        const std::string opt = spec.substr(q + 1);
        if (opt.rfind("max_conc=", 0) == 0)
            max_conc = atoi(opt.c_str() + 9);
    }
    auto impl = std::make_shared<WorkerBackend>(
        path, server.state().settings.timeout_s);
    server.add_backend(impl, "worker:" + path);
    // a worker advertises its own concurrency (continuous batching);
    // an explicit ?max_conc= pins it
    auto& st = server.state();
    std::lock_guard<std::mutex> g(st.backends_mu);
    auto& b = st.backends.back();
    ProbeResult p = impl->probe({});
    apply_probe(b, p);
    b.max_concurrency = max_conc > 0 ? max_conc
                        : (impl->max_concurrency() > 0
                               ? impl->max_concurrency()
                               : 1);
}

}  // namespace omq
