// Core state of the ollamamq_amd dispatcher — native C++ rebuild of the
// reference's AppState (reference src/dispatcher.rs:159-198) with the same
// coarse-mutex discipline (SURVEY.md §5 "Race detection": coarse locks +
// snapshot pattern; lock order: control_ops BEFORE backends, mirroring
// reference src/control.rs:963-965).
//
// Where the reference's backends are external HTTP servers, ours are an
// abstract Backend: HttpBackend (wire-compatible external mode, config 1 of
// BASELINE.json), WorkerBackend (in-process MI355X GPU worker over a unix
// socket) and MockBackend (tests).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

namespace omq {

// ---------------------------------------------------------------------- util
int64_t now_ms();

// --------------------------------------------------------------- log ring
// C13: bounded 300-event IN/OUT/CTL feed (reference src/dispatcher.rs:202-215)
struct LogEvent {
    int64_t ts_ms;
    std::string kind;  // "IN" | "OUT" | "CTL"
    std::string text;
};

class LogRing {
public:
    explicit LogRing(size_t cap = 300) : cap_(cap) {}
    void push(const std::string& kind, const std::string& text);
    std::vector<LogEvent> snapshot() const;
    // L8 parity (reference src/main.rs:199-218): TUI mode appends to
    // ./ollamamq.log; headless logs to stderr.
    void set_file_sink(const std::string& path);
    void set_stderr_sink(bool on) { stderr_sink_ = on; }
    // Debug level (reference RUST_LOG env-filter, src/main.rs:208,215):
    // the scheduler's per-candidate rejection reasons are emitted only
    // when enabled (reference logs them at debug, src/dispatcher.rs:579-615).
    void set_debug(bool on) { debug_ = on; }
    bool debug_on() const { return debug_; }
    void debug(const std::string& text) {
        if (debug_) push("DBG", text);
    }

private:
    mutable std::mutex mu_;
    size_t cap_;
    std::deque<LogEvent> ring_;
    std::string file_path_;
    bool stderr_sink_ = false;
    std::atomic<bool> debug_{false};
};

// ------------------------------------------------------------------ request
// C5/C6: a queued request (reference Task, src/dispatcher.rs:33-46).
// The responder is a bounded channel of body chunks; the HTTP layer drains
// it into a chunked/SSE response.
struct ResponseChannel {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<std::string> chunks;       // body parts
    int status = 0;                       // 0 = not started
    std::vector<std::pair<std::string, std::string>> headers;
    bool started = false;                 // status/headers available
    bool done = false;
    bool client_gone = false;             // client disconnected: stop compute
    size_t cap = 32;                      // bounded (reference cap 32)

    // producer side
    bool send_status(int st,
                     std::vector<std::pair<std::string, std::string>> hdrs);
    bool send_chunk(std::string data);    // false => client gone
    void finish();
    // consumer side (HTTP connection thread)
    bool wait_started(int timeout_ms);
    bool next_chunk(std::string* out, int timeout_ms);  // false when done
    void mark_client_gone();
};

struct Task {
    std::string method;
    std::string path;
    std::string query;
    std::vector<std::pair<std::string, std::string>> headers;  // Host stripped
    std::string body;
    std::string user_id;
    std::string requested_model;          // extracted "model" JSON field
    std::shared_ptr<ResponseChannel> resp;
    bool stuck_warned = false;
    int64_t queued_at_ms = 0;
};

// ------------------------------------------------------------------ backend
enum class ApiType { Unknown, Ollama, OpenAi, Both };
const char* api_type_name(ApiType t);

struct ProbeResult {
    bool online = false;
    ApiType api_type = ApiType::Unknown;
    std::vector<std::string> available_models;
    std::vector<std::string> loaded_models;
    std::map<std::string, int64_t> loaded_ctx;   // model -> context length
    bool lmstudio = false;
    // LM Studio native: key -> display name; key -> instance id
    std::map<std::string, std::string> native_display;
    std::map<std::string, std::string> native_instance;
    std::vector<std::string> good_endpoints, bad_endpoints;
};

// C9: registry entry (reference BackendStatus, src/dispatcher.rs:127-157)
struct BackendStatus {
    std::string url;                      // or worker descriptor
    int active_requests = 0;
    int64_t processed_count = 0;
    bool is_online = false;
    ApiType api_type = ApiType::Unknown;
    std::vector<std::string> available_models;
    std::vector<std::string> loaded_models;
    std::map<std::string, int64_t> loaded_ctx;
    std::string current_model;
    bool lmstudio = false;
    std::map<std::string, std::string> native_display;
    std::map<std::string, std::string> native_instance;
    std::set<std::string> known_bad_endpoints;
    int max_concurrency = 1;              // reference: 1 in-flight/backend
};

// A backend implementation: HTTP proxy target, in-process GPU worker, or
// test mock.  Execute() streams the response into task->resp and returns
// when the request is fully handled (the executor thread calls it).
class Backend {
public:
    virtual ~Backend() = default;
    virtual ProbeResult probe(const std::set<std::string>& skip_endpoints) = 0;
    // returns final status code, or <0 on transport error
    virtual int execute(const Task& task) = 0;
    // control-plane load/unload; returns "" on success else error text
    virtual std::string load_model(const std::string& model, int64_t num_ctx,
                                   int64_t keep_alive,
                                   const BackendStatus& st) = 0;
    virtual std::string unload_model(const std::string& model,
                                     const BackendStatus& st) = 0;
    virtual bool supports_control(const BackendStatus& st) const = 0;
};

// ------------------------------------------------------------- control ops
// C14 (reference src/control.rs:33-101)
enum class ControlAction { Load, Unload };

struct ControlOp {
    ControlAction action;
    std::string model;      // canonical (post-resolution)
    std::string requested;  // as the client asked (admin "operation")
    int64_t started_ms;
};

struct ControlResult {
    ControlAction action;
    std::string model;
    size_t backend;
    bool ok;
    std::string error;
    int64_t finished_ms;
};

// ------------------------------------------------------------- user queues
struct UserState {
    std::deque<Task> queue;
    int64_t processing = 0;
    int64_t processed = 0;
    int64_t dropped = 0;
};

// ---------------------------------------------------------------- settings
struct Settings {
    int port = 11435;
    std::string host = "127.0.0.1";
    int64_t timeout_s = 300;
    int64_t load_keep_alive_s = 86400;
    bool allow_all_routes = false;
    int64_t stuck_timeout_s = 60;        // reference declares but never
                                         // enforces; we DO enforce (503)
    int probe_interval_ms = 10000;       // health loop cadence (ref: 10 s)
    std::string api_key;                 // empty = auth off
};

struct ModelConfigEntry {
    std::string name;
    std::string identifier;
    int64_t max_ctx = 0;
    int64_t keep_alive = 86400;
    int max_concurrent_requests = 1;
    std::vector<std::string> backends;   // URL substrings or indices
};

// ---------------------------------------------------------------- AppState
class AppState {
public:
    Settings settings;

    // lock order: control_ops_mu BEFORE backends_mu (control.rs:963-965)
    mutable std::mutex queues_mu;
    std::map<std::string, UserState> users;          // user -> state
    std::unordered_map<std::string, std::string> user_ips;

    mutable std::mutex backends_mu;
    std::vector<BackendStatus> backends;
    std::vector<std::shared_ptr<Backend>> impls;     // parallel to backends

    mutable std::mutex control_mu;
    std::map<size_t, ControlOp> control_ops;         // backend idx -> op
    std::deque<ControlResult> control_history;       // ring of 20

    mutable std::mutex blocked_mu;
    std::set<std::string> blocked_users, blocked_ips;
    std::string blocked_path = "blocked_items.json";

    mutable std::mutex models_mu;
    std::vector<ModelConfigEntry> model_config;

    // priorities (C18): single VIP + single boost user, mutually exclusive
    mutable std::mutex prio_mu;
    std::string vip_user, boost_user;
    std::atomic<uint64_t> sched_counter{0};
    size_t last_backend_idx = 0;

    LogRing log;

    // metrics surface beyond the reference (tokens/sec and
    // queue-wait percentiles are the BASELINE headline): ring of recent
    // queue waits (enqueue -> dispatch), sampled by the scheduler
    mutable std::mutex waits_mu;
    std::deque<int64_t> wait_samples_ms;   // cap 2048
    int64_t started_ms = 0;

    // scheduler wakeups (reference notify + backend_freed)
    std::mutex wake_mu;
    std::condition_variable wake_cv;
    bool wake_flag = false;
    std::atomic<bool> shutting_down{false};

    void notify();
    void wait_work(int timeout_ms);

    // blocklist persistence (C17)
    void load_blocked();
    void save_blocked() const;
};

}  // namespace omq
