#include "scheduler.h"

#include <algorithm>
#include <chrono>

#include "json.h"
#include "matching.h"

namespace omq {

int64_t now_ms() {
    using namespace std::chrono;
    return duration_cast<milliseconds>(
               steady_clock::now().time_since_epoch())
        .count();
}

void AppState::notify() {
    {
        std::lock_guard<std::mutex> g(wake_mu);
        wake_flag = true;
    }
    wake_cv.notify_all();
}

void AppState::wait_work(int timeout_ms) {
    std::unique_lock<std::mutex> g(wake_mu);
    wake_cv.wait_for(g, std::chrono::milliseconds(timeout_ms),
                     [&] { return wake_flag || shutting_down.load(); });
    wake_flag = false;
}

// ---------------------------------------------------------------- ordering
std::vector<std::string> candidate_order(
    const std::vector<std::pair<std::string, int64_t>>& active_users,
    const std::string& vip, const std::string& boost, uint64_t counter) {
    // stable sort by total processed ascending (least-served first),
    // reference src/dispatcher.rs:518-534
    // boost is pulled out only on the ticks it is actually boosted (every
    // 2nd); on other ticks it competes at its least-served sorted position
    // like any other user (src/dispatcher.rs:538-549)
    const bool boost_tick = counter % 2 == 0;
    std::vector<std::pair<std::string, int64_t>> rest;
    bool has_vip = false, has_boost = false;
    for (const auto& [u, n] : active_users) {
        if (u == vip) { has_vip = true; continue; }
        if (u == boost && boost_tick) { has_boost = true; continue; }
        rest.emplace_back(u, n);
    }
    std::stable_sort(rest.begin(), rest.end(),
                     [](const auto& a, const auto& b) {
                         return a.second < b.second;
                     });
    // rotation: advance the start point each pass (src/dispatcher.rs:551-564)
    std::vector<std::string> order;
    if (!rest.empty()) {
        const size_t rot = counter % rest.size();
        for (size_t i = 0; i < rest.size(); i++)
            order.push_back(rest[(rot + i) % rest.size()].first);
    }
    // boost: prepended on every 2nd tick (src/dispatcher.rs:538-549)
    if (has_boost) order.insert(order.begin(), boost);
    // VIP: absolute head (src/dispatcher.rs:507,538)
    if (has_vip) order.insert(order.begin(), vip);
    return order;
}

// ------------------------------------------------------------- eligibility
static bool family_supported(ApiType t, const std::string& path) {
    // Unknown-type backends are allowed for BOTH families: the health loop
    // will classify them (reference src/dispatcher.rs:609-612)
    if (t == ApiType::Both || t == ApiType::Unknown) return true;
    const bool openai = path.rfind("/v1/", 0) == 0;
    return openai ? t == ApiType::OpenAi : t == ApiType::Ollama;
}

const char* backend_reject_reason(const BackendStatus& b, bool has_control_op,
                                  const std::string& requested_model,
                                  const std::string& path) {
    if (!b.is_online) return "offline";
    if (b.active_requests >= b.max_concurrency) return "busy";
    if (has_control_op) return "control-op";
    // A specific model is the HARD gate when present; the API-family check
    // applies only to model-less requests (reference src/dispatcher.rs:599-616:
    // "If a specific model is requested, backend MUST have it. If no model is
    // requested, fall back to API family check.")
    if (!requested_model.empty())
        return model_routable(requested_model, b.available_models)
                   ? nullptr
                   : "model-not-available";
    return family_supported(b.api_type, path) ? nullptr : "api-family";
}

bool backend_eligible(const BackendStatus& b, bool has_control_op,
                      const std::string& requested_model,
                      const std::string& path) {
    return backend_reject_reason(b, has_control_op, requested_model, path) ==
           nullptr;
}

size_t pick_backend(const std::vector<BackendStatus>& backends,
                    const std::vector<size_t>& eligible, size_t last_idx) {
    // min active_requests, then first index strictly after last_idx
    // (reference src/dispatcher.rs:664-681)
    int min_active = INT32_MAX;
    for (size_t i : eligible)
        min_active = std::min(min_active, backends[i].active_requests);
    std::vector<size_t> least;
    for (size_t i : eligible)
        if (backends[i].active_requests == min_active) least.push_back(i);
    for (size_t i : least)
        if (i > last_idx) return i;
    return least.front();
}

// --------------------------------------------------------------- the pass
bool schedule_once(AppState& st, Dispatch* out) {
    // lock order: control before backends (control.rs:963-965), then queues
    std::scoped_lock lk(st.control_mu, st.backends_mu, st.queues_mu,
                        st.prio_mu);

    // enforce stuck-timeout 503 (declared-but-latent in the reference —
    // config.rs:54-57; we implement the documented behavior)
    const int64_t now = now_ms();
    const int64_t stuck_ms = st.settings.stuck_timeout_s * 1000;
    for (auto& [user, us] : st.users) {
        for (auto it = us.queue.begin(); it != us.queue.end();) {
            if (stuck_ms > 0 && now - it->queued_at_ms > stuck_ms) {
                if (it->resp) {
                    it->resp->send_status(
                        503, {{"Content-Type", "application/json"}});
                    it->resp->send_chunk(
                        "{\"error\":\"no backend available for this request "
                        "(stuck timeout)\"}");
                    it->resp->finish();
                }
                us.dropped++;
                st.log.push("OUT", "drop(stuck) user=" + user +
                                       " path=" + it->path);
                it = us.queue.erase(it);
            } else {
                ++it;
            }
        }
    }

    std::vector<std::pair<std::string, int64_t>> active;
    for (auto& [user, us] : st.users)
        if (!us.queue.empty()) active.emplace_back(user, us.processed);
    if (active.empty()) return false;

    const uint64_t counter = st.sched_counter.load();
    auto order = candidate_order(active, st.vip_user, st.boost_user, counter);

    for (const auto& user : order) {
        auto& us = st.users[user];
        // scan for the FIRST routable task in this user's queue
        for (auto it = us.queue.begin(); it != us.queue.end(); ++it) {
            std::vector<size_t> eligible;
            const bool dbg = st.log.debug_on();
            for (size_t bi = 0; bi < st.backends.size(); bi++) {
                const bool op = st.control_ops.count(bi) > 0;
                const char* why = backend_reject_reason(
                    st.backends[bi], op, it->requested_model, it->path);
                if (!why)
                    eligible.push_back(bi);
                else if (dbg)
                    st.log.debug("sched: user=" + user + " model=" +
                                 it->requested_model + " backend[" +
                                 std::to_string(bi) + "] " +
                                 st.backends[bi].url + " rejected: " + why);
            }
            if (eligible.empty()) {
                if (!it->stuck_warned) {
                    it->stuck_warned = true;
                    st.log.push("OUT", "no backend available yet for user=" +
                                           user + " model=" +
                                           it->requested_model);
                }
                continue;
            }
            // prefer backends that already have the model loaded
            // (src/dispatcher.rs:622-640)
            if (!it->requested_model.empty()) {
                std::vector<size_t> loaded;
                for (size_t bi : eligible)
                    if (smart_model_match(it->requested_model,
                                          st.backends[bi].loaded_models) ||
                        fuzzy_model_match(it->requested_model,
                                          st.backends[bi].loaded_models))
                        loaded.push_back(bi);
                if (!loaded.empty()) eligible = std::move(loaded);
            }
            const size_t bi =
                pick_backend(st.backends, eligible, st.last_backend_idx);
            {
                std::lock_guard<std::mutex> wg(st.waits_mu);
                st.wait_samples_ms.push_back(now - it->queued_at_ms);
                while (st.wait_samples_ms.size() > 2048)
                    st.wait_samples_ms.pop_front();
            }
            st.last_backend_idx = bi;
            st.sched_counter.fetch_add(1);
            out->task = std::move(*it);
            us.queue.erase(it);
            out->backend_idx = bi;
            out->user = user;
            us.processing++;
            st.backends[bi].active_requests++;
            st.backends[bi].current_model = out->task.requested_model;
            return true;
        }
    }
    return false;
}

void finish_dispatch(AppState& st, const Dispatch& d, bool ok,
                     const std::string& outcome) {
    {
        std::scoped_lock lk(st.backends_mu, st.queues_mu);
        auto& b = st.backends[d.backend_idx];
        b.active_requests = std::max(0, b.active_requests - 1);
        b.processed_count++;
        b.current_model.clear();
        auto& us = st.users[d.user];
        us.processing = std::max<int64_t>(0, us.processing - 1);
        if (ok)
            us.processed++;
        else
            us.dropped++;
    }
    st.log.push("OUT", (ok ? "done " : "drop(") + outcome +
                           (ok ? " " : ") ") + "user=" + d.user + " -> b" +
                           std::to_string(d.backend_idx));
    st.notify();  // backend freed: wake the scheduler
}

// ------------------------------------------------------------- log ring
void LogRing::set_file_sink(const std::string& path) {
    std::lock_guard<std::mutex> g(mu_);
    file_path_ = path;
}

void LogRing::push(const std::string& kind, const std::string& text) {
    std::lock_guard<std::mutex> g(mu_);
    ring_.push_back({now_ms(), kind, text});
    while (ring_.size() > cap_) ring_.pop_front();
    if (!file_path_.empty()) {
        FILE* f = fopen(file_path_.c_str(), "a");
        if (f) {
            fprintf(f, "%lld %s %s\n", (long long)now_ms(), kind.c_str(),
                    text.c_str());
            fclose(f);
        }
    }
    if (stderr_sink_)
        fprintf(stderr, "[%s] %s\n", kind.c_str(), text.c_str());
}

std::vector<LogEvent> LogRing::snapshot() const {
    std::lock_guard<std::mutex> g(mu_);
    return {ring_.begin(), ring_.end()};
}

}  // namespace omq
