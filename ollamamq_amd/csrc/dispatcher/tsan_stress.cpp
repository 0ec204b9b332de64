// Thread-sanitizer stress for the dispatcher core (SURVEY.md §5 "race
// detection": the reference leans on Rust's type system; the C++ rebuild
// runs this hammer under -fsanitize=thread in CI instead).
//
// Spawns ingress threads (enqueue tasks), a scheduler thread
// (schedule_once + finish), a health thread (apply_probe flips), a
// control thread (ops in/out) and a TUI-style snapshot thread, all over
// one AppState for ~2 s.  Any lock-discipline violation (the documented
// order: control before backends, reference src/control.rs:963-965)
// surfaces as a TSan report and a non-zero exit.
#include <atomic>
#include <cstdio>
#include <thread>
#include <vector>

#include "control.h"
#include "core.h"
#include "scheduler.h"

using namespace omq;

int main() {
    AppState st;
    st.settings.stuck_timeout_s = 1;
    {
        std::lock_guard<std::mutex> g(st.backends_mu);
        for (int i = 0; i < 4; i++) {
            BackendStatus b;
            b.url = "b" + std::to_string(i);
            b.is_online = true;
            b.api_type = ApiType::Ollama;
            b.available_models = {"llama3:latest", "qwen2.5:7b"};
            b.max_concurrency = 2;
            st.backends.push_back(std::move(b));
        }
    }
    std::atomic<bool> stop{false};
    std::atomic<long> dispatched{0}, enqueued{0};

    std::vector<std::thread> threads;
    // ingress
    for (int t = 0; t < 4; t++) {
        threads.emplace_back([&, t] {
            int n = 0;
            while (!stop) {
                Task task;
                task.method = "POST";
                task.path = (n % 3 == 0) ? "/api/chat" : "/api/generate";
                task.user_id = "user" + std::to_string((t * 7 + n) % 9);
                task.requested_model =
                    (n % 4 == 0) ? "qwen2.5" : "llama3";
                task.queued_at_ms = now_ms();
                {
                    std::lock_guard<std::mutex> g(st.queues_mu);
                    st.users[task.user_id].queue.push_back(std::move(task));
                }
                st.log.push("IN", task.user_id);
                st.notify();
                enqueued++;
                n++;
                std::this_thread::yield();
            }
        });
    }
    // scheduler + executor finish
    threads.emplace_back([&] {
        while (!stop) {
            Dispatch d;
            while (schedule_once(st, &d)) {
                dispatched++;
                finish_dispatch(st, d, true, "ok");
            }
            st.wait_work(5);
        }
    });
    // health flips
    threads.emplace_back([&] {
        int i = 0;
        while (!stop) {
            ProbeResult p;
            p.online = (i % 5 != 0);
            p.api_type = ApiType::Ollama;
            p.available_models = {"llama3:latest", "qwen2.5:7b"};
            p.loaded_models = (i % 2) ? std::vector<std::string>{
                                            "llama3:latest"}
                                      : std::vector<std::string>{};
            {
                std::lock_guard<std::mutex> g(st.backends_mu);
                apply_probe(st.backends[i % st.backends.size()], p);
            }
            st.notify();
            i++;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
    });
    // control ops in/out (lock order: control before backends)
    threads.emplace_back([&] {
        int i = 0;
        while (!stop) {
            {
                std::scoped_lock lk(st.control_mu, st.backends_mu);
                const size_t bi = i % st.backends.size();
                if (st.control_ops.count(bi))
                    st.control_ops.erase(bi);
                else if (st.backends[bi].active_requests == 0)
                    st.control_ops[bi] =
                        ControlOp{ControlAction::Load, "llama3", "llama3", now_ms()};
            }
            st.notify();
            i++;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
    });
    // VIP/boost + blocklist toggles
    threads.emplace_back([&] {
        int i = 0;
        while (!stop) {
            {
                std::lock_guard<std::mutex> g(st.prio_mu);
                st.vip_user = (i % 2) ? "user1" : "";
                st.boost_user = (i % 3) ? "user2" : "";
            }
            {
                std::lock_guard<std::mutex> g(st.blocked_mu);
                if (i % 2)
                    st.blocked_users.insert("user8");
                else
                    st.blocked_users.erase("user8");
            }
            i++;
            std::this_thread::sleep_for(std::chrono::milliseconds(1));
        }
    });
    // snapshot reader (TUI-style) + admin stats
    threads.emplace_back([&] {
        while (!stop) {
            (void)admin_models_state(st);
            (void)admin_stats(st);
            (void)st.log.snapshot();
            std::this_thread::sleep_for(std::chrono::milliseconds(2));
        }
    });

    std::this_thread::sleep_for(std::chrono::seconds(2));
    stop = true;
    st.notify();
    for (auto& t : threads) t.join();
    printf("tsan_stress done: enqueued=%ld dispatched=%ld\n",
           enqueued.load(), dispatched.load());
    return dispatched.load() > 0 ? 0 : 1;
}
