// C7 (+C6/C18): fair-share round-robin scheduler with least-connections
// backend pick — the exact algorithm of the reference's run_worker loop
// (reference src/dispatcher.rs:494-697):
//   candidate order: VIP user first; Boost user prepended on every 2nd
//   pass; remaining active users sorted by total-processed ascending, then
//   rotated round-robin; per user, the queue is scanned for the FIRST task
//   routable to some eligible backend (online, free, no control op in
//   flight, model routable or API-family supported), preferring backends
//   that already have the model loaded; the backend is picked by least
//   active_requests with a rotating next-index tiebreak.
// Additionally enforced here (the reference declares but never enforces —
// SURVEY.md §5 "latent"): tasks older than stuck_timeout with no eligible
// backend are failed 503.
#pragma once

#include <functional>
#include <string>
#include <vector>

#include "core.h"

namespace omq {

struct Dispatch {
    Task task;
    size_t backend_idx;
    std::string user;
};

// One scheduling decision under the state's locks.  Returns true and fills
// `out` when a task was dispatched (caller runs the executor), false when
// nothing is currently schedulable.  Expired tasks are failed 503 inside.
bool schedule_once(AppState& st, Dispatch* out);

// Candidate user ordering for one pass (exposed for tests):
// VIP -> (boost on even counter) -> least-served sort + rotation.
std::vector<std::string> candidate_order(
    const std::vector<std::pair<std::string, int64_t>>& active_users,
    const std::string& vip, const std::string& boost, uint64_t counter);

// Backend eligibility for a task (exposed for tests).
bool backend_eligible(const BackendStatus& b, bool has_control_op,
                      const std::string& requested_model,
                      const std::string& path);

// Why a backend is NOT eligible (nullptr = eligible).  Reasons mirror the
// reference's debug-level candidate log (src/dispatcher.rs:579-615):
// "offline" | "busy" | "control-op" | "model-not-available" | "api-family".
const char* backend_reject_reason(const BackendStatus& b, bool has_control_op,
                                  const std::string& requested_model,
                                  const std::string& path);

// Least-connections + rotating-index pick among eligible indices.
size_t pick_backend(const std::vector<BackendStatus>& backends,
                    const std::vector<size_t>& eligible, size_t last_idx);

// Called by the executor when a request finishes: decrements
// active_requests, bumps counters, notifies the scheduler.
void finish_dispatch(AppState& st, const Dispatch& d, bool ok,
                     const std::string& outcome);

}  // namespace omq
