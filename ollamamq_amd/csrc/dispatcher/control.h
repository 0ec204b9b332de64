// C14/C16: model control plane + admin API.
// Semantics follow the reference (reference src/control.rs:966-1154
// start_model_control; :573-760 apply/reload config; :1160-1350 admin):
// one op per backend, busy/offline/no-control rejections, canonical name
// resolution (never guesses), history ring of 20, post-op re-probe,
// declarative config apply with resident-skip and ctx-mismatch reload.
#pragma once

#include <string>

#include "core.h"
#include "json.h"

namespace omq {

struct ControlRequest {
    ControlAction action;
    std::string model;
    size_t backend_idx;
    int64_t num_ctx = 0;       // 0 = unspecified
    int64_t keep_alive = 0;    // 0 = default
    std::string identifier;    // optional friendly name (echoed in 202)
};

struct ControlOutcome {
    int http_status;           // 202 accepted or 4xx mapped error
    Json body;
};

// Validates + registers the op and runs the executor on a detached thread.
ControlOutcome start_model_control(AppState& st, const ControlRequest& req);

// Admin handlers (return status + JSON body)
ControlOutcome admin_models_state(AppState& st);
ControlOutcome admin_stats(AppState& st);
std::string metrics_text(AppState& st);
ControlOutcome admin_model_load(AppState& st, const std::string& body);
ControlOutcome admin_model_unload(AppState& st, const std::string& body);

// Declarative application of the models section of appconf.yaml:
// group by backend, sequential per backend, live re-probe first,
// resident-skip, reload on resident-ctx mismatch.
void apply_model_config(AppState& st);
void reload_model_config(AppState& st, const std::string& config_path);

// One probe pass over every backend (health loop body); wakes the
// scheduler when something scheduling-relevant changed.
void probe_all(AppState& st, bool full_reprobe);
void apply_probe(BackendStatus& b, const ProbeResult& p);

}  // namespace omq
