// C9/C11/C15: HTTP backend — the external-inference-server compatibility
// mode (config 1 of BASELINE.json).  Probing order and control-call wire
// bodies follow the reference exactly (reference src/control.rs:128-334
// probe; :762-936 load/unload executors) so real Ollama / LM Studio
// servers and the E2E mocks work unchanged.
#pragma once

#include "core.h"

namespace omq {

class HttpBackend : public Backend {
public:
    HttpBackend(std::string url, int64_t timeout_s, int64_t load_keep_alive)
        : url_(std::move(url)), timeout_s_(timeout_s),
          load_keep_alive_(load_keep_alive) {}

    ProbeResult probe(const std::set<std::string>& skip) override;
    int execute(const Task& task) override;
    std::string load_model(const std::string& model, int64_t num_ctx,
                           int64_t keep_alive,
                           const BackendStatus& st) override;
    std::string unload_model(const std::string& model,
                             const BackendStatus& st) override;
    bool supports_control(const BackendStatus& st) const override;

    const std::string& url() const { return url_; }

private:
    std::string url_;
    int64_t timeout_s_;
    int64_t load_keep_alive_;
};

}  // namespace omq
