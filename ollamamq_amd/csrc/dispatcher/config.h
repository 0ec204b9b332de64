// C2: appconf.yaml config file (reference src/config.rs:36-135).
#pragma once

#include <string>
#include <vector>

#include "core.h"

namespace omq {

struct AppConfig {
    std::vector<std::string> backends;
    Settings settings;
    std::vector<ModelConfigEntry> models;
};

// Parses the three-section YAML subset.  Missing file => true with
// defaults; malformed values (keep_alive < -1) => false with *err set.
bool load_config(const std::string& path, AppConfig* out, std::string* err);

// URL normalization (reference main.rs:182-190): strip trailing '/',
// prepend "http://" when no scheme.
std::string normalize_backend_url(std::string url);

}  // namespace omq
