// C10/C12: model-name matching + resolution, native C++.
// Semantics mirror the reference's matching library
// (reference src/dispatcher.rs:355-397, src/control.rs:450-510):
//  * smart match: exact, else tag/case-normalized (strip ":tag", lowercase);
//  * fuzzy match: case-insensitive substring in either direction;
//  * routable: strict (smart) first, then fuzzy;
//  * resolve: exact -> smart (sorted, deterministic) -> unique
//    case-insensitive substring -> LM Studio key/display name (exact, then
//    unique substring); ambiguous resolves to nothing (never guesses).
#pragma once

#include <map>
#include <optional>
#include <string>
#include <vector>

namespace omq {

std::string lower(const std::string& s);
std::string strip_tag(const std::string& s);   // "llama3:latest" -> "llama3"

bool smart_model_match_one(const std::string& requested,
                           const std::string& available);
bool smart_model_match(const std::string& requested,
                       const std::vector<std::string>& available);
bool fuzzy_model_match(const std::string& requested,
                       const std::vector<std::string>& available);
// strict first, then fuzzy (dispatcher.rs:395-397)
bool model_routable(const std::string& requested,
                    const std::vector<std::string>& available);

std::optional<std::string> resolve_model_name(
    const std::string& requested,
    const std::vector<std::string>& available,
    const std::map<std::string, std::string>& native_display);

}  // namespace omq
