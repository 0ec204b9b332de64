// C19: terminal dashboard (reference src/tui.rs) — ANSI-escape renderer.
#pragma once

#include <string>

namespace omq {

class Server;

// Runs the dashboard on the calling thread until 'q'/Esc; 100 ms frames,
// snapshot-per-frame (never holds state locks while rendering).
void run_tui(Server& server);

// Worker-backend factory (spec = unix socket path of a running GPU worker,
// optionally "path?max_conc=N").
void add_worker_backend(Server& server, const std::string& spec);

}  // namespace omq
