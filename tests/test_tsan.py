"""Race detection: the dispatcher core under
ThreadSanitizer — ingress/scheduler/health/control/TUI-snapshot threads
hammering one AppState.  Any data race or lock-order violation fails."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_dispatcher_core_under_tsan():
    from ollamamq_amd.build import build_tsan_stress
    binary = build_tsan_stress()
    r = subprocess.run([binary], capture_output=True, text=True,
                       timeout=120,
                       env={**os.environ, "TSAN_OPTIONS":
                            "halt_on_error=1 exitcode=66"})
    assert r.returncode == 0, (
        f"TSan stress failed (rc={r.returncode}):\n{r.stdout}\n{r.stderr}")
    assert "dispatched=" in r.stdout


def test_dispatcher_core_under_asan():
    """Same hammer under AddressSanitizer+UBSan: heap misuse, overflow,
    UB in the native dispatcher core fail the test."""
    from ollamamq_amd.build import build_asan_stress
    binary = build_asan_stress()
    r = subprocess.run([binary], capture_output=True, text=True,
                       timeout=120,
                       env={**os.environ,
                            "ASAN_OPTIONS": "detect_leaks=0:exitcode=66"})
    assert r.returncode == 0, (
        f"ASan stress failed (rc={r.returncode}):\n{r.stdout}\n{r.stderr}")
    assert "dispatched=" in r.stdout
