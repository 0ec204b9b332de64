"""Loader for the native C++ dispatcher core (csrc/dispatcher/_dispatch.so).

The scheduler, matching and resolution logic used by the server binary is
this exact compiled code — tests import it through here.
"""
import importlib.util
import os

_SO = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                   "csrc", "dispatcher", "_dispatch.so")


def load():
    if not os.path.exists(_SO):
        raise RuntimeError(
            f"native dispatcher not built: {_SO} missing — run "
            "`python -m ollamamq_amd.build`"
        )
    spec = importlib.util.spec_from_file_location("_dispatch", _SO)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod
