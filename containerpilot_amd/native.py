"""ctypes bindings to the in-tree native library (_native.so).

These call directly into the same C++ code the daemon runs (template
engine, JSON5 parser, duration rules, config validation).
"""

import ctypes
import os

from . import REPO_ROOT

_LIB_PATH = os.path.join(REPO_ROOT, "containerpilot_amd", "_native.so")
_lib = None


def _load():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                "native library missing: %s (run __graft_entry__.build())"
                % _LIB_PATH)
        _lib = ctypes.CDLL(_LIB_PATH)
        _lib.cp_version.restype = ctypes.c_char_p
        _lib.cp_render_template.restype = ctypes.c_void_p
        _lib.cp_render_template.argtypes = [
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
        _lib.cp_parse_duration_ns.restype = ctypes.c_longlong
        _lib.cp_parse_duration_ns.argtypes = [ctypes.c_char_p]
        _lib.cp_json5_to_json.restype = ctypes.c_void_p
        _lib.cp_json5_to_json.argtypes = [
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
        _lib.cp_validate_config.restype = ctypes.c_void_p
        _lib.cp_validate_config.argtypes = [ctypes.c_char_p]
        _lib.cp_consul_endpoint.restype = ctypes.c_void_p
        _lib.cp_consul_endpoint.argtypes = [
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
        _lib.cp_free.argtypes = [ctypes.c_void_p]
    return _lib


def _take_string(lib, ptr):
    if not ptr:
        return None
    out = ctypes.string_at(ptr).decode()
    lib.cp_free(ptr)
    return out


def version():
    return _load().cp_version().decode()


def render_template(text):
    lib = _load()
    err = ctypes.c_void_p()
    out = lib.cp_render_template(text.encode(), ctypes.byref(err))
    if not out:
        raise ValueError(_take_string(lib, err.value))
    return _take_string(lib, out)


def parse_duration_ns(text):
    ns = _load().cp_parse_duration_ns(text.encode())
    if ns < 0:
        raise ValueError("invalid duration: %s" % text)
    return ns


def json5_to_json(text):
    lib = _load()
    err = ctypes.c_void_p()
    out = lib.cp_json5_to_json(text.encode(), ctypes.byref(err))
    if not out:
        raise ValueError(_take_string(lib, err.value))
    return _take_string(lib, out)


def validate_config(text):
    """Returns None when valid, else the error message."""
    lib = _load()
    return _take_string(lib, lib.cp_validate_config(text.encode()))


def consul_endpoint(consul_json):
    """Resolve the consul endpoint "scheme://host:port" the given consul
    config value produces under the current CONSUL_* environment.

    Note: the CONSUL_* env vars are read by the calling process's own
    environment, so tests set os.environ before calling.
    """
    lib = _load()
    err = ctypes.c_void_p()
    out = lib.cp_consul_endpoint(consul_json.encode(), ctypes.byref(err))
    if not out:
        raise ValueError(_take_string(lib, err.value))
    return _take_string(lib, out)
