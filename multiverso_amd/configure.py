"""Flag registry for multiverso_amd.

Capability parity with the reference's gflags-clone configure system
(reference: include/multiverso/util/configure.h:21-114,
src/util/configure.cpp:9-54): typed flag registration, ``-key=value``
command-line parsing that strips consumed argv entries, and programmatic
``set_flag`` (the reference's ``MV_SetFlag``, src/multiverso.cpp:48-51).

Redesigned for Python: a single registry with typed defaults instead of
per-type static singletons.
"""

from __future__ import annotations

import threading
from typing import Any, Dict, List


class _FlagRegistry:
    def __init__(self) -> None:
        self._flags: Dict[str, Any] = {}
        self._types: Dict[str, type] = {}
        self._lock = threading.Lock()

    def define(self, name: str, default: Any, help_: str = "") -> None:
        with self._lock:
            if name not in self._flags:
                self._flags[name] = default
                self._types[name] = type(default)

    def get(self, name: str) -> Any:
        return self._flags[name]

    def set(self, name: str, value: Any) -> None:
        with self._lock:
            if name in self._types:
                ty = self._types[name]
                if ty is bool and isinstance(value, str):
                    value = value.lower() in ("1", "true", "yes", "on")
                else:
                    value = ty(value)
            self._flags[name] = value
            self._types.setdefault(name, type(value))

    def known(self, name: str) -> bool:
        return name in self._flags

    def reset(self) -> None:
        """Reset to an empty registry (test helper)."""
        with self._lock:
            self._flags.clear()
            self._types.clear()
        _define_core_flags()


_registry = _FlagRegistry()


def define_flag(name: str, default: Any, help_: str = "") -> None:
    _registry.define(name, default, help_)


def get_flag(name: str) -> Any:
    return _registry.get(name)


def set_flag(name: str, value: Any) -> None:
    """Programmatic flag set (reference MV_SetFlag, multiverso.cpp:48-51)."""
    _registry.set(name, value)


def parse_cmd_flags(argv: List[str]) -> List[str]:
    """Consume ``-key=value`` entries from argv, returning what's left.

    Mirrors reference ParseCMDFlags (src/util/configure.cpp:9-54): only
    entries of the form ``-key=value`` whose key is a registered flag are
    consumed; everything else passes through untouched.
    """
    rest: List[str] = []
    for arg in argv:
        if arg.startswith("-") and "=" in arg:
            key, _, val = arg.lstrip("-").partition("=")
            if _registry.known(key):
                _registry.set(key, val)
                continue
        rest.append(arg)
    return rest


def _define_core_flags() -> None:
    # Core runtime flags; parity list from SURVEY.md §5.6 / reference flags.
    define_flag("ps_role", "default", "node role: default(worker+server)|worker|server|none")
    define_flag("ma", False, "model-average mode (reference zoo.cpp:24): aggregate() is always available here; the flag is accepted so reference launch lines work unchanged")
    define_flag("sync", False, "BSP synchronous server (vector-clocked)")
    define_flag("backup_worker_ratio", 0.0, "vestigial in reference; kept for parity")
    define_flag("updater_type", "default", "default|sgd|momentum|adagrad")
    define_flag("omp_threads", 0, "CPU-fallback intra-op threads (reference updater.cpp:18; 0 = torch default)")
    define_flag("allocator_type", "smart", "kept for parity; caching allocator is torch's")
    define_flag("allocator_alignment", 16, "kept for parity")
    define_flag("logtostderr", False, "log to stderr instead of stdout")
    define_flag("log_level", "info", "debug|info|error|fatal")
    # MI355X-native additions
    define_flag("bucket_mb", 64, "collective bucket size (MiB) for sharded Add/Get over xGMI")
    define_flag("deterministic", False, "pre-aggregate duplicate rows before keyed scatter (fixed reduction order instead of atomics)")
    define_flag("sparse_filter", True, "SparseFilter on stale-row reply payloads "
                "(sparse_matrix_table.cpp:148-153 parity)")


_define_core_flags()
