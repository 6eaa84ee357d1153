"""Logging, plugin-registry and misc utilities.

MI355X-native re-expression of the reference's ``tools`` package
(/root/reference/tools/__init__.py, tools/misc.py): thread-aware colored
``[context]`` log prefixes, a ``UserException`` that terminates cleanly, the
``ClassRegister`` plugin registry, ``parse_keyval`` nested CLI sub-arguments,
and a directory auto-importer used by the aggregator/experiment/attack plugin
packages.
"""

import importlib
import os
import pathlib
import sys
import threading
import traceback

# ---------------------------------------------------------------------------- #
# User exception: raised for user-facing errors, prints without a traceback.


class UserException(Exception):
    """Exception caused by a user mistake (bad flag, bad config, ...)."""


def _excepthook(etype, value, tb):
    if issubclass(etype, UserException):
        print(f"\033[1;31mFATAL\033[0m {value}", file=sys.stderr)
        sys.exit(1)
    traceback.print_exception(etype, value, tb)


sys.excepthook = _excepthook

# ---------------------------------------------------------------------------- #
# Context logging (reference tools/__init__.py:52-223): nested, thread-local
# colored prefixes on every print that goes through `tools` helpers.

_tls = threading.local()

_COLORS = {
    "info": "\033[1;34m",     # blue
    "warning": "\033[1;33m",  # yellow
    "error": "\033[1;31m",    # red
    "success": "\033[1;32m",  # green
    "trace": "\033[1;35m",    # magenta
    None: "\033[1m",
}
_RESET = "\033[0m"


def _ctx_stack():
    stack = getattr(_tls, "stack", None)
    if stack is None:
        stack = []
        _tls.stack = stack
    return stack


class Context:
    """Nested, thread-local log-prefix context manager.

    ``with tools.Context("graph", "info"): print(...)`` prefixes messages
    written through :func:`info`/:func:`warning`/... with ``[graph]``.
    """

    def __init__(self, name, level=None):
        self.name = name
        self.level = level

    def __enter__(self):
        _ctx_stack().append(self)
        return self

    def __exit__(self, *exc):
        stack = _ctx_stack()
        if stack and stack[-1] is self:
            stack.pop()
        return False


def _emit(level, *args, file=None):
    file = file if file is not None else (sys.stderr if level in ("warning", "error") else sys.stdout)
    use_color = hasattr(file, "isatty") and file.isatty()
    names = [c.name for c in _ctx_stack() if c.name]
    prefix = "".join(f"[{n}] " for n in names)
    text = " ".join(str(a) for a in args)
    if use_color:
        color = _COLORS.get(level, _COLORS[None])
        print(f"{color}{prefix}{_RESET}{text}" if prefix else text, file=file)
    else:
        print(f"{prefix}{text}", file=file)
    file.flush()


def info(*args, **kw):
    _emit("info", *args, **kw)


def warning(*args, **kw):
    _emit("warning", "WARNING:", *args, **kw)


def error(*args, **kw):
    _emit("error", "ERROR:", *args, **kw)


def success(*args, **kw):
    _emit("success", *args, **kw)


def trace(*args, **kw):
    _emit("trace", *args, **kw)


# ---------------------------------------------------------------------------- #
# Stdout tee (reference tools/misc.py:45-78): replicate writes to many files.


class MethodCallReplicator:
    """Replicates attribute calls (write/flush/...) onto several objects."""

    def __init__(self, *targets):
        assert len(targets) > 0, "Expected at least one target object"
        self.__targets = targets

    def __getattr__(self, name):
        targets = self.__dict__["_MethodCallReplicator__targets"]
        funcs = [getattr(t, name) for t in targets]

        def call(*args, **kwargs):
            res = funcs[0](*args, **kwargs)
            for fn in funcs[1:]:
                fn(*args, **kwargs)
            return res

        return call


# ---------------------------------------------------------------------------- #
# Class register (reference tools/misc.py:83-135): named plugin registry.


class ClassRegister:
    """Registry mapping names to plugin classes with uniform error texts."""

    def __init__(self, singular, plural=None):
        self.__denom = singular
        self.__plural = plural if plural is not None else singular + "s"
        self.__register = {}

    def itemize(self):
        """List the registered names."""
        return list(self.__register.keys())

    def register(self, name, cls):
        """Register a class under a unique name."""
        if name in self.__register:
            raise UserException(
                f"{self.__denom.capitalize()} {name!r} already registered")
        self.__register[name] = cls

    def instantiate(self, name, *args, **kwargs):
        """Instantiate the class registered under the given name."""
        if name not in self.__register:
            avail = ", ".join(repr(k) for k in sorted(self.__register))
            raise UserException(
                f"Unknown {self.__denom} {name!r}, expected one of: {avail}"
                if avail else f"Unknown {self.__denom} {name!r}, no {self.__plural} available")
        return self.__register[name](*args, **kwargs)

    def get(self, name):
        """Get the class registered under the given name (or raise)."""
        if name not in self.__register:
            raise UserException(f"Unknown {self.__denom} {name!r}")
        return self.__register[name]

    def __contains__(self, name):
        return name in self.__register


# ---------------------------------------------------------------------------- #
# Nested key:value CLI sub-arguments (reference tools/misc.py:140-170).


def parse_keyval(list_keyval, defaults=None):
    """Parse a list of ``"<key>:<value>"`` strings into a dict.

    Values of keys present in ``defaults`` are coerced to the default's type;
    unknown keys stay strings. Duplicate keys are an error, exactly like the
    reference (tools/misc.py:156-158).
    """
    defaults = dict(defaults) if defaults else {}
    parsed = {}
    for entry in list_keyval or []:
        pos = entry.find(":")
        if pos < 0:
            raise UserException(
                f"Expected list of '<key>:<value>', got {entry!r} as one entry")
        key = entry[:pos]
        if key in parsed:
            raise UserException(
                f"Key {key!r} had already been specified with value {parsed[key]!r}")
        val = entry[pos + 1:]
        if key in defaults:
            want = type(defaults[key])
            try:
                if want is bool:
                    val = val.lower() in ("1", "true", "yes", "on")
                else:
                    val = want(val)
            except Exception:
                raise UserException(
                    f"Required key {key!r} expected a value of type {want.__name__!r}")
        parsed[key] = val
    for key, val in defaults.items():
        parsed.setdefault(key, val)
    return parsed


def print_args(name, selected, list_keyval, head="[ARGS] "):
    """Print a plugin selection and its key:value arguments."""
    print(head + "Selected " + name + ": " + (selected if selected else "<none>"))
    for key, val in parse_keyval(list_keyval).items():
        print(f"{head}· {key}: {val}")


# ---------------------------------------------------------------------------- #
# Directory module auto-import (reference tools/__init__.py:292-315): imports
# every sibling module of a plugin package so `register(...)` calls run.


def import_directory(package_name, package_path):
    """Import every ``.py`` module in a package directory (plugin loading)."""
    directory = pathlib.Path(package_path[0] if isinstance(package_path, (list, tuple)) else package_path)
    for item in sorted(directory.iterdir()):
        if item.suffix != ".py" or item.stem.startswith("_"):
            continue
        importlib.import_module(f"{package_name}.{item.stem}")


# ---------------------------------------------------------------------------- #
# Dataset-directory permission check (reference tools/access.py:42-79).


def can_access(path, read=True, write=False, recursive=False):
    """Check read/write permission on a path (optionally recursively)."""
    path = pathlib.Path(path)
    mode = 0
    if read:
        mode |= os.R_OK
    if write:
        mode |= os.W_OK
    if not os.access(str(path), mode):
        return False
    if recursive and path.is_dir():
        for sub in path.iterdir():
            if not can_access(sub, read=read, write=write, recursive=True):
                return False
    return True


from .checkpoint import Checkpoints  # noqa: E402,F401  (re-export)
from .cluster import cluster_parse, cluster_parsers  # noqa: E402,F401
