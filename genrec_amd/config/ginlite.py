"""ginlite — a small gin-config-compatible binding engine.

The reference drives every trainer with gin files
(`/root/reference/genrec/modules/utils.py:85-117` parses
``python trainer.py <cfg.gin> --split S --gin "k=v"``). gin-config is not
installed in this environment, so this module implements the subset of the
gin language the reference's configs use, with the same UX:

  * ``include "config/base.gin"``
  * macros:      ``d_model = 128`` referenced as ``%d_model``
  * bindings:    ``train.learning_rate = 1e-3`` / ``Class.param = value``
  * references:  ``@AmazonSeqDataset`` (returns the configurable itself) and
                 ``@name()`` (called at injection time)
  * enum consts: ``%genrec_amd.models.rqvae.QuantizeForwardMode.STE``
  * ``{split}`` placeholder substitution before parsing (utils.py:109-110)
  * repeatable ``--gin "key=value"`` CLI overrides

Configurables are registered with the :func:`configurable` decorator and
receive bound parameters for any argument the caller did not pass
explicitly — the same injection semantics gin uses.
"""

from __future__ import annotations

import argparse
import ast
import importlib
import os
import re
import threading
from typing import Any, Callable, Dict, Optional

_REGISTRY: Dict[str, Callable] = {}
_BINDINGS: Dict[str, Dict[str, Any]] = {}
_MACROS: Dict[str, Any] = {}
_LOCK = threading.RLock()


class GinError(ValueError):
    pass


def clear_config() -> None:
    """Drop all bindings and macros (registry is kept)."""
    with _LOCK:
        _BINDINGS.clear()
        _MACROS.clear()


def register(fn: Callable, name: Optional[str] = None) -> Callable:
    short = name or fn.__name__
    with _LOCK:
        _REGISTRY[short] = fn
        qual = f"{fn.__module__}.{fn.__name__}"
        _REGISTRY[qual] = fn
    return fn


class _Ref:
    """A ``@name`` reference: resolves to the registered configurable."""

    __slots__ = ("name", "call")

    def __init__(self, name: str, call: bool):
        self.name = name
        self.call = call

    def resolve(self):
        target = _lookup(self.name)
        if self.call:
            return target()
        return target

    def __repr__(self):  # pragma: no cover
        return f"@{self.name}" + ("()" if self.call else "")


def _lookup(name: str) -> Callable:
    with _LOCK:
        if name in _REGISTRY:
            return _REGISTRY[name]
        # try suffix match (gin allows module-qualified or bare names)
        tail = name.split("/")[-1]
        if tail in _REGISTRY:
            return _REGISTRY[tail]
        matches = [k for k in _REGISTRY if k.endswith("." + tail)]
        if len(set(_REGISTRY[m] for m in matches)) == 1:
            return _REGISTRY[matches[0]]
    raise GinError(f"unknown configurable reference: @{name}")


def _resolve_enum_or_attr(path: str) -> Any:
    """Resolve ``pkg.mod.Enum.MEMBER`` style dotted constants."""
    parts = path.split(".")
    for i in range(len(parts) - 1, 0, -1):
        mod_name = ".".join(parts[:i])
        try:
            obj = importlib.import_module(mod_name)
        except ImportError:
            continue
        try:
            for attr in parts[i:]:
                obj = getattr(obj, attr)
            return obj
        except AttributeError:
            continue
    raise GinError(f"cannot resolve constant %{path}")


_TOKEN_MACRO = re.compile(r"%([A-Za-z_][\w.]*)")


def _parse_value(text: str) -> Any:
    text = text.strip()
    if text.startswith("@"):
        body = text[1:].strip()
        call = body.endswith("()")
        if call:
            body = body[:-2].strip()
        return _Ref(body, call)
    if text.startswith("%"):
        key = text[1:].strip()
        if "." in key:
            return _resolve_enum_or_attr(key)
        with _LOCK:
            if key in _MACROS:
                return _MACROS[key]
        raise GinError(f"undefined macro %{key}")
    # Python literal (numbers, strings, bools, None, lists, dicts, tuples)
    try:
        return ast.literal_eval(text)
    except (ValueError, SyntaxError):
        # bare identifiers are treated as strings, matching gin leniency
        if re.fullmatch(r"[A-Za-z_][\w.]*", text):
            return text
        # lists/tuples that contain refs/macros: parse elementwise
        if (text.startswith("[") and text.endswith("]")) or (
            text.startswith("(") and text.endswith(")")
        ):
            inner = text[1:-1]
            items, depth, cur = [], 0, []
            for ch in inner:
                if ch in "[({":
                    depth += 1
                elif ch in "])}":
                    depth -= 1
                if ch == "," and depth == 0:
                    items.append("".join(cur))
                    cur = []
                else:
                    cur.append(ch)
            if cur and "".join(cur).strip():
                items.append("".join(cur))
            vals = [_parse_value(it) for it in items]
            return tuple(vals) if text.startswith("(") else vals
        raise GinError(
            f"cannot parse value: {text!r} (string values need quotes, "
            f'e.g. --gin \'key="{text}"\' — same as gin-config)')


def bind(key: str, value_text_or_value: Any, *, raw: bool = False) -> None:
    """Bind ``target.param`` (or define macro when key has no dot)."""
    if raw:
        value = value_text_or_value
    else:
        value = _parse_value(str(value_text_or_value))
    with _LOCK:
        if "." not in key:
            _MACROS[key] = value
            return
        target, param = key.rsplit(".", 1)
        target = target.split("/")[-1]  # drop scopes
        _BINDINGS.setdefault(target, {})[param] = value


_LINE_RE = re.compile(r"^\s*([\w./%-]+(?:/[\w.]+)?)\s*=\s*(.+?)\s*$")


_PARSE_STACK: list = []


def parse_file(path: str, substitutions: Optional[Dict[str, str]] = None) -> None:
    ap = os.path.abspath(path)
    if ap in _PARSE_STACK:
        raise GinError(
            "circular include: " + " -> ".join(_PARSE_STACK + [ap]))
    with open(path, "r") as f:
        text = f.read()
    if substitutions:
        for k, v in substitutions.items():
            text = text.replace("{" + k + "}", v)
    base_dir = os.path.dirname(os.path.abspath(path))
    _PARSE_STACK.append(ap)
    try:
        _parse_text(text, base_dir, substitutions)
    finally:
        _PARSE_STACK.pop()


def _strip_comment(line: str) -> str:
    out, in_str, q = [], False, ""
    for ch in line:
        if in_str:
            out.append(ch)
            if ch == q:
                in_str = False
        elif ch in "\"'":
            in_str, q = True, ch
            out.append(ch)
        elif ch == "#":
            break
        else:
            out.append(ch)
    return "".join(out)


def _parse_text(text: str, base_dir: str, substitutions=None) -> None:
    # join continued lines (bracket balance)
    logical: list[str] = []
    buf, depth = [], 0
    for raw_line in text.splitlines():
        line = _strip_comment(raw_line).rstrip()
        if not line.strip() and not buf:
            continue
        buf.append(line.strip() if buf else line)
        depth += sum(line.count(c) for c in "[({") - sum(line.count(c) for c in "])}")
        if depth <= 0:
            logical.append(" ".join(buf))
            buf, depth = [], 0
    if buf:
        logical.append(" ".join(buf))

    for line in logical:
        line = line.strip()
        if not line:
            continue
        if line.startswith("include"):
            m = re.match(r'include\s+["\'](.+?)["\']', line)
            if not m:
                raise GinError(f"bad include line: {line}")
            inc = m.group(1)
            cands = [
                inc,
                os.path.join(base_dir, inc),
                os.path.join(os.getcwd(), inc),
                os.path.join(os.path.dirname(base_dir), inc),
                os.path.join(_repo_root(), inc),
            ]
            for c in cands:
                if os.path.exists(c):
                    parse_file(c, substitutions)
                    break
            else:
                raise GinError(f"include not found: {inc}")
            continue
        if line.startswith("import "):
            importlib.import_module(line[len("import "):].strip())
            continue
        m = _LINE_RE.match(line)
        if not m:
            raise GinError(f"cannot parse gin line: {line!r}")
        bind(m.group(1), m.group(2))


def _repo_root() -> str:
    return os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _materialize(v: Any) -> Any:
    if isinstance(v, _Ref):
        return v.resolve()
    if isinstance(v, list):
        return [_materialize(x) for x in v]
    if isinstance(v, tuple):
        return tuple(_materialize(x) for x in v)
    if isinstance(v, dict):
        return {k: _materialize(x) for k, x in v.items()}
    return v


def get_bindings(name: str) -> Dict[str, Any]:
    with _LOCK:
        return dict(_BINDINGS.get(name, {}))


def configurable(fn: Callable = None, *, name: Optional[str] = None):
    """Decorator: injects gin bindings for params the caller omitted."""

    def wrap(target: Callable):
        import functools
        import inspect

        reg_name = name or target.__name__
        if inspect.isclass(target):
            orig_init = target.__init__

            @functools.wraps(orig_init)
            def __init__(self, *args, **kwargs):
                bound = get_bindings(reg_name)
                sig = inspect.signature(orig_init)
                params = list(sig.parameters)[1:]  # drop self
                consumed = set(params[: len(args)])
                for k, v in bound.items():
                    if k not in kwargs and k not in consumed:
                        kwargs[k] = _materialize(v)
                orig_init(self, *args, **kwargs)

            target.__init__ = __init__
            register(target, reg_name)
            return target
        else:

            @functools.wraps(target)
            def wrapper(*args, **kwargs):
                bound = get_bindings(reg_name)
                import inspect as _i

                sig = _i.signature(target)
                params = list(sig.parameters)
                consumed = set(params[: len(args)])
                for k, v in bound.items():
                    if k not in kwargs and k not in consumed:
                        kwargs[k] = _materialize(v)
                return target(*args, **kwargs)

            wrapper.__gin_name__ = reg_name
            register(wrapper, reg_name)
            register(wrapper, f"{target.__module__}.{target.__name__}")
            return wrapper

    if fn is not None:
        return wrap(fn)
    return wrap


def constants_from_enum(enum_cls):
    """Mirror of gin.constants_from_enum: members resolvable via %Mod.Enum.X."""
    return enum_cls


def parse_config(argv=None, description: str = "genrec_amd trainer"):
    """CLI parity with the reference's parse_config (utils.py:85-117).

    ``python trainer.py <config.gin> [--split S] [--gin k=v ...]``
    """
    parser = argparse.ArgumentParser(description=description)
    parser.add_argument("config", type=str, help="path to .gin config file")
    parser.add_argument("--split", type=str, default=None,
                        help="dataset split substituted for {split}")
    parser.add_argument("--gin", action="append", default=[],
                        help="extra binding 'key=value' (repeatable)")
    args, _ = parser.parse_known_args(argv)

    subs = {"split": args.split} if args.split else None
    parse_file(args.config, substitutions=subs)
    for ov in args.gin:
        if "=" not in ov:
            raise GinError(f"--gin expects key=value, got {ov!r}")
        k, v = ov.split("=", 1)
        if args.split:
            v = v.replace("{split}", args.split)
        bind(k.strip(), v.strip())
    return args
