"""Minimal gin-config-compatible dependency-injection system.

The reference framework (google-research/tensor2robot) configures every layer
through gin-config (`utils/train_eval.py:48-55`, `@gin.configurable`
annotations throughout).  gin is not available in this environment, so this is
a from-scratch implementation of the subset of the gin language the framework
uses:

  * ``@configurable`` decorator (with optional name, module, allowlist and
    denylist) for functions and classes.
  * ``parse_config`` / ``parse_config_file`` / ``parse_config_files_and_bindings``
    with the binding syntax ``[scope/]name.param = value``.
  * Values: python literals, ``@reference`` (the configured fn/class itself),
    ``@reference()`` (call at injection time), ``@scope/reference`` forms,
    ``%MACRO`` references and ``MACRO = value`` definitions, enum constants
    registered via ``constants_from_enum``.
  * ``external_configurable`` to wrap third-party callables.
  * Explicit scopes: bindings under ``scope/name.param`` only apply when the
    configurable is invoked under that scope (``config_scope('scope')`` context
    or called through an ``@scope/name`` reference).
  * ``operative_config_str`` returning bindings that were actually consumed,
    and ``REQUIRED`` placeholders enforced at call time.

Semantics follow gin-config's documented behavior; all code is original.
"""

from __future__ import annotations

import ast
import contextlib
import enum
import functools
import inspect
import threading
import os
import re

__all__ = [
    "configurable",
    "external_configurable",
    "parse_config",
    "parse_config_file",
    "parse_config_files_and_bindings",
    "bind_parameter",
    "query_parameter",
    "clear_config",
    "operative_config_str",
    "config_str",
    "constant",
    "constants_from_enum",
    "config_scope",
    "REQUIRED",
    "macro",
]


class GinError(Exception):
  pass


class _RequiredType:
  """Sentinel for parameters that must be provided via config."""

  def __repr__(self):
    return "gin.REQUIRED"


REQUIRED = _RequiredType()


class _Registry:

  def __init__(self):
    # name -> _Configurable
    self.configurables = {}
    # (scope, name, param) -> parsed value (may be _Reference/_Macro wrappers)
    self.bindings = {}
    # macro name -> parsed value
    self.macros = {}
    # constants: dotted name -> value
    self.constants = {}
    # (scope, name, param) -> value actually used at call time
    self.operative = {}
    self.lock = threading.RLock()


_REGISTRY = _Registry()
_SCOPE_STACK = threading.local()


def _current_scopes():
  return getattr(_SCOPE_STACK, "scopes", [])


@contextlib.contextmanager
def config_scope(name):
  """Activates a gin scope for the duration of the with-block."""
  scopes = list(_current_scopes())
  if name:
    scopes = scopes + name.split("/")
  old = _current_scopes()
  _SCOPE_STACK.scopes = scopes
  try:
    yield
  finally:
    _SCOPE_STACK.scopes = old


class _Reference:
  """An ``@name`` or ``@scope/name`` (optionally called) config value."""

  def __init__(self, name, evaluate):
    self.scope, self.name = _split_scope(name)
    self.evaluate = evaluate  # True for '@f()' forms.

  def resolve(self):
    cfg = _lookup(self.name)
    target = cfg.wrapped_for_reference()
    if self.evaluate:
      if self.scope:
        with config_scope(self.scope):
          return target()
      return target()
    if self.scope:
      scope = self.scope

      @functools.wraps(target)
      def scoped(*args, **kwargs):
        with config_scope(scope):
          return target(*args, **kwargs)

      return scoped
    return target

  def __repr__(self):
    prefix = "@" + (self.scope + "/" if self.scope else "") + self.name
    return prefix + ("()" if self.evaluate else "")


class _Macro:

  def __init__(self, name):
    self.name = name

  def resolve(self):
    if self.name in _REGISTRY.macros:
      return _resolve(_REGISTRY.macros[self.name])
    if self.name in _REGISTRY.constants:
      return _REGISTRY.constants[self.name]
    raise GinError(f"Undefined macro/constant %{self.name}")

  def __repr__(self):
    return "%" + self.name


def _resolve(value):
  if isinstance(value, (_Reference, _Macro)):
    return value.resolve()
  if isinstance(value, list):
    return [_resolve(v) for v in value]
  if isinstance(value, tuple):
    return tuple(_resolve(v) for v in value)
  if isinstance(value, dict):
    return {_resolve(k): _resolve(v) for k, v in value.items()}
  return value


def _split_scope(name):
  if "/" in name:
    scope, base = name.rsplit("/", 1)
    return scope, base
  return "", name


def _lookup(name):
  """Finds a configurable by exact or suffix ('module.name') match."""
  reg = _REGISTRY.configurables
  if name in reg:
    return reg[name]
  # Suffix match: binding 'Model.lr' matches registered 'pkg.Model'.
  matches = [c for full, c in reg.items()
             if full == name or full.endswith("." + name)]
  if len(matches) == 1:
    return matches[0]
  if len(matches) > 1:
    raise GinError(f"Ambiguous configurable name '{name}': "
                   f"{[m.name for m in matches]}")
  raise GinError(f"No configurable named '{name}' is registered. "
                 f"Known: {sorted(reg)[:40]}...")


class _Configurable:
  """Wraps a function/class; injects bound parameters on call."""

  def __init__(self, fn, name, module, allowlist, denylist):
    self.fn = fn
    self.base_name = name
    self.name = (module + "." + name) if module else name
    self.allowlist = allowlist
    self.denylist = denylist
    self._signature = self._get_signature(fn)

  @staticmethod
  def _get_signature(fn):
    try:
      if inspect.isclass(fn):
        return inspect.signature(fn.__init__)
      return inspect.signature(fn)
    except (TypeError, ValueError):
      return None

  def _accepts(self, param):
    if self.denylist and param in self.denylist:
      return False
    if self.allowlist is not None and param not in self.allowlist:
      return False
    if self._signature is None:
      return True
    params = self._signature.parameters
    if param in params:
      return True
    return any(p.kind == inspect.Parameter.VAR_KEYWORD
               for p in params.values())

  def bound_params(self):
    """Collects applicable bindings for the current scope stack.

    More specific (deeper) scopes win over the unscoped binding.
    """
    out = {}
    scopes = _current_scopes()
    # Build candidate scope strings from least to most specific.
    candidates = [""]
    for i in range(len(scopes)):
      candidates.append("/".join(scopes[: i + 1]))
    with _REGISTRY.lock:
      for scope in candidates:
        for (bscope, bname, bparam), v in _REGISTRY.bindings.items():
          if bscope != scope:
            continue
          if not self._matches_name(bname):
            continue
          out[bparam] = (scope, v)
    return out

  def _matches_name(self, bname):
    return self.name == bname or self.name.endswith("." + bname) or \
        self.base_name == bname

  def __call__(self, *args, **kwargs):
    bound = self.bound_params()
    injected = {}
    for param, (scope, value) in bound.items():
      if not self._accepts(param):
        continue
      if param in kwargs:
        continue  # explicit caller kwarg wins
      # Positional args that already cover the parameter win too.
      if self._signature is not None and args:
        names = list(self._signature.parameters)
        if inspect.isclass(self.fn):
          names = names[1:]  # drop self
        pos_covered = set()
        for i, pname in enumerate(names[: len(args)]):
          pos_covered.add(pname)
        if param in pos_covered:
          continue
      resolved = _resolve(value)
      injected[param] = resolved
      with _REGISTRY.lock:
        _REGISTRY.operative[(scope, self.name, param)] = resolved
    kwargs = dict(kwargs)
    kwargs.update(injected)
    result_kwargs = self._check_required(args, kwargs)
    return self.fn(*args, **result_kwargs)

  def _check_required(self, args, kwargs):
    for k, v in list(kwargs.items()):
      if isinstance(v, _RequiredType):
        raise GinError(
            f"Required binding '{self.name}.{k}' was not provided.")
    if self._signature is not None:
      params = list(self._signature.parameters.values())
      if inspect.isclass(self.fn):
        params = params[1:]
      for i, p in enumerate(params):
        if isinstance(p.default, _RequiredType):
          covered = i < len(args) or p.name in kwargs
          if not covered:
            raise GinError(
                f"Required parameter '{self.name}.{p.name}' missing: "
                "bind it in gin config or pass explicitly.")
    return kwargs

  def wrapped_for_reference(self):
    return self


def configurable(name_or_fn=None, module=None, allowlist=None, denylist=None,
                 whitelist=None, blacklist=None):
  """Decorator registering a function or class as configurable."""
  allowlist = allowlist or whitelist
  denylist = denylist or blacklist

  def decorate(fn, name=None):
    name = name or fn.__name__
    cfg = _Configurable(fn, name, module, allowlist, denylist)
    with _REGISTRY.lock:
      _REGISTRY.configurables[cfg.name] = cfg
    if inspect.isclass(fn):
      # Keep the class itself usable (isinstance, subclassing): register the
      # configurable wrapper but return a class whose __init__ injects params.
      orig_init = fn.__init__

      @functools.wraps(orig_init)
      def __init__(self, *args, **kwargs):
        bound = cfg.bound_params()
        injected = {}
        for param, (scope, value) in bound.items():
          if param in kwargs or not cfg._accepts(param):
            continue
          names = list(cfg._signature.parameters)[1:] if cfg._signature else []
          if param in names[: len(args)]:
            continue
          resolved = _resolve(value)
          injected[param] = resolved
          with _REGISTRY.lock:
            _REGISTRY.operative[(scope, cfg.name, param)] = resolved
        kwargs.update(injected)
        cfg._check_required(args, kwargs)
        orig_init(self, *args, **kwargs)

      fn.__init__ = __init__
      cfg.fn = fn
      return fn

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
      return cfg(*args, **kwargs)

    wrapper._gin_configurable = cfg
    cfg.fn = fn
    return wrapper

  if callable(name_or_fn):
    return decorate(name_or_fn)

  def with_name(fn):
    return decorate(fn, name=name_or_fn)

  return with_name


def external_configurable(fn, name=None, module=None, allowlist=None,
                          denylist=None):
  """Registers a third-party callable; returns a configured wrapper."""
  name = name or getattr(fn, "__name__", str(fn))
  cfg = _Configurable(fn, name, module, allowlist, denylist)
  with _REGISTRY.lock:
    _REGISTRY.configurables[cfg.name] = cfg

  @functools.wraps(fn, updated=())
  def wrapper(*args, **kwargs):
    return cfg(*args, **kwargs)

  wrapper._gin_configurable = cfg
  return wrapper


def constant(name, value):
  with _REGISTRY.lock:
    _REGISTRY.constants[name] = value


def constants_from_enum(cls=None, module=None):
  def register(cls):
    mod = module or cls.__module__.split(".")[-1]
    for member in cls:
      constant(f"{cls.__name__}.{member.name}", member)
      constant(f"{mod}.{cls.__name__}.{member.name}", member)
    return cls

  if cls is not None:
    return register(cls)
  return register


def macro(name):
  return _Macro(name).resolve()


# ---------------------------------------------------------------------------
# Config language parsing
# ---------------------------------------------------------------------------

_BINDING_RE = re.compile(r"^(?P<target>[\w./]+)\s*=\s*(?P<value>.*)$",
                         re.DOTALL)
_IMPORT_RE = re.compile(r"^import\s+[\w.]+$")
_INCLUDE_RE = re.compile(r"^include\s+['\"](?P<path>[^'\"]+)['\"]$")


class _ValueParser:
  """Recursive-descent parser for gin binding values."""

  def __init__(self, text):
    self.text = text
    self.pos = 0

  def parse(self):
    v = self._value()
    self._skip_ws()
    if self.pos != len(self.text):
      raise GinError(f"Trailing characters in value: {self.text[self.pos:]!r}")
    return v

  def _skip_ws(self):
    while self.pos < len(self.text):
      c = self.text[self.pos]
      if c in " \t\n\r":
        self.pos += 1
      elif c == "#":
        nl = self.text.find("\n", self.pos)
        self.pos = len(self.text) if nl < 0 else nl
      else:
        break

  def _value(self):
    self._skip_ws()
    if self.pos >= len(self.text):
      raise GinError("Empty value")
    c = self.text[self.pos]
    if c == "@":
      return self._reference()
    if c == "%":
      return self._macro()
    if c == "[":
      return self._seq("[", "]", list)
    if c == "(":
      return self._seq("(", ")", tuple)
    if c == "{":
      return self._dict()
    return self._literal()

  def _reference(self):
    self.pos += 1  # consume '@'
    m = re.match(r"[\w./]+", self.text[self.pos:])
    if not m:
      raise GinError(f"Bad reference at {self.text[self.pos:]!r}")
    name = m.group(0)
    self.pos += m.end()
    self._skip_ws()
    evaluate = False
    if self.text[self.pos: self.pos + 2] == "()":
      evaluate = True
      self.pos += 2
    return _Reference(name, evaluate)

  def _macro(self):
    self.pos += 1
    m = re.match(r"[\w./]+", self.text[self.pos:])
    if not m:
      raise GinError(f"Bad macro at {self.text[self.pos:]!r}")
    self.pos += m.end()
    return _Macro(m.group(0))

  def _seq(self, open_c, close_c, typ):
    assert self.text[self.pos] == open_c
    self.pos += 1
    items = []
    while True:
      self._skip_ws()
      if self.pos < len(self.text) and self.text[self.pos] == close_c:
        self.pos += 1
        return typ(items)
      items.append(self._value())
      self._skip_ws()
      if self.pos < len(self.text) and self.text[self.pos] == ",":
        self.pos += 1
      elif self.pos < len(self.text) and self.text[self.pos] == close_c:
        self.pos += 1
        return typ(items)
      else:
        raise GinError(f"Expected ',' or '{close_c}' in sequence: "
                       f"{self.text!r}")

  def _dict(self):
    assert self.text[self.pos] == "{"
    self.pos += 1
    out = {}
    while True:
      self._skip_ws()
      if self.pos < len(self.text) and self.text[self.pos] == "}":
        self.pos += 1
        return out
      key = self._value()
      self._skip_ws()
      if self.text[self.pos] != ":":
        raise GinError(f"Expected ':' in dict: {self.text!r}")
      self.pos += 1
      val = self._value()
      out[key] = val
      self._skip_ws()
      if self.pos < len(self.text) and self.text[self.pos] == ",":
        self.pos += 1

  def _literal(self):
    # Scan a python literal token: string, number, bool, None, or bare name.
    rest = self.text[self.pos:]
    for strre in (r"^[rbuRBU]*'''(?:[^\\]|\\.)*?'''",
                  r'^[rbuRBU]*"""(?:[^\\]|\\.)*?"""',
                  r"^[rbuRBU]*'(?:[^'\\]|\\.)*'",
                  r'^[rbuRBU]*"(?:[^"\\]|\\.)*"'):
      m = re.match(strre, rest)
      if m:
        self.pos += m.end()
        return ast.literal_eval(m.group(0))
    m = re.match(r"^[+-]?(\d+\.?\d*|\.\d+)([eE][+-]?\d+)?[jJ]?", rest)
    if m:
      self.pos += m.end()
      return ast.literal_eval(m.group(0))
    m = re.match(r"^[\w.]+", rest)
    if m:
      tok = m.group(0)
      self.pos += m.end()
      if tok == "True":
        return True
      if tok == "False":
        return False
      if tok == "None":
        return None
      if tok in ("inf", "nan"):
        return float(tok)
      # Enum constant or registered constant.
      if tok in _REGISTRY.constants:
        return _REGISTRY.constants[tok]
      # suffix-match constants (e.g. 'ConditionMode.LANGUAGE')
      matches = [v for k, v in _REGISTRY.constants.items()
                 if k == tok or k.endswith("." + tok)]
      if len(matches) == 1:
        return matches[0]
      raise GinError(f"Unknown bare token {tok!r} in gin value")
    raise GinError(f"Cannot parse value: {rest!r}")


def parse_value(text):
  return _ValueParser(text).parse()


def _logical_lines(text):
  """Splits config text into logical lines, joining bracket continuations."""
  lines = []
  buf = ""
  depth = 0
  for raw in text.splitlines():
    line = raw.split("#", 1)[0].rstrip() if not _in_string(raw) else raw
    if not line.strip() and not buf:
      continue
    buf = (buf + "\n" + line) if buf else line
    depth = _bracket_depth(buf)
    if depth <= 0 and buf.strip():
      lines.append(buf.strip())
      buf = ""
  if buf.strip():
    lines.append(buf.strip())
  return lines


def _in_string(line):
  return line.count("'") % 2 == 1 or line.count('"') % 2 == 1


def _bracket_depth(s):
  depth = 0
  in_str = None
  i = 0
  while i < len(s):
    c = s[i]
    if in_str:
      if c == "\\":
        i += 2
        continue
      if c == in_str:
        in_str = None
    elif c in "'\"":
      in_str = c
    elif c in "([{":
      depth += 1
    elif c in ")]}":
      depth -= 1
    i += 1
  return depth


def parse_config(text):
  """Parses gin binding text (string or iterable of lines)."""
  if not isinstance(text, str):
    text = "\n".join(text)
  for line in _logical_lines(text):
    if _IMPORT_RE.match(line):
      modname = line.split(None, 1)[1]
      try:
        __import__(modname)
      except ImportError as e:
        raise GinError(f"gin config import failed: {modname}: {e}") from e
      continue
    m = _INCLUDE_RE.match(line)
    if m:
      parse_config_file(m.group("path"))
      continue
    m = _BINDING_RE.match(line)
    if not m:
      raise GinError(f"Cannot parse config line: {line!r}")
    target = m.group("target")
    value = parse_value(m.group("value"))
    if "." not in target:
      # Macro definition.
      with _REGISTRY.lock:
        _REGISTRY.macros[target] = value
      continue
    scope, rest = _split_scope(target)
    name, param = rest.rsplit(".", 1)
    with _REGISTRY.lock:
      _REGISTRY.bindings[(scope, name, param)] = value


_SEARCH_PATHS = [""]


def add_config_file_search_path(path):
  _SEARCH_PATHS.append(path)


def parse_config_file(path):
  for base in _SEARCH_PATHS:
    candidate = os.path.join(base, path) if base else path
    if os.path.exists(candidate):
      # Includes resolve relative to the including file's directory
      # (and the repo root two levels up covers the reference-style
      # 'tensor2robot_amd/research/.../x.gin' include paths).
      file_dir = os.path.dirname(os.path.abspath(candidate))
      pushed = []
      for extra in (file_dir,
                    os.path.abspath(os.path.join(file_dir, "..", "..",
                                                 "..", ".."))):
        if extra not in _SEARCH_PATHS:
          _SEARCH_PATHS.append(extra)
          pushed.append(extra)
      try:
        with open(candidate) as f:
          parse_config(f.read())
      finally:
        for extra in pushed:
          _SEARCH_PATHS.remove(extra)
      return
  raise GinError(f"Config file not found: {path}")


def parse_config_files_and_bindings(config_files=None, bindings=None,
                                    finalize_config=True, **unused):
  for f in config_files or []:
    parse_config_file(f)
  if bindings:
    parse_config(bindings)


def bind_parameter(target, value):
  scope, rest = _split_scope(target)
  name, param = rest.rsplit(".", 1)
  with _REGISTRY.lock:
    _REGISTRY.bindings[(scope, name, param)] = value


def query_parameter(target):
  scope, rest = _split_scope(target)
  name, param = rest.rsplit(".", 1)
  with _REGISTRY.lock:
    if (scope, name, param) in _REGISTRY.bindings:
      return _resolve(_REGISTRY.bindings[(scope, name, param)])
  raise GinError(f"No binding for {target}")


def clear_config():
  with _REGISTRY.lock:
    _REGISTRY.bindings.clear()
    _REGISTRY.macros.clear()
    _REGISTRY.operative.clear()


def _format_value(v):
  if isinstance(v, (_Reference, _Macro)):
    return repr(v)
  if isinstance(v, enum.Enum):
    return f"%{type(v).__name__}.{v.name}"
  return repr(v)


def config_str():
  lines = []
  with _REGISTRY.lock:
    for name, v in sorted(_REGISTRY.macros.items()):
      lines.append(f"{name} = {_format_value(v)}")
    for (scope, name, param), v in sorted(_REGISTRY.bindings.items()):
      prefix = f"{scope}/" if scope else ""
      lines.append(f"{prefix}{name}.{param} = {_format_value(v)}")
  return "\n".join(lines) + "\n"


def operative_config_str():
  lines = []
  with _REGISTRY.lock:
    for (scope, name, param), v in sorted(_REGISTRY.operative.items()):
      prefix = f"{scope}/" if scope else ""
      lines.append(f"{prefix}{name}.{param} = {_format_value(v)}")
  return "\n".join(lines) + "\n"
