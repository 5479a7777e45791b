"""Protobuf text-format (pbtxt) serializer for plain dicts.

The framework ships a `t2r_assets.pbtxt` with every export (the reference's
`proto/t2r.proto:39-43` + `utils/tensorspec_utils.py:1685-1732` handshake).
There is no protoc in this environment, so the schema is represented as plain
dicts and serialized with this standalone text-format implementation:

  * dict -> message, list -> repeated field, str/int/float/bool -> scalar.
  * loads() returns dicts; a repeated field becomes a list when a key occurs
    more than once (callers normalize single-occurrence repeated fields).
"""

from __future__ import annotations

from typing import Any, Dict, List, Union


def _dump_value(key, value, indent, out: List[str]):
  pad = "  " * indent
  if isinstance(value, dict):
    out.append(f"{pad}{key} {{")
    for k, v in value.items():
      _dump_value(k, v, indent + 1, out)
    out.append(f"{pad}}}")
  elif isinstance(value, (list, tuple)):
    for item in value:
      _dump_value(key, item, indent, out)
  elif isinstance(value, bool):
    out.append(f"{pad}{key}: {'true' if value else 'false'}")
  elif isinstance(value, (int, float)):
    out.append(f"{pad}{key}: {value}")
  elif isinstance(value, bytes):
    out.append(f"{pad}{key}: \"{value.decode('latin-1')}\"")
  elif isinstance(value, str):
    escaped = value.replace("\\", "\\\\").replace('"', '\\"')
    out.append(f'{pad}{key}: "{escaped}"')
  elif value is None:
    pass
  else:
    raise ValueError(f"Cannot serialize {type(value)} to pbtxt")


def dumps(message: Dict[str, Any]) -> str:
  out: List[str] = []
  for k, v in message.items():
    _dump_value(k, v, 0, out)
  return "\n".join(out) + "\n"


class _Parser:

  def __init__(self, text: str):
    self.text = text
    self.pos = 0

  def parse(self) -> Dict[str, Any]:
    msg = self._message(top=True)
    return msg

  def _skip_ws(self):
    while self.pos < len(self.text):
      c = self.text[self.pos]
      if c in " \t\n\r,;":
        self.pos += 1
      elif c == "#":
        nl = self.text.find("\n", self.pos)
        self.pos = len(self.text) if nl < 0 else nl
      else:
        return

  def _ident(self) -> str:
    self._skip_ws()
    start = self.pos
    while self.pos < len(self.text) and (
        self.text[self.pos].isalnum() or self.text[self.pos] in "_."):
      self.pos += 1
    if start == self.pos:
      raise ValueError(
          f"pbtxt parse error at {self.text[self.pos:self.pos+30]!r}")
    return self.text[start:self.pos]

  def _message(self, top=False) -> Dict[str, Any]:
    msg: Dict[str, Any] = {}
    while True:
      self._skip_ws()
      if self.pos >= len(self.text):
        if not top:
          raise ValueError("Unexpected end of pbtxt (missing '}')")
        return msg
      if self.text[self.pos] == "}":
        if top:
          raise ValueError("Unexpected '}' at top level")
        self.pos += 1
        return msg
      key = self._ident()
      self._skip_ws()
      if self.pos < len(self.text) and self.text[self.pos] == ":":
        self.pos += 1
        self._skip_ws()
        if self.pos < len(self.text) and self.text[self.pos] == "{":
          self.pos += 1
          value = self._message()
        else:
          value = self._scalar()
      elif self.pos < len(self.text) and self.text[self.pos] == "{":
        self.pos += 1
        value = self._message()
      else:
        raise ValueError(f"Expected ':' or '{{' after {key!r}")
      if key in msg:
        if not isinstance(msg[key], list):
          msg[key] = [msg[key]]
        msg[key].append(value)
      else:
        msg[key] = value

  def _scalar(self) -> Union[str, int, float, bool]:
    self._skip_ws()
    c = self.text[self.pos]
    if c in "\"'":
      return self._string(c)
    start = self.pos
    while self.pos < len(self.text) and self.text[self.pos] not in \
        " \t\n\r,;}#":
      self.pos += 1
    tok = self.text[start:self.pos]
    if tok in ("true", "True"):
      return True
    if tok in ("false", "False"):
      return False
    try:
      return int(tok)
    except ValueError:
      pass
    try:
      return float(tok)
    except ValueError:
      pass
    return tok  # enum value name

  def _string(self, quote: str) -> str:
    assert self.text[self.pos] == quote
    self.pos += 1
    out = []
    while self.pos < len(self.text):
      c = self.text[self.pos]
      if c == "\\":
        nxt = self.text[self.pos + 1]
        mapping = {"n": "\n", "t": "\t", "r": "\r", "\\": "\\",
                   '"': '"', "'": "'"}
        out.append(mapping.get(nxt, nxt))
        self.pos += 2
      elif c == quote:
        self.pos += 1
        # Adjacent string concatenation.
        self._skip_ws()
        if self.pos < len(self.text) and self.text[self.pos] in "\"'":
          return "".join(out) + self._string(self.text[self.pos])
        return "".join(out)
      else:
        out.append(c)
        self.pos += 1
    raise ValueError("Unterminated string in pbtxt")


def loads(text: str) -> Dict[str, Any]:
  return _Parser(text).parse()
