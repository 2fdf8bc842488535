"""Protobuf text-format ("prototxt") parser / serializer.

The reference configures everything through Caffe prototxt files
(/root/reference/usage/def.prototxt, usage/solver.prototxt,
caffe.proto:2-23).  This is a small, dependency-free recursive parser for
the protobuf text format subset Caffe uses:

    name: "GoogleNet"
    layer {
        type: "Convolution"
        convolution_param { num_output: 64 }
        loss_weight: 1
        loss_weight: 1          # repeated scalar field
    }

Parsed into a `Message`: an ordered multimap.  Scalar values are decoded to
bool/int/float/str; bare identifiers (enum values like GLOBAL) stay str.
Repeated fields accumulate into lists; `get` returns the first value,
`get_all` the list.
"""

from __future__ import annotations

import re
from typing import Any, Iterator, List, Optional, Tuple


class Message:
    """Ordered multimap of field name -> list of values (scalar or Message)."""

    def __init__(self) -> None:
        self._fields: List[Tuple[str, Any]] = []

    def add(self, name: str, value: Any) -> None:
        self._fields.append((name, value))

    def get(self, name: str, default: Any = None) -> Any:
        for k, v in self._fields:
            if k == name:
                return v
        return default

    def get_all(self, name: str) -> List[Any]:
        return [v for k, v in self._fields if k == name]

    def has(self, name: str) -> bool:
        return any(k == name for k, _ in self._fields)

    def keys(self) -> List[str]:
        return [k for k, _ in self._fields]

    def items(self) -> Iterator[Tuple[str, Any]]:
        return iter(self._fields)

    def __contains__(self, name: str) -> bool:
        return self.has(name)

    def __getitem__(self, name: str) -> Any:
        v = self.get(name, _MISSING)
        if v is _MISSING:
            raise KeyError(name)
        return v

    def __repr__(self) -> str:
        return "Message(%s)" % ", ".join("%s=%r" % kv for kv in self._fields)


_MISSING = object()

_TOKEN_RE = re.compile(
    r"""
    (?P<space>\s+)
  | (?P<comment>\#[^\n]*)
  | (?P<string>"(?:[^"\\]|\\.)*")
  | (?P<punct>[{}:])
  | (?P<atom>[^\s{}:#"]+)
    """,
    re.VERBOSE,
)


def _tokenize(text: str) -> List[str]:
    tokens: List[str] = []
    pos = 0
    while pos < len(text):
        m = _TOKEN_RE.match(text, pos)
        if m is None:
            raise ValueError("prototxt: cannot tokenize at offset %d: %r" % (pos, text[pos : pos + 40]))
        pos = m.end()
        if m.lastgroup in ("space", "comment"):
            continue
        tokens.append(m.group())
    return tokens


_INT_RE = re.compile(r"^[+-]?\d+$")
_FLOAT_RE = re.compile(r"^[+-]?(\d+\.\d*|\.\d+|\d+)([eE][+-]?\d+)?$")


def _decode_scalar(tok: str) -> Any:
    if tok.startswith('"'):
        # Undo the simple escapes Caffe uses.
        body = tok[1:-1]
        return body.replace('\\"', '"').replace("\\\\", "\\").replace("\\n", "\n").replace("\\t", "\t")
    if tok == "true":
        return True
    if tok == "false":
        return False
    if _INT_RE.match(tok):
        return int(tok)
    if _FLOAT_RE.match(tok):
        return float(tok)
    return tok  # enum identifier


class _Parser:
    def __init__(self, tokens: List[str]) -> None:
        self.tokens = tokens
        self.pos = 0

    def peek(self) -> Optional[str]:
        return self.tokens[self.pos] if self.pos < len(self.tokens) else None

    def next(self) -> str:
        tok = self.peek()
        if tok is None:
            raise ValueError("prototxt: unexpected end of input")
        self.pos += 1
        return tok

    def parse_message(self, top_level: bool) -> Message:
        msg = Message()
        while True:
            tok = self.peek()
            if tok is None:
                if top_level:
                    return msg
                raise ValueError("prototxt: unexpected EOF inside message")
            if tok == "}":
                if top_level:
                    raise ValueError("prototxt: unmatched '}'")
                self.next()
                return msg
            name = self.next()
            sep = self.peek()
            if sep == ":":
                self.next()
                if self.peek() == "{":  # "field: { ... }" is also legal text format
                    self.next()
                    msg.add(name, self.parse_message(False))
                else:
                    msg.add(name, _decode_scalar(self.next()))
            elif sep == "{":
                self.next()
                msg.add(name, self.parse_message(False))
            else:
                raise ValueError("prototxt: expected ':' or '{' after field %r, got %r" % (name, sep))


def parse_prototxt(text: str) -> Message:
    return _Parser(_tokenize(text)).parse_message(top_level=True)


def _format_value(v: Any) -> str:
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, str):
        # Heuristic: round-trip enum identifiers unquoted.
        if re.match(r"^[A-Z][A-Z0-9_]*$", v):
            return v
        return '"%s"' % v.replace("\\", "\\\\").replace('"', '\\"')
    if isinstance(v, float):
        return repr(v)
    return str(v)


def format_prototxt(msg: Message, indent: int = 0) -> str:
    pad = "    " * indent
    out = []
    for k, v in msg.items():
        if isinstance(v, Message):
            out.append("%s%s {\n%s%s}\n" % (pad, k, format_prototxt(v, indent + 1), pad))
        else:
            out.append("%s%s: %s\n" % (pad, k, _format_value(v)))
    return "".join(out)
