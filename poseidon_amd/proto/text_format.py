"""Prototxt (protobuf text-format) parser and printer for the caffe schema.

Handles the subset of text format that Caffe model/solver prototxts use:
scalar fields, enum symbols, quoted strings, nested messages (both
``field { ... }`` and ``field: { ... }``), repeated fields, and ``#``
comments. Mirrors the behavior of the reference's
ReadProtoFromTextFile (/root/reference/src/caffe/util/io.cpp:38-46) without
depending on protoc-generated code.
"""

from __future__ import annotations

import re
from typing import Any, List, Tuple

from . import spec
from .message import Message

_TOKEN_RE = re.compile(
    r"""
    \s+
  | \#[^\n]*
  | (?P<brace>[{}])
  | (?P<colon>:)
  | (?P<string>"(?:\\.|[^"\\])*"|'(?:\\.|[^'\\])*')
  | (?P<ident>[A-Za-z_][A-Za-z0-9_]*)
  | (?P<number>[-+]?(?:\d+\.\d*(?:[eE][-+]?\d+)?|\.\d+(?:[eE][-+]?\d+)?|\d+(?:[eE][-+]?\d+)?|nan|inf))
    """,
    re.VERBOSE,
)

_ESCAPES = {"n": "\n", "t": "\t", "r": "\r", "\\": "\\", '"': '"', "'": "'", "0": "\0"}


def _tokenize(text: str) -> List[Tuple[str, str]]:
    tokens: List[Tuple[str, str]] = []
    pos = 0
    while pos < len(text):
        m = _TOKEN_RE.match(text, pos)
        if m is None:
            raise ValueError(f"prototxt parse error near {text[pos:pos+40]!r}")
        pos = m.end()
        for kind in ("brace", "colon", "string", "ident", "number"):
            val = m.group(kind)
            if val is not None:
                tokens.append((kind, val))
                break
    return tokens


def _unquote(s: str) -> str:
    body = s[1:-1]
    out = []
    i = 0
    while i < len(body):
        c = body[i]
        if c == "\\" and i + 1 < len(body):
            out.append(_ESCAPES.get(body[i + 1], body[i + 1]))
            i += 2
        else:
            out.append(c)
            i += 1
    return "".join(out)


class _Parser:
    def __init__(self, tokens: List[Tuple[str, str]]):
        self.tokens = tokens
        self.pos = 0

    def peek(self):
        return self.tokens[self.pos] if self.pos < len(self.tokens) else (None, None)

    def next(self):
        tok = self.peek()
        self.pos += 1
        return tok

    def parse_into(self, msg: Message) -> None:
        while True:
            kind, val = self.peek()
            if kind is None or val == "}":
                return
            self._parse_field(msg)

    def _parse_field(self, msg: Message) -> None:
        kind, name = self.next()
        if kind != "ident":
            raise ValueError(f"expected field name, got {name!r}")
        fields = spec.MESSAGES[msg.type_name]
        if name not in fields:
            raise ValueError(f"{msg.type_name} has no field {name!r}")
        num, fkind, label, default = fields[name]

        kind, val = self.peek()
        has_colon = kind == "colon"
        if has_colon:
            self.next()
            kind, val = self.peek()

        if fkind.startswith("msg:"):
            if val != "{":
                raise ValueError(f"expected '{{' for field {name}")
            self.next()
            sub = Message(fkind[4:])
            self.parse_into(sub)
            kind, val = self.next()
            if val != "}":
                raise ValueError(f"unterminated message for field {name}")
            if label == "rep":
                getattr(msg, name).append(sub)
            else:
                cur = object.__getattribute__(msg, "_values").get(name)
                if cur is None:
                    setattr(msg, name, sub)
                else:
                    cur.merge_from(sub)
            return

        kind, val = self.next()
        parsed = self._scalar(fkind, kind, val, name)
        if label in ("rep", "packed"):
            getattr(msg, name).append(parsed)
        else:
            setattr(msg, name, parsed)

    @staticmethod
    def _scalar(fkind: str, tok_kind: str, val: str, name: str) -> Any:
        if fkind.startswith("enum:"):
            table = spec.ENUMS[fkind[5:]]
            if tok_kind == "ident":
                if val not in table:
                    raise ValueError(f"{name}: unknown enum symbol {val!r}")
                return table[val]
            return int(val)
        if fkind == "bool":
            if tok_kind == "ident":
                return val in ("true", "True")
            return bool(int(val))
        if fkind in ("string", "bytes"):
            if tok_kind != "string":
                raise ValueError(f"{name}: expected quoted string")
            s = _unquote(val)
            return s.encode("utf-8") if fkind == "bytes" else s
        if fkind in ("float", "double"):
            return float(val)
        return int(float(val))  # int kinds; tolerate 1e3 style


def parse_text(type_name: str, text: str) -> Message:
    msg = Message(type_name)
    parser = _Parser(_tokenize(text))
    parser.parse_into(msg)
    if parser.pos != len(parser.tokens):
        raise ValueError("trailing tokens in prototxt")
    return msg


def _fmt_scalar(fkind: str, v: Any, msg_fields, name: str) -> str:
    if fkind.startswith("enum:"):
        table = spec.ENUMS[fkind[5:]]
        for sym, n in table.items():
            if n == v:
                return sym
        return str(v)
    if fkind == "bool":
        return "true" if v else "false"
    if fkind in ("string",):
        escaped = str(v).replace("\\", "\\\\").replace('"', '\\"').replace("\n", "\\n")
        return f'"{escaped}"'
    if fkind == "bytes":
        return '"' + "".join(f"\\{b:03o}" if b < 32 or b > 126 else chr(b) for b in v) + '"'
    if fkind in ("float", "double"):
        return repr(float(v))
    return str(int(v))


def to_text(msg: Message, indent: int = 0) -> str:
    pad = "  " * indent
    lines: List[str] = []
    values = object.__getattribute__(msg, "_values")
    for name, (num, fkind, label, default) in spec.MESSAGES[msg.type_name].items():
        if name not in values:
            continue
        v = values[name]
        items = list(v) if label in ("rep", "packed") else [v]
        for item in items:
            if fkind.startswith("msg:"):
                lines.append(f"{pad}{name} {{")
                lines.append(to_text(item, indent + 1))
                lines.append(f"{pad}}}")
            else:
                lines.append(f"{pad}{name}: {_fmt_scalar(fkind, item, None, name)}")
    return "\n".join(l for l in lines if l != "")
