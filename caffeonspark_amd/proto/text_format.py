"""Caffe prototxt (protobuf text format) parser and printer.

Parses the `name: value` / `name { ... }` syntax used by every solver and
net definition in the reference corpus (reference data/*.prototxt), including
single-quoted strings, `#` comments, enum identifiers, and repeated fields.
"""

from __future__ import annotations

from typing import Any, List

from .pbcodec import Field, Message


class ParseError(ValueError):
    def __init__(self, msg: str, line: int):
        super().__init__(f"line {line}: {msg}")
        self.line = line


class _Lexer:
    def __init__(self, text: str):
        self.text = text
        self.pos = 0
        self.line = 1
        self._peeked = None

    def _skip_ws(self):
        t, n = self.text, len(self.text)
        while self.pos < n:
            c = t[self.pos]
            if c == "#":
                while self.pos < n and t[self.pos] != "\n":
                    self.pos += 1
            elif c == "\n":
                self.line += 1
                self.pos += 1
            elif c in " \t\r,":
                self.pos += 1
            else:
                return

    def peek(self):
        if self._peeked is None:
            self._peeked = self._next()
        return self._peeked

    def next(self):
        tok = self.peek()
        self._peeked = None
        return tok

    def _next(self):
        self._skip_ws()
        t, n = self.text, len(self.text)
        if self.pos >= n:
            return None
        c = t[self.pos]
        if c in "{}:":
            self.pos += 1
            return c
        if c in "\"'":
            quote = c
            self.pos += 1
            out = []
            while self.pos < n and t[self.pos] != quote:
                ch = t[self.pos]
                if ch == "\\" and self.pos + 1 < n:
                    self.pos += 1
                    esc = t[self.pos]
                    out.append({"n": "\n", "t": "\t", "\\": "\\",
                                "'": "'", '"': '"'}.get(esc, esc))
                else:
                    out.append(ch)
                self.pos += 1
            if self.pos >= n:
                raise ParseError("unterminated string", self.line)
            self.pos += 1
            return ("STR", "".join(out))
        # bare token: identifier / number / enum / bool
        start = self.pos
        while self.pos < n and t[self.pos] not in " \t\r\n,:{}#\"'":
            self.pos += 1
        if self.pos == start:
            raise ParseError(f"unexpected character {c!r}", self.line)
        return ("TOK", t[start:self.pos])


def _convert_scalar(f: Field, raw: Any, line: int):
    if isinstance(raw, tuple) and raw[0] == "STR":
        if f.type in ("string", "bytes"):
            return raw[1] if f.type == "string" else raw[1].encode("utf-8")
        raw = ("TOK", raw[1])
    tok = raw[1]
    t = f.type
    if t in ("string", "bytes"):
        return tok if t == "string" else tok.encode("utf-8")
    if t == "bool":
        if tok in ("true", "True", "1"):
            return True
        if tok in ("false", "False", "0"):
            return False
        raise ParseError(f"bad bool {tok!r}", line)
    if t == "enum":
        if f.enum_type is not None and tok in f.enum_type:
            return f.enum_type.by_name[tok]
        try:
            return int(tok)
        except ValueError:
            raise ParseError(f"unknown enum value {tok!r} for {f.name}", line)
    if t in ("float", "double"):
        return float(tok)
    try:
        return int(tok, 0)
    except ValueError:
        raise ParseError(f"bad number {tok!r} for field {f.name}", line)


def _parse_message(lex: _Lexer, msg: Message, terminator):
    cls = type(msg)
    while True:
        tok = lex.next()
        if tok == terminator:
            return
        if tok is None:
            if terminator is None:
                return
            raise ParseError("unexpected end of input (missing '}')", lex.line)
        if not (isinstance(tok, tuple) and tok[0] == "TOK"):
            raise ParseError(f"expected field name, got {tok!r}", lex.line)
        fname = tok[1]
        f = cls._by_name.get(fname)
        nxt = lex.next()
        if nxt == "{" or (nxt == ":" and lex.peek() == "{"):
            if nxt == ":":
                lex.next()
            if f is None:
                _skip_block(lex)
                continue
            if f.type != "message":
                raise ParseError(f"field {fname} is not a message", lex.line)
            sub = f.msg_type()
            _parse_message(lex, sub, "}")
            if f.repeated:
                getattr(msg, fname).append(sub)
            else:
                if fname in msg._values:
                    msg._values[fname]._merge(sub)
                else:
                    setattr(msg, fname, sub)
        elif nxt == ":":
            val_tok = lex.next()
            if f is None:
                continue
            # repeated scalar shorthand `f: [a, b, c]`
            if isinstance(val_tok, tuple) and val_tok[1] == "[":
                raise ParseError("list syntax not supported", lex.line)
            val = _convert_scalar(f, val_tok, lex.line)
            if f.repeated:
                getattr(msg, fname).append(val)
            else:
                setattr(msg, fname, val)
        else:
            raise ParseError(f"expected ':' or '{{' after {fname}", lex.line)


def _skip_block(lex: _Lexer):
    depth = 1
    while depth:
        tok = lex.next()
        if tok is None:
            raise ParseError("unexpected end of input in skipped block", lex.line)
        if tok == "{":
            depth += 1
        elif tok == "}":
            depth -= 1


def parse(text: str, msg_cls) -> Message:
    msg = msg_cls()
    merge(text, msg)
    return msg


def merge(text: str, msg: Message) -> Message:
    lex = _Lexer(text)
    _parse_message(lex, msg, None)
    return msg


def parse_file(path: str, msg_cls) -> Message:
    with open(path, "r") as fh:
        return parse(fh.read(), msg_cls)


# ---------------------------------------------------------------------------


def _fmt_scalar(f: Field, v: Any) -> str:
    t = f.type
    if t == "string":
        escaped = str(v).replace("\\", "\\\\").replace('"', '\\"').replace("\n", "\\n")
        return f'"{escaped}"'
    if t == "bytes":
        return '"' + v.decode("utf-8", errors="replace") + '"'
    if t == "bool":
        return "true" if v else "false"
    if t == "enum":
        if f.enum_type is not None and v in f.enum_type.by_number:
            return f.enum_type.by_number[v]
        return str(int(v))
    if t in ("float", "double"):
        s = repr(float(v))
        return s
    return str(int(v))


def _dump(msg: Message, lines: List[str], indent: int) -> None:
    pad = "  " * indent
    for f in type(msg).FIELDS:
        if f.name not in msg._values:
            continue
        v = msg._values[f.name]
        items = v if f.repeated else [v]
        for item in items:
            if f.type == "message":
                if isinstance(item, Message) and not item._values and not f.repeated:
                    continue
                lines.append(f"{pad}{f.name} {{")
                _dump(item, lines, indent + 1)
                lines.append(f"{pad}}}")
            else:
                lines.append(f"{pad}{f.name}: {_fmt_scalar(f, item)}")


def dumps(msg: Message) -> str:
    lines: List[str] = []
    _dump(msg, lines, 0)
    return "\n".join(lines) + ("\n" if lines else "")
