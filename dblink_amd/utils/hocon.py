"""Minimal HOCON parser for dblink-style configuration files.

Implements the subset of HOCON (Typesafe Config) used by dblink configs
(reference: /root/reference/examples/RLdata500.conf, docs/configuration.md):

- ``//`` and ``#`` comments
- nested objects ``{ ... }`` with ``:`` / ``=`` separators (or none before ``{``)
- arrays ``[ ... ]`` of scalars or objects
- newline- or comma-separated members, trailing commas tolerated
- quoted and unquoted strings, ints, floats, booleans, null
- substitutions ``${a.b.c}`` resolved against the root after parsing
- duplicate object keys merge (later wins for scalars, deep-merge for objects)
- dotted keys ``a.b : v`` create nested objects

The public access API mirrors the small part of the Typesafe Config API the
reference uses (``Project.scala:170-199``, ``ProjectSteps.scala:53-83``):
``get*`` accessors take dotted paths and raise ``ConfigMissingError`` /
``ConfigValueError``.
"""

from __future__ import annotations


class ConfigError(Exception):
    pass


class ConfigMissingError(ConfigError):
    pass


class ConfigValueError(ConfigError):
    pass


class _Substitution:
    __slots__ = ("path", "optional")

    def __init__(self, path, optional=False):
        self.path = path
        self.optional = optional

    def __repr__(self):
        return f"${{{self.path}}}"


class _Tokenizer:
    PUNCT = {"{", "}", "[", "]", ",", ":", "="}

    def __init__(self, text):
        self.text = text
        self.pos = 0
        self.n = len(text)

    def _skip_ws_inline(self):
        while self.pos < self.n:
            c = self.text[self.pos]
            if c in " \t\r":
                self.pos += 1
            elif c == "/" and self.text[self.pos : self.pos + 2] == "//":
                while self.pos < self.n and self.text[self.pos] != "\n":
                    self.pos += 1
            elif c == "#":
                while self.pos < self.n and self.text[self.pos] != "\n":
                    self.pos += 1
            else:
                return

    def tokens(self):
        """Yield (kind, value) tokens; kind in {punct, newline, string, raw, subst}."""
        out = []
        while True:
            self._skip_ws_inline()
            if self.pos >= self.n:
                break
            c = self.text[self.pos]
            if c == "\n":
                out.append(("newline", "\n"))
                self.pos += 1
            elif c in self.PUNCT:
                out.append(("punct", c))
                self.pos += 1
            elif c == '"':
                out.append(("string", self._read_quoted()))
            elif c == "$" and self.text[self.pos : self.pos + 2] == "${":
                out.append(("subst", self._read_subst()))
            else:
                out.append(("raw", self._read_unquoted()))
        return out

    def _read_quoted(self):
        # triple-quoted
        if self.text[self.pos : self.pos + 3] == '"""':
            end = self.text.find('"""', self.pos + 3)
            if end < 0:
                raise ConfigError("unterminated triple-quoted string")
            s = self.text[self.pos + 3 : end]
            self.pos = end + 3
            return s
        self.pos += 1
        buf = []
        while self.pos < self.n:
            c = self.text[self.pos]
            if c == "\\":
                esc = self.text[self.pos + 1]
                mapping = {"n": "\n", "t": "\t", "r": "\r", '"': '"', "\\": "\\", "/": "/", "b": "\b", "f": "\f"}
                if esc == "u":
                    buf.append(chr(int(self.text[self.pos + 2 : self.pos + 6], 16)))
                    self.pos += 6
                    continue
                buf.append(mapping.get(esc, esc))
                self.pos += 2
            elif c == '"':
                self.pos += 1
                return "".join(buf)
            elif c == "\n":
                raise ConfigError("newline in quoted string")
            else:
                buf.append(c)
                self.pos += 1
        raise ConfigError("unterminated string")

    def _read_subst(self):
        end = self.text.find("}", self.pos)
        if end < 0:
            raise ConfigError("unterminated substitution")
        inner = self.text[self.pos + 2 : end]
        self.pos = end + 1
        optional = inner.startswith("?")
        if optional:
            inner = inner[1:]
        return _Substitution(inner.strip(), optional)

    def _read_unquoted(self):
        start = self.pos
        while self.pos < self.n:
            c = self.text[self.pos]
            if c in self.PUNCT or c in ' \t\r\n"$' or c == "#" or self.text[self.pos : self.pos + 2] == "//":
                break
            self.pos += 1
        return self.text[start : self.pos]


def _coerce_scalar(raw):
    if raw == "true":
        return True
    if raw == "false":
        return False
    if raw == "null":
        return None
    try:
        return int(raw)
    except ValueError:
        pass
    try:
        return float(raw)
    except ValueError:
        pass
    return raw


class _Parser:
    def __init__(self, tokens):
        self.toks = tokens
        self.i = 0

    def peek(self):
        return self.toks[self.i] if self.i < len(self.toks) else ("eof", None)

    def next(self):
        t = self.peek()
        self.i += 1
        return t

    def skip_newlines(self):
        while self.peek()[0] == "newline":
            self.i += 1

    def parse_root(self):
        self.skip_newlines()
        if self.peek() == ("punct", "{"):
            obj = self.parse_object()
        else:
            obj = self.parse_members(until=None)
        self.skip_newlines()
        if self.peek()[0] != "eof":
            raise ConfigError(f"trailing content at token {self.peek()}")
        return obj

    def parse_object(self):
        assert self.next() == ("punct", "{")
        obj = self.parse_members(until="}")
        tok = self.next()
        if tok != ("punct", "}"):
            raise ConfigError(f"expected '}}', got {tok}")
        return obj

    def parse_members(self, until):
        obj = {}
        while True:
            self.skip_newlines()
            kind, val = self.peek()
            if kind == "eof" or (kind == "punct" and val == until):
                return obj
            if kind == "punct" and val == ",":
                self.i += 1
                continue
            key = self.parse_key()
            self.skip_newlines_not_needed = None
            kind, val = self.peek()
            if kind == "punct" and val in (":", "="):
                self.i += 1
                value = self.parse_value()
            elif kind == "punct" and val == "{":
                value = self.parse_object()
            else:
                raise ConfigError(f"expected ':', '=' or '{{' after key {key!r}, got {self.peek()}")
            _assign(obj, key.split("."), value)

    def parse_key(self):
        kind, val = self.next()
        if kind == "string":
            return val
        if kind == "raw":
            return val
        raise ConfigError(f"expected key, got {(kind, val)}")

    def parse_value(self):
        self.skip_newlines()
        kind, val = self.peek()
        if kind == "punct" and val == "{":
            return self.parse_object()
        if kind == "punct" and val == "[":
            return self.parse_array()
        # scalar, possibly a multi-token concatenation on one line
        parts = []
        while True:
            kind, val = self.peek()
            if kind in ("raw", "string", "subst"):
                parts.append((kind, val))
                self.i += 1
            else:
                break
        if not parts:
            raise ConfigError(f"expected value, got {self.peek()}")
        if len(parts) == 1:
            kind, val = parts[0]
            if kind == "raw":
                return _coerce_scalar(val)
            return val
        # value concatenation: join as string (substitutions resolved later)
        return _Concat([v if k != "raw" else v for k, v in parts])

    def parse_array(self):
        assert self.next() == ("punct", "[")
        items = []
        while True:
            self.skip_newlines()
            kind, val = self.peek()
            if kind == "punct" and val == "]":
                self.i += 1
                return items
            if kind == "punct" and val == ",":
                self.i += 1
                continue
            items.append(self.parse_value())


class _Concat:
    __slots__ = ("parts",)

    def __init__(self, parts):
        self.parts = parts


def _assign(obj, path, value):
    """Assign value at dotted path inside obj, merging objects on duplicate keys."""
    key = path[0]
    if len(path) == 1:
        if key in obj and isinstance(obj[key], dict) and isinstance(value, dict):
            _merge(obj[key], value)
        else:
            obj[key] = value
        return
    child = obj.setdefault(key, {})
    if not isinstance(child, dict):
        child = {}
        obj[key] = child
    _assign(child, path[1:], value)


def _merge(dst, src):
    for k, v in src.items():
        if k in dst and isinstance(dst[k], dict) and isinstance(v, dict):
            _merge(dst[k], v)
        else:
            dst[k] = v


def _resolve(node, root, seen):
    if isinstance(node, _Substitution):
        key = node.path
        if key in seen:
            raise ConfigError(f"substitution cycle at ${{{key}}}")
        try:
            target = _lookup(root, key)
        except ConfigMissingError:
            if node.optional:
                return None
            raise
        return _resolve(target, root, seen | {key})
    if isinstance(node, _Concat):
        vals = [_resolve(p, root, seen) for p in node.parts]
        return "".join(str(v) for v in vals)
    if isinstance(node, dict):
        return {k: _resolve(v, root, seen) for k, v in node.items()}
    if isinstance(node, list):
        return [_resolve(v, root, seen) for v in node]
    return node


def _lookup(obj, dotted):
    cur = obj
    for part in dotted.split("."):
        if not isinstance(cur, dict) or part not in cur:
            raise ConfigMissingError(f"no config value at path '{dotted}'")
        cur = cur[part]
    return cur


class Config:
    """Resolved configuration tree with typed dotted-path accessors."""

    def __init__(self, data):
        self._data = data

    @property
    def data(self):
        return self._data

    def has_path(self, path):
        try:
            _lookup(self._data, path)
            return True
        except ConfigMissingError:
            return False

    def get(self, path):
        return _lookup(self._data, path)

    def get_or(self, path, default=None):
        try:
            return _lookup(self._data, path)
        except ConfigMissingError:
            return default

    def get_string(self, path):
        v = self.get(path)
        if v is None:
            raise ConfigValueError(f"null value at '{path}'")
        return str(v)

    def get_int(self, path):
        v = self.get(path)
        if isinstance(v, bool) or not isinstance(v, (int, float)):
            raise ConfigValueError(f"expected number at '{path}', got {v!r}")
        return int(v)

    def get_long(self, path):
        return self.get_int(path)

    def get_float(self, path):
        v = self.get(path)
        if isinstance(v, bool) or not isinstance(v, (int, float)):
            raise ConfigValueError(f"expected number at '{path}', got {v!r}")
        return float(v)

    get_double = get_float

    def get_bool(self, path):
        v = self.get(path)
        if not isinstance(v, bool):
            raise ConfigValueError(f"expected boolean at '{path}', got {v!r}")
        return v

    def get_list(self, path):
        v = self.get(path)
        if not isinstance(v, list):
            raise ConfigValueError(f"expected list at '{path}', got {v!r}")
        return v

    def get_string_list(self, path):
        return [str(x) for x in self.get_list(path)]

    def get_config(self, path):
        v = self.get(path)
        if not isinstance(v, dict):
            raise ConfigValueError(f"expected object at '{path}', got {v!r}")
        return Config(v)

    def get_config_list(self, path):
        v = self.get_list(path)
        out = []
        for x in v:
            if not isinstance(x, dict):
                raise ConfigValueError(f"expected list of objects at '{path}'")
            out.append(Config(x))
        return out


def parse_string(text):
    toks = _Tokenizer(text).tokens()
    raw = _Parser(toks).parse_root()
    resolved = _resolve(raw, raw, frozenset())
    return Config(resolved)


def parse_file(path):
    with open(path, "r", encoding="utf-8") as f:
        return parse_string(f.read())
