"""Minimal CEL evaluator for ValidatingAdmissionPolicy expressions.

Covers the CEL subset the FMA admission policies use (reference
config/validating-admission-policies/fma-immutable-fields.yaml:1-33 and
fma-bound-serverreqpod.yaml:1-30; ours in
manifests/validating-admission-policies/):

  - logical ``||`` ``&&`` ``!`` and parentheses
  - equality ``==`` ``!=``
  - member access ``a.b``, optional member access ``a.?b`` (CEL optional
    semantics: absent fields propagate), map index ``a['k']``
  - method calls ``.orValue(x)``, ``.matches(re)``, ``.startsWith(s)``,
    ``.endsWith(s)``, ``.contains(s)``
  - macro ``has(a.b.c)``
  - string / int / bool / null literals, ``in`` membership, list literals

The policies are data, not code: evaluating the *shipped YAML artifacts*
in tests proves the deny rules a real cluster would enforce, instead of
trusting a hand-written Python re-implementation to match them.
"""

from __future__ import annotations

import re
from typing import Any, List, Optional, Tuple


class CelError(Exception):
    pass


class _Absent:
    """CEL optional.none(): propagates through optional chains."""

    _instance: Optional["_Absent"] = None

    def __new__(cls):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __repr__(self):  # pragma: no cover
        return "optional.none()"


ABSENT = _Absent()


class _Opt:
    """A CEL optional value (either a wrapped value or absent)."""

    __slots__ = ("value",)

    def __init__(self, value: Any):
        self.value = value  # may be ABSENT

    @property
    def present(self) -> bool:
        return self.value is not ABSENT


# ---------------------------------------------------------------------------
# tokenizer
# ---------------------------------------------------------------------------

_TOKEN_RE = re.compile(r"""
    \s*(?:
      (?P<op>\|\||&&|==|!=|<=|>=|\.\?|[()\[\].!,<>])
    | (?P<str>'(?:[^'\\]|\\.)*'|"(?:[^"\\]|\\.)*")
    | (?P<num>\d+)
    | (?P<ident>[A-Za-z_][A-Za-z0-9_]*)
    )""", re.VERBOSE)


def _tokenize(src: str) -> List[Tuple[str, str]]:
    toks: List[Tuple[str, str]] = []
    pos = 0
    while pos < len(src):
        m = _TOKEN_RE.match(src, pos)
        if not m:
            if src[pos:].strip() == "":
                break
            raise CelError(f"cannot tokenize at: {src[pos:pos + 20]!r}")
        pos = m.end()
        for kind in ("op", "str", "num", "ident"):
            v = m.group(kind)
            if v is not None:
                toks.append((kind, v))
                break
    toks.append(("eof", ""))
    return toks


# ---------------------------------------------------------------------------
# recursive-descent parser -> closures
# ---------------------------------------------------------------------------

class _Parser:
    def __init__(self, toks: List[Tuple[str, str]]):
        self.toks = toks
        self.i = 0

    def peek(self) -> Tuple[str, str]:
        return self.toks[self.i]

    def next(self) -> Tuple[str, str]:
        t = self.toks[self.i]
        self.i += 1
        return t

    def expect(self, val: str) -> None:
        k, v = self.next()
        if v != val:
            raise CelError(f"expected {val!r}, got {v!r}")

    # or_expr := and_expr ('||' and_expr)*
    def parse(self):
        e = self.or_expr()
        if self.peek()[0] != "eof":
            raise CelError(f"trailing tokens at {self.peek()!r}")
        return e

    def or_expr(self):
        parts = [self.and_expr()]
        while self.peek()[1] == "||":
            self.next()
            parts.append(self.and_expr())
        if len(parts) == 1:
            return parts[0]
        return lambda env: any(_truth(p(env)) for p in parts)

    def and_expr(self):
        parts = [self.rel_expr()]
        while self.peek()[1] == "&&":
            self.next()
            parts.append(self.rel_expr())
        if len(parts) == 1:
            return parts[0]
        return lambda env: all(_truth(p(env)) for p in parts)

    def rel_expr(self):
        left = self.unary_expr()
        op = self.peek()[1]
        if op in ("==", "!="):
            self.next()
            right = self.unary_expr()
            if op == "==":
                return lambda env: _unwrap(left(env)) == _unwrap(right(env))
            return lambda env: _unwrap(left(env)) != _unwrap(right(env))
        if self.peek() == ("ident", "in"):
            self.next()
            right = self.unary_expr()

            def _in(env):
                a, b = _unwrap(left(env)), _unwrap(right(env))
                try:
                    return a in b
                except TypeError as e:
                    raise CelError(f"'in' type error: {e}") from e
            return _in
        return left

    def unary_expr(self):
        if self.peek()[1] == "!":
            self.next()
            inner = self.unary_expr()
            return lambda env: not _truth(inner(env))
        return self.postfix_expr()

    def postfix_expr(self):
        base = self.primary()
        while True:
            k, v = self.peek()
            if v == ".":
                self.next()
                name = self._ident()
                if self.peek()[1] == "(":
                    base = self._call(base, name, optional=False)
                else:
                    base = self._member(base, name, optional=False)
            elif v == ".?":
                self.next()
                name = self._ident()
                base = self._member(base, name, optional=True)
            elif v == "[":
                self.next()
                idx = self.or_expr()
                self.expect("]")
                base = self._index(base, idx)
            else:
                return base

    def _ident(self) -> str:
        k, v = self.next()
        if k != "ident":
            raise CelError(f"expected identifier, got {v!r}")
        return v

    def primary(self):
        k, v = self.peek()
        if v == "(":
            self.next()
            e = self.or_expr()
            self.expect(")")
            return e
        if k == "str":
            self.next()
            s = _unquote(v)
            return lambda env, s=s: s
        if k == "num":
            self.next()
            n = int(v)
            return lambda env, n=n: n
        if v == "[":
            self.next()
            items = []
            if self.peek()[1] != "]":
                items.append(self.or_expr())
                while self.peek()[1] == ",":
                    self.next()
                    items.append(self.or_expr())
            self.expect("]")
            return lambda env: [_unwrap(i(env)) for i in items]
        if k == "ident":
            self.next()
            if v == "true":
                return lambda env: True
            if v == "false":
                return lambda env: False
            if v == "null":
                return lambda env: None
            if v == "has" and self.peek()[1] == "(":
                self.next()
                inner = self.or_expr()
                self.expect(")")

                def _has(env):
                    try:
                        val = inner(env)
                    except CelError:
                        return False
                    if isinstance(val, _Opt):
                        return val.present
                    return val is not ABSENT and val is not None
                return _has
            name = v
            return lambda env, name=name: _lookup_var(env, name)
        raise CelError(f"unexpected token {v!r}")

    def _member(self, base, name: str, optional: bool):
        def access(env):
            obj = base(env)
            return _get_field(obj, name, optional)
        return access

    def _index(self, base, idx):
        def access(env):
            obj = base(env)
            key = _unwrap(idx(env))
            return _get_index(obj, key)
        return access

    def _call(self, base, name: str, optional: bool):
        self.expect("(")
        args = []
        if self.peek()[1] != ")":
            args.append(self.or_expr())
            while self.peek()[1] == ",":
                self.next()
                args.append(self.or_expr())
        self.expect(")")

        def call(env):
            recv = base(env)
            vals = [_unwrap(a(env)) for a in args]
            return _method(recv, name, vals)
        return call


def _unquote(tok: str) -> str:
    body = tok[1:-1]
    return re.sub(r"\\(.)", r"\1", body)


def _lookup_var(env: dict, name: str) -> Any:
    if name in env:
        return env[name]
    raise CelError(f"unknown variable {name!r}")


def _get_field(obj: Any, name: str, optional: bool) -> Any:
    if isinstance(obj, _Opt):
        if not obj.present:
            return obj
        obj = obj.value
    if isinstance(obj, dict):
        if name in obj and obj[name] is not None:
            val = obj[name]
            return _Opt(val) if optional else val
        if optional:
            return _Opt(ABSENT)
        raise CelError(f"no such field {name!r}")
    if optional:
        return _Opt(ABSENT)
    raise CelError(f"cannot access field {name!r} on {type(obj).__name__}")


def _get_index(obj: Any, key: Any) -> Any:
    if isinstance(obj, _Opt):
        if not obj.present:
            return obj
        inner = obj.value
        if isinstance(inner, dict) and key in inner:
            return _Opt(inner[key])
        return _Opt(ABSENT)
    if isinstance(obj, dict):
        if key in obj:
            return obj[key]
        raise CelError(f"no such key {key!r}")
    if isinstance(obj, list):
        try:
            return obj[key]
        except (TypeError, IndexError) as e:
            raise CelError(f"index error: {e}") from e
    raise CelError(f"cannot index {type(obj).__name__}")


def _method(recv: Any, name: str, args: List[Any]) -> Any:
    if name == "orValue":
        if isinstance(recv, _Opt):
            return recv.value if recv.present else args[0]
        return recv if recv is not ABSENT else args[0]
    if name == "optMap":  # pragma: no cover - not used by our policies
        raise CelError("optMap unsupported")
    recv = _unwrap(recv)
    try:
        if name == "matches":
            return re.search(args[0], recv) is not None
        if name == "startsWith":
            return recv.startswith(args[0])
        if name == "endsWith":
            return recv.endswith(args[0])
        if name == "contains":
            return args[0] in recv
        if name == "size":
            return len(recv)
    except (TypeError, AttributeError, re.error) as e:
        raise CelError(f"method {name!r} type error: {e}") from e
    raise CelError(f"unknown method {name!r}")


def _unwrap(v: Any) -> Any:
    if isinstance(v, _Opt):
        if not v.present:
            raise CelError("optional.none() used as a value (use orValue)")
        return v.value
    return v


def _truth(v: Any) -> bool:
    v = _unwrap(v)
    if not isinstance(v, bool):
        raise CelError(f"non-bool in boolean context: {v!r}")
    return v


def compile_expr(src: str):
    """Compile a CEL expression to a callable(env_dict) -> value."""
    return _Parser(_tokenize(src)).parse()


def evaluate(src: str, env: dict) -> Any:
    return compile_expr(src)(env)
