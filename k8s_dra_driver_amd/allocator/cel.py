"""CEL-subset evaluator for DRA device selectors.

In production the kube-scheduler evaluates DeviceClass/request CEL selectors
against published ResourceSlice attributes (reference relies on this
entirely — SURVEY.md §3.5). This driver ships its own evaluator so the
in-repo allocator, bench harness and tests can run the SAME selector
expressions without a cluster.

Supported grammar (covers every expression in the reference's DeviceClasses
and demo specs — deviceclass-gpu.yaml:10, gpu-test4/5/6 — plus quantity
comparisons for capacity):

    expr     := or
    or       := and ("||" and)*
    and      := rel ("&&" rel)*
    rel      := add (("=="|"!="|"<="|">="|"<"|">"|"in") add)?
    add      := unary (("+"|"-") unary)*
    unary    := "!" unary | "-" unary | postfix
    postfix  := primary (("." ident call?) | "[" expr "]")*
    primary  := literal | ident | "(" expr ")" | list | "quantity(" expr ")"

Variables: ``device.driver`` (string), ``device.attributes["domain"].name``,
``device.capacity["domain"].name`` (quantity-valued). String methods:
``lowerAscii upperAscii matches contains startsWith endsWith size``.
A missing attribute raises :class:`CelError`; per the DRA contract a
runtime error means *no match* (the allocator treats it so).
"""

from __future__ import annotations

import re
from typing import Any, Dict, List

from ..api.types import parse_quantity_bytes


class CelError(Exception):
    pass


class _CelErr:
    """Evaluation-error VALUE (not an exception): CEL's ||/&& absorb
    operand errors (true || error -> true), so runtime errors must flow
    as values through the evaluator and only surface if they reach the
    top. Parse errors still raise CelError immediately."""

    __slots__ = ("msg",)

    def __init__(self, msg: str):
        self.msg = msg

    def __repr__(self):
        return f"celerror({self.msg!r})"


def _is_err(v) -> bool:
    return isinstance(v, _CelErr)


_TOKEN_RE = re.compile(
    r"""
    (?P<ws>\s+)
  | (?P<float>\d+\.\d+)
  | (?P<int>\d+)
  | (?P<string>'(?:[^'\\]|\\.)*'|"(?:[^"\\]|\\.)*")
  | (?P<op>\|\||&&|==|!=|<=|>=|[()\[\].,<>!+\-])
  | (?P<ident>[A-Za-z_][A-Za-z0-9_]*)
""",
    re.VERBOSE,
)


def _tokenize(src: str) -> List[tuple]:
    out = []
    pos = 0
    while pos < len(src):
        m = _TOKEN_RE.match(src, pos)
        if not m:
            raise CelError(f"bad token at {pos}: {src[pos:pos+20]!r}")
        pos = m.end()
        kind = m.lastgroup
        if kind == "ws":
            continue
        text = m.group()
        if kind == "string":
            text = text[1:-1].replace("\\'", "'").replace('\\"', '"')
        out.append((kind, text))
    out.append(("eof", ""))
    return out


class Quantity:
    """Comparable quantity (capacity values)."""

    __slots__ = ("bytes",)

    def __init__(self, b: int):
        self.bytes = b

    def _cmp_val(self, other):
        if isinstance(other, Quantity):
            return other.bytes
        if isinstance(other, (int, float)):
            return other
        if isinstance(other, str):
            return parse_quantity_bytes(other)
        raise CelError(f"cannot compare quantity with {type(other).__name__}")

    def __eq__(self, other):
        try:
            return self.bytes == self._cmp_val(other)
        except CelError:
            return NotImplemented

    def __lt__(self, other):
        return self.bytes < self._cmp_val(other)

    def __le__(self, other):
        return self.bytes <= self._cmp_val(other)

    def __gt__(self, other):
        return self.bytes > self._cmp_val(other)

    def __ge__(self, other):
        return self.bytes >= self._cmp_val(other)

    def __hash__(self):
        return hash(self.bytes)

    def __repr__(self):
        return f"quantity({self.bytes})"


class _AttrMap:
    """``device.attributes['domain']`` — fields via member access."""

    def __init__(self, fields: Dict[str, Any], domain: str, kind: str):
        self._fields = fields
        self._domain = domain
        self._kind = kind

    def get(self, name: str) -> Any:
        if name not in self._fields:
            raise CelError(
                f"device.{self._kind}[{self._domain!r}] has no field {name!r}"
            )
        return self._fields[name]


class _Parser:
    def __init__(self, tokens: List[tuple], env: Dict[str, Any]):
        self.toks = tokens
        self.i = 0
        self.env = env

    def peek(self):
        return self.toks[self.i]

    def next(self):
        t = self.toks[self.i]
        self.i += 1
        return t

    def expect(self, text: str):
        kind, t = self.next()
        if t != text:
            raise CelError(f"expected {text!r}, got {t!r}")

    # grammar ----------------------------------------------------------------
    def parse(self) -> Any:
        v = self.or_()
        if self.peek()[0] != "eof":
            raise CelError(f"trailing tokens at {self.peek()[1]!r}")
        if _is_err(v):
            raise CelError(v.msg)
        return v

    @staticmethod
    def _as_bool(v):
        """CEL logical operand: bool or error value."""
        if _is_err(v) or isinstance(v, bool):
            return v
        return _CelErr(f"logical operand is {type(v).__name__}, not bool")

    def or_(self) -> Any:
        first = self.and_()
        if self.peek()[1] != "||":
            return first  # not a logical expression: pass through untyped
        operands = [self._as_bool(first)]
        while self.peek()[1] == "||":
            self.next()
            operands.append(self._as_bool(self.and_()))
        # CEL commutative-or: any true -> true; else any error -> error
        if any(v is True for v in operands):
            return True
        for v in operands:
            if _is_err(v):
                return v
        return False

    def and_(self) -> Any:
        first = self.rel()
        if self.peek()[1] != "&&":
            return first  # not a logical expression: pass through untyped
        operands = [self._as_bool(first)]
        while self.peek()[1] == "&&":
            self.next()
            operands.append(self._as_bool(self.rel()))
        # CEL commutative-and: any false -> false; else any error -> error
        if any(v is False for v in operands):
            return False
        for v in operands:
            if _is_err(v):
                return v
        return True

    def rel(self) -> Any:
        v = self.add()
        op = self.peek()[1]
        if op in ("==", "!=", "<", "<=", ">", ">=", "in"):
            self.next()
            rhs = self.add()
            if _is_err(v):
                return v
            if _is_err(rhs):
                return rhs
            if op == "==":
                return v == rhs
            if op == "!=":
                return v != rhs
            if op == "in":
                if not isinstance(rhs, list):
                    return _CelErr("'in' requires a list on the right")
                return v in rhs
            try:
                if op == "<":
                    return v < rhs
                if op == "<=":
                    return v <= rhs
                if op == ">":
                    return v > rhs
                return v >= rhs
            except TypeError as e:
                return _CelErr(str(e))
        return v

    def add(self) -> Any:
        v = self.unary()
        while self.peek()[1] in ("+", "-"):
            op = self.next()[1]
            rhs = self.unary()
            if _is_err(v):
                continue
            if _is_err(rhs):
                v = rhs
                continue
            try:
                v = v + rhs if op == "+" else v - rhs
            except TypeError as e:
                v = _CelErr(str(e))
        return v

    def unary(self) -> Any:
        t = self.peek()[1]
        if t == "!":
            self.next()
            v = self.unary()
            if _is_err(v):
                return v
            if not isinstance(v, bool):
                return _CelErr(f"! on {type(v).__name__}")
            return not v
        if t == "-":
            self.next()
            v = self.unary()
            if _is_err(v):
                return v
            if not isinstance(v, (int, float)) or isinstance(v, bool):
                return _CelErr("unary - on non-number")
            return -v
        return self.postfix()

    def postfix(self) -> Any:
        v = self.primary()
        while True:
            kind, t = self.peek()
            if t == ".":
                self.next()
                k2, name = self.next()
                if k2 != "ident":
                    raise CelError(f"expected member name, got {name!r}")
                if self.peek()[1] == "(":
                    v = self.call_method(v, name)
                else:
                    v = self.member(v, name)
            elif t == "[":
                self.next()
                idx = self.or_()
                self.expect("]")
                v = self.index(v, idx)
            else:
                return v

    def member(self, v: Any, name: str) -> Any:
        if _is_err(v):
            return v
        if isinstance(v, _AttrMap):
            try:
                return v.get(name)
            except CelError as e:
                return _CelErr(str(e))
        if isinstance(v, dict):
            if name not in v:
                return _CelErr(f"no field {name!r}")
            return v[name]
        return _CelErr(f"member access .{name} on {type(v).__name__}")

    def index(self, v: Any, idx: Any) -> Any:
        if _is_err(v):
            return v
        if _is_err(idx):
            return idx
        if isinstance(v, dict):
            if idx not in v:
                return _CelErr(f"no key {idx!r}")
            return v[idx]
        if isinstance(v, list):
            try:
                return v[int(idx)]
            except (IndexError, ValueError, TypeError) as e:
                return _CelErr(str(e))
        return _CelErr(f"indexing {type(v).__name__}")

    def call_method(self, v: Any, name: str) -> Any:
        self.expect("(")
        args = []
        if self.peek()[1] != ")":
            args.append(self.or_())
            while self.peek()[1] == ",":
                self.next()
                args.append(self.or_())
        self.expect(")")
        if _is_err(v):
            return v
        for a in args:
            if _is_err(a):
                return a
        if not isinstance(v, str):
            return _CelErr(f"method .{name}() on {type(v).__name__}")
        if name == "lowerAscii":
            return v.lower()
        if name == "upperAscii":
            return v.upper()
        if name == "size":
            return len(v)
        if name == "matches":
            return re.search(str(args[0]), v) is not None
        if name == "contains":
            return str(args[0]) in v
        if name == "startsWith":
            return v.startswith(str(args[0]))
        if name == "endsWith":
            return v.endswith(str(args[0]))
        return _CelErr(f"unknown method {name!r}")

    def primary(self) -> Any:
        kind, t = self.next()
        if kind == "int":
            return int(t)
        if kind == "float":
            return float(t)
        if kind == "string":
            return t
        if t == "(":
            v = self.or_()
            self.expect(")")
            return v
        if t == "[":
            items = []
            if self.peek()[1] != "]":
                items.append(self.or_())
                while self.peek()[1] == ",":
                    self.next()
                    items.append(self.or_())
            self.expect("]")
            return items
        if kind == "ident":
            if t == "true":
                return True
            if t == "false":
                return False
            if t == "quantity" and self.peek()[1] == "(":
                self.next()
                arg = self.or_()
                self.expect(")")
                if _is_err(arg):
                    return arg
                try:
                    return Quantity(parse_quantity_bytes(str(arg)))
                except Exception as e:
                    return _CelErr(f"quantity(): {e}")
            if t in self.env:
                return self.env[t]
            return _CelErr(f"unknown identifier {t!r}")
        raise CelError(f"unexpected token {t!r}")


def _typed_attr_value(v: dict) -> Any:
    for key in ("string", "version"):
        if key in v:
            return v[key]
    if "int" in v:
        return int(v["int"])
    if "bool" in v:
        return bool(v["bool"])
    raise CelError(f"unrecognized attribute value {v!r}")


def device_env(device: dict, driver: str) -> Dict[str, Any]:
    """Build the CEL ``device`` variable from a v1beta1 Device dict."""
    basic = device.get("basic", device)
    attrs_by_domain: Dict[str, Dict[str, Any]] = {}
    for qname, val in (basic.get("attributes") or {}).items():
        domain, _, name = qname.rpartition("/")
        attrs_by_domain.setdefault(domain, {})[name] = _typed_attr_value(val)
    caps_by_domain: Dict[str, Dict[str, Any]] = {}
    for qname, val in (basic.get("capacity") or {}).items():
        domain, _, name = qname.rpartition("/")
        caps_by_domain.setdefault(domain, {})[name] = Quantity(
            parse_quantity_bytes(val.get("value", "0"))
        )
    return {
        "device": {
            "driver": driver,
            "attributes": {
                d: _AttrMap(f, d, "attributes")
                for d, f in attrs_by_domain.items()
            },
            "capacity": {
                d: _AttrMap(f, d, "capacity") for d, f in caps_by_domain.items()
            },
        }
    }


def evaluate(expression: str, device: dict, driver: str) -> bool:
    """True iff the device matches. CelError propagates to the caller,
    which must treat it as non-match (DRA semantics)."""
    tokens = _tokenize(expression)
    result = _Parser(tokens, device_env(device, driver)).parse()
    if not isinstance(result, bool):
        raise CelError(
            f"selector must evaluate to bool, got {type(result).__name__}"
        )
    return result


def matches(expression: str, device: dict, driver: str) -> bool:
    """evaluate() with DRA error semantics: errors -> False."""
    try:
        return evaluate(expression, device, driver)
    except CelError:
        return False
