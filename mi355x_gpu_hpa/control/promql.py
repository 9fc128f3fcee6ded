"""promql.py — a small PromQL instant-query evaluator.

Covers the PromQL subset the stack's recording rules use (reference rule:
``cuda-test-prometheusrule.yaml:13``)::

    avg(
      max by(node, pod, namespace) (dcgm_gpu_utilization)
      * on(pod) group_left(label_app)
      max by(pod, label_app) (kube_pod_labels{label_app="cuda-test"})
    )

Supported grammar:
  * instant vector selectors with label matchers (=, !=, =~, !~)
  * aggregations: sum / avg / max / min / count, with optional ``by (...)``
    (prefix or suffix position)
  * binary operators: ``*``, ``+``, ``-``, ``/`` with * / binding
    tighter than + - (Prometheus precedence), and optional
    ``on(...)``/``ignoring(...)`` and ``group_left(...)`` modifiers
  * scalar literals (e.g. ``expr * 100``)
  * parentheses

The evaluator follows Prometheus vector-matching semantics: one-to-one by
default (error on duplicate match keys), many-to-one with ``group_left``
(extra labels copied from the "one" side onto the result).

This is used by the in-process control-loop harness (loop.py) and by the
rule unit tests; in a real cluster the recording rule is evaluated by stock
Prometheus (SURVEY.md C12 "reuse as-is") — parity between the two is what
the kind-cluster integration harness checks.
"""

from __future__ import annotations

import math
import re
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

Labels = Tuple[Tuple[str, str], ...]  # sorted, excludes __name__


def mklabels(d: Dict[str, str]) -> Labels:
    return tuple(sorted((k, v) for k, v in d.items() if k != "__name__"))


@dataclass
class Sample:
    name: str
    labels: Dict[str, str]
    value: float

    def label_key(self, names: Optional[List[str]] = None) -> Labels:
        if names is None:
            return mklabels(self.labels)
        return tuple((n, self.labels.get(n, "")) for n in names)


Vector = List[Sample]


class PromQLError(ValueError):
    pass


# --- tokenizer -------------------------------------------------------------

_TOKEN_RE = re.compile(
    r"""\s*(?:
        (?P<num>\d+(?:\.\d+)?(?:[eE][+-]?\d+)?)
      | (?P<id>[a-zA-Z_:][a-zA-Z0-9_:]*)
      | (?P<str>"(?:[^"\\]|\\.)*"|'(?:[^'\\]|\\.)*')
      | (?P<op>=~|!~|!=|=|\{|\}|\(|\)|,|\*|/|\+|-)
    )""",
    re.X,
)


def _tokenize(s: str) -> List[Tuple[str, str]]:
    out = []
    pos = 0
    while pos < len(s):
        m = _TOKEN_RE.match(s, pos)
        if not m or m.end() == pos:
            rest = s[pos:].strip()
            if not rest:
                break
            raise PromQLError(f"cannot tokenize at: {rest[:30]!r}")
        pos = m.end()
        for kind in ("num", "id", "str", "op"):
            if m.group(kind) is not None:
                out.append((kind, m.group(kind)))
                break
    return out


# --- AST -------------------------------------------------------------------

AGG_FUNCS: Dict[str, Callable[[List[float]], float]] = {
    "sum": sum,
    "avg": lambda v: sum(v) / len(v),
    "max": max,
    "min": min,
    "count": len,
}


@dataclass
class Selector:
    name: str
    matchers: List[Tuple[str, str, str]] = field(default_factory=list)


@dataclass
class Agg:
    func: str
    by: Optional[List[str]]
    expr: "Expr"


@dataclass
class BinOp:
    op: str
    on: Optional[List[str]]
    ignoring: Optional[List[str]]
    group_left: Optional[List[str]]
    lhs: "Expr"
    rhs: "Expr"


@dataclass
class Scalar:
    value: float


Expr = object  # Selector | Agg | BinOp | Scalar


class _Parser:
    def __init__(self, tokens: List[Tuple[str, str]]):
        self.toks = tokens
        self.i = 0

    def peek(self) -> Optional[Tuple[str, str]]:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> Tuple[str, str]:
        t = self.peek()
        if t is None:
            raise PromQLError("unexpected end of expression")
        self.i += 1
        return t

    def expect(self, val: str):
        t = self.next()
        if t[1] != val:
            raise PromQLError(f"expected {val!r}, got {t[1]!r}")

    def parse(self) -> Expr:
        e = self.parse_binary()
        if self.peek() is not None:
            raise PromQLError(f"trailing input at {self.peek()[1]!r}")
        return e

    # Precedence follows Prometheus: * / bind tighter than + - (both
    # levels left-associative), so e.g. `a + b * c` is `a + (b * c)`.
    def parse_binary(self) -> Expr:
        return self.parse_additive()

    def parse_additive(self) -> Expr:
        lhs = self.parse_multiplicative()
        while True:
            t = self.peek()
            if t is None or t[1] not in ("+", "-"):
                return lhs
            op = self.next()[1]
            on, ignoring, group_left = self.parse_match_modifiers()
            rhs = self.parse_multiplicative()
            lhs = BinOp(op, on, ignoring, group_left, lhs, rhs)

    def parse_multiplicative(self) -> Expr:
        lhs = self.parse_primary()
        while True:
            t = self.peek()
            if t is None or t[1] not in ("*", "/"):
                return lhs
            op = self.next()[1]
            on, ignoring, group_left = self.parse_match_modifiers()
            rhs = self.parse_primary()
            lhs = BinOp(op, on, ignoring, group_left, lhs, rhs)

    def parse_match_modifiers(self):
        on = ignoring = group_left = None
        t = self.peek()
        if t and t[0] == "id" and t[1] in ("on", "ignoring"):
            kind = self.next()[1]
            names = self.parse_name_list()
            if kind == "on":
                on = names
            else:
                ignoring = names
        t = self.peek()
        if t and t[0] == "id" and t[1] in ("group_left", "group_right"):
            kind = self.next()[1]
            if kind == "group_right":
                raise PromQLError("group_right not supported")
            group_left = []
            if self.peek() and self.peek()[1] == "(":
                group_left = self.parse_name_list()
        return on, ignoring, group_left

    def parse_name_list(self) -> List[str]:
        self.expect("(")
        names = []
        while True:
            t = self.next()
            if t[1] == ")":
                break
            if t[0] != "id":
                raise PromQLError(f"expected label name, got {t[1]!r}")
            names.append(t[1])
            t = self.peek()
            if t and t[1] == ",":
                self.next()
        return names

    def parse_primary(self) -> Expr:
        t = self.peek()
        if t is None:
            raise PromQLError("unexpected end of expression")
        if t[1] == "(":
            self.next()
            e = self.parse_binary()
            self.expect(")")
            return e
        if t[0] == "num":
            self.next()
            return Scalar(float(t[1]))
        if t[0] == "id" and t[1] in AGG_FUNCS:
            # could still be a metric named e.g. "sum" — disambiguate by
            # lookahead for '(' or 'by'
            nxt = self.toks[self.i + 1] if self.i + 1 < len(self.toks) else None
            if nxt and (nxt[1] == "(" or nxt[1] == "by"):
                func = self.next()[1]
                by = None
                if self.peek() and self.peek()[1] == "by":
                    self.next()
                    by = self.parse_name_list()
                self.expect("(")
                inner = self.parse_binary()
                self.expect(")")
                if by is None and self.peek() and self.peek()[1] == "by":
                    self.next()
                    by = self.parse_name_list()
                return Agg(func, by, inner)
        if t[0] == "id":
            self.next()
            sel = Selector(t[1])
            if self.peek() and self.peek()[1] == "{":
                self.next()
                while True:
                    t2 = self.next()
                    if t2[1] == "}":
                        break
                    if t2[0] != "id":
                        raise PromQLError(f"expected label name, got {t2[1]!r}")
                    op = self.next()[1]
                    if op not in ("=", "!=", "=~", "!~"):
                        raise PromQLError(f"bad matcher op {op!r}")
                    v = self.next()
                    if v[0] != "str":
                        raise PromQLError("matcher value must be a string")
                    sel.matchers.append((t2[1], op, v[1][1:-1]))
                    if self.peek() and self.peek()[1] == ",":
                        self.next()
            return sel
        raise PromQLError(f"unexpected token {t[1]!r}")


def parse(expr: str) -> Expr:
    return _Parser(_tokenize(expr)).parse()


# --- evaluation ------------------------------------------------------------


def _match(s: Sample, matchers) -> bool:
    for name, op, val in matchers:
        have = s.name if name == "__name__" else s.labels.get(name, "")
        if op == "=" and have != val:
            return False
        if op == "!=" and have == val:
            return False
        if op == "=~" and not re.fullmatch(val, have):
            return False
        if op == "!~" and re.fullmatch(val, have):
            return False
    return True


def _eval(node: Expr, samples: Vector):
    if isinstance(node, Scalar):
        return node.value
    if isinstance(node, Selector):
        return [s for s in samples if s.name == node.name and _match(s, node.matchers)]
    if isinstance(node, Agg):
        vec = _eval(node.expr, samples)
        if isinstance(vec, float):
            raise PromQLError("aggregation over scalar")
        groups: Dict[Labels, List[float]] = {}
        gl: Dict[Labels, Dict[str, str]] = {}
        for s in vec:
            key = s.label_key(node.by) if node.by is not None else ()
            groups.setdefault(key, []).append(s.value)
            if key not in gl:
                gl[key] = (
                    {n: s.labels.get(n, "") for n in node.by}
                    if node.by is not None
                    else {}
                )
        f = AGG_FUNCS[node.func]
        return [Sample("", {k: v for k, v in gl[key].items() if v != ""}, float(f(vals)))
                for key, vals in groups.items()]
    if isinstance(node, BinOp):
        lhs = _eval(node.lhs, samples)
        rhs = _eval(node.rhs, samples)
        opf = {
            "*": lambda a, b: a * b,
            "/": lambda a, b: a / b if b != 0 else math.nan,
            "+": lambda a, b: a + b,
            "-": lambda a, b: a - b,
        }[node.op]
        if isinstance(lhs, float) and isinstance(rhs, float):
            return opf(lhs, rhs)
        if isinstance(rhs, float):
            return [Sample(s.name, s.labels, opf(s.value, rhs)) for s in lhs]
        if isinstance(lhs, float):
            return [Sample(s.name, s.labels, opf(lhs, s.value)) for s in rhs]
        # vector-vector matching: key = on-labels, or all labels minus the
        # ignoring-set, or (default) the full label set
        on = node.on
        ignoring = node.ignoring

        def match_key(s: Sample) -> Labels:
            if on is not None:
                return s.label_key(on)
            if ignoring is not None:
                return tuple(sorted(
                    (k, v) for k, v in s.labels.items()
                    if k != "__name__" and k not in ignoring))
            return mklabels(s.labels)

        right_index: Dict[Labels, Sample] = {}
        for s in rhs:
            key = match_key(s)
            if key in right_index:
                raise PromQLError(
                    f"many-to-many matching: duplicate right-side key {key}"
                )
            right_index[key] = s
        out: Vector = []
        seen_left: Dict[Labels, int] = {}
        for s in lhs:
            key = match_key(s)
            r = right_index.get(key)
            if r is None:
                continue
            if node.group_left is None:
                # one-to-one: left keys must be unique too
                seen_left[key] = seen_left.get(key, 0) + 1
                if seen_left[key] > 1:
                    raise PromQLError(
                        f"many-to-one matching without group_left on key {key}"
                    )
            labels = dict(s.labels)
            if node.group_left:
                for n in node.group_left:
                    if n in r.labels:
                        labels[n] = r.labels[n]
            out.append(Sample("", labels, opf(s.value, r.value)))
        return out
    raise PromQLError(f"unknown node {node!r}")


def evaluate(expr: str, samples: Vector) -> Vector:
    """Evaluate an instant query; returns a vector (scalars are wrapped)."""
    res = _eval(parse(expr), samples)
    if isinstance(res, float):
        return [Sample("", {}, res)]
    return res


def evaluate_scalar(expr: str, samples: Vector) -> Optional[float]:
    """Evaluate an expression expected to yield one value (or None if the
    vector is empty — e.g. no pods matched the join)."""
    vec = evaluate(expr, samples)
    if not vec:
        return None
    if len(vec) > 1:
        raise PromQLError(f"expected scalar result, got {len(vec)} series")
    return vec[0].value
