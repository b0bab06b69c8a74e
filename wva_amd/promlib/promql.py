"""PromQL-subset evaluator.

Grammar (the shapes the controller and integrations emit —
wva_amd/controller/collector.py and the KEDA/HPA queries):

    expr     := term ( '/' term )?
    term     := AGG '(' inner ')' | inner
    inner    := 'rate' '(' selector '[' DURATION ']' ')' | selector
    selector := NAME ( '{' NAME '=' '"' VALUE '"' ( ',' ... )* '}' )?
    AGG      := sum | avg | min | max | count

Semantics follow Prometheus: instant selectors return the latest sample
per series within a 5-minute lookback (with its original timestamp);
rate() needs >=2 samples in the window and divides the reset-corrected
sum of sample deltas by the time delta (a mid-window counter reset
contributes the post-reset value, Prometheus-style); aggregations collapse matching
series; '/' divides two single-sample vectors and returns empty when
either side is empty.
"""

from __future__ import annotations

import math
import re
import time
from typing import Dict, List, Optional, Tuple

from ..controller.promclient import Sample
from .store import TimeSeriesStore

LOOKBACK_SECONDS = 5 * 60

AGGREGATIONS = {"sum", "avg", "min", "max", "count"}

_NAME = r"[a-zA-Z_:][a-zA-Z0-9_:]*"
_TOKEN_RE = re.compile(
    rf"""\s*(?:
        (?P<name>{_NAME})
      | (?P<lbrace>\{{) | (?P<rbrace>\}})
      | (?P<lparen>\() | (?P<rparen>\))
      | (?P<lbrack>\[) | (?P<rbrack>\])
      | (?P<string>"(?:[^"\\]|\\.)*")
      | (?P<duration>\d+[smhd])
      | (?P<eq>=) | (?P<comma>,) | (?P<slash>/)
    )""",
    re.VERBOSE,
)


class PromQLError(ValueError):
    pass


def _tokenize(q: str) -> List[Tuple[str, str]]:
    out = []
    pos = 0
    while pos < len(q):
        m = _TOKEN_RE.match(q, pos)
        if m is None or m.end() == pos:
            rest = q[pos:].strip()
            if not rest:
                break
            raise PromQLError(f"cannot tokenize {q!r} at {pos}")
        pos = m.end()
        for kind, val in m.groupdict().items():
            if val is not None:
                out.append((kind, val))
                break
    return out


_DUR_UNITS = {"s": 1.0, "m": 60.0, "h": 3600.0, "d": 86400.0}


class _Parser:
    def __init__(self, tokens: List[Tuple[str, str]]) -> None:
        self.tokens = tokens
        self.i = 0

    def peek(self) -> Optional[Tuple[str, str]]:
        return self.tokens[self.i] if self.i < len(self.tokens) else None

    def next(self, kind: Optional[str] = None) -> Tuple[str, str]:
        tok = self.peek()
        if tok is None:
            raise PromQLError("unexpected end of query")
        if kind is not None and tok[0] != kind:
            raise PromQLError(f"expected {kind}, got {tok}")
        self.i += 1
        return tok

    # expr := term ('/' term)?
    def parse_expr(self):
        left = self.parse_term()
        if self.peek() and self.peek()[0] == "slash":
            self.next("slash")
            right = self.parse_term()
            return ("div", left, right)
        return left

    def parse_term(self):
        tok = self.peek()
        if tok is None:
            raise PromQLError("empty query")
        if tok[0] == "name" and tok[1] in AGGREGATIONS:
            agg = self.next("name")[1]
            self.next("lparen")
            inner = self.parse_inner()
            self.next("rparen")
            return ("agg", agg, inner)
        return self.parse_inner()

    def parse_inner(self):
        tok = self.peek()
        if tok is None:
            raise PromQLError("empty inner expression")
        if tok[0] == "name" and tok[1] == "rate":
            self.next("name")
            self.next("lparen")
            sel = self.parse_selector()
            self.next("lbrack")
            dur = self.next("duration")[1]
            self.next("rbrack")
            self.next("rparen")
            window = float(dur[:-1]) * _DUR_UNITS[dur[-1]]
            return ("rate", sel, window)
        return self.parse_selector()

    def parse_selector(self):
        name = self.next("name")[1]
        matchers: Dict[str, str] = {}
        if self.peek() and self.peek()[0] == "lbrace":
            self.next("lbrace")
            while True:
                if self.peek() and self.peek()[0] == "rbrace":
                    break
                label = self.next("name")[1]
                self.next("eq")
                raw = self.next("string")[1]
                matchers[label] = raw[1:-1].replace('\\"', '"').replace("\\\\", "\\")
                if self.peek() and self.peek()[0] == "comma":
                    self.next("comma")
                    continue
                break
            self.next("rbrace")
        return ("selector", name, matchers)


def _eval_node(node, store: TimeSeriesStore, now: float) -> List[Sample]:
    kind = node[0]
    if kind == "selector":
        _, name, matchers = node
        out = []
        for series in store.select(name, matchers):
            hit = series.latest_in(now - LOOKBACK_SECONDS, now)
            if hit is not None:
                out.append(Sample(value=hit[1], timestamp=hit[0], labels=dict(series.labels)))
        return out
    if kind == "rate":
        _, sel, window = node
        _, name, matchers = sel
        out = []
        for series in store.select(name, matchers):
            pts = series.range(now - window, now)
            if len(pts) < 2:
                continue
            t0, t1 = pts[0][0], pts[-1][0]
            if t1 <= t0:
                continue
            # per-adjacent-pair counter-reset correction, as Prometheus's
            # extrapolatedRate does: a mid-window reset contributes the
            # post-reset value instead of poisoning the endpoint delta
            # (matters when an emulator/vLLM pod restarts inside the
            # rate window during fleet resizes)
            delta = 0.0
            prev = pts[0][1]
            for _, v in pts[1:]:
                delta += (v - prev) if v >= prev else v
                prev = v
            out.append(Sample(value=delta / (t1 - t0), timestamp=now, labels=dict(series.labels)))
        return out
    if kind == "agg":
        _, agg, inner = node
        vec = _eval_node(inner, store, now)
        if not vec:
            return []
        values = [s.value for s in vec]
        if agg == "sum":
            v = sum(values)
        elif agg == "avg":
            v = sum(values) / len(values)
        elif agg == "min":
            v = min(values)
        elif agg == "max":
            v = max(values)
        else:  # count
            v = float(len(values))
        return [Sample(value=v, timestamp=now, labels={})]
    if kind == "div":
        _, l, r = node
        lv = _eval_node(l, store, now)
        rv = _eval_node(r, store, now)
        if not lv or not rv:
            return []
        denom = rv[0].value
        value = lv[0].value / denom if denom != 0 else math.nan
        return [Sample(value=value, timestamp=now, labels={})]
    raise PromQLError(f"unknown node {node!r}")


def evaluate(query: str, store: TimeSeriesStore, now: Optional[float] = None) -> List[Sample]:
    now = time.time() if now is None else now
    parser = _Parser(_tokenize(query))
    node = parser.parse_expr()
    if parser.peek() is not None:
        raise PromQLError(f"trailing tokens in {query!r}")
    return _eval_node(node, store, now)
