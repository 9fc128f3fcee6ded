"""scraper.py — Prometheus text-format scrape client + parser.

The harness's stand-in for Prometheus's scrape loop (reference scrape
config: kube-prometheus-stack-values.yaml:3-16, 1 s interval with the
``node`` relabel). Scrapes exporter /metrics endpoints over HTTP, parses the
text exposition format into Samples, and applies the node relabel the
reference does in Prometheus config (``__meta_kubernetes_pod_node_name`` ->
``node``; here: a static node label per target, same effect).
"""

from __future__ import annotations

import re
import time
import urllib.request
from dataclasses import dataclass
from typing import Dict, List, Optional

from .promql import Sample

_LINE_RE = re.compile(
    r"^(?P<name>[a-zA-Z_:][a-zA-Z0-9_:]*)"
    r"(?:\{(?P<labels>[^}]*)\})?\s+"
    r"(?P<value>[+-]?(?:\d+\.?\d*(?:[eE][+-]?\d+)?|Inf|NaN))"
    r"(?:\s+(?P<ts>-?\d+))?$"
)
_LABEL_RE = re.compile(r'([a-zA-Z_][a-zA-Z0-9_]*)="((?:[^"\\]|\\.)*)"')


def parse_prometheus_text(text: str) -> List[Sample]:
    out: List[Sample] = []
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        m = _LINE_RE.match(line)
        if not m:
            continue
        labels = {}
        if m.group("labels"):
            for lm in _LABEL_RE.finditer(m.group("labels")):
                labels[lm.group(1)] = lm.group(2).replace('\\"', '"').replace(
                    "\\\\", "\\"
                )
        v = m.group("value")
        value = float("inf") if v == "Inf" else float("nan") if v == "NaN" else float(v)
        out.append(Sample(m.group("name"), labels, value))
    return out


@dataclass
class ScrapeTarget:
    url: str                      # http://host:port/metrics
    node: str = "node0"           # the relabel's node label
    extra_labels: Optional[Dict[str, str]] = None


class Scraper:
    """Scrapes a set of targets; keeps the latest sample set per target."""

    def __init__(self, targets: List[ScrapeTarget], timeout_s: float = 2.0):
        self.targets = targets
        self.timeout_s = timeout_s
        self.last: Dict[str, List[Sample]] = {}
        self.last_scrape_duration_s: float = 0.0

    def scrape_once(self) -> List[Sample]:
        t0 = time.monotonic()
        merged: List[Sample] = []
        for t in self.targets:
            try:
                with urllib.request.urlopen(t.url, timeout=self.timeout_s) as r:
                    text = r.read().decode()
            except Exception:
                # target down: keep serving its last samples (staleness is
                # handled upstream; Prometheus marks them stale after 5 min)
                merged.extend(self.last.get(t.url, []))
                continue
            samples = parse_prometheus_text(text)
            for s in samples:
                s.labels.setdefault("node", t.node)
                if t.extra_labels:
                    for k, v in t.extra_labels.items():
                        s.labels.setdefault(k, v)
            self.last[t.url] = samples
            merged.extend(samples)
        self.last_scrape_duration_s = time.monotonic() - t0
        return merged
