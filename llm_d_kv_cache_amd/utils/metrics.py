"""Prometheus metrics for the index + events plane.

Capability parity with the reference pkg/kvcache/metrics/collector.go:
kvcache_index_{admissions,evictions,lookup_requests,lookup_hits}_total and
the periodic "metrics beat" log line. Implemented as a custom collector
that reads the native stats counters on scrape — the C++ hot paths carry
plain atomics and never touch Python.
"""
from __future__ import annotations

import logging
import threading
from typing import Optional

log = logging.getLogger(__name__)

try:
    from prometheus_client import REGISTRY
    from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily

    HAVE_PROMETHEUS = True
except ImportError:  # pragma: no cover
    HAVE_PROMETHEUS = False


class KVCacheMetricsCollector:
    """Scrape-time collector over native index/pool/engine stats."""

    def __init__(self, indexer=None, events_pool=None, offload_engine=None,
                 peer_service=None):
        self.indexer = indexer
        self.events_pool = events_pool
        self.offload_engine = offload_engine
        self.peer_service = peer_service

    def collect(self):
        if self.indexer is not None:
            s = self.indexer.index.stats()
            yield CounterMetricFamily(
                "kvcache_index_admissions_total",
                "pod entries admitted to the index", value=s.admissions)
            yield CounterMetricFamily(
                "kvcache_index_evictions_total",
                "pod entries evicted from the index", value=s.evictions)
            yield CounterMetricFamily(
                "kvcache_index_lookup_requests_total",
                "index lookups", value=s.lookups)
            yield CounterMetricFamily(
                "kvcache_index_lookup_hits_total",
                "index lookups with at least one hit", value=s.hits)
            yield GaugeMetricFamily(
                "kvcache_index_keys", "request keys resident", value=s.keys)
            yield CounterMetricFamily(
                "kvcache_index_admission_rejects_total",
                "adds rejected by the byte-budget admission sketch",
                value=s.rejections)
        if self.events_pool is not None:
            p = self.events_pool.stats()
            yield CounterMetricFamily(
                "kvcache_events_enqueued_total", "raw messages enqueued",
                value=p.enqueued)
            yield CounterMetricFamily(
                "kvcache_events_processed_total", "messages processed",
                value=p.processed)
            yield CounterMetricFamily(
                "kvcache_events_parse_failures_total", "undecodable payloads",
                value=p.parse_failures)
            yield CounterMetricFamily(
                "kvcache_events_parent_misses_total",
                "BlockStored dropped on unknown parent chain",
                value=p.dropped_parent_misses)
            yield CounterMetricFamily(
                "kvcache_events_handler_failures_total",
                "events dropped on backend errors (outages, bad fields)",
                value=p.handler_failures)
        if self.offload_engine is not None:
            e = self.offload_engine.stats()
            yield CounterMetricFamily(
                "kv_offload_files_written_total", "KV files written",
                value=e.files_written)
            yield CounterMetricFamily(
                "kv_offload_files_read_total", "KV files read",
                value=e.files_read)
            yield CounterMetricFamily(
                "kv_offload_files_deduped_total", "stores skipped (exists)",
                value=e.files_deduped)
            yield CounterMetricFamily(
                "kv_offload_writes_dropped_total",
                "stores dropped by the queue limit", value=e.writes_dropped)
            yield CounterMetricFamily(
                "kv_offload_bytes_stored_total", "bytes offloaded",
                value=e.bytes_stored)
            yield CounterMetricFamily(
                "kv_offload_bytes_loaded_total", "bytes loaded back",
                value=e.bytes_loaded)
            yield CounterMetricFamily(
                "kv_offload_d2h_lane_busy_seconds_total",
                "PCIe D2H lane busy wall", value=e.d2h_lane_busy_ms / 1e3)
            yield CounterMetricFamily(
                "kv_offload_h2d_lane_busy_seconds_total",
                "PCIe H2D lane busy wall", value=e.h2d_lane_busy_ms / 1e3)
            yield CounterMetricFamily(
                "kv_offload_d2h_lane_bytes_total",
                "bytes through the D2H lane", value=e.d2h_lane_bytes)
            yield CounterMetricFamily(
                "kv_offload_h2d_lane_bytes_total",
                "bytes through the H2D lane", value=e.h2d_lane_bytes)
        if self.peer_service is not None:
            ps = self.peer_service.stats()
            yield CounterMetricFamily(
                "kv_peer_pulls_requested_total", "peer chunk pulls requested",
                value=ps.pulls_requested)
            yield CounterMetricFamily(
                "kv_peer_pulls_served_total", "peer chunk pulls served",
                value=ps.pulls_served)
            yield CounterMetricFamily(
                "kv_peer_pulls_served_dram_total",
                "peer pulls served from the host-DRAM cache",
                value=ps.pulls_served_dram)
            yield CounterMetricFamily(
                "kv_peer_pulls_failed_total", "peer pulls failed/denied",
                value=ps.pulls_failed)
            yield CounterMetricFamily(
                "kv_peer_bytes_sent_total", "bytes served to peers",
                value=ps.bytes_sent)
            yield CounterMetricFamily(
                "kv_peer_bytes_received_total", "bytes pulled from peers",
                value=ps.bytes_received)


def register(indexer=None, events_pool=None, offload_engine=None,
             peer_service=None, registry=None):
    if not HAVE_PROMETHEUS:  # pragma: no cover
        raise RuntimeError("prometheus_client is not installed")
    collector = KVCacheMetricsCollector(indexer, events_pool,
                                        offload_engine, peer_service)
    (registry or REGISTRY).register(collector)
    return collector


def start_metrics_logging(indexer, interval_s: float = 60.0,
                          stop_event: Optional[threading.Event] = None):
    """Periodic 'metrics beat' log line (reference collector.go:97-165)."""
    stop_event = stop_event or threading.Event()

    def beat():
        while not stop_event.wait(interval_s):
            s = indexer.index.stats()
            hit_pct = 100.0 * s.hits / s.lookups if s.lookups else 0.0
            log.info(
                "kvcache metrics beat: keys=%d admissions=%d evictions=%d "
                "lookups=%d hit%%=%.1f", s.keys, s.admissions, s.evictions,
                s.lookups, hit_pct,
            )

    t = threading.Thread(target=beat, daemon=True, name="kvcache-metrics-beat")
    t.start()
    return stop_event


class MeteredIndexer:
    """Decorator over KVCacheIndexer exporting the reference's scoring
    metrics (collector.go:28-75): `kvcache_lookup_latency_seconds`
    histogram and `kvcache_max_pod_hit_count` counter of blocks on the
    best pod per request. Composes with TracedIndexer (wrap either way)."""

    def __init__(self, indexer, registry=None):
        if not HAVE_PROMETHEUS:  # pragma: no cover
            raise RuntimeError("prometheus_client is not installed")
        import time as _time

        from prometheus_client import Counter, Histogram

        self._time = _time
        self._ix = indexer
        kw = {"registry": registry} if registry is not None else {}
        self._lat = Histogram(
            "kvcache_lookup_latency_seconds",
            "score_tokens end-to-end latency",
            buckets=(1e-5, 3e-5, 1e-4, 3e-4, 1e-3, 3e-3, 1e-2, 0.1), **kw)
        self._max_hits = Counter(
            "kvcache_max_pod_hit_count",
            "sum over requests of the best pod's matched block count", **kw)

    def __getattr__(self, name):
        return getattr(self._ix, name)

    def score_tokens(self, tokens, model_name, pod_identifiers=(),
                     extra_features=None):
        t0 = self._time.perf_counter()
        scores = self._ix.score_tokens(tokens, model_name, pod_identifiers,
                                       extra_features)
        self._lat.observe(self._time.perf_counter() - t0)
        if scores:
            self._max_hits.inc(int(max(scores.values())))
        return scores
