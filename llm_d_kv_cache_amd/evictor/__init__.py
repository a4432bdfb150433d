"""Storage-tier disk-space manager ("PVC evictor").

Capability parity with the reference kv_connectors/pvc_evictor
(ARCHITECTURE.md N+2 design): N crawler processes partitioned by the
hash-prefix shard directories walk the content-addressed layout and queue
cold files (atime threshold), one activator toggles deletion on a
filesystem-utilization hysteresis (default 85% on / 70% off), and one
deleter batch-unlinks queued files, optionally publishing BlockRemoved
storage events so the global indexer drops the entries; a folder cleaner
prunes directories the deleter emptied. The coordinator restarts failed
children.
"""
from __future__ import annotations

import logging
import multiprocessing as mp
import os
import time
from dataclasses import dataclass
from typing import Callable, Optional

log = logging.getLogger(__name__)


@dataclass
class EvictorConfig:
    root: str = "/mnt/kvcache"
    crawlers: int = 2
    atime_threshold_s: float = 3600.0      # colder than this is evictable
    crawl_interval_s: float = 30.0
    activate_utilization: float = 0.85     # start deleting above this
    deactivate_utilization: float = 0.70   # stop below this
    check_interval_s: float = 5.0
    delete_batch: int = 64
    clean_interval_s: float = 300.0        # empty-dir pruning cadence
    # ZMQ endpoint to publish BlockRemoved storage events (optional)
    events_endpoint: Optional[str] = None
    events_model: str = ""

    @staticmethod
    def from_env() -> "EvictorConfig":
        e = os.environ
        return EvictorConfig(
            root=e.get("KVC_EVICTOR_ROOT", "/mnt/kvcache"),
            crawlers=int(e.get("KVC_EVICTOR_CRAWLERS", "2")),
            atime_threshold_s=float(e.get("KVC_EVICTOR_ATIME_S", "3600")),
            crawl_interval_s=float(e.get("KVC_EVICTOR_CRAWL_INTERVAL_S", "30")),
            activate_utilization=float(e.get("KVC_EVICTOR_ACTIVATE", "0.85")),
            deactivate_utilization=float(e.get("KVC_EVICTOR_DEACTIVATE", "0.70")),
            check_interval_s=float(e.get("KVC_EVICTOR_CHECK_S", "5")),
            events_endpoint=e.get("KVC_EVICTOR_EVENTS_ENDPOINT") or None,
            events_model=e.get("KVC_EVICTOR_EVENTS_MODEL", ""),
        )


def default_utilization(root: str) -> float:
    st = os.statvfs(root)
    if st.f_blocks == 0:
        return 0.0
    return 1.0 - st.f_bavail / st.f_blocks


# ---- child processes --------------------------------------------------------

def crawler_proc(cfg: EvictorConfig, shard: int, candidates: mp.Queue,
                 active: "mp.Event", stop: "mp.Event") -> None:
    """Walk this crawler's shard of the layout; queue cold files while
    deletion is active. Shards partition on the first hash-hex directory
    (<run>/<h[0:3]>) modulo crawler count."""
    while not stop.is_set():
        if active.is_set():
            now = time.time()
            for run_dir in _list_dirs(cfg.root):
                for shard_dir in _list_dirs(run_dir):
                    base = os.path.basename(shard_dir)
                    try:
                        shard_id = int(base, 16)
                    except ValueError:
                        continue
                    if shard_id % cfg.crawlers != shard:
                        continue
                    for sub in _list_dirs(shard_dir):
                        for name in _safe_listdir(sub):
                            if not name.endswith(".bin"):
                                continue
                            path = os.path.join(sub, name)
                            try:
                                st = os.stat(path)
                            except OSError:
                                continue
                            if now - st.st_atime > cfg.atime_threshold_s:
                                candidates.put(path)
                    if stop.is_set() or not active.is_set():
                        break
        stop.wait(cfg.crawl_interval_s)


def activator_proc(cfg: EvictorConfig, active: "mp.Event", stop: "mp.Event",
                   utilization: Callable[[str], float] = default_utilization
                   ) -> None:
    """Hysteresis switch over filesystem utilization."""
    while not stop.is_set():
        try:
            u = utilization(cfg.root)
        except OSError:
            u = 0.0
        if u >= cfg.activate_utilization and not active.is_set():
            log.warning("evictor: utilization %.1f%% >= %.1f%%, deleting ON",
                        u * 100, cfg.activate_utilization * 100)
            active.set()
        elif u <= cfg.deactivate_utilization and active.is_set():
            log.info("evictor: utilization %.1f%% <= %.1f%%, deleting OFF",
                     u * 100, cfg.deactivate_utilization * 100)
            active.clear()
        stop.wait(cfg.check_interval_s)


def deleter_proc(cfg: EvictorConfig, candidates: mp.Queue, active: "mp.Event",
                 stop: "mp.Event", deleted_counter) -> None:
    """Batch-unlink queued candidates; publish BlockRemoved when wired."""
    publisher = None
    if cfg.events_endpoint:
        from ..offload.events import StorageEventPublisher

        publisher = StorageEventPublisher(cfg.events_endpoint, cfg.events_model,
                                          bind=False)
    while not stop.is_set():
        batch = []
        try:
            batch.append(candidates.get(timeout=0.25))
            while len(batch) < cfg.delete_batch:
                batch.append(candidates.get_nowait())
        except Exception:
            pass
        if not batch:
            continue
        if not active.is_set():
            continue  # deletion switched off: drop stale candidates
        removed_hashes = []
        for path in batch:
            try:
                os.unlink(path)
                with deleted_counter.get_lock():
                    deleted_counter.value += 1
                name = os.path.basename(path)
                if name.endswith(".bin"):
                    try:
                        removed_hashes.append(int(name[:-4], 16))
                    except ValueError:
                        pass
            except OSError:
                pass
        if publisher is not None and removed_hashes:
            publisher.publish_block_removed(removed_hashes)


def clean_empty_dirs(root: str) -> int:
    """Remove empty shard/sub directories left behind by deletions
    (reference folder_cleaner process). Run directories themselves stay:
    they hold the layout manifest (config.json). Returns dirs removed."""
    removed = 0
    for run_dir in _list_dirs(root):
        for shard_dir in _list_dirs(run_dir):
            for sub in _list_dirs(shard_dir):
                try:
                    os.rmdir(sub)  # fails (correctly) unless empty
                    removed += 1
                except OSError:
                    pass
            try:
                os.rmdir(shard_dir)
                removed += 1
            except OSError:
                pass
    return removed


def folder_cleaner_proc(cfg: EvictorConfig, stop: "mp.Event") -> None:
    while not stop.is_set():
        stop.wait(cfg.clean_interval_s)
        if stop.is_set():
            return
        n = clean_empty_dirs(cfg.root)
        if n:
            log.info("evictor: pruned %d empty directories", n)


def _list_dirs(path):
    try:
        return [os.path.join(path, d) for d in sorted(os.listdir(path))
                if os.path.isdir(os.path.join(path, d))]
    except OSError:
        return []


def _safe_listdir(path):
    try:
        return os.listdir(path)
    except OSError:
        return []


# ---- coordinator ------------------------------------------------------------

class PvcEvictor:
    """N crawlers + activator + deleter, restarted on failure."""

    def __init__(self, cfg: EvictorConfig,
                 utilization: Callable[[str], float] = default_utilization):
        self.cfg = cfg
        self._utilization = utilization
        ctx = mp.get_context("spawn")
        self._ctx = ctx
        self.candidates = ctx.Queue()
        self.active = ctx.Event()
        self.stop_event = ctx.Event()
        self.deleted = ctx.Value("q", 0)
        self._procs = {}

    def _spec(self):
        spec = {}
        for i in range(self.cfg.crawlers):
            spec[f"crawler-{i}"] = (crawler_proc,
                                    (self.cfg, i, self.candidates, self.active,
                                     self.stop_event))
        spec["activator"] = (activator_proc,
                             (self.cfg, self.active, self.stop_event,
                              self._utilization))
        spec["deleter"] = (deleter_proc,
                           (self.cfg, self.candidates, self.active,
                            self.stop_event, self.deleted))
        spec["folder-cleaner"] = (folder_cleaner_proc,
                                  (self.cfg, self.stop_event))
        return spec

    def start(self) -> None:
        for name, (fn, args) in self._spec().items():
            p = self._ctx.Process(target=fn, args=args, name=name, daemon=True)
            p.start()
            self._procs[name] = p

    def supervise_once(self) -> None:
        """Restart any dead child (reference: main restarts failed
        children, ARCHITECTURE.md:55-60)."""
        spec = self._spec()
        for name, p in list(self._procs.items()):
            if not p.is_alive() and not self.stop_event.is_set():
                log.warning("evictor child %s died (exit %s); restarting",
                            name, p.exitcode)
                fn, args = spec[name]
                np_ = self._ctx.Process(target=fn, args=args, name=name,
                                        daemon=True)
                np_.start()
                self._procs[name] = np_

    def run_forever(self, supervise_interval_s: float = 5.0):  # pragma: no cover
        self.start()
        while not self.stop_event.is_set():
            time.sleep(supervise_interval_s)
            self.supervise_once()

    def shutdown(self, timeout: float = 10.0) -> None:
        self.stop_event.set()
        for p in self._procs.values():
            p.join(timeout=timeout)
            if p.is_alive():
                p.terminate()
        self._procs.clear()


def main():  # pragma: no cover - operational entrypoint
    logging.basicConfig(level=logging.INFO)
    PvcEvictor(EvictorConfig.from_env()).run_forever()


if __name__ == "__main__":  # pragma: no cover
    main()
