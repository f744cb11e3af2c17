"""Engine + request statistics.

Behavioural parity:
  * EngineStatsScraper — reference stats/engine_stats.py: background thread
    scrapes each engine's /metrics every interval and parses the vllm:*
    series (num_requests_running/waiting, gpu_cache_usage_perc,
    gpu_prefix_cache_hits/queries).
  * RequestStatsMonitor — reference stats/request_stats.py: per-engine
    sliding-window QPS / TTFT / latency via on_new_request /
    on_request_response / on_request_complete lifecycle hooks.
"""

from __future__ import annotations

import logging
import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Deque, Dict, Optional, Tuple

import requests

logger = logging.getLogger("router.stats")


# ---------------------------------------------------------------------------
# Engine stats
# ---------------------------------------------------------------------------
@dataclass
class EngineStats:
    num_running_requests: int = 0
    num_queuing_requests: int = 0
    gpu_cache_usage_perc: float = 0.0
    gpu_prefix_cache_hits_total: float = 0.0
    gpu_prefix_cache_queries_total: float = 0.0
    gpu_prefix_cache_hit_rate: float = 0.0

    @staticmethod
    def from_prometheus_text(text: str) -> "EngineStats":
        vals: Dict[str, float] = {}
        for line in text.splitlines():
            if not line or line.startswith("#"):
                continue
            try:
                name, value = line.rsplit(" ", 1)
            except ValueError:
                continue
            name = name.split("{", 1)[0].strip()
            if name.startswith("vllm:"):
                try:
                    vals[name[5:]] = vals.get(name[5:], 0.0) + float(value)
                except ValueError:
                    pass
        hits = vals.get("gpu_prefix_cache_hits_total", 0.0)
        queries = vals.get("gpu_prefix_cache_queries_total", 0.0)
        return EngineStats(
            num_running_requests=int(vals.get("num_requests_running", 0)),
            num_queuing_requests=int(vals.get("num_requests_waiting", 0)),
            gpu_cache_usage_perc=vals.get("gpu_cache_usage_perc", 0.0),
            gpu_prefix_cache_hits_total=hits,
            gpu_prefix_cache_queries_total=queries,
            gpu_prefix_cache_hit_rate=(
                vals.get("gpu_prefix_cache_hit_rate", hits / queries)
                if queries
                else 0.0
            ),
        )


class EngineStatsScraper:
    def __init__(self, interval: float = 10.0, start: bool = True) -> None:
        self.interval = interval
        self._stats: Dict[str, EngineStats] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        if start:
            self._thread = threading.Thread(
                target=self._worker, daemon=True
            )
            self._thread.start()

    def _scrape_one(self, url: str) -> Optional[EngineStats]:
        try:
            r = requests.get(url + "/metrics", timeout=5)
            if r.status_code == 200:
                return EngineStats.from_prometheus_text(r.text)
        except requests.RequestException:
            pass
        return None

    def scrape_once(self) -> None:
        from production_stack_amd.router.service_discovery import (
            get_service_discovery,
        )

        try:
            endpoints = get_service_discovery().get_endpoint_info()
        except RuntimeError:
            return
        new: Dict[str, EngineStats] = {}
        for ep in endpoints:
            s = self._scrape_one(ep.url)
            if s is not None:
                new[ep.url] = s
        with self._lock:
            self._stats = new

    def _worker(self) -> None:
        while not self._stop.wait(self.interval):
            try:
                self.scrape_once()
            except Exception as e:  # pragma: no cover
                logger.warning("engine stats scrape failed: %s", e)

    def get_engine_stats(self) -> Dict[str, EngineStats]:
        with self._lock:
            return dict(self._stats)

    def get_health(self) -> bool:
        return self._thread is None or self._thread.is_alive()

    def close(self) -> None:
        self._stop.set()


# ---------------------------------------------------------------------------
# Request stats
# ---------------------------------------------------------------------------
class MovingAverageMonitor:
    def __init__(self, window: float) -> None:
        self.window = window
        self.timestamps: Deque[float] = deque()
        self.values: Deque[float] = deque()

    def update(self, ts: float, value: float) -> None:
        self.timestamps.append(ts)
        self.values.append(value)
        self._expire(ts)

    def _expire(self, now: float) -> None:
        while self.timestamps and self.timestamps[0] < now - self.window:
            self.timestamps.popleft()
            self.values.popleft()

    def get_average(self) -> float:
        if not self.values:
            return -1.0
        return sum(self.values) / len(self.values)

    def get_rate(self, now: Optional[float] = None) -> float:
        now = now or time.time()
        self._expire(now)
        if not self.timestamps:
            return 0.0
        span = max(now - self.timestamps[0], 1e-6)
        return len(self.timestamps) / span


@dataclass
class RequestStats:
    qps: float = 0.0
    ttft: float = -1.0
    in_prefill_requests: int = 0
    in_decoding_requests: int = 0
    finished_requests: int = 0
    uncomputed_latency: float = -1.0
    avg_decoding_length: float = -1.0
    avg_latency: float = -1.0
    avg_itl: float = -1.0


class RequestStatsMonitor:
    def __init__(self, window: float = 60.0) -> None:
        self.window = window
        self.qps: Dict[str, MovingAverageMonitor] = {}
        self.ttft: Dict[str, MovingAverageMonitor] = {}
        self.latency: Dict[str, MovingAverageMonitor] = {}
        self.in_prefill: Dict[str, int] = {}
        self.in_decoding: Dict[str, int] = {}
        self.finished: Dict[str, int] = {}
        # (engine, request_id) -> submit ts
        self._start: Dict[Tuple[str, str], float] = {}
        self._first_token: Dict[Tuple[str, str], float] = {}
        self._lock = threading.Lock()

    def _mon(self, d: Dict[str, MovingAverageMonitor], url: str):
        if url not in d:
            d[url] = MovingAverageMonitor(self.window)
        return d[url]

    def on_new_request(self, url: str, request_id: str, ts: float) -> None:
        with self._lock:
            self._start[(url, request_id)] = ts
            self._mon(self.qps, url).update(ts, 1.0)
            self.in_prefill[url] = self.in_prefill.get(url, 0) + 1

    def on_request_response(
        self, url: str, request_id: str, ts: float
    ) -> None:
        with self._lock:
            start = self._start.get((url, request_id))
            if start is None or (url, request_id) in self._first_token:
                return
            self._first_token[(url, request_id)] = ts
            self._mon(self.ttft, url).update(ts, ts - start)
            if self.in_prefill.get(url, 0) > 0:
                self.in_prefill[url] -= 1
            self.in_decoding[url] = self.in_decoding.get(url, 0) + 1

    def on_request_complete(
        self, url: str, request_id: str, ts: float
    ) -> None:
        with self._lock:
            start = self._start.pop((url, request_id), None)
            had_first = (url, request_id) in self._first_token
            self._first_token.pop((url, request_id), None)
            if had_first:
                if self.in_decoding.get(url, 0) > 0:
                    self.in_decoding[url] -= 1
            elif self.in_prefill.get(url, 0) > 0:
                self.in_prefill[url] -= 1
            self.finished[url] = self.finished.get(url, 0) + 1
            if start is not None:
                self._mon(self.latency, url).update(ts, ts - start)

    def on_request_failed(self, url: str, request_id: str) -> None:
        with self._lock:
            self._start.pop((url, request_id), None)
            had_first = (url, request_id) in self._first_token
            self._first_token.pop((url, request_id), None)
            if had_first:
                if self.in_decoding.get(url, 0) > 0:
                    self.in_decoding[url] -= 1
            elif self.in_prefill.get(url, 0) > 0:
                self.in_prefill[url] -= 1

    def get_request_stats(
        self, now: Optional[float] = None
    ) -> Dict[str, RequestStats]:
        now = now or time.time()
        with self._lock:
            urls = (
                set(self.qps)
                | set(self.in_prefill)
                | set(self.in_decoding)
            )
            out: Dict[str, RequestStats] = {}
            for u in urls:
                qps = (
                    self.qps[u].get_rate(now) if u in self.qps else 0.0
                )
                ttft = (
                    self.ttft[u].get_average() if u in self.ttft else -1.0
                )
                lat = (
                    self.latency[u].get_average()
                    if u in self.latency
                    else -1.0
                )
                out[u] = RequestStats(
                    qps=qps,
                    ttft=ttft,
                    in_prefill_requests=self.in_prefill.get(u, 0),
                    in_decoding_requests=self.in_decoding.get(u, 0),
                    finished_requests=self.finished.get(u, 0),
                    avg_latency=lat,
                )
            return out


_scraper: Optional[EngineStatsScraper] = None
_monitor: Optional[RequestStatsMonitor] = None


def initialize_engine_stats_scraper(
    interval: float = 10.0, start: bool = True
) -> EngineStatsScraper:
    global _scraper
    if _scraper is not None:
        _scraper.close()
    _scraper = EngineStatsScraper(interval, start=start)
    return _scraper


def get_engine_stats_scraper() -> Optional[EngineStatsScraper]:
    return _scraper


def initialize_request_stats_monitor(
    window: float = 60.0,
) -> RequestStatsMonitor:
    global _monitor
    _monitor = RequestStatsMonitor(window)
    return _monitor


def get_request_stats_monitor() -> RequestStatsMonitor:
    global _monitor
    if _monitor is None:
        _monitor = RequestStatsMonitor()
    return _monitor


def reset_stats() -> None:
    global _scraper, _monitor
    if _scraper is not None:
        _scraper.close()
    _scraper = None
    _monitor = None
