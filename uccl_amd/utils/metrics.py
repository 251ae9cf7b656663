"""Prometheus metrics export for engine/transport/p2p statistics.

Reference parity: the stats threads that print per-engine counters
every 2s (RDMAEndpoint::stats_thread_fn) — here exposed as a
scrapeable endpoint instead of log lines, for production serving.

    from uccl_amd.utils import metrics
    metrics.track_transport("tp0", transport_endpoint)
    metrics.track_p2p("ep0", p2p_endpoint)
    metrics.start_server(port=9464)      # /metrics, standard exposition

Collectors pull the stats lazily at scrape time (no background thread,
no overhead between scrapes).
"""

from __future__ import annotations

import threading
from typing import Callable, Dict, Tuple

_tracked: Dict[Tuple[str, str], Callable[[], dict]] = {}
_lock = threading.Lock()


def _transport_stats(tp) -> dict:
    st = tp.stats()
    return {
        "data_sent": st.data_sent, "data_recv": st.data_recv,
        "acks_sent": st.acks_sent, "acks_recv": st.acks_recv,
        "retransmits": st.retransmits,
        "rto_retransmits": st.rto_retransmits,
        "msgs_sent": st.msgs_sent, "msgs_recv": st.msgs_recv,
        "srtt_us": st.srtt_us, "cwnd": st.cwnd,
        "rtt_p50_us": st.rtt_p50_us, "rtt_p99_us": st.rtt_p99_us,
    }


def _p2p_stats(ep) -> dict:
    out = {}
    for op, st in ep.stats().items():
        for k, v in st.items():
            out[f"{op}_{k}"] = v
    return out


def track_transport(name: str, tp) -> None:
    with _lock:
        _tracked[("transport", name)] = lambda: _transport_stats(tp)


def track_p2p(name: str, ep) -> None:
    with _lock:
        _tracked[("p2p", name)] = lambda: _p2p_stats(ep)


def track_custom(kind: str, name: str, fn: Callable[[], dict]) -> None:
    with _lock:
        _tracked[(kind, name)] = fn


def untrack(kind: str, name: str) -> None:
    with _lock:
        _tracked.pop((kind, name), None)


class _Collector:
    def collect(self):
        from prometheus_client.core import GaugeMetricFamily

        with _lock:
            snapshot = dict(_tracked)
        families: Dict[str, GaugeMetricFamily] = {}
        for (kind, name), fn in snapshot.items():
            try:
                stats = fn()
            except Exception:  # endpoint torn down; skip this scrape
                continue
            for key, val in stats.items():
                mname = f"uccl_{kind}_{key}"
                fam = families.get(mname)
                if fam is None:
                    fam = GaugeMetricFamily(mname, f"uccl {kind} {key}",
                                            labels=["name"])
                    families[mname] = fam
                fam.add_metric([name], float(val))
        yield from families.values()


_collector_registered = False


def registry():
    """A fresh-scrape registry containing only uccl collectors."""
    import prometheus_client

    global _collector_registered
    reg = prometheus_client.CollectorRegistry()
    reg.register(_Collector())
    return reg


def render() -> bytes:
    """One exposition-format snapshot of every tracked object."""
    from prometheus_client import generate_latest

    return generate_latest(registry())


def start_server(port: int = 9464, addr: str = "0.0.0.0"):
    """Serve /metrics; returns the underlying server object."""
    from prometheus_client import start_http_server

    srv, _ = start_http_server(port, addr, registry=registry())
    return srv
