"""Prometheus metrics exporter.

The reference has no observability beyond log.Printf (SURVEY.md §5 —
prometheus libs are indirect deps only, never wired). Here: a real
/metrics endpoint with allocation counters/latency, device health and
discovery stats, enabled with --metrics-port.
"""
from __future__ import annotations

from prometheus_client import CollectorRegistry, Gauge, start_http_server
from prometheus_client.core import CounterMetricFamily

from .utils.log import get_logger

log = get_logger(__name__)


class MetricsExporter:
    def __init__(self, manager):
        self.manager = manager
        self.registry = CollectorRegistry()
        self._server = None
        # Registered FIRST so each scrape refreshes the gauges before the
        # registry collects them (collectors run in registration order).
        # The *_total series are Counter-typed and emitted by the collector
        # itself (monotonic cumulative values read straight off the plugin
        # objects) so rate()/increase() reset-handling works — a Gauge-typed
        # *_total violates Prometheus conventions (advisor finding r1).
        self.registry.register(_Refresher(self))

        self.g_devices = Gauge(
            "kxdp_devices", "Discovered schedulable xPU devices",
            ["resource"], registry=self.registry,
        )
        self.g_healthy = Gauge(
            "kxdp_devices_healthy", "Healthy xPU devices",
            ["resource"], registry=self.registry,
        )
        self.g_last_alloc = Gauge(
            "kxdp_last_allocate_seconds", "Duration of the last Allocate()",
            ["resource"], registry=self.registry,
        )
        self.g_scan_wall = Gauge(
            "kxdp_discovery_seconds", "Wall time of the last discovery scan",
            registry=self.registry,
        )
        self.g_xgmi_up = Gauge(
            "kxdp_xgmi_link_up",
            "xGMI link state (1 up, 0 down/disabled) per GPU link "
            "(amd-smi; amdgpu-bound GPUs only)",
            ["bdf", "link"], registry=self.registry,
        )
        self.g_hive_degraded = Gauge(
            "kxdp_xgmi_degraded",
            "GPU excluded from hive placement due to sick xGMI links",
            ["bdf"], registry=self.registry,
        )

    def refresh(self) -> None:
        m = self.manager
        if m.inventory is not None:
            self.g_scan_wall.set(m.inventory.scan_wall_s)
        for rname, plugin in list(m.plugins.items()):
            state = m.states.get(rname)
            if state is None:
                continue
            self.g_devices.labels(rname).set(len(state.device_ids()))
            self.g_healthy.labels(rname).set(len(state.healthy_ids()))
            self.g_last_alloc.labels(rname).set(plugin.last_allocate_s)
        poller = getattr(m, "_amdsmi", None)
        if poller is not None:
            for bdf, dh in poller.last_snapshot.items():
                for link in dh.xgmi_links:
                    self.g_xgmi_up.labels(bdf, str(link.index)).set(
                        1.0 if link.status == "up" else 0.0)
        if m.topology is not None:
            for bdf in m.topology.degraded:
                self.g_hive_degraded.labels(bdf).set(1.0)

    def counter_families(self):
        """Counter-typed *_total families, one sample per resource."""
        c_allocs = CounterMetricFamily(
            "kxdp_allocations", "Allocate() calls served", labels=["resource"])
        c_fail = CounterMetricFamily(
            "kxdp_allocation_failures", "Allocate() calls rejected",
            labels=["resource"])
        c_secs = CounterMetricFamily(
            "kxdp_allocate_seconds",
            "Cumulative server-side Allocate() handler time",
            labels=["resource"])
        for rname, plugin in list(self.manager.plugins.items()):
            c_allocs.add_metric([rname], plugin.allocations)
            c_fail.add_metric([rname], plugin.allocate_failures)
            c_secs.add_metric([rname], plugin.allocate_seconds_total)
        c_xgmi_err = CounterMetricFamily(
            "kxdp_xgmi_link_errors",
            "Cumulative xGMI link errors per GPU link (amd-smi)",
            labels=["bdf", "link"])
        c_ecc = CounterMetricFamily(
            "kxdp_ecc_uncorrectable",
            "Uncorrectable RAS errors per IP block (amd-smi)",
            labels=["bdf", "block"])
        poller = getattr(self.manager, "_amdsmi", None)
        if poller is not None:
            for bdf, dh in poller.last_snapshot.items():
                for link in dh.xgmi_links:
                    c_xgmi_err.add_metric([bdf, str(link.index)], link.errors)
                for block, count in dh.ecc_by_block.items():
                    c_ecc.add_metric([bdf, block], count)
        return [c_allocs, c_fail, c_secs, c_xgmi_err, c_ecc]

    def start(self, port: int) -> None:
        self._server, _ = start_http_server(port, registry=self.registry)
        log.info("metrics on :%d/metrics", port)

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server = None


class _Refresher:
    """Collector that refreshes gauges before each scrape and emits the
    Counter-typed *_total families."""

    def __init__(self, exporter: MetricsExporter):
        self.exporter = exporter

    def collect(self):
        try:
            self.exporter.refresh()
            return self.exporter.counter_families()
        except Exception:
            log.exception("metrics refresh failed")
            return []
