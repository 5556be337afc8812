"""Prometheus metrics of the POST surface — the same metric names/labels the
reference exports, emitted around this engine (SURVEY §5 observability row):

  smh_init_size{step="start"|"complete"}   metrics/public/public.go:15-20
  smh_post_seconds                         metrics/public/public.go:21-24
  activation_post_duration                 activation/metrics/metrics.go:13-18
  activation_post_verification_waiting_total          :27-32 (pool queue)
  activation_post_verification_seconds (histogram,
      exponential buckets 1..2^19)                    :46-52

Wired from PostSetupManager (init start/complete, post duration), the
prove entry points (post duration/seconds) and the OffloadingVerifier
(queue gauge, verification latency) — matching the reference call sites
activation/post.go:293-331, validation.go:216-220, activation.go:398-399,
post_verifier.go:319-320."""
from __future__ import annotations

from prometheus_client import CollectorRegistry, Gauge, Histogram

registry = CollectorRegistry()

init_size = Gauge("smh_init_size", "init size by step", ["step"],
                  registry=registry)
post_seconds = Gauge("smh_post_seconds", "duration of last PoST in seconds",
                     registry=registry)
post_duration = Gauge("activation_post_duration",
                      "duration of last PoST in nanoseconds",
                      registry=registry)
post_verification_queue = Gauge(
    "activation_post_verification_waiting_total",
    "the number of POSTs waiting to be verified", registry=registry)
post_verification_latency = Histogram(
    "activation_post_verification_seconds", "post verification in seconds",
    buckets=[2.0 ** i for i in range(20)], registry=registry)
