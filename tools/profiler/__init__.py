"""MI355X profiling tooling: fit the autoscaler's linear perf parameters
(alpha/beta for decode ITL, gamma/delta for prefill TTFT) from measured
CDNA4 latency curves."""
