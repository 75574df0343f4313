"""client_amd.perf — perf_analyzer-class load generator."""

from .analyzer import ConcurrencyDriver, LatencyRecorder, PerfAnalyzer

__all__ = ["PerfAnalyzer", "ConcurrencyDriver", "LatencyRecorder"]
