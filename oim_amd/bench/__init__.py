"""Benchmark harness package: fio-shaped workloads + perfdash emitter."""

from .perftype import DataItem, PerfData, emit_perf_data

__all__ = ["DataItem", "PerfData", "emit_perf_data"]
