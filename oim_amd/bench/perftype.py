"""perfdash-compatible result schema (reference test/e2e/perftype/
perftype.go:26-53 — defined there but never emitted; here the harness
actually emits it)."""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List, TextIO


@dataclass
class DataItem:
    data: Dict[str, float]
    unit: str
    labels: Dict[str, str] = field(default_factory=dict)


@dataclass
class PerfData:
    version: str
    data_items: List[DataItem]
    labels: Dict[str, str] = field(default_factory=dict)


def emit_perf_data(perf: PerfData, out: TextIO) -> None:
    """The perfdash framing: a tagged JSON block the dashboard scrapes."""
    payload = {
        "version": perf.version,
        "dataItems": [
            {"data": item.data, "unit": item.unit, "labels": item.labels}
            for item in perf.data_items
        ],
        "labels": perf.labels,
    }
    out.write("[Finished:Performance] " + json.dumps(payload) + "\n")


def perf_result_to_data_item(result: dict, labels: Dict[str, str]) -> DataItem:
    """Map a hipstored perf result to a perfdash item."""
    return DataItem(
        data={
            "iops": result["iops"],
            "throughput_mbps": result["throughput_mbps"],
            "lat_avg_us": result["lat_avg_us"],
            "lat_p50_us": result["lat_p50_us"],
            "lat_p99_us": result["lat_p99_us"],
            "lat_p999_us": result["lat_p999_us"],
        },
        unit="mixed",
        labels=labels,
    )
