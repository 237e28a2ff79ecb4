"""PCI BDF helpers (reference pkg/oim-common/pci.go:19-90).

A PCI address has domain/bus/device/function components; the sentinel
0xFFFF marks a component as "unset" so partial addresses can be merged
(spec.md:150-162).  On MI355X nodes each GPU "card" contributes its own
BDF (from /sys/class/kfd or rocm-smi) as the registry ``<id>/pci`` entry.
"""

from __future__ import annotations

import re
from dataclasses import dataclass

UNSET = 0xFFFF

# "[domain:]bus:device.function", hex components.
_BDF_RE = re.compile(
    r"^(?:(?P<domain>[0-9a-fA-F]{1,4}):)?"
    r"(?P<bus>[0-9a-fA-F]{1,2}):"
    r"(?P<device>[0-9a-fA-F]{1,2})\.(?P<function>[0-9a-fA-F])$"
)


@dataclass
class PCIAddress:
    domain: int = UNSET
    bus: int = UNSET
    device: int = UNSET
    function: int = UNSET

    def is_complete(self) -> bool:
        return UNSET not in (self.domain, self.bus, self.device, self.function)


def parse_bdf_string(text: str) -> PCIAddress:
    """Parse "[dddd:]bb:dd.f"; missing domain stays UNSET."""
    m = _BDF_RE.match(text.strip())
    if not m:
        raise ValueError(f"not a PCI BDF: {text!r}")
    addr = PCIAddress(
        bus=int(m.group("bus"), 16),
        device=int(m.group("device"), 16),
        function=int(m.group("function"), 16),
    )
    if m.group("domain") is not None:
        addr.domain = int(m.group("domain"), 16)
    return addr


def complete_pci_address(primary: PCIAddress, fallback: PCIAddress) -> PCIAddress:
    """Merge two partial addresses, primary components winning.

    A still-unset domain defaults to 0 (reference remote.go:173-190:
    "domain 0xFFFF -> 0").
    """
    merged = PCIAddress(
        domain=primary.domain if primary.domain != UNSET else fallback.domain,
        bus=primary.bus if primary.bus != UNSET else fallback.bus,
        device=primary.device if primary.device != UNSET else fallback.device,
        function=primary.function if primary.function != UNSET else fallback.function,
    )
    if merged.domain == UNSET:
        merged.domain = 0
    return merged


def pretty_pci_address(addr: PCIAddress) -> str:
    """Render as dddd:bb:dd.f, printing unset components as '*'."""

    def fmt(value: int, width: int) -> str:
        return "*" * width if value == UNSET else f"{value:0{width}x}"

    return (
        f"{fmt(addr.domain, 4)}:{fmt(addr.bus, 2)}:"
        f"{fmt(addr.device, 2)}.{fmt(addr.function, 1)}"
    )
