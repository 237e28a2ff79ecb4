"""Minimal vhost-user master + virtio-scsi/blk initiator.

Plays QEMU's role against hipstored's vhost-user targets
(native/src/vhost.cpp): guest memory is a memfd shared with the slave
via SET_MEM_TABLE, a virtio split ring lives inside it, kick/call are
eventfds passed over SCM_RIGHTS. Used by the conformance tests
(tests/test_vhost.py) and by the vhost benchmark harness
(oim_amd.bench.vhost_harness), which drives multiple outstanding
chains through the same ring primitives.
"""

from __future__ import annotations

import mmap
import os
import select
import socket
import struct

GET_FEATURES = 1
SET_FEATURES = 2
SET_OWNER = 3
SET_MEM_TABLE = 5
SET_VRING_NUM = 8
SET_VRING_ADDR = 9
SET_VRING_BASE = 10
GET_VRING_BASE = 11
SET_VRING_KICK = 12
SET_VRING_CALL = 13
GET_PROTOCOL_FEATURES = 15
SET_PROTOCOL_FEATURES = 16
GET_QUEUE_NUM = 17
SET_VRING_ENABLE = 18

FEAT_VERSION_1 = 1 << 32
FEAT_INDIRECT = 1 << 28
FEAT_PROTOCOL = 1 << 30

DESC_NEXT = 1
DESC_WRITE = 2
DESC_INDIRECT = 4

QUEUE = 2  # first virtio-scsi request queue
QSIZE = 16  # default ring size (tests); the harness passes more

# guest-memory layout (offsets into the shared region)
DESC_OFF = 0x1000
AVAIL_OFF = 0x2000
USED_OFF = 0x3000
INDIRECT_OFF = 0x4000
REQ_OFF = 0x10000
RESP_OFF = 0x11000
DATA_OUT_OFF = 0x20000
DATA_IN_OFF = 0x100000

GPA_BASE = 0x10000000
UADDR_BASE = 0x7F0000000000

RESP_FMT = "<IIHBB"  # sense_len, resid, status_qualifier, status, response


class ScsiResult:
    def __init__(self, raw_resp: bytes, data: bytes):
        (self.sense_len, self.resid, self.status_qualifier, self.status,
         self.response) = struct.unpack(RESP_FMT, raw_resp[:12])
        self.sense = raw_resp[12:12 + self.sense_len]
        self.data = data

    @property
    def sense_key(self):
        return self.sense[2] & 0x0F if len(self.sense) > 2 else None

    @property
    def asc(self):
        return self.sense[12] if len(self.sense) > 12 else None


class VhostUserMaster:
    def __init__(self, path: str, mem_size: int = 4 << 20,
                 queue: int = QUEUE, qsize: int = QSIZE):
        self.queue = queue
        self.qsize = qsize
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(10)
        self.sock.connect(path)
        self.memfd = os.memfd_create("vhost-guest-mem")
        os.ftruncate(self.memfd, mem_size)
        self.mem = mmap.mmap(self.memfd, mem_size)
        self.mem_size = mem_size
        self.kick = os.eventfd(0)
        self.call = os.eventfd(0, os.EFD_NONBLOCK)
        self.avail_idx = 0
        self.used_idx = 0
        self.tag = 0

    # -- protocol plumbing -------------------------------------------------

    def _send(self, request: int, payload: bytes = b"", fds=()):
        header = struct.pack("<III", request, 0x1, len(payload))
        if fds:
            rights = (socket.SOL_SOCKET, socket.SCM_RIGHTS,
                      struct.pack(f"<{len(fds)}i", *fds))
            self.sock.sendmsg([header + payload], [rights])
        else:
            self.sock.sendall(header + payload)

    def _recv_reply(self) -> bytes:
        header = b""
        while len(header) < 12:
            chunk = self.sock.recv(12 - len(header))
            if not chunk:
                raise ConnectionError("vhost slave closed the socket")
            header += chunk
        _, _, size = struct.unpack("<III", header)
        payload = b""
        while len(payload) < size:
            payload += self.sock.recv(size - len(payload))
        return payload

    def query(self, request: int, payload: bytes = b"") -> bytes:
        self._send(request, payload)
        return self._recv_reply()

    def handshake(self, features_extra: int = 0):
        """Feature negotiation + guest memory; no rings yet."""
        features, = struct.unpack("<Q", self.query(GET_FEATURES))
        assert features & FEAT_VERSION_1, hex(features)
        self._send(SET_FEATURES, struct.pack(
            "<Q", FEAT_VERSION_1 | FEAT_PROTOCOL | features_extra))
        proto, = struct.unpack("<Q", self.query(GET_PROTOCOL_FEATURES))
        self._send(SET_PROTOCOL_FEATURES, struct.pack("<Q", proto & 0x1))
        self._send(SET_OWNER)
        region = struct.pack("<II", 1, 0) + struct.pack(
            "<QQQQ", GPA_BASE, self.mem_size, UADDR_BASE, 0)
        self._send(SET_MEM_TABLE, region, fds=[self.memfd])

    def setup_ring(self, queue: int, desc_off: int, avail_off: int,
                   used_off: int, qsize: int, kick: int, call: int):
        """Configure + enable one vring at the given guest offsets."""
        self._send(SET_VRING_NUM, struct.pack("<II", queue, qsize))
        self._send(SET_VRING_BASE, struct.pack("<II", queue, 0))
        self._send(SET_VRING_ADDR, struct.pack(
            "<IIQQQQ", queue, 0, UADDR_BASE + desc_off,
            UADDR_BASE + used_off, UADDR_BASE + avail_off, 0))
        self._send(SET_VRING_CALL, struct.pack("<Q", queue), fds=[call])
        self._send(SET_VRING_KICK, struct.pack("<Q", queue), fds=[kick])
        self._send(SET_VRING_ENABLE, struct.pack("<II", queue, 1))

    def add_ring(self, queue: int, region_base: int,
                 qsize: int) -> "VhostRing":
        """Independent ring with its own eventfds and memory region
        (multi-queue benchmarking)."""
        ring = VhostRing(self, queue, region_base, qsize)
        self.setup_ring(queue, ring.desc_off, ring.avail_off,
                        ring.used_off, qsize, ring.kick, ring.call)
        return ring

    def negotiate(self, features_extra: int = 0):
        self.handshake(features_extra)
        self.setup_ring(self.queue, DESC_OFF, AVAIL_OFF, USED_OFF,
                        self.qsize, self.kick, self.call)

    def stop_ring(self) -> int:
        """GET_VRING_BASE stops the ring; returns last_avail."""
        reply = self.query(GET_VRING_BASE, struct.pack("<II", self.queue, 0))
        _, base = struct.unpack("<II", reply)
        return base

    def close(self):
        if getattr(self, "_closed", False):
            return
        self._closed = True
        for fd in (self.kick, self.call, self.memfd):
            os.close(fd)
        self.mem.close()
        self.sock.close()

    # -- ring operations ---------------------------------------------------

    def _write_desc(self, slot: int, gpa: int, length: int, flags: int,
                    nxt: int = 0, table_off: int = DESC_OFF):
        self.mem[table_off + 16 * slot:table_off + 16 * (slot + 1)] = (
            struct.pack("<QIHH", gpa, length, flags, nxt))

    def _submit(self, head: int):
        ring_slot = AVAIL_OFF + 4 + 2 * (self.avail_idx % self.qsize)
        self.mem[ring_slot:ring_slot + 2] = struct.pack("<H", head)
        self.avail_idx += 1
        self.mem[AVAIL_OFF + 2:AVAIL_OFF + 4] = struct.pack(
            "<H", self.avail_idx & 0xFFFF)
        os.eventfd_write(self.kick, 1)

    def _wait_used(self, timeout: float = 10.0) -> int:
        """Wait for the next used entry; returns its len field."""
        import time
        deadline = time.time() + timeout
        while True:
            idx, = struct.unpack(
                "<H", self.mem[USED_OFF + 2:USED_OFF + 4])
            if idx != self.used_idx & 0xFFFF:
                break
            remaining = deadline - time.time()
            if remaining <= 0:
                raise TimeoutError("no completion from vhost target")
            select.select([self.call], [], [], min(remaining, 0.5))
            try:
                os.eventfd_read(self.call)
            except BlockingIOError:
                pass
        slot = USED_OFF + 4 + 8 * (self.used_idx % self.qsize)
        head, used_len = struct.unpack("<II", self.mem[slot:slot + 8])
        self.used_idx += 1
        self.last_used_head = head
        return used_len

    # -- SCSI --------------------------------------------------------------

    def scsi(self, target: int, cdb: bytes, data_out: bytes = b"",
             data_in_len: int = 0, lun: int = 0,
             indirect: bool = False) -> ScsiResult:
        self.tag += 1
        req = (bytes([1, target, 0x40 | (lun >> 8), lun & 0xFF, 0, 0, 0, 0])
               + struct.pack("<Q", self.tag) + bytes(3)
               + cdb.ljust(32, b"\0"))
        assert len(req) == 51
        self.mem[REQ_OFF:REQ_OFF + 51] = req
        if data_out:
            self.mem[DATA_OUT_OFF:DATA_OUT_OFF + len(data_out)] = data_out
        self.mem[RESP_OFF:RESP_OFF + 108] = bytes(108)

        chain = [(GPA_BASE + REQ_OFF, 51, 0)]
        if data_out:
            chain.append((GPA_BASE + DATA_OUT_OFF, len(data_out), 0))
        chain.append((GPA_BASE + RESP_OFF, 108, DESC_WRITE))
        if data_in_len:
            chain.append((GPA_BASE + DATA_IN_OFF, data_in_len, DESC_WRITE))

        if indirect:
            table = b"".join(
                struct.pack("<QIHH", gpa, length,
                            flags | (DESC_NEXT if i < len(chain) - 1 else 0),
                            i + 1 if i < len(chain) - 1 else 0)
                for i, (gpa, length, flags) in enumerate(chain))
            self.mem[INDIRECT_OFF:INDIRECT_OFF + len(table)] = table
            self._write_desc(0, GPA_BASE + INDIRECT_OFF, len(table),
                             DESC_INDIRECT)
        else:
            for i, (gpa, length, flags) in enumerate(chain):
                last = i == len(chain) - 1
                self._write_desc(i, gpa, length,
                                 flags | (0 if last else DESC_NEXT),
                                 0 if last else i + 1)
        self._submit(0)
        self._wait_used()
        raw_resp = bytes(self.mem[RESP_OFF:RESP_OFF + 108])
        data = (bytes(self.mem[DATA_IN_OFF:DATA_IN_OFF + data_in_len])
                if data_in_len else b"")
        return ScsiResult(raw_resp, data)

    # -- virtio-blk ---------------------------------------------------------

    def blk(self, blk_type: int, sector: int, data_out: bytes = b"",
            data_in_len: int = 0):
        """Submit a virtio-blk request; returns (status_byte, data)."""
        header = struct.pack("<IIQ", blk_type, 0, sector)
        self.mem[REQ_OFF:REQ_OFF + 16] = header
        if data_out:
            self.mem[DATA_OUT_OFF:DATA_OUT_OFF + len(data_out)] = data_out
        status_off = RESP_OFF
        self.mem[status_off] = 0xAA  # sentinel
        chain = [(GPA_BASE + REQ_OFF, 16, 0)]
        if data_out:
            chain.append((GPA_BASE + DATA_OUT_OFF, len(data_out), 0))
        if data_in_len:
            chain.append((GPA_BASE + DATA_IN_OFF, data_in_len, DESC_WRITE))
        chain.append((GPA_BASE + status_off, 1, DESC_WRITE))
        for i, (gpa, length, flags) in enumerate(chain):
            last = i == len(chain) - 1
            self._write_desc(i, gpa, length,
                             flags | (0 if last else DESC_NEXT),
                             0 if last else i + 1)
        self._submit(0)
        self._wait_used()
        status = self.mem[status_off]
        data = (bytes(self.mem[DATA_IN_OFF:DATA_IN_OFF + data_in_len])
                if data_in_len else b"")
        return status, data

    def blk_read(self, sector: int, length: int):
        return self.blk(0, sector, data_in_len=length)

    def blk_write(self, sector: int, data: bytes):
        return self.blk(1, sector, data_out=data)

    # convenience wrappers

    def inquiry(self, target: int, evpd_page: int | None = None):
        cdb = bytearray(6)
        cdb[0] = 0x12
        if evpd_page is not None:
            cdb[1] = 1
            cdb[2] = evpd_page
        cdb[3:5] = struct.pack(">H", 255)
        return self.scsi(target, bytes(cdb), data_in_len=255)

    def read_capacity10(self, target: int):
        result = self.scsi(target, bytes([0x25]) + bytes(9), data_in_len=8)
        if result.status == 0:
            last_lba, block = struct.unpack(">II", result.data[:8])
            return last_lba, block
        return result

    def read_capacity16(self, target: int):
        cdb = bytearray(16)
        cdb[0] = 0x9E
        cdb[1] = 0x10
        cdb[10:14] = struct.pack(">I", 32)
        result = self.scsi(target, bytes(cdb), data_in_len=32)
        last_lba, block = struct.unpack(">QI", result.data[:12])
        return last_lba, block

    def read10(self, target: int, lba: int, count: int, block: int,
               **kw) -> ScsiResult:
        cdb = bytearray(10)
        cdb[0] = 0x28
        cdb[2:6] = struct.pack(">I", lba)
        cdb[7:9] = struct.pack(">H", count)
        return self.scsi(target, bytes(cdb), data_in_len=count * block, **kw)

    def write10(self, target: int, lba: int, data: bytes, block: int,
                **kw) -> ScsiResult:
        cdb = bytearray(10)
        cdb[0] = 0x2A
        cdb[2:6] = struct.pack(">I", lba)
        cdb[7:9] = struct.pack(">H", len(data) // block)
        return self.scsi(target, bytes(cdb), data_out=data, **kw)

    def read16(self, target: int, lba: int, count: int, block: int):
        cdb = bytearray(16)
        cdb[0] = 0x88
        cdb[2:10] = struct.pack(">Q", lba)
        cdb[10:14] = struct.pack(">I", count)
        return self.scsi(target, bytes(cdb), data_in_len=count * block)

    def write16(self, target: int, lba: int, data: bytes, block: int):
        cdb = bytearray(16)
        cdb[0] = 0x8A
        cdb[2:10] = struct.pack(">Q", lba)
        cdb[10:14] = struct.pack(">I", len(data) // block)
        return self.scsi(target, bytes(cdb), data_out=data)

    def report_luns(self, target: int):
        cdb = bytearray(12)
        cdb[0] = 0xA0
        cdb[6:10] = struct.pack(">I", 256)
        return self.scsi(target, bytes(cdb), data_in_len=256)

GET_CONFIG = 24


def read_blk_config(master) -> dict:
    """VHOST_USER_GET_CONFIG -> virtio_blk_config fields."""
    payload = struct.pack("<III", 0, 24, 0) + bytes(24)
    reply = master.query(GET_CONFIG, payload)
    region = reply[12:]
    capacity, = struct.unpack("<Q", region[0:8])
    blk_size, = struct.unpack("<I", region[20:24])
    return {"capacity_sectors": capacity, "blk_size": blk_size}


class VhostRing:
    """One vring with private eventfds inside a master's guest memory
    (multi-queue benchmarking; the legacy single-ring helpers on
    VhostUserMaster stay as-is for the conformance tests).

    Layout relative to region_base: descriptor table, avail ring, then
    the used ring page-aligned; everything past header_bytes is the
    caller's for request/data buffers."""

    def __init__(self, master: VhostUserMaster, queue: int,
                 region_base: int, qsize: int):
        self.m = master
        self.queue = queue
        self.base = region_base
        self.qsize = qsize
        self.desc_off = region_base
        self.avail_off = region_base + 16 * qsize
        used_unaligned = self.avail_off + 4 + 2 * qsize
        self.used_off = (used_unaligned + 0xFFF) & ~0xFFF
        self.header_bytes = self.used_off + 4 + 8 * qsize - region_base
        self.kick = os.eventfd(0)
        self.call = os.eventfd(0, os.EFD_NONBLOCK)
        self.avail_idx = 0
        self.used_idx = 0
        self.last_used_head = 0

    def close(self):
        os.close(self.kick)
        os.close(self.call)

    def write_desc(self, slot: int, gpa: int, length: int, flags: int,
                   nxt: int = 0):
        off = self.desc_off + 16 * slot
        self.m.mem[off:off + 16] = struct.pack("<QIHH", gpa, length,
                                               flags, nxt)

    def submit(self, head: int):
        slot = self.avail_off + 4 + 2 * (self.avail_idx % self.qsize)
        self.m.mem[slot:slot + 2] = struct.pack("<H", head)
        self.avail_idx += 1
        self.m.mem[self.avail_off + 2:self.avail_off + 4] = struct.pack(
            "<H", self.avail_idx & 0xFFFF)
        os.eventfd_write(self.kick, 1)

    def wait_used(self, timeout: float = 10.0) -> int:
        import time
        deadline = time.time() + timeout
        while True:
            idx, = struct.unpack(
                "<H", self.m.mem[self.used_off + 2:self.used_off + 4])
            if idx != self.used_idx & 0xFFFF:
                break
            remaining = deadline - time.time()
            if remaining <= 0:
                raise TimeoutError("no completion from vhost target")
            select.select([self.call], [], [], min(remaining, 0.5))
            try:
                os.eventfd_read(self.call)
            except BlockingIOError:
                pass
        slot = self.used_off + 4 + 8 * (self.used_idx % self.qsize)
        head, used_len = struct.unpack("<II", self.m.mem[slot:slot + 8])
        self.used_idx += 1
        self.last_used_head = head
        return used_len
