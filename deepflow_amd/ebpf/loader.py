"""Raw bpf(2) loader + tracepoint attach + perf-buffer reader.

No libbpf, no BTF: maps are created with BPF_MAP_CREATE, programs loaded
with BPF_PROG_LOAD (map refs patched to fds in the ld_imm64 stream), and
attached to the stable raw_syscalls tracepoints via
perf_event_open(PERF_TYPE_TRACEPOINT) + PERF_EVENT_IOC_SET_BPF.
Reference counterpart: agent/src/ebpf/user/tracer.c + load/attach in
user/socket.c — built on libbpf/BTF there.

Degrades gracefully: `available()` probes for tracefs + CAP_BPF; in
environments without them (this container) the same programs run under
ebpf/vm.py instead.
"""
from __future__ import annotations

import ctypes as ct
import mmap
import os
import struct
from typing import Callable, Dict, List, Optional

from .insn import Asm

SYS_bpf = 321           # x86_64
SYS_perf_event_open = 298

BPF_MAP_CREATE = 0
BPF_MAP_LOOKUP_ELEM = 1
BPF_MAP_UPDATE_ELEM = 2
BPF_MAP_GET_NEXT_KEY = 4
BPF_PROG_LOAD = 5

BPF_PROG_TYPE_TRACEPOINT = 5
BPF_PROG_TYPE_PERF_EVENT = 7

PERF_TYPE_TRACEPOINT = 2
PERF_TYPE_SOFTWARE = 1
PERF_COUNT_SW_CPU_CLOCK = 0
PERF_EVENT_IOC_ENABLE = 0x2400
PERF_EVENT_IOC_SET_BPF = 0x40042408

_libc = ct.CDLL(None, use_errno=True)


def _bpf(cmd: int, attr: bytes) -> int:
    buf = ct.create_string_buffer(attr, len(attr))
    rc = _libc.syscall(SYS_bpf, cmd, buf, len(attr))
    if rc < 0:
        err = ct.get_errno()
        raise OSError(err, f"bpf(cmd={cmd}): {os.strerror(err)}")
    return rc


def map_create(mtype: int, key_size: int, value_size: int,
               max_entries: int) -> int:
    if mtype == 4 and max_entries == 0:       # perf array: one per cpu
        max_entries = os.cpu_count() or 1
    attr = struct.pack("<IIIII", mtype, key_size, value_size, max_entries, 0)
    return _bpf(BPF_MAP_CREATE, attr + b"\x00" * 48)

def map_update(fd: int, key: bytes, value: bytes, flags: int = 0) -> None:
    k = ct.create_string_buffer(key, len(key))
    v = ct.create_string_buffer(value, len(value))
    attr = struct.pack("<IIQQQ", fd, 0, ct.addressof(k), ct.addressof(v),
                       flags)
    _bpf(BPF_MAP_UPDATE_ELEM, attr)


def map_lookup(fd: int, key: bytes, value_size: int) -> Optional[bytes]:
    k = ct.create_string_buffer(key, len(key))
    v = ct.create_string_buffer(value_size)
    attr = struct.pack("<IIQQQ", fd, 0, ct.addressof(k), ct.addressof(v), 0)
    try:
        _bpf(BPF_MAP_LOOKUP_ELEM, attr)
    except OSError:
        return None
    return v.raw


def map_next_key(fd: int, key: Optional[bytes], key_size: int) -> Optional[bytes]:
    k = ct.create_string_buffer(key or b"\x00" * key_size, key_size)
    n = ct.create_string_buffer(key_size)
    attr = struct.pack("<IIQQQ", fd, 0,
                       ct.addressof(k) if key else 0, ct.addressof(n), 0)
    try:
        _bpf(BPF_MAP_GET_NEXT_KEY, attr)
    except OSError:
        return None
    return n.raw


def prog_load(prog_type: int, insns: bytes, license_: bytes = b"GPL",
              log: bool = False) -> int:
    """Load a program; on failure re-load with the verifier log for a
    useful error. NOTE: a verbose (level-2) log that overflows its
    buffer FAILS the load with ENOSPC even for valid programs — the
    buffer here is 8 MB, enough for ~500-insn programs."""
    lic = ct.create_string_buffer(license_, len(license_) + 1)
    log_buf = ct.create_string_buffer(1 << 23) if log else None
    # the insns buffer must outlive the syscall (a temporary here once
    # handed the kernel freed memory -> garbage verifier errors)
    ibuf = ct.create_string_buffer(insns, len(insns))
    attr = struct.pack(
        "<IIQQIIQI4x", prog_type, len(insns) // 8,
        ct.addressof(ibuf),
        ct.addressof(lic), 2 if log else 0,
        len(log_buf) if log else 0,
        ct.addressof(log_buf) if log else 0, 0)
    try:
        return _bpf(BPF_PROG_LOAD, attr + b"\x00" * 24)
    except OSError:
        if log and log_buf is not None:
            raise OSError(f"prog_load failed; verifier log:\n"
                          f"{log_buf.value.decode(errors='replace')}")
        raise


BPF_RAW_TRACEPOINT_OPEN = 17
BPF_PROG_TYPE_RAW_TRACEPOINT = 17


def raw_tracepoint_open(name: str, prog_fd: int) -> int:
    """Attach a RAW_TRACEPOINT program by name — no tracefs needed
    (works in unprivileged-ish containers where
    /sys/kernel/tracing is not mounted)."""
    nb = ct.create_string_buffer(name.encode() + b"\x00")
    attr = struct.pack("<QI4x", ct.addressof(nb), prog_fd)
    return _bpf(BPF_RAW_TRACEPOINT_OPEN, attr)


def raw_tracepoint_available() -> bool:
    """Can this process attach raw tracepoints here?"""
    try:
        # r0 = 0; exit
        prog = (b"\xb7\x00\x00\x00\x00\x00\x00\x00"
                b"\x95\x00\x00\x00\x00\x00\x00\x00")
        pfd = prog_load(BPF_PROG_TYPE_RAW_TRACEPOINT, prog)
        try:
            tfd = raw_tracepoint_open("sys_enter", pfd)
            os.close(tfd)
            return True
        finally:
            os.close(pfd)
    except OSError:
        return False


def _tracefs() -> Optional[str]:
    for p in ("/sys/kernel/tracing", "/sys/kernel/debug/tracing"):
        if os.path.isdir(os.path.join(p, "events")):
            return p
    return None


def tracepoint_id(category: str, name: str) -> int:
    base = _tracefs()
    if base is None:
        raise FileNotFoundError("tracefs not mounted")
    with open(f"{base}/events/{category}/{name}/id") as f:
        return int(f.read().strip())


def perf_event_open(attr: bytes, pid: int, cpu: int, group_fd: int = -1,
                    flags: int = 0) -> int:
    buf = ct.create_string_buffer(attr, len(attr))
    fd = _libc.syscall(SYS_perf_event_open, buf, pid, cpu, group_fd, flags)
    if fd < 0:
        err = ct.get_errno()
        raise OSError(err, f"perf_event_open: {os.strerror(err)}")
    return fd


def attach_tracepoint(prog_fd: int, category: str, name: str) -> int:
    tp_id = tracepoint_id(category, name)
    # struct perf_event_attr: type, size, config, sample fields...
    attr = struct.pack("<IIQQQ", PERF_TYPE_TRACEPOINT, 112, tp_id, 0, 0)
    attr = attr.ljust(112, b"\x00")
    fd = perf_event_open(attr, -1, 0)
    import fcntl
    fcntl.ioctl(fd, PERF_EVENT_IOC_SET_BPF, prog_fd)
    fcntl.ioctl(fd, PERF_EVENT_IOC_ENABLE, 0)
    return fd


def elf_sym_file_offset(path: str, name: str) -> int:
    """Resolve `name` in an ELF shared object to its FILE offset (what
    the uprobe PMU wants): st_value mapped through the containing
    PT_LOAD segment. Reads .dynsym then .symtab. Raises KeyError."""
    with open(path, "rb") as f:
        data = f.read()
    if data[:4] != b"\x7fELF" or data[4] != 2:
        raise ValueError(f"{path}: not a 64-bit ELF")
    e_phoff, = struct.unpack_from("<Q", data, 0x20)
    e_shoff, = struct.unpack_from("<Q", data, 0x28)
    e_phentsize, e_phnum = struct.unpack_from("<HH", data, 0x36)
    e_shentsize, e_shnum = struct.unpack_from("<HH", data, 0x3A)
    loads = []
    for i in range(e_phnum):
        off = e_phoff + i * e_phentsize
        p_type, = struct.unpack_from("<I", data, off)
        if p_type == 1:  # PT_LOAD
            p_offset, p_vaddr = struct.unpack_from("<QQ", data, off + 8)
            p_filesz, = struct.unpack_from("<Q", data, off + 0x20)
            loads.append((p_vaddr, p_filesz, p_offset))
    shdrs = []
    for i in range(e_shnum):
        off = e_shoff + i * e_shentsize
        sh_type, = struct.unpack_from("<I", data, off + 4)
        sh_offset, sh_size = struct.unpack_from("<QQ", data, off + 0x18)
        sh_link, = struct.unpack_from("<I", data, off + 0x28)
        sh_entsize, = struct.unpack_from("<Q", data, off + 0x38)
        shdrs.append((sh_type, sh_offset, sh_size, sh_link, sh_entsize))
    for sh_type in (11, 2):  # SHT_DYNSYM, SHT_SYMTAB
        for st, soff, ssize, slink, sent in shdrs:
            if st != sh_type or not sent:
                continue
            stroff = shdrs[slink][1]
            for j in range(ssize // sent):
                base = soff + j * sent
                st_name, = struct.unpack_from("<I", data, base)
                st_value, = struct.unpack_from("<Q", data, base + 8)
                if not st_name or not st_value:
                    continue
                end = data.index(b"\x00", stroff + st_name)
                if data[stroff + st_name:end].decode("latin1") == name:
                    for v, sz, o in loads:
                        if v <= st_value < v + sz:
                            return st_value - v + o
                    return st_value
    raise KeyError(f"{name} not found in {path}")


def _uprobe_pmu_type() -> int:
    with open("/sys/bus/event_source/devices/uprobe/type") as f:
        return int(f.read().strip())


def _uprobe_retprobe_bit() -> int:
    try:
        with open("/sys/bus/event_source/devices/uprobe/format/retprobe"
                  ) as f:
            # "config:N"
            return int(f.read().strip().split(":")[1])
    except OSError:
        return 0


def attach_uprobe(prog_fd: int, path: str, offset: int,
                  retprobe: bool = False) -> int:
    """Attach a KPROBE-type program to a userspace probe point
    (file + offset) via the uprobe PMU (no tracefs writes needed)."""
    cfg = (1 << _uprobe_retprobe_bit()) if retprobe else 0
    pathb = path.encode() + b"\x00"
    pbuf = ct.create_string_buffer(pathb, len(pathb))
    # perf_event_attr: type, size, config, sample_period, sample_type,
    # read_format, flags, ..., config1 (uprobe_path), config2 (offset)
    attr = bytearray(112)
    struct.pack_into("<IIQ", attr, 0, _uprobe_pmu_type(), 112, cfg)
    # perf_event_attr layout: type@0 size@4 config@8 sample_period@16
    # sample_type@24 read_format@32 flags@40 wakeup@48 bp_type@52
    # config1@56 (uprobe path ptr) config2@64 (file offset)
    struct.pack_into("<Q", attr, 56, ct.addressof(pbuf))
    struct.pack_into("<Q", attr, 64, offset)
    fd = perf_event_open(bytes(attr), -1, 0)
    import fcntl
    fcntl.ioctl(fd, PERF_EVENT_IOC_SET_BPF, prog_fd)
    fcntl.ioctl(fd, PERF_EVENT_IOC_ENABLE, 0)
    return fd


class ProfilerTracer:
    """Continuous OnCPU profiler, live: a PERF_TYPE_SOFTWARE/CPU_CLOCK
    sampling event per cpu with the assembled BPF program attached
    (PERF_EVENT_IOC_SET_BPF) counting (tgid, ustack, kstack) into the
    stack maps — no tracefs needed. Reference: perf_profiler.bpf.c."""

    BPF_PROG_TYPE_PERF_EVENT = 7
    PERF_COUNT_SW_CPU_CLOCK = 0

    def __init__(self, sample_freq: int = 99):
        from .progs import PROFILER_MAPS, build_profiler
        self.sample_freq = sample_freq
        self.map_fds = {n: map_create(*spec)
                        for n, spec in PROFILER_MAPS.items()}
        self.prog_fd = prog_load(
            self.BPF_PROG_TYPE_PERF_EVENT,
            build_profiler().to_bytes(self.map_fds), log=True)
        self.fds: List[int] = []

    def attach(self) -> None:
        import fcntl
        for cpu in range(os.cpu_count() or 1):
            attr = bytearray(112)
            struct.pack_into("<IIQQ", attr, 0, PERF_TYPE_SOFTWARE, 112,
                             self.PERF_COUNT_SW_CPU_CLOCK,
                             self.sample_freq)   # sample_freq (freq=1)
            struct.pack_into("<Q", attr, 40, 1 << 10)  # flags: freq
            fd = perf_event_open(bytes(attr), -1, cpu)
            fcntl.ioctl(fd, PERF_EVENT_IOC_SET_BPF, self.prog_fd)
            fcntl.ioctl(fd, PERF_EVENT_IOC_ENABLE, 0)
            self.fds.append(fd)

    def close(self) -> None:
        for fd in self.fds:
            os.close(fd)
        os.close(self.prog_fd)
        for fd in self.map_fds.values():
            os.close(fd)


def find_libssl() -> Optional[str]:
    import glob as _glob
    for pat in ("/lib/x86_64-linux-gnu/libssl.so.*",
                "/usr/lib/x86_64-linux-gnu/libssl.so.*",
                "/usr/lib64/libssl.so.*",
                "/usr/lib/x86_64-linux-gnu/libssl.so*"):
        hits = sorted(_glob.glob(pat))
        if hits:
            return hits[0]
    return None


def available() -> bool:
    """Can this process load + attach BPF programs here? True with
    tracefs OR raw-tracepoint attach (container-friendly)."""
    try:
        fd = map_create(1, 8, 8, 4)
        os.close(fd)
    except OSError:
        return False
    return _tracefs() is not None or raw_tracepoint_available()


class SocketTracer:
    """Load the socket-trace programs into the kernel and stream events.

    perf ring reading uses one mmap'd page-set per CPU
    (PERF_EVENT_IOC_SET_BPF on the tracepoint event delivers
    bpf_perf_event_output records into the same buffers)."""

    def __init__(self, with_tls: bool = True, raw: Optional[bool] = None):
        from .progs import (MAPS, SSL_MAPS, build_sys_enter,
                            build_sys_exit, build_ssl_write,
                            build_ssl_read_enter, build_ssl_read_exit)
        if raw is None:
            # raw tracepoints need no tracefs and attach in containers
            # where /sys/kernel/tracing is absent — prefer them
            raw = _tracefs() is None
        self.raw = raw
        self.map_fds: Dict[str, int] = {}
        for name, spec in MAPS.items():
            t_, k_, v_, n_ = spec
            if n_ == 0:  # perf-event arrays size to the cpu count
                n_ = os.cpu_count() or 1
            self.map_fds[name] = map_create(t_, k_, v_, n_)
        ptype = BPF_PROG_TYPE_RAW_TRACEPOINT if raw \
            else BPF_PROG_TYPE_TRACEPOINT
        self.enter_fd = prog_load(
            ptype, build_sys_enter(raw=raw).to_bytes(self.map_fds),
            log=True)
        self.exit_fd = prog_load(
            ptype, build_sys_exit(raw=raw).to_bytes(self.map_fds),
            log=True)
        self.ssl_fds: List[int] = []
        self.libssl = find_libssl() if with_tls else None
        if self.libssl is not None:
            for name, spec in SSL_MAPS.items():
                self.map_fds[name] = map_create(*spec)
            BPF_PROG_TYPE_KPROBE = 2
            # order: write, read_enter, read_exit, read_enter_ex,
            # read_exit_ex — OpenSSL 3 callers (CPython included) go
            # through SSL_write_ex/SSL_read_ex
            for blob in (build_ssl_write().to_bytes(self.map_fds),
                         build_ssl_read_enter().to_bytes(self.map_fds),
                         build_ssl_read_exit().to_bytes(self.map_fds),
                         build_ssl_read_enter(ex=True).to_bytes(
                             self.map_fds),
                         build_ssl_read_exit(ex=True).to_bytes(
                             self.map_fds)):
                self.ssl_fds.append(prog_load(BPF_PROG_TYPE_KPROBE,
                                              blob, log=True))
        self.tp_fds: List[int] = []
        self.rings: List[tuple] = []

    def attach(self) -> None:
        if self.raw:
            self.tp_fds.append(raw_tracepoint_open("sys_enter",
                                                   self.enter_fd))
            self.tp_fds.append(raw_tracepoint_open("sys_exit",
                                                   self.exit_fd))
        else:
            self.tp_fds.append(attach_tracepoint(
                self.enter_fd, "raw_syscalls", "sys_enter"))
            self.tp_fds.append(attach_tracepoint(
                self.exit_fd, "raw_syscalls", "sys_exit"))
        if self.libssl is not None and self.ssl_fds:
            # TLS plaintext capture: uprobes on OpenSSL entry points
            # (reference kernel/openssl.bpf.c)
            for sym, prog_i, ret in (("SSL_write", 0, False),
                                      ("SSL_write_ex", 0, False),
                                      ("SSL_read", 1, False),
                                      ("SSL_read", 2, True),
                                      ("SSL_read_ex", 3, False),
                                      ("SSL_read_ex", 4, True)):
                try:
                    off = elf_sym_file_offset(self.libssl, sym)
                    self.tp_fds.append(attach_uprobe(
                        self.ssl_fds[prog_i], self.libssl, off,
                        retprobe=ret))
                except (OSError, KeyError):
                    pass  # symbol absent / no uprobe PMU: syscalls on
        self._open_rings()

    def _open_rings(self, pages: int = 64) -> None:
        page = mmap.PAGESIZE
        n_cpu = os.cpu_count() or 1
        for cpu in range(n_cpu):
            # PERF_TYPE_SOFTWARE/BPF_OUTPUT per-cpu ring the program
            # writes into via the events map
            PERF_COUNT_SW_BPF_OUTPUT = 10
            attr = struct.pack("<IIQQQ", PERF_TYPE_SOFTWARE, 112,
                               PERF_COUNT_SW_BPF_OUTPUT, 1, 0)
            # sample_type = PERF_SAMPLE_RAW (bit 10 = 0x400)
            attr = attr[:24] + struct.pack("<Q", 0x400) + attr[32:]
            attr = attr.ljust(112, b"\x00")
            fd = perf_event_open(attr, -1, cpu)
            buf = mmap.mmap(fd, page * (pages + 1))
            map_update(self.map_fds["events"],
                       struct.pack("<I", cpu), struct.pack("<I", fd))
            import fcntl
            fcntl.ioctl(fd, PERF_EVENT_IOC_ENABLE, 0)
            self.rings.append((fd, buf, pages))

    def poll(self, on_event: Callable[[bytes], None]) -> int:
        """Drain all per-cpu rings; returns records delivered."""
        page = mmap.PAGESIZE
        n = 0
        for fd, buf, pages in self.rings:
            head = struct.unpack_from("<Q", buf, 1024)[0]   # data_head
            tail = struct.unpack_from("<Q", buf, 1032)[0]   # data_tail
            size = page * pages
            while tail < head:
                off = page + (tail % size)
                hdr = bytes(buf[off: off + 8]) if off + 8 <= page + size \
                    else bytes(buf[off: page + size]) + \
                    bytes(buf[page: page + 8 - (page + size - off)])
                etype, _misc, esize = struct.unpack("<IHH", hdr)
                body_off = (off + 8 - page) % size + page
                end = body_off + (esize - 8)
                if end <= page + size:
                    rec = bytes(buf[body_off: end])
                else:  # record wraps the ring
                    rec = bytes(buf[body_off: page + size]) + \
                        bytes(buf[page: page + end - (page + size)])
                if etype == 9:  # PERF_RECORD_SAMPLE: u32 size + raw
                    raw_size = struct.unpack_from("<I", rec, 0)[0]
                    on_event(rec[4:4 + raw_size])
                    n += 1
                tail += esize
            struct.pack_into("<Q", buf, 1032, head)
        return n

    def close(self) -> None:
        for fd, buf, _ in self.rings:
            buf.close()
            os.close(fd)
        for fd in self.tp_fds:
            os.close(fd)
        for fd in (self.enter_fd, self.exit_fd):
            os.close(fd)
        for fd in self.map_fds.values():
            os.close(fd)
