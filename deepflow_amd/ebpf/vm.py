"""Userspace eBPF interpreter — executes the exact bytecode the loader
would hand to the kernel, with emulated helpers and maps.

This is the CI story for an image with no BPF toolchain and no BPF
privileges: the assembled programs (progs.py) run here against synthetic
syscall streams, producing the same perf events a kernel would, which
then flow through the real userspace runtime into the agent's FlowMap/L7
parsers (tests/test_ebpf.py). Caller-saved registers are poisoned across
helper calls so clobber bugs fail loudly instead of passing by luck.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Optional, Tuple

from . import insn as I

M64 = (1 << 64) - 1
POISON = 0xDEAD00000000BEEF

STACK_SIZE = 512
STACK_BASE = 0x7FF0_0000_0000
CTX_BASE = 0x7FE0_0000_0000
USER_BASE = 0x7FD0_0000_0000
VAL_BASE = 0x7FC0_0000_0000
MAP_HANDLE_BASE = 0x6D00_0000


def _s64(v: int) -> int:
    v &= M64
    return v - (1 << 64) if v >= (1 << 63) else v


class BpfMap:
    def __init__(self, mtype: int, key_size: int, value_size: int,
                 max_entries: int):
        self.mtype = mtype
        self.key_size = key_size
        self.value_size = value_size
        self.max_entries = max_entries
        self.data: Dict[bytes, bytearray] = {}
        self.regions: Dict[bytes, int] = {}  # key -> value base addr


class Vm:
    def __init__(self, maps_spec: Dict[str, tuple]):
        self.maps: Dict[str, BpfMap] = {
            name: BpfMap(*spec) for name, spec in maps_spec.items()}
        self.map_handles: Dict[int, str] = {}
        for i, name in enumerate(self.maps):
            self.map_handles[MAP_HANDLE_BASE + i] = name
        self.handle_for = {v: k for k, v in self.map_handles.items()}
        # flat regions: addr -> (bytearray, base)
        self.regions: List[Tuple[int, bytearray]] = []
        self._next_val = VAL_BASE
        self.user_mem: Dict[int, bytes] = {}   # user addr -> bytes
        self.kernel_mem: Dict[int, bytes] = {}  # kernel addr -> bytes
        self.events: List[bytes] = []          # perf_event_output captures
        self.clock = 1_700_000_000_000_000_000
        self.pid_tgid = (1234 << 32) | 1234
        self.cpu = 0
        self.stackid_seq: Dict[int, int] = {}  # flags -> next id
        self.helper_calls = 0

    # ------------------------------------------------------------ memory
    def add_region(self, base: int, buf: bytearray) -> None:
        self.regions.append((base, buf))

    def _find(self, addr: int, size: int):
        for base, buf in self.regions:
            if base <= addr and addr + size <= base + len(buf):
                return buf, addr - base
        raise RuntimeError(f"bad memory access addr=0x{addr:x} size={size}")

    def load(self, addr: int, size: int) -> int:
        buf, off = self._find(addr, size)
        return int.from_bytes(buf[off:off + size], "little")

    def store(self, addr: int, size: int, val: int) -> None:
        buf, off = self._find(addr, size)
        buf[off:off + size] = (val & ((1 << (8 * size)) - 1)).to_bytes(
            size, "little")

    def read_bytes(self, addr: int, size: int) -> bytes:
        buf, off = self._find(addr, size)
        return bytes(buf[off:off + size])

    def _val_region(self, m: BpfMap, key: bytes) -> int:
        if key not in m.regions:
            m.data[key] = bytearray(m.value_size)
            base = self._next_val
            self._next_val += (m.value_size + 63) & ~63
            m.regions[key] = base
            self.add_region(base, m.data[key])
        return m.regions[key]

    # ----------------------------------------------------------- helpers
    def _map_by_handle(self, h: int) -> BpfMap:
        name = self.map_handles.get(h)
        if name is None:
            raise RuntimeError(f"bad map handle 0x{h:x}")
        return self.maps[name]

    def _helper(self, hid: int, r: List[int]) -> int:
        self.helper_calls += 1
        if hid == I.H_MAP_LOOKUP:
            m = self._map_by_handle(r[1])
            key = self.read_bytes(r[2], m.key_size)
            if m.mtype == 6:  # percpu array: always present
                return self._val_region(m, key)
            if key not in m.data:
                return 0
            return self._val_region(m, key)
        if hid == I.H_MAP_UPDATE:
            m = self._map_by_handle(r[1])
            key = self.read_bytes(r[2], m.key_size)
            val = self.read_bytes(r[3], m.value_size)
            base = self._val_region(m, key)
            m.data[key][:] = val
            return 0
        if hid == I.H_MAP_DELETE:
            m = self._map_by_handle(r[1])
            key = self.read_bytes(r[2], m.key_size)
            # region stays mapped (kernel defers free too); entry gone
            if key in m.data:
                m.regions.pop(key, None)
                buf = m.data.pop(key)
                # keep the region list entry so stale pointers read zeros
                buf[:] = b"\x00" * len(buf)
            return 0
        if hid == I.H_KTIME_GET_NS:
            return self.clock
        if hid == I.H_GET_SMP_PROC_ID:
            return self.cpu
        if hid == I.H_GET_PID_TGID:
            return self.pid_tgid
        if hid == I.H_PERF_EVENT_OUTPUT:
            size = r[5]
            data = self.read_bytes(r[4], size)
            self.events.append(data)
            return 0
        if hid in (I.H_PROBE_READ_USER, I.H_PROBE_READ_KERNEL):
            dst, size, src = r[1], r[2], r[3]
            pool = self.user_mem if hid == I.H_PROBE_READ_USER \
                else self.kernel_mem
            blob = pool.get(src)
            if blob is None:
                # offset into a registered base block (pt_regs fields)
                for base, b in pool.items():
                    if base <= src < base + len(b):
                        blob = b[src - base:]
                        break
            if blob is None:
                return -14  # -EFAULT
            chunk = blob[:size].ljust(size, b"\x00")
            buf, off = self._find(dst, size)
            buf[off:off + size] = chunk
            return 0
        if hid == I.H_GET_STACKID:
            flags = r[3]
            nxt = self.stackid_seq.get(flags, 1)
            self.stackid_seq[flags] = nxt + 1
            return nxt
        raise RuntimeError(f"unimplemented helper {hid}")

    # --------------------------------------------------------------- run
    def run(self, asm: "I.Asm", ctx: bytes, max_insns: int = 1_000_000) -> int:
        insns = asm.assemble()
        # slot index -> insn index (jump offsets are in slots)
        slot_of = []
        slots = {}
        s = 0
        for idx, ins in enumerate(insns):
            slots[s] = idx
            slot_of.append(s)
            s += 2 if ins.imm64_hi is not None else 1
        end_slot = s
        ctx_buf = bytearray(ctx)
        stack = bytearray(STACK_SIZE)
        self.add_region(CTX_BASE, ctx_buf)
        self.add_region(STACK_BASE - STACK_SIZE, stack)
        r = [0] * 11
        r[1] = CTX_BASE
        r[10] = STACK_BASE
        pc_slot = 0
        steps = 0
        try:
            while True:
                steps += 1
                if steps > max_insns:
                    raise RuntimeError("instruction budget exceeded")
                idx = slots.get(pc_slot)
                if idx is None:
                    raise RuntimeError(f"jump into ld_imm64 hole {pc_slot}")
                ins = insns[idx]
                width = 2 if ins.imm64_hi is not None else 1
                nxt = pc_slot + width
                cls = ins.op & 0x07
                if ins.imm64_hi is not None:
                    if ins.map_ref is not None:
                        r[ins.dst] = self.handle_for[ins.map_ref]
                    else:
                        r[ins.dst] = (ins.imm & 0xFFFFFFFF) | \
                            (ins.imm64_hi << 32)
                elif cls in (I.BPF_ALU64, I.BPF_ALU):
                    op = ins.op & 0xF0
                    srcv = r[ins.src] if (ins.op & 0x08) else ins.imm & M64
                    if cls == I.BPF_ALU:
                        srcv &= 0xFFFFFFFF
                    d = r[ins.dst]
                    if op == I.BPF_MOV:
                        d = srcv
                    elif op == I.BPF_ADD:
                        d = (d + srcv) & M64
                    elif op == I.BPF_SUB:
                        d = (d - srcv) & M64
                    elif op == I.BPF_MUL:
                        d = (d * srcv) & M64
                    elif op == I.BPF_OR:
                        d |= srcv
                    elif op == I.BPF_AND:
                        d &= srcv
                    elif op == I.BPF_XOR:
                        d ^= srcv
                    elif op == I.BPF_LSH:
                        d = (d << (srcv & 63)) & M64
                    elif op == I.BPF_RSH:
                        d = (d & M64) >> (srcv & 63)
                    elif op == I.BPF_ARSH:
                        d = (_s64(d) >> (srcv & 63)) & M64
                    elif op == I.BPF_DIV:
                        d = (d // srcv) & M64 if srcv else 0
                    elif op == I.BPF_MOD:
                        d = (d % srcv) & M64 if srcv else d
                    elif op == I.BPF_NEG:
                        d = (-d) & M64
                    else:
                        raise RuntimeError(f"alu op 0x{op:x}")
                    if cls == I.BPF_ALU:
                        d &= 0xFFFFFFFF
                    r[ins.dst] = d
                elif cls == I.BPF_LDX:
                    size = {I.BPF_B: 1, I.BPF_H: 2, I.BPF_W: 4,
                            I.BPF_DW: 8}[ins.op & 0x18]
                    r[ins.dst] = self.load((r[ins.src] + ins.off) & M64, size)
                elif cls == I.BPF_STX:
                    size = {I.BPF_B: 1, I.BPF_H: 2, I.BPF_W: 4,
                            I.BPF_DW: 8}[ins.op & 0x18]
                    self.store((r[ins.dst] + ins.off) & M64, size,
                               r[ins.src])
                elif cls == I.BPF_ST:
                    size = {I.BPF_B: 1, I.BPF_H: 2, I.BPF_W: 4,
                            I.BPF_DW: 8}[ins.op & 0x18]
                    self.store((r[ins.dst] + ins.off) & M64, size, ins.imm)
                elif cls == I.BPF_JMP:
                    op = ins.op & 0xF0
                    if op == I.BPF_CALL:
                        r[0] = self._helper(ins.imm, r) & M64
                        for i in range(1, 6):
                            r[i] = POISON  # caller-saved die at calls
                        pc_slot = nxt
                        continue
                    if op == I.BPF_EXIT:
                        return r[0]
                    srcv = r[ins.src] if (ins.op & 0x08) else ins.imm & M64
                    d = r[ins.dst]
                    take = {
                        I.BPF_JA: lambda: True,
                        I.BPF_JEQ: lambda: d == srcv,
                        I.BPF_JNE: lambda: d != srcv,
                        I.BPF_JGT: lambda: d > srcv,
                        I.BPF_JGE: lambda: d >= srcv,
                        I.BPF_JLT: lambda: d < srcv,
                        I.BPF_JLE: lambda: d <= srcv,
                        I.BPF_JSGT: lambda: _s64(d) > _s64(srcv),
                        I.BPF_JSGE: lambda: _s64(d) >= _s64(srcv),
                        I.BPF_JSLT: lambda: _s64(d) < _s64(srcv),
                        I.BPF_JSLE: lambda: _s64(d) <= _s64(srcv),
                        I.BPF_JSET: lambda: bool(d & srcv),
                    }[op]()
                    pc_slot = nxt + ins.off if take else nxt
                    continue
                else:
                    raise RuntimeError(f"unhandled insn class 0x{cls:x}")
                pc_slot = nxt
                if pc_slot >= end_slot:
                    raise RuntimeError("fell off program end")
        finally:
            # drop the per-run ctx/stack regions
            self.regions = [x for x in self.regions
                            if x[0] not in (CTX_BASE,
                                            STACK_BASE - STACK_SIZE)]


# ----------------------------------------------------------------- harness
def sys_enter_ctx(syscall: int, args: List[int]) -> bytes:
    a = (args + [0] * 6)[:6]
    return struct.pack("<8xq6Q", syscall, *[x & M64 for x in a])


def sys_exit_ctx(syscall: int, ret: int) -> bytes:
    return struct.pack("<8xqq", syscall, ret)


class SyscallSim:
    """Drive the socket-trace programs with synthetic syscalls."""

    def __init__(self):
        from .progs import MAPS, build_sys_enter, build_sys_exit
        self.vm = Vm(MAPS)
        self.enter = build_sys_enter()
        self.exit = build_sys_exit()
        self._next_user = USER_BASE

    def syscall(self, tgid: int, pid: int, syscall: int, fd: int,
                payload: bytes, ret: Optional[int] = None) -> None:
        """One traced syscall: enter with a user buffer, exit with ret."""
        self.vm.pid_tgid = ((tgid << 32) | pid) & M64
        ubuf = self._next_user
        self._next_user += (len(payload) + 4095) & ~4095 or 4096
        self.vm.user_mem[ubuf] = payload
        self.vm.run(self.enter, sys_enter_ctx(syscall, [fd, ubuf,
                                                        len(payload)]))
        self.vm.clock += 1000
        self.vm.run(self.exit, sys_exit_ctx(
            syscall, len(payload) if ret is None else ret))
        self.vm.clock += 1000

    def events(self) -> List[bytes]:
        return self.vm.events


def raw_tp_sys_enter_ctx(vm: "Vm", syscall: int, fd: int,
                         buf: int, count: int,
                         regs_addr: int = 0x6B00_0000_0000) -> bytes:
    """BPF_RAW_TRACEPOINT sys_enter ctx: {pt_regs*, id}; the pt_regs
    block is registered as VM kernel memory for probe_read_kernel."""
    regs = [0] * 21
    regs[14] = fd & M64     # rdi
    regs[13] = buf & M64    # rsi
    regs[12] = count & M64  # rdx
    vm.kernel_mem[regs_addr] = struct.pack("<21Q", *regs)
    return struct.pack("<Qq", regs_addr, syscall)


def raw_tp_sys_exit_ctx(ret: int) -> bytes:
    return struct.pack("<Qq", 0, ret)


class RawSyscallSim(SyscallSim):
    """SyscallSim over the RAW-tracepoint program variants (the ones
    BPF_RAW_TRACEPOINT_OPEN attaches in production)."""

    def __init__(self):
        from .progs import MAPS, build_sys_enter, build_sys_exit
        self.vm = Vm(MAPS)
        self.enter = build_sys_enter(raw=True)
        self.exit = build_sys_exit(raw=True)
        self._next_user = USER_BASE

    def syscall(self, tgid: int, pid: int, syscall: int, fd: int,
                payload: bytes, ret: Optional[int] = None) -> None:
        self.vm.pid_tgid = ((tgid << 32) | pid) & M64
        ubuf = self._next_user
        self._next_user += (len(payload) + 4095) & ~4095 or 4096
        self.vm.user_mem[ubuf] = payload
        self.vm.run(self.enter, raw_tp_sys_enter_ctx(
            self.vm, syscall, fd, ubuf, len(payload)))
        self.vm.clock += 1000
        self.vm.run(self.exit, raw_tp_sys_exit_ctx(
            len(payload) if ret is None else ret))
        self.vm.clock += 1000


def pt_regs_ctx(rdi: int = 0, rsi: int = 0, rdx: int = 0,
                rax: int = 0) -> bytes:
    """x86_64 pt_regs blob for uprobe programs (ptrace.h order)."""
    regs = [0] * 21
    regs[10] = rax & M64
    regs[12] = rdx & M64
    regs[13] = rsi & M64
    regs[14] = rdi & M64
    return struct.pack("<21Q", *regs)


class SslSim:
    """Drive the OpenSSL uprobe programs (TLS plaintext capture) with
    synthetic SSL_read/SSL_write calls."""

    def __init__(self):
        from .progs import (MAPS, SSL_MAPS, build_ssl_write,
                            build_ssl_read_enter, build_ssl_read_exit)
        allmaps = dict(MAPS)
        allmaps.update(SSL_MAPS)
        self.vm = Vm(allmaps)
        self.w = build_ssl_write()
        self.r_enter = build_ssl_read_enter()
        self.r_exit = build_ssl_read_exit()
        self._next_user = USER_BASE

    def _ubuf(self, payload: bytes) -> int:
        ubuf = self._next_user
        self._next_user += (len(payload) + 4095) & ~4095 or 4096
        self.vm.user_mem[ubuf] = payload
        return ubuf

    def ssl_write(self, tgid: int, pid: int, payload: bytes) -> None:
        self.vm.pid_tgid = ((tgid << 32) | pid) & M64
        ubuf = self._ubuf(payload)
        self.vm.run(self.w, pt_regs_ctx(rsi=ubuf, rdx=len(payload)))
        self.vm.clock += 1000

    def ssl_read(self, tgid: int, pid: int, payload: bytes,
                 ret: Optional[int] = None) -> None:
        self.vm.pid_tgid = ((tgid << 32) | pid) & M64
        ubuf = self._ubuf(payload)
        self.vm.run(self.r_enter, pt_regs_ctx(rsi=ubuf))
        self.vm.run(self.r_exit, pt_regs_ctx(
            rsi=ubuf, rax=len(payload) if ret is None else ret))
        self.vm.clock += 1000

    def events(self) -> List[bytes]:
        return self.vm.events
