"""BPF program builders: socket tracer + on-CPU profiler.

Design vs the reference (agent/src/ebpf/kernel/socket_trace.bpf.c):
the reference attaches 79 per-syscall probes and digs struct sock out of
the fd table with CO-RE offsets. This build attaches TWO programs to the
stable raw_syscalls tracepoints (sys_enter/sys_exit — fixed ABI:
{trace_entry[8]; long id; long args[6]} / {...; long ret}) and dispatches
on the syscall id in-kernel; the socket 4-tuple is resolved in userspace
from (tgid, fd) via /proc (runtime.py), which removes every kernel-struct
offset dependency — no BTF, no CO-RE, loads unmodified on any kernel
with raw_syscalls tracepoints.

Event flow: sys_enter records (syscall, fd, buf) keyed by pid_tgid;
sys_exit reads the result, copies up to CAP_LEN payload bytes from the
user buffer, runs the generated protocol-inference matcher (inference.py
SPEC) with a per-socket verdict cache, assigns a syscall trace id linking
an ingress read to the egress writes that follow it on the same thread
(the reference's syscall_trace_id join key), and perf_event_outputs one
SkEvent record.
"""
from __future__ import annotations

import struct

from .insn import (Asm, BPF_ADD, BPF_AND, BPF_B, BPF_DW, BPF_H, BPF_JEQ,
                   BPF_JGE, BPF_JGT, BPF_JLE, BPF_JLT, BPF_JNE, BPF_JSGE,
                   BPF_JSLE, BPF_LSH, BPF_MOV, BPF_OR, BPF_RSH, BPF_SUB,
                   BPF_W, H_GET_PID_TGID, H_GET_SMP_PROC_ID, H_GET_STACKID,
                   H_KTIME_GET_NS, H_MAP_DELETE, H_MAP_LOOKUP, H_MAP_UPDATE,
                   H_PERF_EVENT_OUTPUT, H_PROBE_READ_KERNEL,
                   H_PROBE_READ_USER, R0, R1, R2, R3,
                   R4, R5, R6, R7, R8, R9, R10)
from .inference import SPEC

# ---------------------------------------------------------------- layout
CAP_LEN = 192
EV_HDR = 48
EV_SIZE = EV_HDR + CAP_LEN

# SkEvent offsets
EV_TS = 0
EV_TGID = 8
EV_PID = 12
EV_FD = 16
EV_LEN = 20
EV_CAP = 24
EV_DIR = 28
EV_PROTO = 29
EV_SYSCALL = 30
EV_TRACE = 32
EV_SOCKKEY = 40
EV_PAYLOAD = 48

SK_EVENT_FMT = "<QIIIIIBBHQQ"  # ts,tgid,pid,fd,len,cap,dir,proto,sc,trace,sockkey
assert struct.calcsize(SK_EVENT_FMT) == EV_HDR

# x86_64 syscall ids
SC_READ, SC_WRITE, SC_CLOSE = 0, 1, 3
SC_SENDTO, SC_RECVFROM = 44, 45
INGRESS = {SC_READ, SC_RECVFROM}
EGRESS = {SC_WRITE, SC_SENDTO}
TRACED = sorted(INGRESS | EGRESS)

# map specs: name -> (type, key_size, value_size, max_entries)
BPF_MAP_TYPE_HASH = 1
BPF_MAP_TYPE_PERF_EVENT_ARRAY = 4
BPF_MAP_TYPE_PERCPU_ARRAY = 6
BPF_MAP_TYPE_STACK_TRACE = 7

MAPS = {
    # pid_tgid -> {u64 syscall, u64 fd, u64 buf}
    "active": (BPF_MAP_TYPE_HASH, 8, 24, 65536),
    # (tgid<<32|fd) -> {u32 proto, u32 pad, u64 last_trace}
    "sockinfo": (BPF_MAP_TYPE_HASH, 8, 16, 131072),
    # per-cpu event scratch
    "scratch": (BPF_MAP_TYPE_PERCPU_ARRAY, 4, EV_SIZE, 1),
    # per-cpu trace-id counter
    "seq": (BPF_MAP_TYPE_PERCPU_ARRAY, 4, 8, 1),
    "events": (BPF_MAP_TYPE_PERF_EVENT_ARRAY, 4, 4, 0),
}

PROFILER_MAPS = {
    "stacks": (BPF_MAP_TYPE_STACK_TRACE, 4, 127 * 8, 16384),
    # {u32 tgid, s32 ustack, s32 kstack, u32 pad} -> u64 count
    "counts": (BPF_MAP_TYPE_HASH, 16, 8, 65536),
}


def build_sys_enter(raw: bool = False) -> Asm:
    """raw=True targets BPF_RAW_TRACEPOINT_OPEN (no tracefs needed):
    ctx = {args[0]=struct pt_regs*, args[1]=syscall id}; syscall args
    read from pt_regs with bpf_probe_read_kernel. raw=False is the
    classic tracefs sys_enter layout (id@8, args@16)."""
    a = Asm()
    a.mov64(R6, R1)                       # r6 = ctx
    a.ldx(BPF_DW, R8, R6, 8)              # r8 = syscall id (both layouts)
    for sc in TRACED:
        a.jmp_imm(BPF_JEQ, R8, sc, "trace")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("trace")
    a.call(H_GET_PID_TGID)
    a.mov64(R7, R0)
    a.stx(BPF_DW, R10, -8, R7)            # key = pid_tgid
    # value {syscall, fd, buf} at fp-32
    a.stx(BPF_DW, R10, -32, R8)
    if raw:
        a.ldx(BPF_DW, R9, R6, 0)          # struct pt_regs *
        # fd = regs->rdi -> fp-24
        a.mov64(R1, R10)
        a.alu64_imm(BPF_ADD, R1, -24)
        a.mov64_imm(R2, 8)
        a.mov64(R3, R9)
        a.alu64_imm(BPF_ADD, R3, 112)     # offsetof(pt_regs, rdi)
        a.call(H_PROBE_READ_KERNEL)
        # buf = regs->rsi -> fp-16
        a.mov64(R1, R10)
        a.alu64_imm(BPF_ADD, R1, -16)
        a.mov64_imm(R2, 8)
        a.mov64(R3, R9)
        a.alu64_imm(BPF_ADD, R3, 104)     # offsetof(pt_regs, rsi)
        a.call(H_PROBE_READ_KERNEL)
    else:
        a.ldx(BPF_DW, R2, R6, 16)         # args[0] = fd
        a.stx(BPF_DW, R10, -24, R2)
        a.ldx(BPF_DW, R2, R6, 24)         # args[1] = buf
        a.stx(BPF_DW, R10, -16, R2)
    a.ld_map_fd(R1, "active")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.mov64(R3, R10)
    a.alu64_imm(BPF_ADD, R3, -32)
    a.mov64_imm(R4, 0)                    # BPF_ANY
    a.call(H_MAP_UPDATE)
    a.mov64_imm(R0, 0)
    a.exit()
    return a


def _emit_inference(a: Asm) -> None:
    """Generated matcher over the captured payload.

    In: r9 = event buffer (payload at EV_PAYLOAD), r8 = cap_len,
        r7 = full syscall len. Out: proto in r0 (0 = unknown).
    Each SPEC alternative becomes a straight-line compare chain; a failed
    check jumps to the next alternative.
    """
    for pi, (proto, name, alts) in enumerate(SPEC):
        for ai, alt in enumerate(alts):
            nxt = f"alt_{pi}_{ai + 1}" if ai + 1 < len(alts) \
                else f"proto_{pi + 1}"
            a.label(f"alt_{pi}_{ai}" if ai else f"proto_{pi}")
            # implicit: every byte check needs cap > idx
            max_idx = 0
            for r in alt:
                if r[0] == "prefix":
                    max_idx = max(max_idx, len(r[1]) - 1)
                elif r[0] == "prefix_at":
                    max_idx = max(max_idx, r[1] + len(r[2]) - 1)
                elif r[0] in ("byte_in", "byte_eq", "byte_range",
                              "byte_range_or_digit"):
                    max_idx = max(max_idx, r[1])
                elif r[0] == "min_len":
                    max_idx = max(max_idx, r[1] - 1)
                elif r[0] == "u32be0_lenm4":
                    max_idx = max(max_idx, 3)
            a.jmp_imm(BPF_JLT, R8, max_idx + 1, nxt)
            for r in alt:
                kind = r[0]
                if kind == "min_len":
                    continue  # covered by the max_idx guard
                if kind == "prefix":
                    for i, ch in enumerate(r[1]):
                        a.ldx(BPF_B, R0, R9, EV_PAYLOAD + i)
                        a.jmp_imm(BPF_JNE, R0, ch, nxt)
                elif kind == "prefix_at":
                    for i, ch in enumerate(r[2]):
                        a.ldx(BPF_B, R0, R9, EV_PAYLOAD + r[1] + i)
                        a.jmp_imm(BPF_JNE, R0, ch, nxt)
                elif kind == "byte_eq":
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + r[1])
                    a.jmp_imm(BPF_JNE, R0, r[2], nxt)
                elif kind == "byte_in":
                    ok = f"in_ok_{pi}_{ai}_{r[1]}"
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + r[1])
                    for ch in r[2]:
                        a.jmp_imm(BPF_JEQ, R0, ch, ok)
                    a.jmp(nxt)
                    a.label(ok)
                elif kind == "byte_range":
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + r[1])
                    a.jmp_imm(BPF_JLT, R0, r[2], nxt)
                    a.jmp_imm(BPF_JGT, R0, r[3], nxt)
                elif kind == "byte_range_or_digit":
                    ok = f"rd_ok_{pi}_{ai}"
                    dig = f"rd_dig_{pi}_{ai}"
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + r[1])
                    a.jmp_imm(BPF_JLT, R0, 0x30, dig)
                    a.jmp_imm(BPF_JLE, R0, 0x39, ok)
                    a.label(dig)
                    for ch in b"+-OE":
                        a.jmp_imm(BPF_JEQ, R0, ch, ok)
                    a.jmp(nxt)
                    a.label(ok)
                elif kind == "u32be0_lenm4":
                    # BE u32 at payload[0] == full_len - 4
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + 0)
                    a.alu64_imm(BPF_LSH, R0, 24)
                    a.mov64(R5, R0)
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + 1)
                    a.alu64_imm(BPF_LSH, R0, 16)
                    a.alu64(BPF_OR, R5, R0)
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + 2)
                    a.alu64_imm(BPF_LSH, R0, 8)
                    a.alu64(BPF_OR, R5, R0)
                    a.ldx(BPF_B, R0, R9, EV_PAYLOAD + 3)
                    a.alu64(BPF_OR, R5, R0)
                    a.mov64(R0, R7)
                    a.alu64_imm(BPF_SUB, R0, 4)
                    a.jmp_reg(BPF_JNE, R5, R0, nxt)
                else:
                    raise ValueError(f"no BPF codegen for rule {kind}")
            a.mov64_imm(R0, proto)
            a.jmp("infer_done")
    a.label(f"proto_{len(SPEC)}")
    a.mov64_imm(R0, 0)
    a.label("infer_done")


def build_sys_exit(raw: bool = False) -> Asm:
    a = Asm()
    a.mov64(R6, R1)                       # r6 = ctx
    a.call(H_GET_PID_TGID)
    a.mov64(R7, R0)
    a.stx(BPF_DW, R10, -8, R7)
    a.ld_map_fd(R1, "active")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JNE, R0, 0, "have_entry")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("have_entry")
    a.mov64(R8, R0)
    a.ldx(BPF_DW, R2, R8, 0)              # syscall id
    a.stx(BPF_DW, R10, -16, R2)
    a.ldx(BPF_DW, R2, R8, 8)              # fd
    a.stx(BPF_DW, R10, -24, R2)
    a.ldx(BPF_DW, R2, R8, 16)             # buf
    a.stx(BPF_DW, R10, -32, R2)
    # delete the active entry (key still at fp-8)
    a.ld_map_fd(R1, "active")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.call(H_MAP_DELETE)
    # ret <= 0 -> done (raw tp: {pt_regs*, ret}; tracefs tp: {id, ret})
    a.ldx(BPF_DW, R2, R6, 8 if raw else 16)
    a.jmp_imm(BPF_JSGE, R2, 1, "have_ret")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("have_ret")
    a.stx(BPF_DW, R10, -40, R2)           # ret (full len)
    # r9 = scratch event buffer
    a.st_imm(BPF_W, R10, -44, 0)
    a.ld_map_fd(R1, "scratch")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -44)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JNE, R0, 0, "have_buf")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("have_buf")
    a.mov64(R9, R0)
    # ---- header
    a.call(H_KTIME_GET_NS)
    a.stx(BPF_DW, R9, EV_TS, R0)
    a.mov64(R1, R7)
    a.alu64_imm(BPF_RSH, R1, 32)
    a.stx(BPF_W, R9, EV_TGID, R1)
    a.mov64(R2, R7)
    a.alu64_imm(BPF_LSH, R2, 32)
    a.alu64_imm(BPF_RSH, R2, 32)
    a.stx(BPF_W, R9, EV_PID, R2)
    a.ldx(BPF_DW, R2, R10, -24)           # fd
    a.stx(BPF_W, R9, EV_FD, R2)
    # sockkey = tgid<<32 | fd
    a.alu64_imm(BPF_LSH, R1, 32)
    a.alu64(BPF_OR, R1, R2)
    a.stx(BPF_DW, R9, EV_SOCKKEY, R1)
    a.stx(BPF_DW, R10, -56, R1)           # sockinfo key
    a.ldx(BPF_DW, R2, R10, -40)           # ret
    a.stx(BPF_W, R9, EV_LEN, R2)
    a.ldx(BPF_DW, R2, R10, -16)           # syscall id
    a.stx(BPF_H, R9, EV_SYSCALL, R2)
    # direction: ingress (read/recvfrom) = 1
    a.mov64_imm(R3, 0)
    for sc in sorted(INGRESS):
        a.jmp_imm(BPF_JNE, R2, sc, f"not_in_{sc}")
        a.mov64_imm(R3, 1)
        a.label(f"not_in_{sc}")
    a.stx(BPF_B, R9, EV_DIR, R3)
    a.stx(BPF_DW, R10, -64, R3)           # direction
    # ---- capture payload: cap = min(ret, CAP_LEN)
    a.ldx(BPF_DW, R8, R10, -40)
    a.jmp_imm(BPF_JLE, R8, CAP_LEN, "cap_ok")
    a.mov64_imm(R8, CAP_LEN)
    a.label("cap_ok")
    a.stx(BPF_W, R9, EV_CAP, R8)
    a.mov64(R1, R9)
    a.alu64_imm(BPF_ADD, R1, EV_PAYLOAD)
    a.mov64(R2, R8)
    a.ldx(BPF_DW, R3, R10, -32)           # user buf
    a.call(H_PROBE_READ_USER)
    # ---- trace id: per-cpu counter; ingress mints, egress inherits
    a.st_imm(BPF_W, R10, -44, 0)
    a.ld_map_fd(R1, "seq")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -44)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JEQ, R0, 0, "no_seq")
    a.stx(BPF_DW, R10, -112, R0)          # counter ptr (r0-r5 die at call)
    a.call(H_GET_SMP_PROC_ID)
    a.ldx(BPF_DW, R1, R10, -112)
    a.ldx(BPF_DW, R2, R1, 0)
    a.alu64_imm(BPF_ADD, R2, 1)
    a.stx(BPF_DW, R1, 0, R2)
    a.alu64_imm(BPF_LSH, R0, 48)
    a.alu64(BPF_OR, R2, R0)
    a.jmp("have_tid")
    a.label("no_seq")
    a.mov64_imm(R2, 0)
    a.label("have_tid")
    a.stx(BPF_DW, R10, -72, R2)           # fresh trace id
    # ---- sockinfo: cached proto verdict + last ingress trace id
    a.ld_map_fd(R1, "sockinfo")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -56)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JEQ, R0, 0, "infer_fresh")
    a.mov64(R7, R0)                       # sockinfo value ptr
    a.ldx(BPF_W, R1, R7, 0)               # cached proto
    a.jmp_imm(BPF_JEQ, R1, 0, "infer_fresh2")
    # cached: proto = cached; trace: ingress -> store fresh, egress -> load
    a.stx(BPF_B, R9, EV_PROTO, R1)
    a.ldx(BPF_DW, R3, R10, -64)
    a.jmp_imm(BPF_JEQ, R3, 0, "egress_tid")
    a.ldx(BPF_DW, R2, R10, -72)
    a.stx(BPF_DW, R7, 8, R2)              # last_trace = fresh
    a.stx(BPF_DW, R9, EV_TRACE, R2)
    a.jmp("emit")
    a.label("egress_tid")
    a.ldx(BPF_DW, R2, R7, 8)              # inherit last ingress id
    a.stx(BPF_DW, R9, EV_TRACE, R2)
    a.jmp("emit")
    # ---- inference path (no/unknown verdict yet)
    a.label("infer_fresh")
    a.mov64_imm(R7, 0)
    a.label("infer_fresh2")
    a.ldx(BPF_DW, R5, R10, -40)           # full len for framed checks
    a.mov64(R1, R5)                       # keep in r7? r7 may hold ptr
    a.stx(BPF_DW, R10, -80, R7)           # save sockinfo ptr (may be 0)
    a.mov64(R7, R1)                       # r7 = full len (matcher input)
    _emit_inference(a)                    # -> r0 = proto
    a.stx(BPF_B, R9, EV_PROTO, R0)
    a.stx(BPF_DW, R10, -88, R0)
    # trace id handling (same as cached path)
    a.ldx(BPF_DW, R2, R10, -72)
    a.stx(BPF_DW, R9, EV_TRACE, R2)
    # write sockinfo {proto, pad, last_trace} at fp-104..  (16B at -104)
    a.ldx(BPF_DW, R1, R10, -88)
    a.stx(BPF_W, R10, -104, R1)
    a.st_imm(BPF_W, R10, -100, 0)
    a.stx(BPF_DW, R10, -96, R2)
    a.ld_map_fd(R1, "sockinfo")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -56)
    a.mov64(R3, R10)
    a.alu64_imm(BPF_ADD, R3, -104)
    a.mov64_imm(R4, 0)
    a.call(H_MAP_UPDATE)
    # ---- emit
    a.label("emit")
    a.mov64(R1, R6)
    a.ld_map_fd(R2, "events")
    a.ld_imm64(R3, 0xFFFFFFFF)            # BPF_F_CURRENT_CPU
    a.mov64(R4, R9)
    a.ldx(BPF_W, R5, R9, EV_CAP)
    # re-bound after the map round-trip: the verifier forgets the
    # earlier clamp once the value passes through the per-cpu scratch
    a.alu64_imm(BPF_AND, R5, 0xFF)
    a.jmp_imm(BPF_JLE, R5, CAP_LEN, "evsz_ok")
    a.mov64_imm(R5, CAP_LEN)
    a.label("evsz_ok")
    a.alu64_imm(BPF_ADD, R5, EV_HDR)
    a.call(H_PERF_EVENT_OUTPUT)
    a.mov64_imm(R0, 0)
    a.exit()
    return a


# ---------------------------------------------------------------- TLS
# OpenSSL uprobes (reference: kernel/openssl.bpf.c): plaintext capture
# for TLS connections at SSL_write / SSL_read, where the syscall tracer
# only sees ciphertext. Zero struct offsets are read from the SSL
# object (version-proof); the socket 4-tuple is paired in userspace
# with the thread's most recent socket syscall (runtime.record_events).
# x86_64 pt_regs offsets (ptrace.h order):
PT_RAX, PT_RDX, PT_RSI, PT_RDI = 80, 96, 104, 112
SC_SSL_READ, SC_SSL_WRITE = 0xFFF0, 0xFFF1
TLS_FD = 0xFFFFFFFF

SSL_MAPS = {
    # pid_tgid -> {u64 buf ptr, u64 out-len ptr (SSL_read_ex) or 0}
    "ssl_args": (BPF_MAP_TYPE_HASH, 8, 16, 65536),
}
PT_RCX = 88


def _ssl_emit(a: Asm, syscall_marker: int, direction: int) -> None:
    """Shared tail: expects fp-32 = user buf, fp-40 = data len.
    Builds the event in the per-cpu scratch buffer and emits it."""
    a.st_imm(BPF_W, R10, -44, 0)
    a.ld_map_fd(R1, "scratch")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -44)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JNE, R0, 0, "have_buf")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("have_buf")
    a.mov64(R9, R0)
    a.call(H_KTIME_GET_NS)
    a.stx(BPF_DW, R9, EV_TS, R0)
    a.call(H_GET_PID_TGID)
    a.mov64(R1, R0)
    a.alu64_imm(BPF_RSH, R1, 32)
    a.stx(BPF_W, R9, EV_TGID, R1)
    a.mov64(R2, R0)
    a.alu64_imm(BPF_LSH, R2, 32)
    a.alu64_imm(BPF_RSH, R2, 32)
    a.stx(BPF_W, R9, EV_PID, R2)
    a.ld_imm64(R2, TLS_FD)
    a.stx(BPF_W, R9, EV_FD, R2)
    a.st_imm(BPF_DW, R9, EV_SOCKKEY, 0)
    a.st_imm(BPF_DW, R9, EV_TRACE, 0)
    a.ldx(BPF_DW, R2, R10, -40)
    a.stx(BPF_W, R9, EV_LEN, R2)
    a.st_imm(BPF_H, R9, EV_SYSCALL, syscall_marker)
    a.st_imm(BPF_B, R9, EV_DIR, direction)
    # cap = min(len, CAP_LEN); copy plaintext
    a.ldx(BPF_DW, R8, R10, -40)
    a.jmp_imm(BPF_JLE, R8, CAP_LEN, "cap_ok")
    a.mov64_imm(R8, CAP_LEN)
    a.label("cap_ok")
    a.stx(BPF_W, R9, EV_CAP, R8)
    a.mov64(R1, R9)
    a.alu64_imm(BPF_ADD, R1, EV_PAYLOAD)
    a.mov64(R2, R8)
    a.ldx(BPF_DW, R3, R10, -32)
    a.call(H_PROBE_READ_USER)
    # protocol inference over the captured plaintext
    a.ldx(BPF_DW, R7, R10, -40)
    _emit_inference(a)
    a.stx(BPF_B, R9, EV_PROTO, R0)
    a.mov64(R1, R6)
    a.ld_map_fd(R2, "events")
    a.ld_imm64(R3, 0xFFFFFFFF)            # BPF_F_CURRENT_CPU
    a.mov64(R4, R9)
    a.ldx(BPF_W, R5, R9, EV_CAP)
    a.alu64_imm(BPF_AND, R5, 0xFF)
    a.jmp_imm(BPF_JLE, R5, CAP_LEN, "evsz_ok")
    a.mov64_imm(R5, CAP_LEN)
    a.label("evsz_ok")
    a.alu64_imm(BPF_ADD, R5, EV_HDR)
    a.call(H_PERF_EVENT_OUTPUT)
    a.mov64_imm(R0, 0)
    a.exit()


def build_ssl_write() -> Asm:
    """uprobe SSL_write(ssl, buf, num): plaintext is in hand at entry."""
    a = Asm()
    a.mov64(R6, R1)                       # r6 = pt_regs ctx
    a.ldx(BPF_DW, R2, R6, PT_RSI)         # buf
    a.stx(BPF_DW, R10, -32, R2)
    a.ldx(BPF_DW, R2, R6, PT_RDX)         # num
    a.jmp_imm(BPF_JSGE, R2, 1, "len_ok")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("len_ok")
    a.stx(BPF_DW, R10, -40, R2)
    _ssl_emit(a, SC_SSL_WRITE, 0)
    return a


def build_ssl_read_enter(ex: bool = False) -> Asm:
    """uprobe SSL_read / SSL_read_ex entry: stash the destination
    buffer (and for _ex the out-length pointer) per thread."""
    a = Asm()
    a.mov64(R6, R1)
    a.call(H_GET_PID_TGID)
    a.stx(BPF_DW, R10, -8, R0)
    a.ldx(BPF_DW, R2, R6, PT_RSI)         # buf
    a.stx(BPF_DW, R10, -24, R2)
    if ex:
        a.ldx(BPF_DW, R2, R6, PT_RCX)     # size_t *readbytes
    else:
        a.mov64_imm(R2, 0)
    a.stx(BPF_DW, R10, -16, R2)
    a.ld_map_fd(R1, "ssl_args")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.mov64(R3, R10)
    a.alu64_imm(BPF_ADD, R3, -24)
    a.mov64_imm(R4, 0)
    a.call(H_MAP_UPDATE)
    a.mov64_imm(R0, 0)
    a.exit()
    return a


def build_ssl_read_exit(ex: bool = False) -> Asm:
    """uretprobe SSL_read / SSL_read_ex: plaintext length comes from
    the return value (SSL_read) or the stashed out-pointer
    (SSL_read_ex, where ret is just a success flag); emit the event."""
    a = Asm()
    a.mov64(R6, R1)
    a.call(H_GET_PID_TGID)
    a.stx(BPF_DW, R10, -8, R0)
    a.ld_map_fd(R1, "ssl_args")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JNE, R0, 0, "have_args")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("have_args")
    a.ldx(BPF_DW, R2, R0, 0)              # stashed buf
    a.stx(BPF_DW, R10, -32, R2)
    a.ldx(BPF_DW, R2, R0, 8)              # stashed out-len ptr (_ex)
    a.stx(BPF_DW, R10, -48, R2)
    a.ld_map_fd(R1, "ssl_args")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -8)
    a.call(H_MAP_DELETE)
    a.ldx(BPF_DW, R2, R6, PT_RAX)         # return value
    a.jmp_imm(BPF_JSGE, R2, 1, "ret_ok")
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("ret_ok")
    if ex:
        # length lives behind the out-pointer: *readbytes
        a.mov64(R1, R10)
        a.alu64_imm(BPF_ADD, R1, -40)
        a.mov64_imm(R2, 8)
        a.ldx(BPF_DW, R3, R10, -48)
        a.call(H_PROBE_READ_USER)
        a.ldx(BPF_DW, R2, R10, -40)
        a.jmp_imm(BPF_JSGE, R2, 1, "len_ok2")
        a.mov64_imm(R0, 0)
        a.exit()
        a.label("len_ok2")
    else:
        a.stx(BPF_DW, R10, -40, R2)
    _ssl_emit(a, SC_SSL_READ, 1)
    return a


def build_profiler() -> Asm:
    """On-CPU sampling profiler: perf_event program — count per
    (tgid, user stack id, kernel stack id). Reference:
    kernel/perf_profiler.bpf.c PERF_EVENT_PROG."""
    a = Asm()
    a.mov64(R6, R1)
    a.call(H_GET_PID_TGID)
    a.alu64_imm(BPF_RSH, R0, 32)
    a.stx(BPF_W, R10, -16, R0)            # key.tgid
    # user stack id
    a.mov64(R1, R6)
    a.ld_map_fd(R2, "stacks")
    a.ld_imm64(R3, 0x100)                 # BPF_F_USER_STACK
    a.call(H_GET_STACKID)
    a.stx(BPF_W, R10, -12, R0)
    # kernel stack id
    a.mov64(R1, R6)
    a.ld_map_fd(R2, "stacks")
    a.mov64_imm(R3, 0)
    a.call(H_GET_STACKID)
    a.stx(BPF_W, R10, -8, R0)
    a.st_imm(BPF_W, R10, -4, 0)
    # counts[key]++ (init on miss)
    a.ld_map_fd(R1, "counts")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -16)
    a.call(H_MAP_LOOKUP)
    a.jmp_imm(BPF_JEQ, R0, 0, "miss")
    a.ldx(BPF_DW, R1, R0, 0)
    a.alu64_imm(BPF_ADD, R1, 1)
    a.stx(BPF_DW, R0, 0, R1)
    a.mov64_imm(R0, 0)
    a.exit()
    a.label("miss")
    a.st_imm(BPF_DW, R10, -24, 1)
    a.ld_map_fd(R1, "counts")
    a.mov64(R2, R10)
    a.alu64_imm(BPF_ADD, R2, -16)
    a.mov64(R3, R10)
    a.alu64_imm(BPF_ADD, R3, -24)
    a.mov64_imm(R4, 0)
    a.call(H_MAP_UPDATE)
    a.mov64_imm(R0, 0)
    a.exit()
    return a
