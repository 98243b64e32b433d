"""eBPF instruction encoder + assembler.

Encodes the classic 64-bit eBPF instruction format:
  {u8 opcode; u8 dst:4; u8 src:4; s16 off; s32 imm}
(LD_IMM64 takes two slots). The assembler supports labels, forward jumps,
map references (patched to fds at load time or to VM handles in tests),
and emits bytes identical to what clang -target bpf would produce for the
same instruction stream — the kernel ABI is the instruction encoding, not
the C source.

Reference counterpart: the clang-compiled programs under
/root/reference/agent/src/ebpf/kernel/; this build has no BPF compiler in
the image, so the programs are generated directly (see ebpf/__init__).
"""
from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple, Union

# instruction classes
BPF_LD, BPF_LDX, BPF_ST, BPF_STX = 0x00, 0x01, 0x02, 0x03
BPF_ALU, BPF_JMP, BPF_JMP32, BPF_ALU64 = 0x04, 0x05, 0x06, 0x07
# sizes
BPF_W, BPF_H, BPF_B, BPF_DW = 0x00, 0x08, 0x10, 0x18
# modes
BPF_IMM, BPF_MEM = 0x00, 0x60
# alu ops
BPF_ADD, BPF_SUB, BPF_MUL, BPF_DIV = 0x00, 0x10, 0x20, 0x30
BPF_OR, BPF_AND, BPF_LSH, BPF_RSH = 0x40, 0x50, 0x60, 0x70
BPF_NEG, BPF_MOD, BPF_XOR, BPF_MOV = 0x80, 0x90, 0xA0, 0xB0
BPF_ARSH = 0xC0
# jmp ops
BPF_JA, BPF_JEQ, BPF_JGT, BPF_JGE = 0x00, 0x10, 0x20, 0x30
BPF_JSET, BPF_JNE, BPF_JSGT, BPF_JSGE = 0x40, 0x50, 0x60, 0x70
BPF_CALL, BPF_EXIT = 0x80, 0x90
BPF_JLT, BPF_JLE, BPF_JSLT, BPF_JSLE = 0xA0, 0xB0, 0xC0, 0xD0
# src
BPF_K, BPF_X = 0x00, 0x08
# ld_imm64 src pseudo
BPF_PSEUDO_MAP_FD = 1

# helper ids (include/uapi/linux/bpf.h — stable ABI)
H_MAP_LOOKUP = 1
H_MAP_UPDATE = 2
H_MAP_DELETE = 3
H_KTIME_GET_NS = 5
H_GET_PRANDOM = 7
H_GET_SMP_PROC_ID = 8
H_GET_PID_TGID = 14
H_GET_COMM = 16
H_PERF_EVENT_OUTPUT = 25
H_GET_STACKID = 27
H_PROBE_READ_USER = 112
H_PROBE_READ_KERNEL = 113

R0, R1, R2, R3, R4, R5, R6, R7, R8, R9, R10 = range(11)


@dataclass
class Insn:
    op: int
    dst: int = 0
    src: int = 0
    off: int = 0
    imm: int = 0
    # label this insn jumps to (resolved at assemble time)
    target: Optional[str] = None
    # second slot of ld_imm64 carries imm = high 32 bits
    imm64_hi: Optional[int] = None
    # symbolic map reference (patched by loader/vm)
    map_ref: Optional[str] = None

    def pack(self) -> bytes:
        imm = self.imm & 0xFFFFFFFF
        if imm >= 1 << 31:
            imm -= 1 << 32
        return struct.pack("<BBhi", self.op & 0xFF,
                           (self.dst & 0xF) | ((self.src & 0xF) << 4),
                           self.off, imm)


class Asm:
    """Tiny assembler: append instructions, mark labels, assemble."""

    def __init__(self):
        self.insns: List[Insn] = []
        self._labels: Dict[str, int] = {}

    # ------------------------------------------------------------ core
    def label(self, name: str) -> None:
        if name in self._labels:
            raise ValueError(f"duplicate label {name}")
        self._labels[name] = len(self.insns)

    def emit(self, insn: Insn) -> None:
        self.insns.append(insn)

    # ------------------------------------------------------------ alu
    def mov64_imm(self, dst, imm):
        self.emit(Insn(BPF_ALU64 | BPF_MOV | BPF_K, dst, 0, 0, imm))

    def mov64(self, dst, src):
        self.emit(Insn(BPF_ALU64 | BPF_MOV | BPF_X, dst, src))

    def alu64_imm(self, op, dst, imm):
        self.emit(Insn(BPF_ALU64 | op | BPF_K, dst, 0, 0, imm))

    def alu64(self, op, dst, src):
        self.emit(Insn(BPF_ALU64 | op | BPF_X, dst, src))

    def alu32_imm(self, op, dst, imm):
        self.emit(Insn(BPF_ALU | op | BPF_K, dst, 0, 0, imm))

    def ld_imm64(self, dst, imm):
        lo = imm & 0xFFFFFFFF
        hi = (imm >> 32) & 0xFFFFFFFF
        self.emit(Insn(BPF_LD | BPF_IMM | BPF_DW, dst, 0, 0, lo,
                       imm64_hi=hi))

    def ld_map_fd(self, dst, map_name: str):
        self.emit(Insn(BPF_LD | BPF_IMM | BPF_DW, dst, BPF_PSEUDO_MAP_FD,
                       0, 0, imm64_hi=0, map_ref=map_name))

    # ------------------------------------------------------------ mem
    def ldx(self, size, dst, src, off):
        self.emit(Insn(BPF_LDX | BPF_MEM | size, dst, src, off))

    def stx(self, size, dst, off, src):
        self.emit(Insn(BPF_STX | BPF_MEM | size, dst, src, off))

    def st_imm(self, size, dst, off, imm):
        self.emit(Insn(BPF_ST | BPF_MEM | size, dst, 0, off, imm))

    # ------------------------------------------------------------ jumps
    def jmp(self, target: str):
        self.emit(Insn(BPF_JMP | BPF_JA, target=target))

    def jmp_imm(self, op, dst, imm, target: str):
        self.emit(Insn(BPF_JMP | op | BPF_K, dst, 0, 0, imm, target=target))

    def jmp_reg(self, op, dst, src, target: str):
        self.emit(Insn(BPF_JMP | op | BPF_X, dst, src, target=target))

    def call(self, helper_id: int):
        self.emit(Insn(BPF_JMP | BPF_CALL, 0, 0, 0, helper_id))

    def exit(self):
        self.emit(Insn(BPF_JMP | BPF_EXIT))

    # ------------------------------------------------------------ out
    def assemble(self) -> List[Insn]:
        """Resolve labels -> slot-relative offsets (counting the extra
        slot of each ld_imm64 before the jump)."""
        # slot index of each insn
        slot = []
        s = 0
        for i in self.insns:
            slot.append(s)
            s += 2 if i.imm64_hi is not None else 1
        label_slot = {}
        for name, idx in self._labels.items():
            label_slot[name] = slot[idx] if idx < len(self.insns) else s
        out: List[Insn] = []
        for n, i in enumerate(self.insns):
            if i.target is not None:
                if i.target not in label_slot:
                    raise ValueError(f"undefined label {i.target}")
                i.off = label_slot[i.target] - (slot[n] + 1)
            out.append(i)
        return out

    def to_bytes(self, map_fds: Optional[Dict[str, int]] = None) -> bytes:
        """Kernel-loadable bytecode; map_refs patched to fds."""
        out = bytearray()
        for i in self.assemble():
            if i.map_ref is not None:
                if map_fds is None or i.map_ref not in map_fds:
                    raise ValueError(f"unpatched map ref {i.map_ref}")
                i = Insn(i.op, i.dst, BPF_PSEUDO_MAP_FD, 0,
                         map_fds[i.map_ref], imm64_hi=0)
            out += i.pack()
            if i.imm64_hi is not None:
                hi = i.imm64_hi - (1 << 32) if i.imm64_hi >= 1 << 31 \
                    else i.imm64_hi
                out += struct.pack("<BBhi", 0, 0, 0, hi)
        return bytes(out)

    def n_insns(self) -> int:
        return sum(2 if i.imm64_hi is not None else 1 for i in self.insns)
