"""MI355X-native eBPF collection stack.

The reference ships ~54k LoC of clang-compiled BPF C
(/root/reference/agent/src/ebpf/kernel/socket_trace.bpf.c — 79 probes —
plus perf_profiler.bpf.c) and a libbpf userspace runtime. This build
targets an environment with NO BPF compiler (the ROCm LLVM registers only
amdgcn/x86 backends), so the stack is self-contained:

  insn.py      eBPF instruction encoder + assembler (labels, maps, helpers)
  progs.py     program builders: socket tracer (raw_syscalls tracepoints,
               stable ABI — no CO-RE/BTF needed) + on-CPU perf profiler
  inference.py in-kernel protocol inference spec — ONE table drives the
               generated BPF matcher and the test oracle, cross-checked
               against the C++ packet parsers (reference keeps two
               hand-written copies: protocol_inference.h + check_payload)
  vm.py        userspace eBPF interpreter with helper/map emulation: the
               exact bytecode that loads into a kernel is executed against
               synthetic syscall streams in CI (stronger than compile-only)
  loader.py    raw bpf(2)/perf_event_open loader — no libbpf; degrades
               gracefully where BPF is unavailable (this container)
  runtime.py   EbpfCollector: perf events -> fd->socket resolution ->
               agent_core FlowMap/L7 parse (dfa_syscall_event)
"""
from .insn import Asm  # noqa: F401
