"""splitmix64 PRNG — shared deterministic stream for Python and C++ generators.

The C++ twin lives in ops/csrc/gen_cpu.cpp (df_splitmix64); golden tests
assert byte-identical payloads between the two, which is what lets the fast
C++ generator stand in for this reference implementation in benchmarks.
"""

MASK = (1 << 64) - 1
GAMMA = 0x9E3779B97F4A7C15


class SplitMix64:
    __slots__ = ("state",)

    def __init__(self, seed: int):
        self.state = seed & MASK

    def next(self) -> int:
        self.state = (self.state + GAMMA) & MASK
        z = self.state
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & MASK
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & MASK
        return (z ^ (z >> 31)) & MASK

    def below(self, n: int) -> int:
        """Uniform-ish in [0, n) via modulo (bias irrelevant for synth data)."""
        return self.next() % n
