// Shared protobuf (proto3) encode helpers for host-side native code.
// Semantics match deepflow_amd/wire/pb.py encode(): ascending field order is
// the caller's responsibility; zero/empty values are skipped.
#pragma once
#include <cstdint>
#include <cstring>

namespace dfpb {

struct Buf {
    uint8_t* p;
    size_t len;
    size_t cap;
    void put(uint8_t b) { if (len < cap) p[len] = b; len++; }
    void bytes(const void* src, size_t n) {
        if (len + n <= cap) memcpy(p + len, src, n);
        len += n;
    }
};

inline void varint(Buf& b, uint64_t v) {
    while (true) {
        uint8_t x = v & 0x7F;
        v >>= 7;
        if (v) b.put(x | 0x80); else { b.put(x); return; }
    }
}

inline void f_u(Buf& b, uint32_t num, uint64_t v) {
    if (!v) return;
    varint(b, (uint64_t(num) << 3) | 0);
    varint(b, v);
}

inline void f_i(Buf& b, uint32_t num, int64_t v) {
    if (!v) return;
    varint(b, (uint64_t(num) << 3) | 0);
    varint(b, uint64_t(v));
}

inline void f_s(Buf& b, uint32_t num, const char* s, size_t n) {
    if (!n) return;
    varint(b, (uint64_t(num) << 3) | 2);
    varint(b, n);
    b.bytes(s, n);
}

inline void f_s(Buf& b, uint32_t num, const char* s) {
    f_s(b, num, s, strlen(s));
}

// length-delimited sub-message via scratch buffer
template <size_t CAP = 4096, typename F>
inline void f_m(Buf& b, uint32_t num, F&& fill) {
    uint8_t scratch[CAP];
    Buf sub{scratch, 0, sizeof scratch};
    fill(sub);
    varint(b, (uint64_t(num) << 3) | 2);
    varint(b, sub.len);
    b.bytes(scratch, sub.len < sizeof scratch ? sub.len : sizeof scratch);
}

}  // namespace dfpb
