// dfcpu — host-side native library: fast synthetic-stream generators and the
// ingest pre-segmentation scan.
//
// The generators are byte-identical twins of deepflow_amd/gen/*.py (golden
// tests in tests/test_gen_native.py assert equality); they exist because the
// benchmark drives millions of spans per second and the pure-Python encoder
// is a test fixture, not a data plane.
//
// Build: g++ -O3 -shared -fPIC -march=native -fopenmp dfcpu.cpp -o libdfcpu.so
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <dlfcn.h>

namespace {

constexpr uint64_t GAMMA = 0x9E3779B97F4A7C15ull;

struct SplitMix64 {
    uint64_t state;
    explicit SplitMix64(uint64_t seed) : state(seed) {}
    uint64_t next() {
        state += GAMMA;
        uint64_t z = state;
        z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
        z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
        return z ^ (z >> 31);
    }
    uint64_t below(uint64_t n) { return next() % n; }
};

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

void varint(Buf& b, uint64_t v) {
    while (true) {
        uint8_t x = v & 0x7F;
        v >>= 7;
        if (v) b.put(x | 0x80); else { b.put(x); return; }
    }
}

// field helpers: skip proto3 zero-defaults, ascending field order is the
// caller's responsibility (matches pb.py encode())
void f_u(Buf& b, uint32_t num, uint64_t v) {
    if (!v) return;
    varint(b, (uint64_t(num) << 3) | 0);
    varint(b, v);
}
void f_i(Buf& b, uint32_t num, int64_t v) {
    if (!v) return;
    varint(b, (uint64_t(num) << 3) | 0);
    varint(b, uint64_t(v));
}
void f_s(Buf& b, uint32_t num, const char* s, size_t n) {
    if (!n) return;
    varint(b, (uint64_t(num) << 3) | 2);
    varint(b, n);
    b.bytes(s, n);
}
void f_s(Buf& b, uint32_t num, const char* s) { f_s(b, num, s, strlen(s)); }
// length-delimited sub-message: encode into scratch, then emit
template <typename F>
void f_m(Buf& b, uint32_t num, F&& fill) {
    uint8_t scratch[4096];
    Buf sub{scratch, 0, sizeof scratch};
    fill(sub);
    varint(b, (uint64_t(num) << 3) | 2);
    varint(b, sub.len);
    b.bytes(scratch, sub.len);
}

void hex_str(char* out, uint64_t v, int width) {
    static const char* H = "0123456789abcdef";
    for (int i = 0; i < width; i++)
        out[i] = H[(v >> (4 * (width - 1 - i))) & 0xF];
    out[width] = 0;
}

struct SpanCfg {
    uint64_t seed;
    uint64_t base_time_ns;
    uint64_t dt_ns;
    uint32_t n_agents;
    uint32_t n_ips;
    uint32_t n_epcs;
    uint32_t n_services;
    uint32_t n_resources;
    uint32_t tag_cardinality;
    uint32_t n_attrs;
    uint32_t err_rate_pct;
    uint32_t ip6_rate_pct;
};

// Byte-identical twin of gen/spans.py:gen_span_dict + pb.encode
void encode_span(Buf& b, const SpanCfg& c, uint64_t i) {
    SplitMix64 rng(c.seed * 0x9E3779B9ull + i);
    uint64_t r0 = rng.next();
    uint64_t start = c.base_time_ns + i * c.dt_ns + rng.below(1000) * 1000;
    uint64_t rrt_us = 100 + rng.below(200000);
    uint64_t end = start + rrt_us * 1000;
    uint32_t svc = (uint32_t)rng.below(c.n_services);
    uint32_t res = (uint32_t)rng.below(c.n_resources);
    uint32_t ip_c = 0x0A000000u | (uint32_t)rng.below(c.n_ips);
    uint32_t ip_s = 0x0A000000u | ((svc * 7u) % c.n_ips);
    // v6 decision draws nothing from the rng stream (gen/spans.py twin)
    bool v6 = c.ip6_rate_pct > 0 &&
              ((i * 2654435761ull + c.seed) % 100) < c.ip6_rate_pct;
    bool err = rng.below(100) < c.err_rate_pct;
    uint64_t trace_hi = rng.next(), trace_lo = rng.next();
    uint64_t span_id_v = rng.next();
    // remaining draws in the exact order gen_span_dict evaluates them:
    // base-dict literal draws, then the attr loop, then span-dict literal.
    uint32_t port_src = 32768 + (uint32_t)(r0 % 28000);
    uint32_t process_id_0 = 1000 + (uint32_t)rng.below(64);
    uint64_t syscall_req = rng.next() & 0x7FFFFFFFFFFFFFFFull;
    uint32_t gpid_0 = 1 + (uint32_t)rng.below(1u << 16);
    uint32_t attr_vals[64];
    for (uint32_t a = 0; a < c.n_attrs && a < 64; a++)
        attr_vals[a] = (uint32_t)rng.below(c.tag_cardinality);
    int64_t req_len = 128 + (int64_t)rng.below(1024);
    int64_t resp_len = 256 + (int64_t)rng.below(8192);
    uint64_t request_id = rng.below(1u << 30);

    // base (field 1)
    f_m(b, 1, [&](Buf& s) {
        f_u(s, 1, start);
        f_u(s, 2, end);
        f_u(s, 3, r0 & 0x7FFFFFFFFFFFFFFFull);
        // 4 tap_port = 0 skipped
        f_u(s, 5, 1 + (r0 % c.n_agents));
        f_u(s, 6, 3);            // tap_type
        if (v6) f_u(s, 7, 1);    // is_ipv6
        f_u(s, 8, 1);            // tap_side
        f_m(s, 9, [&](Buf& h) {  // head
            f_u(h, 1, 20);       // proto = HTTP_1
            f_u(h, 2, 2);        // msg_type
            f_u(h, 5, rrt_us);
        });
        if (!v6) {
            f_u(s, 12, ip_c);
            f_u(s, 13, ip_s);
        } else {
            uint8_t a6[16] = {0x20, 0x01, 0x0d, 0xb8, 0, 0, 0, 0,
                              0, 0, 0, 0, 0, 0, 0, 0};
            a6[12] = (uint8_t)(ip_c >> 24); a6[13] = (uint8_t)(ip_c >> 16);
            a6[14] = (uint8_t)(ip_c >> 8);  a6[15] = (uint8_t)ip_c;
            f_s(s, 14, (const char*)a6, 16);
            a6[12] = (uint8_t)(ip_s >> 24); a6[13] = (uint8_t)(ip_s >> 16);
            a6[14] = (uint8_t)(ip_s >> 8);  a6[15] = (uint8_t)ip_s;
            f_s(s, 15, (const char*)a6, 16);
        }
        f_i(s, 16, 1 + (int64_t)(ip_c % c.n_epcs));
        f_i(s, 17, 1 + (int64_t)(ip_s % c.n_epcs));
        f_u(s, 18, port_src);
        f_u(s, 19, 8080);
        f_u(s, 20, 6);
        f_u(s, 23, r0 & 0xFFFFFFFFull);
        f_u(s, 24, (r0 >> 16) & 0xFFFFFFFFull);
        f_u(s, 25, process_id_0);
        f_u(s, 26, 2000 + (svc % 64));
        f_u(s, 29, syscall_req);
        f_u(s, 35, gpid_0);
        f_u(s, 36, 1 + (svc % (1u << 16)));
        f_u(s, 41, 1 + (ip_c % c.n_ips));
        f_u(s, 42, 1 + (ip_s % c.n_ips));
    });
    f_i(b, 9, req_len);
    f_i(b, 10, resp_len);
    char tmp[64];
    f_m(b, 11, [&](Buf& s) {  // req
        f_s(s, 1, ((r0 >> 8) % 4 == 0) ? "POST" : "GET");
        snprintf(tmp, sizeof tmp, "svc-%03u.example.com", svc);
        f_s(s, 2, tmp);
        snprintf(tmp, sizeof tmp, "/api/v1/r/%05u", res);
        f_s(s, 3, tmp);
        f_s(s, 4, "/api/v1/r");
    });
    f_m(b, 12, [&](Buf& s) {  // resp
        f_u(s, 1, err ? 3 : 0);       // status
        f_i(s, 2, err ? 500 : 200);   // code
    });
    f_s(b, 13, "1.1");  // version
    f_m(b, 14, [&](Buf& s) {  // trace_info
        char tid[40], sid[24];
        hex_str(tid, trace_hi, 16);
        hex_str(tid + 16, trace_lo, 16);
        hex_str(sid, span_id_v, 16);
        f_s(s, 1, tid, 32);
        f_s(s, 2, sid, 16);
    });
    f_m(b, 15, [&](Buf& s) {  // ext_info
        snprintf(tmp, sizeof tmp, "svc-%03u", svc);
        f_s(s, 1, tmp);
        f_u(s, 3, request_id);
        for (uint32_t a = 0; a < c.n_attrs && a < 64; a++) {
            snprintf(tmp, sizeof tmp, "attr_%u", a);
            f_s(s, 16, tmp);
        }
        for (uint32_t a = 0; a < c.n_attrs && a < 64; a++) {
            snprintf(tmp, sizeof tmp, "v%07u", attr_vals[a]);
            f_s(s, 17, tmp);
        }
    });
    f_u(b, 17, 255);  // direction_score
    f_u(b, 19, 128);  // captured_request_byte
    f_u(b, 20, 256);  // captured_response_byte
}

}  // namespace

extern "C" {

// Generate spans [i0, i0+n) as a length-prefixed record payload into out.
// Returns total bytes needed (call with cap=0 to size, then again to fill).
uint64_t df_gen_spans(const SpanCfg* cfg, uint64_t i0, uint64_t n,
                      uint8_t* out, uint64_t cap) {
    Buf b{out, 0, cap};
    uint8_t scratch[16384];
    for (uint64_t i = i0; i < i0 + n; i++) {
        Buf rec{scratch, 0, sizeof scratch};
        encode_span(rec, *cfg, i);
        uint32_t ln = (uint32_t)rec.len;
        b.bytes(&ln, 4);
        b.bytes(scratch, rec.len < sizeof scratch ? rec.len : sizeof scratch);
    }
    return b.len;
}

// Parallel variant: fills per-chunk into a caller-provided buffer using a
// first sizing pass. Returns bytes written; also writes record offsets/lens.
uint64_t df_gen_spans_indexed(const SpanCfg* cfg, uint64_t i0, uint64_t n,
                              uint8_t* out, uint64_t cap,
                              uint32_t* offs, uint32_t* lens) {
    Buf b{out, 0, cap};
    uint8_t scratch[16384];
    for (uint64_t i = 0; i < n; i++) {
        Buf rec{scratch, 0, sizeof scratch};
        encode_span(rec, *cfg, i0 + i);
        uint32_t ln = (uint32_t)rec.len;
        b.bytes(&ln, 4);
        if (offs) offs[i] = (uint32_t)b.len;
        if (lens) lens[i] = ln;
        b.bytes(scratch, rec.len);
    }
    return b.len;
}

// OpenMP-parallel generator: pass 1 sizes every record, prefix-sums, pass 2
// encodes in place. Byte-identical to df_gen_spans. Call with out=NULL to
// get the total size, then again with the buffer (offs/lens optional).
uint64_t df_gen_spans_parallel(const SpanCfg* cfg, uint64_t i0, uint64_t n,
                               uint8_t* out, uint64_t cap,
                               uint32_t* offs, uint32_t* lens) {
    static thread_local int dummy = 0; (void)dummy;
    uint64_t* starts = (uint64_t*)malloc((n + 1) * sizeof(uint64_t));
    if (!starts) return 0;
#pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)n; i++) {
        Buf b{nullptr, 0, 0};
        encode_span(b, *cfg, i0 + i);
        starts[i + 1] = b.len + 4;
    }
    starts[0] = 0;
    for (uint64_t i = 0; i < n; i++) starts[i + 1] += starts[i];
    uint64_t total = starts[n];
    if (out) {
#pragma omp parallel for schedule(static)
        for (int64_t i = 0; i < (int64_t)n; i++) {
            uint64_t pos = starts[i];
            uint64_t rec_len = starts[i + 1] - pos - 4;
            if (pos + 4 + rec_len <= cap) {
                uint32_t ln = (uint32_t)rec_len;
                memcpy(out + pos, &ln, 4);
                Buf b{out + pos + 4, 0, rec_len};
                encode_span(b, *cfg, i0 + i);
            }
            if (offs) offs[i] = (uint32_t)(pos + 4);
            if (lens) lens[i] = (uint32_t)rec_len;
        }
    }
    free(starts);
    return total;
}

// Pre-segmentation scan of a [u32 LE len][pb bytes]* payload.
// Writes up to max_n (offset, len) pairs; returns record count (may exceed
// max_n to signal truncation). Offsets point at the pb bytes.
uint64_t df_scan_offsets(const uint8_t* payload, uint64_t len,
                         uint32_t* offs, uint32_t* lens, uint64_t max_n) {
    uint64_t pos = 0, n = 0;
    while (pos + 4 <= len) {
        uint32_t ln;
        memcpy(&ln, payload + pos, 4);
        pos += 4;
        if (pos + ln > len) break;
        if (n < max_n) { offs[n] = (uint32_t)pos; lens[n] = ln; }
        n++;
        pos += ln;
    }
    return n;
}

// Shard routing peek: owner shard of each AppProtoLogsData record from its
// base.vtap_id (field 1 -> 5) without a full decode — the data-plane
// all-to-all key (reference hashes frames to queues by agent,
// server/libs/receiver/receiver.go:519-566). OpenMP over records.
static inline uint64_t route_mix(uint64_t z) {
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

uint64_t df_route_spans(const uint8_t* payload, uint64_t total,
                        const uint32_t* offs, const uint32_t* lens,
                        uint64_t n, uint32_t world, uint32_t org_id,
                        uint8_t* out_shard) {
    (void)total;
    uint64_t errs = 0;
#pragma omp parallel for schedule(static) reduction(+ : errs)
    for (uint64_t i = 0; i < n; i++) {
        uint32_t pos = offs[i], end = offs[i] + lens[i];
        uint32_t vtap = 0;
        bool found = false;
        // top level: find field 1 (base)
        while (pos < end && !found) {
            uint64_t key = 0;
            int sh = 0;
            while (pos < end) {
                uint8_t b = payload[pos++];
                key |= (uint64_t)(b & 0x7F) << sh;
                if (!(b & 0x80)) break;
                sh += 7;
            }
            uint32_t num = (uint32_t)(key >> 3), wt = (uint32_t)(key & 7);
            if (wt == 2) {
                uint64_t ln = 0;
                sh = 0;
                while (pos < end) {
                    uint8_t b = payload[pos++];
                    ln |= (uint64_t)(b & 0x7F) << sh;
                    if (!(b & 0x80)) break;
                    sh += 7;
                }
                if (num == 1) {
                    uint32_t p2 = pos, e2 = pos + (uint32_t)ln;
                    while (p2 < e2) {
                        uint64_t k2 = 0;
                        sh = 0;
                        while (p2 < e2) {
                            uint8_t b = payload[p2++];
                            k2 |= (uint64_t)(b & 0x7F) << sh;
                            if (!(b & 0x80)) break;
                            sh += 7;
                        }
                        uint32_t n2 = (uint32_t)(k2 >> 3),
                                 w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = 0;
                            sh = 0;
                            while (p2 < e2) {
                                uint8_t b = payload[p2++];
                                v |= (uint64_t)(b & 0x7F) << sh;
                                if (!(b & 0x80)) break;
                                sh += 7;
                            }
                            if (n2 == 5) { vtap = (uint32_t)v; found = true; break; }
                        } else if (w2 == 2) {
                            uint64_t l2 = 0;
                            sh = 0;
                            while (p2 < e2) {
                                uint8_t b = payload[p2++];
                                l2 |= (uint64_t)(b & 0x7F) << sh;
                                if (!(b & 0x80)) break;
                                sh += 7;
                            }
                            p2 += (uint32_t)l2;
                        } else if (w2 == 1) {
                            p2 += 8;
                        } else {
                            p2 += 4;
                        }
                    }
                    break;
                }
                pos += (uint32_t)ln;
            } else if (wt == 0) {
                uint64_t v = 0;
                sh = 0;
                while (pos < end) {
                    uint8_t b = payload[pos++];
                    v |= (uint64_t)(b & 0x7F) << sh;
                    if (!(b & 0x80)) break;
                    sh += 7;
                }
                (void)v;
            } else if (wt == 1) {
                pos += 8;
            } else {
                pos += 4;
            }
        }
        if (!found) errs++;
        out_shard[i] =
            (uint8_t)(route_mix(((uint64_t)org_id << 32) | vtap) % world);
    }
    return errs;
}

// ---- zstd / lz4 codecs via dlopen (runtime libs are present without dev
// headers; the trident frame encoder byte 3 = whole-payload zstd,
// agent/src/trident.rs:416-431) ----

typedef size_t (*zstd_fn4)(void*, size_t, const void*, size_t);
typedef size_t (*zstd_fn5)(void*, size_t, const void*, size_t, int);
typedef unsigned (*zstd_iserr)(size_t);

static void* zstd_sym(const char* name) {
    static void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    return h ? dlsym(h, name) : nullptr;
}

// Returns decompressed size, or -1 on failure.
int64_t df_zstd_decompress(const uint8_t* src, uint64_t n,
                           uint8_t* dst, uint64_t cap) {
    static zstd_fn4 dec = (zstd_fn4)zstd_sym("ZSTD_decompress");
    static zstd_iserr iserr = (zstd_iserr)zstd_sym("ZSTD_isError");
    if (!dec || !iserr) return -1;
    size_t r = dec(dst, cap, src, n);
    return iserr(r) ? -1 : (int64_t)r;
}

int64_t df_zstd_compress(const uint8_t* src, uint64_t n,
                         uint8_t* dst, uint64_t cap, int level) {
    static zstd_fn5 comp = (zstd_fn5)zstd_sym("ZSTD_compress");
    static zstd_iserr iserr = (zstd_iserr)zstd_sym("ZSTD_isError");
    if (!comp || !iserr) return -1;
    size_t r = comp(dst, cap, src, n, level);
    return iserr(r) ? -1 : (int64_t)r;
}

}  // extern "C"
