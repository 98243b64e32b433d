// dfgpu — CDNA4 (gfx950) HIP kernels for the MI355X-native ingest + query
// hot path. This replaces the reference's CPU-side ingester decode
// (server/ingester/flow_log/decoder/decoder.go), PlatformInfoTable hash join
// (server/libs/grpc/grpc_platformdata.go), flow_tag LRU dedup
// (server/ingester/flow_tag/flow_tag_writer.go), time-window metric
// aggregation, and the ClickHouse group-by the reference's querier pushes
// down (server/querier/engine/clickhouse) — redesigned as GPU kernels over
// HBM-resident columnar segments.
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes; block sizes are multiples of 64.
//  - decode is thread-per-record (protobuf varint streams are sequential per
//    record); records land in L2/L3 so byte-granularity reads amortize.
//  - all cross-workgroup state (hash tables, counters) uses device-scope
//    atomics; no inter-workgroup ordering is assumed (XCD L2s not coherent).
//  - dictionary/table IDs are SLOT INDICES: a claim is one atomicCAS on the
//    64-bit key; no ID counter, no spin-waiting on a second word.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC
#include <hip/hip_runtime.h>
#include <rocprim/device/device_radix_sort.hpp>
#include "l7_layout.h"

#define DEV __device__ __forceinline__

namespace {

constexpr uint64_t EMPTY_KEY = 0ull;       // hash tables reserve 0 = empty
constexpr uint32_t BLOCK = 256;

// Word-buffered byte stream: protobuf parsing is byte-granular and
// sequential per record; buffering an aligned u64 per 8 bytes turns 8
// byte-loads into one dwordx2 load (the payload tensor base is
// 256B-aligned, so aligned word indexing is safe).
struct ByteStream {
    const uint64_t* w;
    uint64_t cur;
    uint32_t wpos;
    DEV void init(const uint8_t* base) {
        w = (const uint64_t*)base;
        wpos = 0xFFFFFFFFu;
        cur = 0;
    }
    DEV uint8_t get(uint32_t i) {
        uint32_t wp = i >> 3;
        if (wp != wpos) {
            wpos = wp;
            cur = w[wp];
        }
        return (uint8_t)(cur >> ((i & 7) * 8));
    }
};

DEV uint64_t rd_varint(ByteStream& bs, uint32_t& pos, uint32_t end) {
    uint64_t v = 0;
    int sh = 0;
    while (pos < end) {
        uint8_t b = bs.get(pos++);
        v |= (uint64_t)(b & 0x7F) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
        if (sh >= 70) break;
    }
    return v;
}

DEV void skip_field(ByteStream& bs, uint32_t& pos, uint32_t end, uint32_t wt) {
    switch (wt) {
        case 0: rd_varint(bs, pos, end); break;
        case 1: pos += 8; break;
        case 2: { uint64_t ln = rd_varint(bs, pos, end); pos += (uint32_t)ln; } break;
        case 5: pos += 4; break;
        default: pos = end; break;  // malformed
    }
}

DEV uint64_t rd_varint(const uint8_t* p, uint32_t& pos, uint32_t end) {
    uint64_t v = 0;
    int sh = 0;
    while (pos < end) {
        uint8_t b = p[pos++];
        v |= (uint64_t)(b & 0x7F) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
        if (sh >= 70) break;
    }
    return v;
}

DEV void skip_field(const uint8_t* p, uint32_t& pos, uint32_t end, uint32_t wt) {
    switch (wt) {
        case 0: rd_varint(p, pos, end); break;
        case 1: pos += 8; break;
        case 2: { uint64_t ln = rd_varint(p, pos, end); pos += (uint32_t)ln; } break;
        case 5: pos += 4; break;
        default: pos = end; break;  // malformed
    }
}

// xxhash64-flavoured string hash (seeded); only needs to be collision-safe
// at 64 bits, not xxh-compatible.
DEV uint64_t str_hash(const uint8_t* s, uint32_t len, uint64_t seed) {
    uint64_t h = seed ^ 0x27d4eb2f165667c5ull ^ (uint64_t)len * 0x9e3779b97f4a7c15ull;
    uint32_t i = 0;
    for (; i + 8 <= len; i += 8) {
        uint64_t k;
        __builtin_memcpy(&k, s + i, 8);
        h ^= k * 0xc2b2ae3d27d4eb4full;
        h = (h << 31) | (h >> 33);
        h *= 0x9e3779b185ebca87ull;
    }
    uint64_t tail = 0;
    for (uint32_t j = 0; i + j < len; j++) tail |= (uint64_t)s[i + j] << (8 * j);
    h ^= tail * 0x165667b19e3779f9ull;
    h ^= h >> 33; h *= 0xff51afd7ed558ccdull;
    h ^= h >> 29; h *= 0xc4ceb9fe1a85ec53ull;
    h ^= h >> 32;
    return h ? h : 1ull;  // never return EMPTY_KEY
}

// parse 16 lowercase/uppercase hex chars at pos -> u64; false on non-hex
DEV bool hex_parse64(ByteStream& bs, uint32_t pos, uint64_t& out) {
    uint64_t v = 0;
    for (uint32_t i = 0; i < 16; i++) {
        uint8_t c = bs.get(pos + i);
        uint8_t d;
        if (c >= '0' && c <= '9') d = c - '0';
        else if (c >= 'a' && c <= 'f') d = c - 'a' + 10;
        else if (c >= 'A' && c <= 'F') d = c - 'A' + 10;
        else return false;
        v = (v << 4) | d;
    }
    out = v;
    return true;
}

DEV uint64_t mix64(uint64_t z) {
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    return z ^ (z >> 31);
}

// ----------------------------------------------------------------------
// K1: AppProtoLogsData protobuf decode, thread-per-record.
// ----------------------------------------------------------------------

struct L7Cols {
    uint64_t* u64c;       // [L7_U64_N, stride]
    uint32_t* u32c;       // [L7_U32_N, stride]
    uint8_t* u8c;         // [L7_U8_N, stride]
    uint64_t* strc;       // SCRATCH [L7_STR_N, scratch_stride], row = rid
    uint64_t* attrc;      // SCRATCH [2*L7_MAX_ATTRS, scratch_stride]
    uint8_t* attr_cnt;    // [stride] (segment)
    uint64_t stride;
    uint64_t base_row;
    uint64_t scratch_stride;
};

#define W64(c, v) cols.u64c[(uint64_t)(c) * cols.stride + row] = (v)
#define W32(c, v) cols.u32c[(uint64_t)(c) * cols.stride + row] = (uint32_t)(v)
#define W8(c, v)  cols.u8c[(uint64_t)(c) * cols.stride + row] = (uint8_t)(v)
#define WSTR(c, off, len) cols.strc[(uint64_t)(c) * cols.scratch_stride + rid] = STR_REF_PACK(off, len)

__global__ void k_decode_l7(const uint8_t* __restrict__ payload,
                            const uint32_t* __restrict__ offs,
                            const uint32_t* __restrict__ lens,
                            uint32_t n, L7Cols cols) {
    uint32_t rid = blockIdx.x * blockDim.x + threadIdx.x;
    if (rid >= n) return;
    uint64_t row = cols.base_row + rid;
    uint32_t pos = offs[rid];
    ByteStream bs;
    bs.init(payload);
    uint32_t end = pos + lens[rid];
    uint32_t n_names = 0, n_vals = 0;

    while (pos < end) {
        uint64_t key = rd_varint(bs, pos, end);
        uint32_t num = (uint32_t)(key >> 3), wt = (uint32_t)(key & 7);
        if (wt == 0) {
            uint64_t v = rd_varint(bs, pos, end);
            switch (num) {
                case 9: W32(L7_U32_REQ_LEN, v); break;
                case 10: W32(L7_U32_RESP_LEN, v); break;
                case 17: W8(L7_U8_DIR_SCORE, v); break;
                case 18: W32(L7_U32_FLAGS, v); break;
                case 19: W32(L7_U32_CAP_REQ_BYTE, v); break;
                case 20: W32(L7_U32_CAP_RESP_BYTE, v); break;
                default: break;
            }
        } else if (wt == 2) {
            uint32_t ln = (uint32_t)rd_varint(bs, pos, end);
            uint32_t sub = pos, send = pos + ln;
            pos = send;
            switch (num) {
                case 1: {  // AppProtoLogsBaseInfo
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            switch (n2) {
                                case 1: W64(L7_U64_START_TIME, v); break;
                                case 2: W64(L7_U64_END_TIME, v); break;
                                case 3: W64(L7_U64_FLOW_ID, v); break;
                                case 5: W32(L7_U32_VTAP_ID, v); break;
                                case 6: W8(L7_U8_TAP_TYPE, v); break;
                                case 7: W8(L7_U8_IS_IPV6, v); break;
                                case 8: W8(L7_U8_TAP_SIDE, v); break;
                                case 12: W32(L7_U32_IP4_0, v); break;
                                case 13: W32(L7_U32_IP4_1, v); break;
                                case 16: W32(L7_U32_EPC_0, v); break;
                                case 17: W32(L7_U32_EPC_1, v); break;
                                case 18: W32(L7_U32_PORT_0, v); break;
                                case 19: W32(L7_U32_PORT_1, v); break;
                                case 20: W8(L7_U8_PROTOCOL, v); break;
                                case 23: W32(L7_U32_REQ_TCP_SEQ, v); break;
                                case 24: W32(L7_U32_RESP_TCP_SEQ, v); break;
                                case 25: W32(L7_U32_PID_0, v); break;
                                case 26: W32(L7_U32_PID_1, v); break;
                                case 29: W64(L7_U64_SYSCALL_REQ, v); break;
                                case 30: W64(L7_U64_SYSCALL_RESP, v); break;
                                case 35: W32(L7_U32_GPID_0, v); break;
                                case 36: W32(L7_U32_GPID_1, v); break;
                                case 41: W32(L7_U32_POD_0, v); break;
                                case 42: W32(L7_U32_POD_1, v); break;
                                case 43: W32(L7_U32_BIZ_TYPE, v); break;
                                default: break;
                            }
                        } else if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            uint32_t s3 = p2, e3 = p2 + l3;
                            p2 = e3;
                            if (n2 == 9) {  // AppProtoHead
                                uint32_t p3 = s3;
                                while (p3 < e3) {
                                    uint64_t k3 = rd_varint(bs, p3, e3);
                                    if ((k3 & 7) == 0) {
                                        uint64_t v = rd_varint(bs, p3, e3);
                                        switch ((uint32_t)(k3 >> 3)) {
                                            case 1: W8(L7_U8_L7_PROTOCOL, v); break;
                                            case 2: W8(L7_U8_MSG_TYPE, v); break;
                                            case 5: W64(L7_U64_RRT, v); break;
                                            default: break;
                                        }
                                    } else {
                                        uint32_t w3 = (uint32_t)(k3 & 7);
                                        skip_field(bs, p3, e3, w3);
                                    }
                                }
                            } else if (n2 == 27) {
                                WSTR(L7_STR_PKNAME_0, s3, l3);
                            } else if (n2 == 28) {
                                WSTR(L7_STR_PKNAME_1, s3, l3);
                            } else if (n2 == 14) {   // ip6_src (16 raw bytes,
                                WSTR(L7_STR_IP6_0, s3, l3);  // pooled)
                            } else if (n2 == 15) {   // ip6_dst
                                WSTR(L7_STR_IP6_1, s3, l3);
                            }
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 11: {  // L7Request
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            switch (n2) {
                                case 1: WSTR(L7_STR_REQ_TYPE, p2, l3); break;
                                case 2: WSTR(L7_STR_DOMAIN, p2, l3); break;
                                case 3: WSTR(L7_STR_RESOURCE, p2, l3); break;
                                case 4: WSTR(L7_STR_ENDPOINT, p2, l3); break;
                                default: break;
                            }
                            p2 += l3;
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 12: {  // L7Response
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            if (n2 == 1) W8(L7_U8_STATUS, v);
                            else if (n2 == 2) W32(L7_U32_CODE, v);
                        } else if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            if (n2 == 3) WSTR(L7_STR_EXCEPTION, p2, l3);
                            else if (n2 == 4) WSTR(L7_STR_RESULT, p2, l3);
                            p2 += l3;
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 13: WSTR(L7_STR_VERSION, sub, ln); break;
                case 14: {  // TraceInfo: hex ids transcode to binary
                    // columns (SmartEncoding: a 32-hex trace id stores as
                    // 16 B of u64s, not 48 B of pool+len); non-hex ids
                    // fall back to the pool columns
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            if (n2 == 1) {
                                uint64_t hi, lo;
                                if (l3 == 32 &&
                                    hex_parse64(bs, p2, hi) &&
                                    hex_parse64(bs, p2 + 16, lo)) {
                                    W64(L7_U64_TRACE_HI, hi);
                                    W64(L7_U64_TRACE_LO, lo);
                                } else {
                                    WSTR(L7_STR_TRACE_ID, p2, l3);
                                }
                            } else if (n2 == 2) {
                                uint64_t sv;
                                if (l3 == 16 && hex_parse64(bs, p2, sv)) {
                                    W64(L7_U64_SPAN_ID_B, sv);
                                } else {
                                    WSTR(L7_STR_SPAN_ID, p2, l3);
                                }
                            } else if (n2 == 3) {
                                WSTR(L7_STR_PARENT_SPAN_ID, p2, l3);
                            }
                            p2 += l3;
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 15: {  // ExtendedInfo
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            if (n2 == 3) W32(L7_U32_REQUEST_ID, v);
                        } else if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            switch (n2) {
                                case 1: WSTR(L7_STR_SERVICE_NAME, p2, l3); break;
                                case 4: WSTR(L7_STR_XREQ_0, p2, l3); break;
                                case 6: WSTR(L7_STR_UA, p2, l3); break;
                                case 7: WSTR(L7_STR_REFERER, p2, l3); break;
                                case 10: WSTR(L7_STR_XREQ_1, p2, l3); break;
                                case 16:
                                    if (n_names < L7_MAX_ATTRS)
                                        cols.attrc[(uint64_t)n_names * cols.scratch_stride + rid] =
                                            STR_REF_PACK(p2, l3);
                                    n_names++;
                                    break;
                                case 17:
                                    if (n_vals < L7_MAX_ATTRS)
                                        cols.attrc[(uint64_t)(L7_MAX_ATTRS + n_vals) * cols.scratch_stride + rid] =
                                            STR_REF_PACK(p2, l3);
                                    n_vals++;
                                    break;
                                default: break;
                            }
                            p2 += l3;
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 21: WSTR(L7_STR_BIZ_CODE, sub, ln); break;
                default: break;
            }
        } else {
            skip_field(bs, pos, end, wt);
        }
    }
    uint32_t na = n_names < n_vals ? n_names : n_vals;
    cols.attr_cnt[row] = (uint8_t)(na > L7_MAX_ATTRS ? L7_MAX_ATTRS : na);
}

// ----------------------------------------------------------------------
// K1b: TaggedFlow (L4 flow log) protobuf decode, thread-per-record.
// Wire schema: message/flow_log.proto:14-120; columns: l4_layout.h.
// ----------------------------------------------------------------------
#include "l4_layout.h"

struct L4Cols {
    uint64_t* u64c;
    uint32_t* u32c;
    uint8_t* u8c;
    uint64_t* strc;   // SCRATCH [L4_STR_N, scratch_stride], row = rid
    uint64_t stride;
    uint64_t base_row;
    uint64_t scratch_stride;
};

#define L4W64(c, v) cols.u64c[(uint64_t)(c) * cols.stride + row] = (v)
#define L4W32(c, v) cols.u32c[(uint64_t)(c) * cols.stride + row] = (uint32_t)(v)
#define L4W8(c, v)  cols.u8c[(uint64_t)(c) * cols.stride + row] = (uint8_t)(v)

__global__ void k_decode_l4(const uint8_t* __restrict__ payload,
                            const uint32_t* __restrict__ offs,
                            const uint32_t* __restrict__ lens,
                            uint32_t n, L4Cols cols) {
    uint32_t rid = blockIdx.x * blockDim.x + threadIdx.x;
    if (rid >= n) return;
    uint64_t row = cols.base_row + rid;
    uint32_t pos = offs[rid];
    ByteStream bs;
    bs.init(payload);
    uint32_t end = pos + lens[rid];
    // locate flow submessage (TaggedFlow field 1)
    while (pos < end) {
        uint64_t key = rd_varint(bs, pos, end);
        uint32_t num = (uint32_t)(key >> 3), wt = (uint32_t)(key & 7);
        if (num == 1 && wt == 2) {
            uint32_t ln = (uint32_t)rd_varint(bs, pos, end);
            end = pos + ln;  // narrow to Flow
            break;
        }
        skip_field(bs, pos, end, wt);
    }
    while (pos < end) {
        uint64_t key = rd_varint(bs, pos, end);
        uint32_t num = (uint32_t)(key >> 3), wt = (uint32_t)(key & 7);
        if (wt == 0) {
            uint64_t v = rd_varint(bs, pos, end);
            switch (num) {
                case 5: L4W64(L4_U64_FLOW_ID, v); break;
                case 6: L4W64(L4_U64_START_TIME, v); break;
                case 7: L4W64(L4_U64_END_TIME, v); break;
                case 8: L4W64(L4_U64_DURATION, v); break;
                case 10: L4W32(L4_U32_VLAN, v); break;
                case 11: L4W32(L4_U32_ETH_TYPE, v); break;
                case 14: L4W8(L4_U8_CLOSE_TYPE, v); break;
                case 15: L4W8(L4_U8_SIGNAL_SOURCE, v); break;
                case 16: L4W8(L4_U8_IS_ACTIVE_SERVICE, v); break;
                case 18: L4W8(L4_U8_IS_NEW_FLOW, v); break;
                case 19: L4W8(L4_U8_TAP_SIDE, v); break;
                case 25: L4W8(L4_U8_DIRECTION_SCORE, v); break;
                default: break;
            }
        } else if (wt == 2) {
            uint32_t ln = (uint32_t)rd_varint(bs, pos, end);
            uint32_t sub = pos, send = pos + ln;
            pos = send;
            switch (num) {
                case 1: {  // FlowKey
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            switch (n2) {
                                case 1: L4W32(L4_U32_VTAP_ID, v); break;
                                case 2: L4W8(L4_U8_TAP_TYPE, v); break;
                                case 4: L4W64(L4_U64_MAC_SRC, v); break;
                                case 5: L4W64(L4_U64_MAC_DST, v); break;
                                case 6: L4W32(L4_U32_IP4_0, v); break;
                                case 7: L4W32(L4_U32_IP4_1, v); break;
                                case 10: L4W32(L4_U32_PORT_SRC, v); break;
                                case 11: L4W32(L4_U32_PORT_DST, v); break;
                                case 12: L4W8(L4_U8_PROTOCOL, v); break;
                                default: break;
                            }
                        } else if (w2 == 2 && (n2 == 8 || n2 == 9)) {
                            // ip6_src/dst: raw 16 bytes into the str pool
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            cols.strc[(uint64_t)(n2 == 8 ? L4_STR_IP6_0
                                                         : L4_STR_IP6_1) *
                                      cols.scratch_stride + rid] =
                                STR_REF_PACK(p2, l3);
                            p2 += l3;
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 24: {  // acl_gids (packed repeated u32): keep first
                    uint32_t p2 = sub;
                    if (p2 < send) {
                        uint64_t v = rd_varint(bs, p2, send);
                        L4W32(L4_U32_ACL_GID, v);
                    }
                    break;
                }
                case 2: case 3: {  // FlowMetricsPeer src/dst
                    bool tx = num == 2;
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            switch (n2) {
                                case 1: L4W64(tx ? L4_U64_BYTE_TX : L4_U64_BYTE_RX, v); break;
                                case 2: L4W64(tx ? L4_U64_L3_BYTE_TX : L4_U64_L3_BYTE_RX, v); break;
                                case 3: L4W64(tx ? L4_U64_L4_BYTE_TX : L4_U64_L4_BYTE_RX, v); break;
                                case 4: L4W64(tx ? L4_U64_PACKET_TX : L4_U64_PACKET_RX, v); break;
                                case 5: L4W64(tx ? L4_U64_TOTAL_BYTE_TX : L4_U64_TOTAL_BYTE_RX, v); break;
                                case 6: L4W64(tx ? L4_U64_TOTAL_PACKET_TX : L4_U64_TOTAL_PACKET_RX, v); break;
                                case 9: L4W32(tx ? L4_U32_TCP_FLAGS_SRC : L4_U32_TCP_FLAGS_DST, v); break;
                                case 10: L4W32(tx ? L4_U32_EPC_0 : L4_U32_EPC_1, v); break;
                                case 20: L4W32(tx ? L4_U32_NAT_REAL_IP_0 : L4_U32_NAT_REAL_IP_1, v); break;
                                case 21: L4W32(tx ? L4_U32_NAT_REAL_PORT_0 : L4_U32_NAT_REAL_PORT_1, v); break;
                                case 22: L4W32(tx ? L4_U32_GPID_0 : L4_U32_GPID_1, v); break;
                                default: break;
                            }
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 13: {  // FlowPerfStats
                    uint32_t p2 = sub;
                    while (p2 < send) {
                        uint64_t k2 = rd_varint(bs, p2, send);
                        uint32_t n2 = (uint32_t)(k2 >> 3), w2 = (uint32_t)(k2 & 7);
                        if (w2 == 0) {
                            uint64_t v = rd_varint(bs, p2, send);
                            if (n2 == 3) L4W8(L4_U8_L4_PROTOCOL, v);
                            else if (n2 == 4) L4W8(L4_U8_L7_PROTOCOL, v);
                        } else if (w2 == 2) {
                            uint32_t l3 = (uint32_t)rd_varint(bs, p2, send);
                            uint32_t s3 = p2, e3 = p2 + l3;
                            p2 = e3;
                            if (n2 == 1) {  // TCPPerfStats
                                uint32_t p3 = s3;
                                while (p3 < e3) {
                                    uint64_t k3 = rd_varint(bs, p3, e3);
                                    uint32_t n3 = (uint32_t)(k3 >> 3), w3 = (uint32_t)(k3 & 7);
                                    if (w3 == 0) {
                                        uint64_t v = rd_varint(bs, p3, e3);
                                        switch (n3) {
                                            case 3: L4W32(L4_U32_SRT_MAX, v); break;
                                            case 4: L4W32(L4_U32_ART_MAX, v); break;
                                            case 5: L4W32(L4_U32_RTT, v); break;
                                            case 8: L4W32(L4_U32_SRT_SUM, v); break;
                                            case 9: L4W32(L4_U32_ART_SUM, v); break;
                                            case 12: L4W32(L4_U32_SRT_COUNT, v); break;
                                            case 13: L4W32(L4_U32_ART_COUNT, v); break;
                                            case 16: L4W32(L4_U32_RETRANS_TOTAL, v); break;
                                            case 17: L4W32(L4_U32_SYN_COUNT, v); break;
                                            case 18: L4W32(L4_U32_SYNACK_COUNT, v); break;
                                            case 19: L4W32(L4_U32_CIT_MAX, v); break;
                                            case 20: L4W32(L4_U32_CIT_SUM, v); break;
                                            case 21: L4W32(L4_U32_CIT_COUNT, v); break;
                                            default: break;
                                        }
                                    } else if (w3 == 2) {
                                        uint32_t l4b = (uint32_t)rd_varint(bs, p3, e3);
                                        uint32_t s4 = p3, e4 = p3 + l4b;
                                        p3 = e4;
                                        if (n3 == 14 || n3 == 15) {  // TcpPerfCountsPeer
                                            bool ptx = n3 == 14;
                                            uint32_t p4 = s4;
                                            while (p4 < e4) {
                                                uint64_t k4 = rd_varint(bs, p4, e4);
                                                if ((k4 & 7) == 0) {
                                                    uint64_t v = rd_varint(bs, p4, e4);
                                                    uint32_t n4 = (uint32_t)(k4 >> 3);
                                                    if (n4 == 1) L4W32(ptx ? L4_U32_RETRANS_TX : L4_U32_RETRANS_RX, v);
                                                    else if (n4 == 2) L4W32(ptx ? L4_U32_ZERO_WIN_TX : L4_U32_ZERO_WIN_RX, v);
                                                    else if (n4 == 3) L4W32(ptx ? L4_U32_OOO_TX : L4_U32_OOO_RX, v);
                                                } else {
                                                    uint32_t w4 = (uint32_t)(k4 & 7);
                                                    skip_field(bs, p4, e4, w4);
                                                }
                                            }
                                        }
                                    } else {
                                        skip_field(bs, p3, e3, w3);
                                    }
                                }
                            } else if (n2 == 2) {  // L7PerfStats
                                uint32_t p3 = s3;
                                while (p3 < e3) {
                                    uint64_t k3 = rd_varint(bs, p3, e3);
                                    if ((k3 & 7) == 0) {
                                        uint64_t v = rd_varint(bs, p3, e3);
                                        switch ((uint32_t)(k3 >> 3)) {
                                            case 1: L4W32(L4_U32_L7_REQUEST, v); break;
                                            case 2: L4W32(L4_U32_L7_RESPONSE, v); break;
                                            case 3: L4W32(L4_U32_L7_ERR_CLIENT, v); break;
                                            case 4: L4W32(L4_U32_L7_ERR_SERVER, v); break;
                                            case 5: L4W32(L4_U32_L7_ERR_TIMEOUT, v); break;
                                            case 6: L4W32(L4_U32_L7_RRT_COUNT, v); break;
                                            case 7: L4W64(L4_U64_L7_RRT_SUM, v); break;
                                            case 8: L4W32(L4_U32_L7_RRT_MAX, v); break;
                                            default: break;
                                        }
                                    } else {
                                        uint32_t w3 = (uint32_t)(k3 & 7);
                                        skip_field(bs, p3, e3, w3);
                                    }
                                }
                            }
                        } else {
                            skip_field(bs, p2, send, w2);
                        }
                    }
                    break;
                }
                case 26:
                    cols.strc[(uint64_t)L4_STR_REQUEST_DOMAIN * cols.scratch_stride + rid] =
                        STR_REF_PACK(sub, ln);
                    break;
                default: break;
            }
        } else {
            skip_field(bs, pos, end, wt);
        }
    }
}

// ----------------------------------------------------------------------
// K5: flow_metrics table family (reference libs/flow-metrics/tag.go
// 443-523: network{,_map}.{1s,1m}, application{,_map}.{1s,1m},
// traffic_policy.1m — per-table Code bitmask selects tag columns).
//
// Exact keys: each table keeps three arrays — tkeys[cap] holds the 64-bit
// mixed hash of the key tuple (the CAS claim word), traw[cap][RU_MAX_KEYS]
// holds the raw tuple words for verification/harvest, tvals[cap][nv] the
// accumulators. A reader that matches the hash verifies the raw words; if
// the claimant has not published them yet it simply probes on and claims a
// fresh slot with the same tuple (no spin — intra-wave spinning on another
// lane's store can deadlock the wave on CDNA's exec-mask divergence
// model); duplicate slots for one tuple are merged exactly at harvest.
// This removes the round-1 masked-pack collisions (vtap&0xFFF etc.).
// ----------------------------------------------------------------------

#define RU_MAX_KEYS 8
constexpr uint64_t RU_SENTINEL = 0xFFFFFFFFFFFFFFFFull;

// host-built per-table key spec (the Code-bitmask analog)
struct RuSpec {
    uint32_t interval_s;            // time bucket width (1, 60, ...)
    uint32_t n_keys;                // key sources after the time bucket
    uint8_t fam[RU_MAX_KEYS];       // 0=u64 1=u32 2=u8
    uint8_t idx[RU_MAX_KEYS];
    uint8_t require_nonzero;        // 1-based key index that must be != 0
                                    // (traffic_policy: acl_gid), 0 = off
    uint8_t use_lds;                // per-block LDS pre-aggregation: on for
                                    // low-cardinality tables where global
                                    // atomics would serialize on hot slots
};

enum { NAGG_BYTE_TX = 0, NAGG_BYTE_RX, NAGG_PKT_TX, NAGG_PKT_RX,
       NAGG_NEW_FLOW, NAGG_CLOSED_FLOW, NAGG_RTT_SUM, NAGG_RTT_CNT,
       NAGG_RTT_MAX, NAGG_RETRANS, NAGG_NVALS };

template <typename ColsT>
DEV uint64_t ru_src(const ColsT& c, uint64_t row, uint8_t fam, uint8_t idx) {
    switch (fam) {
        case 0: return c.u64c[(uint64_t)idx * c.stride + row];
        case 1: return c.u32c[(uint64_t)idx * c.stride + row];
        default: return c.u8c[(uint64_t)idx * c.stride + row];
    }
}

// claim a slot for the raw tuple kw[0..nw); returns slot or ~0u.
// Probes are bounded: once a table region is saturated the row is DROPPED
// and counted (drops tensor) instead of walking the whole table — a full
// table must degrade to a watermark drop, not an O(cap) scan per row
// (the reference throttles/evicts at capacity too).
#define RU_MAX_PROBES 128u
DEV uint64_t ru_hash(const uint64_t* kw, uint32_t nw) {
    uint64_t h = 0x9E3779B97F4A7C15ull;
    for (uint32_t w = 0; w < nw; w++) h = mix64(h ^ kw[w]);
    return h == EMPTY_KEY ? 1 : h;
}

DEV uint32_t ru_claim_h(const uint64_t* kw, uint32_t nw, uint64_t h,
                        uint64_t* tkeys, uint64_t* traw, uint32_t cap_mask) {
    uint32_t slot = (uint32_t)(h & cap_mask);
    uint32_t max_probes = cap_mask < RU_MAX_PROBES ? cap_mask : RU_MAX_PROBES;
    for (uint32_t probe = 0; probe <= max_probes; probe++) {
        uint64_t cur = tkeys[slot];
        if (cur == EMPTY_KEY) {
            uint64_t old = atomicCAS((unsigned long long*)&tkeys[slot],
                                     EMPTY_KEY, h);
            if (old == EMPTY_KEY) {
                uint64_t* r = &traw[(uint64_t)slot * RU_MAX_KEYS];
                for (uint32_t w = nw; w-- > 1;)
                    __hip_atomic_store(&r[w], kw[w], __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                // word 0 published last with release: a reader that sees
                // it non-sentinel sees the full tuple
                __hip_atomic_store(&r[0], kw[0], __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
                return slot;
            }
            cur = old;
        }
        if (cur == h) {
            const uint64_t* r = &traw[(uint64_t)slot * RU_MAX_KEYS];
            uint64_t w0 = __hip_atomic_load(&r[0], __ATOMIC_ACQUIRE,
                                            __HIP_MEMORY_SCOPE_AGENT);
            // w0 is a bucketed relative time — never the sentinel. If the
            // claimant has not published yet, fall through and claim a
            // duplicate slot (merged at harvest) instead of spinning.
            if (w0 == kw[0]) {
                bool match = true;
                for (uint32_t w = 1; w < nw; w++)
                    match = match &&
                        __hip_atomic_load(&r[w], __ATOMIC_RELAXED,
                                          __HIP_MEMORY_SCOPE_AGENT) == kw[w];
                if (match) return slot;
            }
        }
        slot = (slot + 1) & cap_mask;
    }
    return 0xFFFFFFFFu;
}

DEV uint32_t ru_claim(const uint64_t* kw, uint32_t nw,
                      uint64_t* tkeys, uint64_t* traw, uint32_t cap_mask) {
    return ru_claim_h(kw, nw, ru_hash(kw, nw), tkeys, traw, cap_mask);
}

// build the raw key tuple for one row; false = row not eligible
template <typename ColsT>
DEV bool ru_build_key(const ColsT& cols, uint64_t row, uint64_t time_base_s,
                      const RuSpec& ru, uint64_t* kw) {
    uint64_t t_s = cols.u64c[0 * cols.stride + row] / 1000000000ull;
    uint64_t rel = t_s > time_base_s ? t_s - time_base_s : 0;
    if (ru.interval_s > 1) rel = (rel / ru.interval_s) * ru.interval_s;
    kw[0] = rel;
    for (uint32_t k = 0; k < ru.n_keys; k++)
        kw[1 + k] = ru_src(cols, row, ru.fam[k], ru.idx[k]);
    return !(ru.require_nonzero && kw[ru.require_nonzero] == 0);
}

template <typename ColsT>
DEV uint32_t ru_key_claim(const ColsT& cols, uint64_t row, uint64_t time_base_s,
                          const RuSpec& ru, uint64_t* tkeys, uint64_t* traw,
                          uint32_t cap_mask, unsigned long long* drops) {
    uint64_t kw[RU_MAX_KEYS + 1];
    if (!ru_build_key(cols, row, time_base_s, ru, kw))
        return 0xFFFFFFFEu;  // row not eligible for this table
    uint32_t slot = ru_claim(kw, ru.n_keys + 1, tkeys, traw, cap_mask);
    if (slot == 0xFFFFFFFFu) atomicAdd(drops, 1ull);
    return slot;
}

// multi-table spec: one launch updates every table of a family
// (one pass over the row columns instead of one launch per table —
// k_rollup_l7 was 63% of device time as 4 launches, r2 profile)
#define RU_MAX_TABLES 6
struct RuMulti {
    RuSpec ru[RU_MAX_TABLES];
    uint32_t n_tables;
};
struct RuTablePtrs {
    uint64_t* tkeys[RU_MAX_TABLES];
    uint64_t* traw[RU_MAX_TABLES];
    unsigned long long* tvals[RU_MAX_TABLES];
    uint32_t cap_mask[RU_MAX_TABLES];
    unsigned long long* drops[RU_MAX_TABLES];
};

// ----------------------------------------------------------------------
// LDS pre-aggregation pass: realistic streams concentrate millions of
// rows into a handful of 1s/1m rollup groups, so straight global atomics
// serialize on hot slots (measured: 63% of device time). Each workgroup
// aggregates its stripe into a 256-slot LDS table (claimant stores the
// row id so the flush can rebuild the raw tuple), then flushes once —
// global traffic drops from per-row to per-(block x live-slot). Fn maps
// (acc, row) -> atomic accumulate and works on both LDS and global
// pointers (generic address space).
// ----------------------------------------------------------------------
#define RU_NSLOT 256
#define RU_LDS_PROBES 8

template <typename ColsT, int NV, typename Fn, typename Merge>
DEV void ru_pass(const ColsT& cols, uint32_t n, uint64_t time_base_s,
                 const RuSpec& ru, uint64_t* tkeys, uint64_t* traw,
                 unsigned long long* tvals, uint32_t cap_mask,
                 unsigned long long* drops, Fn&& accum, Merge&& merge,
                 uint64_t* lkey, int* lrow, unsigned long long* lagg) {
    const uint32_t stride_g = gridDim.x * blockDim.x;
    if (ru.use_lds) {
        for (uint32_t s = threadIdx.x; s < RU_NSLOT; s += blockDim.x) {
            lkey[s] = 0;
            lrow[s] = -1;
            for (int v = 0; v < NV; v++) lagg[s * NV + v] = 0;
        }
        __syncthreads();
    }
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride_g) {
        uint64_t row = cols.base_row + i;
        uint64_t kw[RU_MAX_KEYS + 1];
        if (!ru_build_key(cols, row, time_base_s, ru, kw)) continue;
        uint64_t h = ru_hash(kw, ru.n_keys + 1);
        bool done = false;
        if (ru.use_lds) {
            uint32_t s = (uint32_t)(h & (RU_NSLOT - 1));
            for (uint32_t probe = 0; probe < RU_LDS_PROBES; probe++) {
                uint64_t cur = lkey[s];
                if (cur == 0) {
                    uint64_t old = atomicCAS((unsigned long long*)&lkey[s],
                                             0ull, h);
                    if (old == 0) lrow[s] = (int)i;
                    cur = old == 0 ? h : old;
                }
                if (cur == h) {
                    accum(&lagg[(uint64_t)s * NV], row);
                    done = true;
                    break;
                }
                s = (s + 1) & (RU_NSLOT - 1);
            }
        }
        if (!done) {
            uint32_t slot = ru_claim_h(kw, ru.n_keys + 1, h, tkeys, traw,
                                       cap_mask);
            if (slot == 0xFFFFFFFFu) {
                atomicAdd(drops, 1ull);
                continue;
            }
            accum(&tvals[(uint64_t)slot * NV], row);
        }
    }
    if (ru.use_lds) {
        __syncthreads();
        for (uint32_t s = threadIdx.x; s < RU_NSLOT; s += blockDim.x) {
            if (lkey[s] == 0) continue;
            uint64_t row = cols.base_row + (uint32_t)lrow[s];
            uint64_t kw[RU_MAX_KEYS + 1];
            ru_build_key(cols, row, time_base_s, ru, kw);
            uint32_t slot = ru_claim_h(kw, ru.n_keys + 1, lkey[s], tkeys,
                                       traw, cap_mask);
            if (slot == 0xFFFFFFFFu) {
                atomicAdd(drops, 1ull);
                continue;
            }
            merge(&tvals[(uint64_t)slot * NV], &lagg[(uint64_t)s * NV]);
        }
        __syncthreads();
    }
}

__global__ void __launch_bounds__(256)
k_rollup_l4(const L4Cols cols, uint32_t n, uint64_t time_base_s,
            RuMulti mu, RuTablePtrs tp) {
    __shared__ uint64_t lkey[RU_NSLOT];
    __shared__ int lrow[RU_NSLOT];
    __shared__ unsigned long long lagg[RU_NSLOT * NAGG_NVALS];
    auto accum = [&](unsigned long long* acc, uint64_t row) {
        atomicAdd(&acc[NAGG_BYTE_TX],
                  cols.u64c[L4_U64_BYTE_TX * cols.stride + row]);
        atomicAdd(&acc[NAGG_BYTE_RX],
                  cols.u64c[L4_U64_BYTE_RX * cols.stride + row]);
        atomicAdd(&acc[NAGG_PKT_TX],
                  cols.u64c[L4_U64_PACKET_TX * cols.stride + row]);
        atomicAdd(&acc[NAGG_PKT_RX],
                  cols.u64c[L4_U64_PACKET_RX * cols.stride + row]);
        if (cols.u8c[L4_U8_IS_NEW_FLOW * cols.stride + row])
            atomicAdd(&acc[NAGG_NEW_FLOW], 1ull);
        if (cols.u8c[L4_U8_CLOSE_TYPE * cols.stride + row])
            atomicAdd(&acc[NAGG_CLOSED_FLOW], 1ull);
        uint32_t rtt = cols.u32c[L4_U32_RTT * cols.stride + row];
        if (rtt) {
            atomicAdd(&acc[NAGG_RTT_SUM], (unsigned long long)rtt);
            atomicAdd(&acc[NAGG_RTT_CNT], 1ull);
            atomicMax(&acc[NAGG_RTT_MAX], (unsigned long long)rtt);
        }
        uint64_t retrans = cols.u32c[L4_U32_RETRANS_TX * cols.stride + row] +
                           cols.u32c[L4_U32_RETRANS_RX * cols.stride + row];
        if (retrans) atomicAdd(&acc[NAGG_RETRANS], retrans);
    };
    auto merge = [&](unsigned long long* acc, unsigned long long* part) {
        for (int v = 0; v < NAGG_NVALS; v++) {
            if (part[v] == 0) continue;
            if (v == NAGG_RTT_MAX) atomicMax(&acc[v], part[v]);
            else atomicAdd(&acc[v], part[v]);
        }
    };
    for (uint32_t t = 0; t < mu.n_tables; t++)
        ru_pass<L4Cols, NAGG_NVALS>(cols, n, time_base_s, mu.ru[t],
                                    tp.tkeys[t], tp.traw[t], tp.tvals[t],
                                    tp.cap_mask[t], tp.drops[t], accum,
                                    merge, lkey, lrow, lagg);
}

// generic raw-tuple batch insert (agent Document ingest; ops: 0=sum 1=max)
__global__ void k_rollup_insert(const uint64_t* __restrict__ kws,  // [n, nw]
                                const unsigned long long* __restrict__ vals,  // [n, nv]
                                const uint8_t* __restrict__ ops,   // [nv]
                                uint32_t n, uint32_t nw, uint32_t nv,
                                uint64_t* __restrict__ tkeys,
                                uint64_t* __restrict__ traw,
                                unsigned long long* __restrict__ tvals,
                                uint32_t cap_mask,
                                unsigned long long* __restrict__ drops) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t kw[RU_MAX_KEYS + 1];
    for (uint32_t w = 0; w < nw; w++) kw[w] = kws[(uint64_t)i * nw + w];
    uint32_t slot = ru_claim(kw, nw, tkeys, traw, cap_mask);
    if (slot == 0xFFFFFFFFu) { atomicAdd(drops, 1ull); return; }
    unsigned long long* acc = &tvals[(uint64_t)slot * nv];
    for (uint32_t v = 0; v < nv; v++) {
        unsigned long long x = vals[(uint64_t)i * nv + v];
        if (!x) continue;
        if (ops[v] == 0) atomicAdd(&acc[v], x);
        else atomicMax(&acc[v], x);
    }
}

// ----------------------------------------------------------------------
// Data-plane shard routing: gather selected records into a packed
// per-destination buffer for the RCCL all-to-all (one wave per record,
// lanes copy bytes strided — records are 100-500 B, so a wave64 gets
// coalesced 64 B segments).
// ----------------------------------------------------------------------
__global__ void k_gather_records(const uint8_t* __restrict__ src,
                                 const uint32_t* __restrict__ offs,
                                 const uint32_t* __restrict__ lens,
                                 const uint32_t* __restrict__ sel,
                                 const uint64_t* __restrict__ dst_off,
                                 uint32_t m, uint8_t* __restrict__ out) {
    uint32_t waves = blockDim.x / 64;
    uint32_t rec = blockIdx.x * waves + (threadIdx.x / 64);
    uint32_t lane = threadIdx.x % 64;
    if (rec >= m) return;
    uint32_t r = sel[rec];
    const uint8_t* s = src + offs[r];
    uint8_t* d = out + dst_off[rec];
    uint32_t n = lens[r];
    for (uint32_t b = lane; b < n; b += 64) d[b] = s[b];
}

// ----------------------------------------------------------------------
// K2: KnowledgeGraph (epc,ip) -> resource-id join.
//     Open-addressing table: keys u64 ((epc<<32)|ip), vals KG_VALS_N x u32.
// ----------------------------------------------------------------------

__global__ void k_kg_build(const uint64_t* __restrict__ keys,
                           const uint32_t* __restrict__ vals,  // [n, KG_VALS_N]
                           uint32_t n,
                           uint64_t* __restrict__ tkeys,
                           uint32_t* __restrict__ tvals,       // [cap, KG_VALS_N]
                           uint32_t cap_mask) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t k = keys[i];
    if (k == EMPTY_KEY) return;
    uint32_t slot = (uint32_t)(mix64(k) & cap_mask);
    for (uint32_t probe = 0; probe <= cap_mask; probe++) {
        uint64_t old = atomicCAS((unsigned long long*)&tkeys[slot], EMPTY_KEY, k);
        if (old == EMPTY_KEY || old == k) {
            // last-writer-wins value update (platform data refresh semantics)
            for (int j = 0; j < KG_VALS_N; j++)
                tvals[(uint64_t)slot * KG_VALS_N + j] = vals[(uint64_t)i * KG_VALS_N + j];
            return;
        }
        slot = (slot + 1) & cap_mask;
    }
}

__global__ void k_kg_probe(const uint32_t* __restrict__ epc0,
                           const uint32_t* __restrict__ ip0,
                           const uint32_t* __restrict__ epc1,
                           const uint32_t* __restrict__ ip1,
                           uint32_t n,
                           const uint64_t* __restrict__ tkeys,
                           const uint32_t* __restrict__ tvals,
                           uint32_t cap_mask,
                           uint32_t* __restrict__ out,  // [2*KG_VALS_N, stride]
                           uint64_t stride, uint64_t base_row) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t row = base_row + i;
    for (int side = 0; side < 2; side++) {
        uint64_t k = side == 0
            ? (((uint64_t)epc0[i] << 32) | ip0[i])
            : (((uint64_t)epc1[i] << 32) | ip1[i]);
        uint32_t slot = (uint32_t)(mix64(k) & cap_mask);
        bool found = false;
        for (uint32_t probe = 0; probe <= cap_mask; probe++) {
            uint64_t tk = tkeys[slot];
            if (tk == k) { found = true; break; }
            if (tk == EMPTY_KEY) break;
            slot = (slot + 1) & cap_mask;
        }
        for (int j = 0; j < KG_VALS_N; j++)
            out[((uint64_t)(side * KG_VALS_N + j)) * stride + row] =
                found ? tvals[(uint64_t)slot * KG_VALS_N + j] : 0u;
    }
}

// ----------------------------------------------------------------------
// K3: string interning (SmartEncoding dictionary). One shared table for all
// domains; key = hash(domain, bytes); DICT ID == slot index (stable within a
// shard). New entries are emitted as (slot, packed batch ref) for the host to
// harvest into the id->string dictionary + flow_tag write path.
// ----------------------------------------------------------------------

// probe-or-insert: plain loads on the hot (hit) path, atomicCAS only on
// empty slots. Newly claimed slots are emitted for host harvest.
DEV uint32_t intern_probe(uint64_t h, uint64_t ref, uint8_t domain,
                          uint64_t* tkeys, uint32_t cap_mask,
                          uint64_t* emit, uint32_t* emit_ctr,
                          uint32_t emit_cap) {
    uint32_t slot = (uint32_t)(h & cap_mask);
    for (uint32_t probe = 0; probe <= cap_mask; probe++) {
        uint64_t cur = tkeys[slot];
        if (cur == h) return slot;
        if (cur == EMPTY_KEY) {
            uint64_t old = atomicCAS((unsigned long long*)&tkeys[slot],
                                     EMPTY_KEY, h);
            if (old == EMPTY_KEY) {
                uint32_t e = atomicAdd(emit_ctr, 1u);
                if (e < emit_cap) {
                    emit[(uint64_t)e * 2] = ((uint64_t)domain << 56) | slot;
                    emit[(uint64_t)e * 2 + 1] = ref;
                }
                return slot;
            }
            if (old == h) return slot;
            // lost the race to a different key: fall through, keep probing
        }
        slot = (slot + 1) & cap_mask;
    }
    return DICT_ID_INVALID;
}

// wave-cooperative variant: when every active lane holds the same hash
// (common for low-cardinality columns: req_type, version, attr names),
// lane 0 probes once and broadcasts.
DEV uint32_t intern_probe_wave(uint64_t h, uint64_t ref, uint8_t domain,
                               bool active,
                               uint64_t* tkeys, uint32_t cap_mask,
                               uint64_t* emit, uint32_t* emit_ctr,
                               uint32_t emit_cap) {
    uint64_t h0 = __shfl(h, 0);
    bool uniform = __all(!active || h == h0);
    if (uniform) {
        uint32_t slot = DICT_ID_INVALID;
        // some lane that is active must probe; lowest active lane does
        uint64_t act_mask = __ballot(active);
        if (act_mask == 0) return DICT_ID_INVALID;
        uint32_t leader = (uint32_t)__ffsll((unsigned long long)act_mask) - 1;
        if ((threadIdx.x & 63) == leader)
            slot = intern_probe(h, ref, domain, tkeys, cap_mask, emit,
                                emit_ctr, emit_cap);
        slot = __shfl((int)slot, leader);
        return slot;
    }
    if (!active) return DICT_ID_INVALID;
    return intern_probe(h, ref, domain, tkeys, cap_mask, emit, emit_ctr,
                        emit_cap);
}

__global__ void k_intern_many(const uint8_t* __restrict__ payload,
                              const uint64_t* __restrict__ refs,  // [*, stride]
                              const uint16_t* __restrict__ ref_rows,  // [C] row in refs per column
                              const uint8_t* __restrict__ domains,  // [C]
                              uint32_t C, uint32_t n,
                              uint64_t ref_stride, uint64_t ref_base_row,
                              uint64_t* __restrict__ tkeys, uint32_t cap_mask,
                              uint64_t* __restrict__ emit,
                              uint32_t* __restrict__ emit_ctr, uint32_t emit_cap,
                              uint32_t* __restrict__ out_ids,  // [C, out_stride]
                              uint64_t out_stride, uint64_t out_base_row) {
    uint64_t gid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t total = (uint64_t)C * n;
    if (gid >= total) return;
    uint32_t c = (uint32_t)(gid / n);
    uint32_t i = (uint32_t)(gid % n);
    uint64_t ref = refs[(uint64_t)ref_rows[c] * ref_stride + ref_base_row + i];
    uint32_t len = STR_REF_LEN(ref);
    uint64_t off = STR_REF_OFF(ref);
    bool active = len != 0;
    uint64_t h = 0;
    uint8_t dom = domains[c];
    if (active)
        h = str_hash(payload + off, len, 0x9E3779B97F4A7C15ull * (dom + 1));
    uint32_t slot = intern_probe_wave(h, ref, dom, active, tkeys, cap_mask,
                                      emit, emit_ctr, emit_cap);
    // out block is pre-initialized to DICT_ID_INVALID; only write hits
    if (active)
        out_ids[c * out_stride + out_base_row + i] = slot;
    else
        out_ids[c * out_stride + out_base_row + i] = DICT_ID_INVALID;
}

// attr interning: one thread per row, looping over the row's attr_cnt
// (name, value) pairs — avoids launching MAX_ATTRS*2 threads per row when
// the typical count is ~4. Names are wave-uniform per iteration.
__global__ void k_intern_attrs(const uint8_t* __restrict__ payload,
                               const uint64_t* __restrict__ attr_refs,  // SCRATCH [2*MAX, ref_stride]
                               const uint8_t* __restrict__ attr_cnt,
                               uint32_t n, uint64_t stride, uint64_t base_row,
                               uint64_t ref_stride,
                               uint64_t* __restrict__ tkeys, uint32_t cap_mask,
                               uint64_t* __restrict__ emit,
                               uint32_t* __restrict__ emit_ctr, uint32_t emit_cap,
                               const uint32_t* __restrict__ attr_start,  // [n] row block offsets (global)
                               int32_t* __restrict__ attr_pool) {  // segment attr-id pool
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    bool in_range = i < n;
    uint64_t row = base_row + (in_range ? i : 0);
    uint32_t cnt = in_range ? attr_cnt[row] : 0;
    uint32_t pbase = in_range ? attr_start[i] : 0;
    // wave-max so all lanes iterate together for the uniform-name fast path
    uint32_t maxc = cnt;
    for (int d = 32; d > 0; d >>= 1)
        maxc = max(maxc, (uint32_t)__shfl_xor((int)maxc, d));
    for (uint32_t a = 0; a < maxc; a++) {
        bool act = in_range && a < cnt;
        // names (wave-uniform in the common schema-stable case)
        uint64_t nref = act ? attr_refs[(uint64_t)a * ref_stride + i] : 0;
        uint32_t nlen = STR_REF_LEN(nref);
        uint64_t h = 0;
        if (act && nlen)
            h = str_hash(payload + STR_REF_OFF(nref), nlen,
                         0x9E3779B97F4A7C15ull * (DICT_DOM_ATTR_NAME + 1));
        uint32_t slot = intern_probe_wave(h, nref, DICT_DOM_ATTR_NAME,
                                          act && nlen, tkeys, cap_mask,
                                          emit, emit_ctr, emit_cap);
        if (act)
            attr_pool[pbase + a] = nlen ? (int32_t)slot : -1;
        // values (high cardinality -> per-lane probes)
        uint64_t vref = act
            ? attr_refs[(uint64_t)(L7_MAX_ATTRS + a) * ref_stride + i] : 0;
        uint32_t vlen = STR_REF_LEN(vref);
        if (act) {
            uint32_t vslot = DICT_ID_INVALID;
            if (vlen) {
                uint64_t vh = str_hash(payload + STR_REF_OFF(vref), vlen,
                    0x9E3779B97F4A7C15ull * (DICT_DOM_ATTR_VALUE + 1));
                vslot = intern_probe(vh, vref, DICT_DOM_ATTR_VALUE, tkeys,
                                     cap_mask, emit, emit_ctr, emit_cap);
            }
            attr_pool[pbase + cnt + a] = (int32_t)vslot;
        }
    }
}

// ----------------------------------------------------------------------
// K4: string pool gather. Pass 1 computes per-row pooled byte count; host
// runs an exclusive cumsum (torch); pass 2 copies bytes into the segment
// pool and rewrites refs to pool-relative offsets.
// ----------------------------------------------------------------------

__global__ void k_pool_lens(const uint64_t* __restrict__ strc,  // [L7_STR_N, stride]
                            const uint8_t* __restrict__ pool_cols,  // [npc]
                            uint32_t npc, uint32_t n,
                            uint64_t stride, uint64_t base_row,
                            uint32_t* __restrict__ row_len) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint32_t total = 0;
    for (uint32_t c = 0; c < npc; c++)
        total += STR_REF_LEN(strc[(uint64_t)pool_cols[c] * stride + base_row + i]);
    row_len[i] = total;
}

__global__ void k_pool_gather(const uint8_t* __restrict__ payload,
                              const uint64_t* __restrict__ strc,  // scratch
                              const uint8_t* __restrict__ pool_cols,
                              uint32_t npc, uint32_t n,
                              uint64_t stride, uint64_t base_row,
                              const uint64_t* __restrict__ row_start,  // exclusive cumsum
                              uint8_t* __restrict__ pool, uint64_t pool_base,
                              uint64_t* __restrict__ out_rowref,  // [out_stride] segment
                              int16_t* __restrict__ out_lens,     // [npc, out_stride]
                              uint64_t out_stride, uint64_t out_base_row) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t dst0 = pool_base + row_start[i];
    uint64_t dst = dst0;
    for (uint32_t c = 0; c < npc; c++) {
        uint64_t r = strc[(uint64_t)pool_cols[c] * stride + base_row + i];
        uint32_t len = STR_REF_LEN(r);
        uint64_t src = STR_REF_OFF(r);
        for (uint32_t b = 0; b < len; b++) pool[dst + b] = payload[src + b];
        out_lens[(uint64_t)c * out_stride + out_base_row + i] = (int16_t)len;
        dst += len;
    }
    out_rowref[out_base_row + i] = STR_REF_PACK(dst0, dst - dst0);
}

// ----------------------------------------------------------------------
// K5 (application side): flow_metrics application{,_map} rollups from L7
// rows (reference: flow_metrics unmarshaller -> application tables).
// Exact keys via ru_key_claim (see the table-family section above).
// ----------------------------------------------------------------------

enum { AGG_REQ = 0, AGG_RESP, AGG_ERR_C, AGG_ERR_S, AGG_RRT_SUM, AGG_RRT_CNT, AGG_RRT_MAX, AGG_NVALS };

__global__ void __launch_bounds__(256)
k_rollup_l7(const L7Cols cols, uint32_t n, uint64_t time_base_s,
            RuMulti mu, RuTablePtrs tp) {
    __shared__ uint64_t lkey[RU_NSLOT];
    __shared__ int lrow[RU_NSLOT];
    __shared__ unsigned long long lagg[RU_NSLOT * AGG_NVALS];
    auto accum = [&](unsigned long long* acc, uint64_t row) {
        uint8_t status = cols.u8c[L7_U8_STATUS * cols.stride + row];
        uint8_t mtype = cols.u8c[L7_U8_MSG_TYPE * cols.stride + row];
        uint64_t rrt = cols.u64c[L7_U64_RRT * cols.stride + row];
        // msg_type: 0=request,1=response,2=session(both)
        if (mtype == 0 || mtype == 2) atomicAdd(&acc[AGG_REQ], 1ull);
        if (mtype == 1 || mtype == 2) atomicAdd(&acc[AGG_RESP], 1ull);
        if (status == 4) atomicAdd(&acc[AGG_ERR_C], 1ull);
        if (status == 3) atomicAdd(&acc[AGG_ERR_S], 1ull);
        if (rrt) {
            atomicAdd(&acc[AGG_RRT_SUM], (unsigned long long)rrt);
            atomicAdd(&acc[AGG_RRT_CNT], 1ull);
            atomicMax(&acc[AGG_RRT_MAX], (unsigned long long)rrt);
        }
    };
    auto merge = [&](unsigned long long* acc, unsigned long long* part) {
        for (int v = 0; v < AGG_NVALS; v++) {
            if (part[v] == 0) continue;
            if (v == AGG_RRT_MAX) atomicMax(&acc[v], part[v]);
            else atomicAdd(&acc[v], part[v]);
        }
    };
    for (uint32_t t = 0; t < mu.n_tables; t++)
        ru_pass<L7Cols, AGG_NVALS>(cols, n, time_base_s, mu.ru[t],
                                   tp.tkeys[t], tp.traw[t], tp.tvals[t],
                                   tp.cap_mask[t], tp.drops[t], accum,
                                   merge, lkey, lrow, lagg);
}

// ----------------------------------------------------------------------
// K7: generic filtered hash group-by over a segment (the querier hot path;
// replaces ClickHouse-side GROUP BY in the reference design).
// ----------------------------------------------------------------------

// key/agg source families
enum {
    SRC_U64 = 0, SRC_U32, SRC_U8, SRC_DID, SRC_KG, SRC_ATTR_VAL,
    SRC_TIME_BUCKET, SRC_CONST0, SRC_STR_HASH,
    // filter-only: exists(slot): name_id == v0 && value_id == v1
    SRC_ATTR_MATCH,
    // binary OTel ids (idx 0 = trace_id, 1 = span_id): mix64(hi)^lo of the
    // binary columns, or the pooled-string hash for non-hex fallbacks
    SRC_TRACE128,
};
// seed for SRC_STR_HASH terms (host twin: store/dictionary.py STR_FILTER_SEED)
#define STR_FILTER_SEED 0x5157A15E5EEDull
// filter ops
enum { OP_EQ = 0, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE, OP_BETWEEN };
// agg ops
enum { AGGOP_COUNT = 0, AGGOP_SUM, AGGOP_MIN, AGGOP_MAX };

// group 0 terms AND together; groups >= 1 are OR-clauses: the row must
// satisfy at least one term of every present group (CNF — covers
// `(a=1 OR a=2)` and `x IN (...)`)
struct QTerm { uint8_t family; uint8_t op; uint16_t idx; uint8_t group;
               uint64_t v0, v1; };
struct QKey  { uint8_t family; uint16_t idx; uint32_t bucket; };  // bucket: seconds per bucket for SRC_TIME_BUCKET
struct QAgg  { uint8_t op; uint8_t family; uint16_t idx; };

#define QMAX_TERMS 16
#define QMAX_KEYS 4
#define QMAX_AGGS 8

struct QuerySpec {
    QTerm terms[QMAX_TERMS];
    QKey keys[QMAX_KEYS];
    QAgg aggs[QMAX_AGGS];
    uint32_t n_terms, n_keys, n_aggs;
    uint64_t time_base_s;
};

struct SegView {
    const uint64_t* u64c;
    const uint32_t* u32c;
    const uint8_t* u8c;
    const uint32_t* didc;      // [L7_DID_N, stride]
    // KnowledgeGraph table for query-time join (SmartEncoding: the 24
    // per-row KG id columns are NOT materialized — they are a pure
    // function of the row's (epc, ip) key, resolved here by probing the
    // GPU-resident platform table instead of spending 96 B/span of HBM)
    const uint64_t* kg_tk;
    const uint32_t* kg_tv;     // [cap, KG_VALS_N]
    uint32_t kg_mask;
    const int32_t* attr_pool;  // variable attr-id pool
    const uint32_t* attr_start;  // [stride] row block offsets into attr_pool
    const uint8_t* attr_cnt;   // [stride]
    const uint64_t* str_rowref;  // [stride] (pool_off<<16 | total_len)
    const int16_t* str_lens;   // [n_pool_cols, stride]
    const uint8_t* pool;       // segment string pool
    uint64_t stride;
    uint64_t n_rows;
};

// epc/ip column indices are identical in the L7 and L4 u32 layouts
// (vtap, ip4_0, ip4_1, epc_0, epc_1 — asserted in the layout headers)
DEV uint64_t kg_join(const SegView& s, uint64_t row, uint16_t idx) {
    if (s.kg_tk == nullptr) return 0;
    uint32_t side = idx / KG_VALS_N, j = idx % KG_VALS_N;
    uint32_t epc = s.u32c[(uint64_t)(3 + side) * s.stride + row];
    uint32_t ip = s.u32c[(uint64_t)(1 + side) * s.stride + row];
    uint64_t k = ((uint64_t)epc << 32) | ip;
    uint32_t slot = (uint32_t)(mix64(k) & s.kg_mask);
    for (uint32_t probe = 0; probe <= s.kg_mask; probe++) {
        uint64_t tk = s.kg_tk[slot];
        if (tk == k) return s.kg_tv[(uint64_t)slot * KG_VALS_N + j];
        if (tk == EMPTY_KEY) return 0;
        slot = (slot + 1) & s.kg_mask;
    }
    return 0;
}

DEV uint64_t pool_str_hash(const SegView& s, uint64_t row, uint16_t idx) {
    uint64_t rr = s.str_rowref[row];
    uint64_t off = STR_REF_OFF(rr);
    for (uint16_t c = 0; c < (uint16_t)idx; c++)
        off += (uint16_t)s.str_lens[(uint64_t)c * s.stride + row];
    uint32_t len = (uint16_t)s.str_lens[(uint64_t)idx * s.stride + row];
    if (len == 0) return 0;
    return str_hash(s.pool + off, len, STR_FILTER_SEED);
}

DEV uint64_t src_value(const SegView& s, uint64_t row, uint8_t family,
                       uint16_t idx, uint32_t bucket, uint64_t time_base_s) {
    switch (family) {
        case SRC_U64: return s.u64c[(uint64_t)idx * s.stride + row];
        case SRC_U32: return s.u32c[(uint64_t)idx * s.stride + row];
        case SRC_U8:  return s.u8c[(uint64_t)idx * s.stride + row];
        case SRC_DID: return s.didc[(uint64_t)idx * s.stride + row];
        case SRC_KG:  return kg_join(s, row, idx);
        case SRC_ATTR_VAL: {
            // idx is the attr slot (0..cnt-1); value ids follow name ids in
            // the row's attr-pool block
            uint32_t cnt = s.attr_cnt[row];
            if (idx >= cnt) return DICT_ID_INVALID;
            return (uint32_t)s.attr_pool[s.attr_start[row] + cnt + idx];
        }
        case SRC_TIME_BUCKET: {
            uint64_t t_s = s.u64c[L7_U64_START_TIME * s.stride + row] / 1000000000ull;
            uint64_t rel = t_s > time_base_s ? t_s - time_base_s : 0;
            return bucket ? (rel / bucket) * bucket : rel;
        }
        case SRC_STR_HASH:
            return pool_str_hash(s, row, idx);
        case SRC_TRACE128: {
            if (idx == 0) {
                uint64_t hi = s.u64c[L7_U64_TRACE_HI * s.stride + row];
                uint64_t lo = s.u64c[L7_U64_TRACE_LO * s.stride + row];
                if (hi | lo) return mix64(hi) ^ lo;
                return pool_str_hash(s, row, L7_POOL_TRACE_ID);
            }
            uint64_t sv = s.u64c[L7_U64_SPAN_ID_B * s.stride + row];
            if (sv) return sv;
            return pool_str_hash(s, row, L7_POOL_SPAN_ID);
        }
        default: return 0;
    }
}

DEV bool eval_terms(const SegView& s, uint64_t row, const QuerySpec& q) {
    uint32_t need = 0, have = 0;
    for (uint32_t t = 0; t < q.n_terms; t++) {
        const QTerm& term = q.terms[t];
        if (term.family == SRC_ATTR_MATCH) {
            uint32_t cnt = s.attr_cnt[row];
            uint32_t start = s.attr_start[row];
            bool okm = false;
            for (uint32_t a = 0; a < cnt && !okm; a++)
                okm = (uint32_t)s.attr_pool[start + a] == (uint32_t)term.v0 &&
                      (uint32_t)s.attr_pool[start + cnt + a] ==
                          (uint32_t)term.v1;
            if (term.op == OP_NE) okm = !okm;
            if (term.group == 0) {
                if (!okm) return false;
            } else {
                uint32_t bit = 1u << (term.group & 31);
                need |= bit;
                if (okm) have |= bit;
            }
            continue;
        }
        uint64_t v = src_value(s, row, term.family, term.idx, 0, q.time_base_s);
        bool ok;
        switch (term.op) {
            case OP_EQ: ok = v == term.v0; break;
            case OP_NE: ok = v != term.v0; break;
            case OP_LT: ok = v < term.v0; break;
            case OP_LE: ok = v <= term.v0; break;
            case OP_GT: ok = v > term.v0; break;
            case OP_GE: ok = v >= term.v0; break;
            case OP_BETWEEN: ok = v >= term.v0 && v <= term.v1; break;
            default: ok = true;
        }
        if (term.group == 0) {
            if (!ok) return false;
        } else {
            uint32_t bit = 1u << (term.group & 31);
            need |= bit;
            if (ok) have |= bit;
        }
    }
    return (have & need) == need;
}

// group table: gkeys u64 hash (claim word), graw [cap, QMAX_KEYS] raw key
// values, gvals [cap, QMAX_AGGS] u64 accumulators (min encoded as ~v).
// group claim: load-first probe, CAS on empty (shared by both agg paths)
DEV uint32_t group_claim(uint64_t h, const uint64_t* kraw, uint32_t n_keys,
                         uint64_t* gkeys, uint64_t* graw,
                         uint32_t cap_mask) {
    uint32_t slot = (uint32_t)(h & cap_mask);
    for (uint32_t probe = 0; probe <= cap_mask; probe++) {
        uint64_t cur = gkeys[slot];
        if (cur == h) return slot;
        if (cur == EMPTY_KEY) {
            uint64_t old = atomicCAS((unsigned long long*)&gkeys[slot],
                                     EMPTY_KEY, h);
            if (old == EMPTY_KEY) {
                for (uint32_t k = 0; k < n_keys; k++)
                    graw[(uint64_t)slot * QMAX_KEYS + k] = kraw[k];
                return slot;
            }
            if (old == h) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return 0;
}

constexpr uint32_t QAGG_NSLOT = 512;

// One row's LDS-table aggregation (shared by the direct and the
// partitioned kernels): probe/claim the block-local table, accumulate;
// on LDS saturation fall through to the global table directly.
DEV void agg_vals_lds(const QuerySpec& q, uint64_t h,
                      const uint64_t* kraw, const uint64_t* varr,
                      uint64_t* lkey, uint32_t* lgslot,
                      unsigned long long (*lagg)[QMAX_AGGS],
                      uint64_t* gkeys, uint64_t* graw,
                      unsigned long long* gvals, uint32_t cap_mask) {
    uint32_t slot = (uint32_t)h & (QAGG_NSLOT - 1);
    uint32_t gslot = 0xFFFFFFFFu;
    for (uint32_t probe = 0; probe < 16; probe++) {
        uint64_t cur = lkey[slot];
        if (cur == EMPTY_KEY) {
            uint64_t old = atomicCAS((unsigned long long*)&lkey[slot],
                                     (unsigned long long)EMPTY_KEY,
                                     (unsigned long long)h);
            if (old == EMPTY_KEY) {
                lgslot[slot] = group_claim(h, kraw, q.n_keys, gkeys, graw,
                                           cap_mask);
                __threadfence_block();
                cur = h;
            } else {
                cur = old;
            }
        }
        if (cur == h) {
            gslot = slot;
            break;
        }
        slot = (slot + 1) & (QAGG_NSLOT - 1);
    }
    // two separate accumulate loops: merging them behind one pointer
    // forces FLAT atomics on the LDS path (the compiler can no longer
    // prove addrspace(3)) — measured 2.7x slower on the direct kernel
    if (gslot != 0xFFFFFFFFu) {
        for (uint32_t a = 0; a < q.n_aggs; a++) {
            uint32_t op = q.aggs[a].op;
            if (op == AGGOP_COUNT || op == AGGOP_SUM)
                atomicAdd(&lagg[gslot][a], (unsigned long long)varr[a]);
            else if (op == AGGOP_MIN)
                atomicMin(&lagg[gslot][a], (unsigned long long)varr[a]);
            else
                atomicMax(&lagg[gslot][a], (unsigned long long)varr[a]);
        }
    } else {  // LDS table saturated for this key: go global directly
        uint32_t g = group_claim(h, kraw, q.n_keys, gkeys, graw, cap_mask);
        unsigned long long* acc = &gvals[(uint64_t)g * QMAX_AGGS];
        for (uint32_t a = 0; a < q.n_aggs; a++) {
            uint32_t op = q.aggs[a].op;
            if (op == AGGOP_COUNT || op == AGGOP_SUM)
                atomicAdd(&acc[a], (unsigned long long)varr[a]);
            else if (op == AGGOP_MIN)
                atomicMin(&acc[a], (unsigned long long)varr[a]);
            else
                atomicMax(&acc[a], (unsigned long long)varr[a]);
        }
    }
}

DEV void agg_row_lds(const SegView& s, const QuerySpec& q, uint64_t row,
                     uint64_t* lkey, uint32_t* lgslot,
                     unsigned long long (*lagg)[QMAX_AGGS],
                     uint64_t* gkeys, uint64_t* graw,
                     unsigned long long* gvals, uint32_t cap_mask) {
    uint64_t kraw[QMAX_KEYS];
    uint64_t h = 0x243F6A8885A308D3ull;
    for (uint32_t k = 0; k < q.n_keys; k++) {
        kraw[k] = src_value(s, row, q.keys[k].family, q.keys[k].idx,
                            q.keys[k].bucket, q.time_base_s);
        h = mix64(h ^ kraw[k] ^ ((uint64_t)k << 56));
    }
    if (h == EMPTY_KEY) h = 1;
    uint64_t varr[QMAX_AGGS];
    for (uint32_t a = 0; a < q.n_aggs; a++)
        varr[a] = q.aggs[a].op == AGGOP_COUNT ? 1
            : src_value(s, row, q.aggs[a].family, q.aggs[a].idx, 0,
                        q.time_base_s);
    agg_vals_lds(q, h, kraw, varr, lkey, lgslot, lagg, gkeys, graw,
                 gvals, cap_mask);
}

DEV void agg_lds_init(const QuerySpec& q, uint64_t* lkey,
                      unsigned long long (*lagg)[QMAX_AGGS]) {
    for (uint32_t sl = threadIdx.x; sl < QAGG_NSLOT; sl += blockDim.x) {
        lkey[sl] = EMPTY_KEY;
        for (uint32_t a = 0; a < q.n_aggs; a++)
            lagg[sl][a] = q.aggs[a].op == AGGOP_MIN ? ~0ull : 0ull;
    }
}

DEV void agg_lds_flush(const QuerySpec& q, const uint64_t* lkey,
                       const uint32_t* lgslot,
                       unsigned long long (*lagg)[QMAX_AGGS],
                       unsigned long long* gvals) {
    for (uint32_t sl = threadIdx.x; sl < QAGG_NSLOT; sl += blockDim.x) {
        if (lkey[sl] == EMPTY_KEY) continue;
        unsigned long long* acc = &gvals[(uint64_t)lgslot[sl] * QMAX_AGGS];
        for (uint32_t a = 0; a < q.n_aggs; a++) {
            uint32_t op = q.aggs[a].op;
            unsigned long long v = lagg[sl][a];
            if (op == AGGOP_COUNT || op == AGGOP_SUM) {
                if (v) atomicAdd(&acc[a], v);
            } else if (op == AGGOP_MIN) {
                if (v != ~0ull) atomicMin(&acc[a], v);
            } else {
                if (v) atomicMax(&acc[a], v);
            }
        }
    }
}

__global__ void k_query_agg(SegView s, QuerySpec q, uint32_t n, uint64_t base_row,
                            uint64_t* __restrict__ gkeys,
                            uint64_t* __restrict__ graw,
                            unsigned long long* __restrict__ gvals,
                            uint32_t cap_mask) {
    // Grid-strided scan with an LDS-resident group table: each workgroup
    // aggregates its stripe into shared memory (~37 KB of the 160 KB/CU
    // LDS: 512 slots x (key + global-slot + 8 accumulators)) and flushes
    // once at the end — global atomics drop from per-row to
    // per-(block x live slot). Groups are identified by the 64-bit mixed
    // key hash (same convention as the global table); the LDS claimant
    // registers the group globally immediately, so raw keys never stage
    // in LDS. Rows whose key misses the table (very high per-block
    // cardinality) fall back to per-lane global accumulation.
    __shared__ uint64_t lkey[QAGG_NSLOT];
    __shared__ uint32_t lgslot[QAGG_NSLOT];
    __shared__ unsigned long long lagg[QAGG_NSLOT][QMAX_AGGS];
    agg_lds_init(q, lkey, lagg);
    __syncthreads();
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += (uint64_t)gridDim.x * blockDim.x) {
        uint64_t row = base_row + i;
        if (!eval_terms(s, row, q)) continue;
        agg_row_lds(s, q, row, lkey, lgslot, lagg, gkeys, graw, gvals,
                    cap_mask);
    }
    __syncthreads();
    agg_lds_flush(q, lkey, lgslot, lagg, gvals);
}

// ---- radix-partitioned group-by (high cardinality) ----
// When the key cardinality is far above QAGG_NSLOT, the direct kernel
// spills most rows to per-row global atomics. The partitioned path
// buckets rows by 8 hash bits first, then aggregates one bucket per
// workgroup group: each block then sees ~1/256th of the distinct keys,
// so the LDS table holds them all (zero spill up to ~128k groups).
constexpr uint32_t QPART_NB = 256;
constexpr uint32_t QPART_SHIFT = 40;

DEV uint64_t qpart_hash(const SegView& s, const QuerySpec& q,
                        uint64_t row) {
    uint64_t h = 0x243F6A8885A308D3ull;
    for (uint32_t k = 0; k < q.n_keys; k++) {
        uint64_t kv = src_value(s, row, q.keys[k].family, q.keys[k].idx,
                                q.keys[k].bucket, q.time_base_s);
        h = mix64(h ^ kv ^ ((uint64_t)k << 56));
    }
    return h == EMPTY_KEY ? 1 : h;
}

__global__ void k_qpart_count(SegView s, QuerySpec q, uint32_t n,
                              uint64_t base_row,
                              uint32_t* __restrict__ counts) {
    __shared__ uint32_t lc[QPART_NB];
    for (uint32_t i = threadIdx.x; i < QPART_NB; i += blockDim.x)
        lc[i] = 0;
    __syncthreads();
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += (uint64_t)gridDim.x * blockDim.x) {
        uint64_t row = base_row + i;
        if (!eval_terms(s, row, q)) continue;
        uint64_t h = qpart_hash(s, q, row);
        atomicAdd(&lc[(h >> QPART_SHIFT) & (QPART_NB - 1)], 1u);
    }
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < QPART_NB; i += blockDim.x)
        if (lc[i]) atomicAdd(&counts[i], lc[i]);
}

// The scatter stores the COMPUTED per-row payload (key values +
// aggregate operands), not row ids: pass 3 then reads each bucket as a
// dense sequential stripe. Scattering ids instead was measured SLOWER
// than the direct kernel (the bucketed pass degenerates into random
// 8-byte gathers across the whole segment).
__global__ void k_qpart_scatter(SegView s, QuerySpec q, uint32_t n,
                                uint64_t base_row,
                                uint32_t* __restrict__ cursors,
                                uint64_t* __restrict__ out_pay) {
    // two-pass block staging over a CONTIGUOUS per-block stripe: count
    // the stripe's bucket histogram in LDS, reserve one contiguous
    // chunk per (block, bucket) with a single global atomic each, then
    // write. A naive per-row atomicAdd on 256 global cursors serializes
    // 30M RMWs onto 4 cache lines.
    __shared__ uint32_t lc[QPART_NB];
    __shared__ uint32_t lbase[QPART_NB];
    uint32_t w = q.n_keys + q.n_aggs;
    uint64_t per = ((uint64_t)n + gridDim.x - 1) / gridDim.x;
    uint64_t lo = (uint64_t)blockIdx.x * per;
    uint64_t hi = lo + per < n ? lo + per : n;
    for (uint32_t i = threadIdx.x; i < QPART_NB; i += blockDim.x)
        lc[i] = 0;
    __syncthreads();
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint64_t row = base_row + i;
        if (!eval_terms(s, row, q)) continue;
        uint64_t h = qpart_hash(s, q, row);
        atomicAdd(&lc[(h >> QPART_SHIFT) & (QPART_NB - 1)], 1u);
    }
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < QPART_NB; i += blockDim.x) {
        uint32_t c = lc[i];
        lbase[i] = c ? atomicAdd(&cursors[i], c) : 0u;
        lc[i] = 0;
    }
    __syncthreads();
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint64_t row = base_row + i;
        if (!eval_terms(s, row, q)) continue;
        uint64_t kraw[QMAX_KEYS];
        uint64_t h = 0x243F6A8885A308D3ull;
        for (uint32_t k = 0; k < q.n_keys; k++) {
            kraw[k] = src_value(s, row, q.keys[k].family, q.keys[k].idx,
                                q.keys[k].bucket, q.time_base_s);
            h = mix64(h ^ kraw[k] ^ ((uint64_t)k << 56));
        }
        if (h == EMPTY_KEY) h = 1;
        uint32_t b = (uint32_t)(h >> QPART_SHIFT) & (QPART_NB - 1);
        uint32_t pos = lbase[b] + atomicAdd(&lc[b], 1u);
        uint64_t* dst = out_pay + (uint64_t)pos * w;
        for (uint32_t k = 0; k < q.n_keys; k++) dst[k] = kraw[k];
        for (uint32_t a = 0; a < q.n_aggs; a++)
            dst[q.n_keys + a] = q.aggs[a].op == AGGOP_COUNT ? 1
                : src_value(s, row, q.aggs[a].family, q.aggs[a].idx, 0,
                            q.time_base_s);
    }
}

// single-block exclusive scan over the QPART_NB counters:
// counts[i] (in: per-bucket count) -> counts[i] = start offset (out);
// cursors[i] = start offset too (the scatter pass advances cursors to
// the bucket ends, which k_qpart_agg uses as `hi`... but scatter order
// is nondeterministic, so ends must be start+count: write both).
__global__ void k_qpart_prefix(uint32_t* counts, uint32_t* cursors) {
    __shared__ uint32_t vals[QPART_NB];
    uint32_t i = threadIdx.x;
    vals[i] = counts[i];
    __syncthreads();
    // simple serial scan by lane 0 (256 values; latency-trivial)
    if (i == 0) {
        uint32_t acc = 0;
        for (uint32_t k = 0; k < QPART_NB; k++) {
            uint32_t c = vals[k];
            counts[k] = acc;        // start
            cursors[k] = acc;       // scatter cursor starts here
            acc += c;
            vals[k] = acc;          // end (reused below)
        }
    }
    __syncthreads();
}

__global__ void k_qpart_agg(QuerySpec q,
                            const uint64_t* __restrict__ pay,
                            const uint32_t* __restrict__ starts,
                            const uint32_t* __restrict__ ends,
                            uint64_t* __restrict__ gkeys,
                            uint64_t* __restrict__ graw,
                            unsigned long long* __restrict__ gvals,
                            uint32_t cap_mask,
                            uint32_t blocks_per_bucket) {
    __shared__ uint64_t lkey[QAGG_NSLOT];
    __shared__ uint32_t lgslot[QAGG_NSLOT];
    __shared__ unsigned long long lagg[QAGG_NSLOT][QMAX_AGGS];
    agg_lds_init(q, lkey, lagg);
    __syncthreads();
    uint32_t w = q.n_keys + q.n_aggs;
    uint32_t bucket = blockIdx.x / blocks_per_bucket;
    uint32_t sub = blockIdx.x % blocks_per_bucket;
    uint32_t lo = starts[bucket], hi = ends[bucket];
    // rows were pre-filtered and their payload materialized by the
    // scatter pass: this is a dense sequential read, no SegView access
    for (uint64_t j = lo + (uint64_t)sub * blockDim.x + threadIdx.x;
         j < hi; j += (uint64_t)blocks_per_bucket * blockDim.x) {
        const uint64_t* row = pay + j * w;
        uint64_t h = 0x243F6A8885A308D3ull;
        for (uint32_t k = 0; k < q.n_keys; k++)
            h = mix64(h ^ row[k] ^ ((uint64_t)k << 56));
        if (h == EMPTY_KEY) h = 1;
        agg_vals_lds(q, h, row, row + q.n_keys, lkey, lgslot, lagg,
                     gkeys, graw, gvals, cap_mask);
    }
    __syncthreads();
    agg_lds_flush(q, lkey, lgslot, lagg, gvals);
}

// non-aggregated SELECT: emit matching row ids (bounded)
__global__ void k_query_select(SegView s, QuerySpec q, uint32_t n, uint64_t base_row,
                               uint64_t* __restrict__ out_rows,
                               uint32_t* __restrict__ out_ctr, uint32_t out_cap) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t row = base_row + i;
    if (!eval_terms(s, row, q)) return;
    uint32_t e = atomicAdd(out_ctr, 1u);
    if (e < out_cap) out_rows[e] = row;
}

// ----------------------------------------------------------------------
// cold-segment bit packing: fixed-width pack of (value - base) streams.
// Word-centric pack (each thread owns one output u32 — no atomics, fully
// coalesced writes); value-centric unpack. A constant column packs to
// bits==0 (no payload at all). Extends the HBM hot window several-fold
// for demoted segments (reference analog: ClickHouse column codecs
// T64/DoubleDelta on cold parts).
// ----------------------------------------------------------------------
__global__ void k_pack_bits(const uint32_t* __restrict__ src, uint32_t n,
                            uint32_t base, uint32_t bits,
                            uint32_t* __restrict__ out,
                            uint32_t out_words) {
    uint32_t w = blockIdx.x * blockDim.x + threadIdx.x;
    if (w >= out_words) return;
    uint64_t bit0 = (uint64_t)w * 32u;
    uint32_t i = (uint32_t)(bit0 / bits);
    uint32_t acc = 0;
    for (; i < n; i++) {
        uint64_t vb = (uint64_t)i * bits;
        if (vb >= bit0 + 32u) break;
        if (vb + bits <= bit0) continue;  // first value may start earlier
        uint64_t v = (uint64_t)(src[i] - base);
        int64_t sh = (int64_t)vb - (int64_t)bit0;
        if (sh >= 0)
            acc |= (uint32_t)(v << sh);
        else
            acc |= (uint32_t)(v >> (uint32_t)(-sh));
    }
    out[w] = acc;
}

__global__ void k_unpack_bits(const uint32_t* __restrict__ packed,
                              uint32_t n, uint32_t base, uint32_t bits,
                              uint32_t* __restrict__ out) {
    uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint64_t vb = (uint64_t)i * bits;
    uint32_t w = (uint32_t)(vb >> 5), off = (uint32_t)(vb & 31u);
    uint64_t lo = packed[w];
    uint64_t hi = (off + bits > 32u) ? packed[w + 1] : 0ull;
    uint64_t v = ((hi << 32) | lo) >> off;
    uint32_t mask = bits >= 32u ? 0xFFFFFFFFu : ((1u << bits) - 1u);
    out[i] = base + (uint32_t)(v & mask);
}

inline uint32_t grid_for(uint64_t total) {
    return (uint32_t)((total + BLOCK - 1) / BLOCK);
}

}  // namespace

// ----------------------------------------------------------------------
// C API (ctypes). All functions take the HIP stream as uint64 and return
// hipError_t as int; launches are async on that stream.
// ----------------------------------------------------------------------

#define STREAM(s) reinterpret_cast<hipStream_t>(s)

extern "C" {

int df_gpu_ready() { int n = 0; return hipGetDeviceCount(&n) == hipSuccess && n > 0; }

int df_decode_l7(const void* payload, const void* offs, const void* lens,
                 uint32_t n,
                 void* u64c, void* u32c, void* u8c, void* strc,
                 void* attrc, void* attr_cnt,
                 uint64_t stride, uint64_t base_row, uint64_t scratch_stride,
                 uint64_t stream) {
    L7Cols cols{(uint64_t*)u64c, (uint32_t*)u32c, (uint8_t*)u8c, (uint64_t*)strc,
                (uint64_t*)attrc, (uint8_t*)attr_cnt, stride, base_row,
                scratch_stride};
    hipLaunchKernelGGL(k_decode_l7, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint8_t*)payload, (const uint32_t*)offs,
                       (const uint32_t*)lens, n, cols);
    return (int)hipGetLastError();
}

int df_decode_l4(const void* payload, const void* offs, const void* lens,
                 uint32_t n, void* u64c, void* u32c, void* u8c, void* strc,
                 uint64_t stride, uint64_t base_row, uint64_t scratch_stride,
                 uint64_t stream) {
    L4Cols cols{(uint64_t*)u64c, (uint32_t*)u32c, (uint8_t*)u8c,
                (uint64_t*)strc, stride, base_row, scratch_stride};
    hipLaunchKernelGGL(k_decode_l4, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint8_t*)payload, (const uint32_t*)offs,
                       (const uint32_t*)lens, n, cols);
    return (int)hipGetLastError();
}

// specs: n_tables x RuSpec bytes; ptrs: per-table
// [tkeys, traw, tvals, drops] device pointers + caps
static void ru_multi_fill(const void* specs, uint32_t n_tables,
                          const uint64_t* ptrs, const uint32_t* caps,
                          RuMulti& mu, RuTablePtrs& tp) {
    mu.n_tables = n_tables;
    for (uint32_t t = 0; t < n_tables; t++) {
        __builtin_memcpy(&mu.ru[t], (const uint8_t*)specs + t * sizeof(RuSpec),
                         sizeof(RuSpec));
        tp.tkeys[t] = (uint64_t*)ptrs[t * 4 + 0];
        tp.traw[t] = (uint64_t*)ptrs[t * 4 + 1];
        tp.tvals[t] = (unsigned long long*)ptrs[t * 4 + 2];
        tp.drops[t] = (unsigned long long*)ptrs[t * 4 + 3];
        tp.cap_mask[t] = caps[t] - 1;
    }
}

int df_rollup_l4(void* u64c, void* u32c, void* u8c, uint64_t stride,
                 uint64_t base_row, uint32_t n, uint64_t time_base_s,
                 const void* specs, uint32_t n_tables, const void* ptrs,
                 const void* caps, uint64_t stream) {
    L4Cols cols{(uint64_t*)u64c, (uint32_t*)u32c, (uint8_t*)u8c,
                nullptr, stride, base_row};
    RuMulti mu;
    RuTablePtrs tp;
    ru_multi_fill(specs, n_tables, (const uint64_t*)ptrs,
                  (const uint32_t*)caps, mu, tp);
    uint32_t blocks = grid_for(n);
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_rollup_l4, dim3(blocks), dim3(BLOCK), 0, STREAM(stream),
                       cols, n, time_base_s, mu, tp);
    return (int)hipGetLastError();
}

int df_rollup_insert(const void* kws, const void* vals, const void* ops,
                     uint32_t n, uint32_t nw, uint32_t nv,
                     void* tkeys, void* traw, void* tvals, uint32_t cap,
                     void* drops, uint64_t stream) {
    hipLaunchKernelGGL(k_rollup_insert, dim3(grid_for(n)), dim3(BLOCK), 0,
                       STREAM(stream),
                       (const uint64_t*)kws, (const unsigned long long*)vals,
                       (const uint8_t*)ops, n, nw, nv, (uint64_t*)tkeys,
                       (uint64_t*)traw, (unsigned long long*)tvals, cap - 1,
                       (unsigned long long*)drops);
    return (int)hipGetLastError();
}

int df_gather_records(const void* src, const void* offs, const void* lens,
                      const void* sel, const void* dst_off, uint32_t m,
                      void* out, uint64_t stream) {
    uint32_t waves = BLOCK / 64;
    uint32_t blocks = (m + waves - 1) / waves;
    hipLaunchKernelGGL(k_gather_records, dim3(blocks), dim3(BLOCK), 0,
                       STREAM(stream), (const uint8_t*)src,
                       (const uint32_t*)offs, (const uint32_t*)lens,
                       (const uint32_t*)sel, (const uint64_t*)dst_off, m,
                       (uint8_t*)out);
    return (int)hipGetLastError();
}

int df_kg_build(const void* keys, const void* vals, uint32_t n,
                void* tkeys, void* tvals, uint32_t cap, uint64_t stream) {
    hipLaunchKernelGGL(k_kg_build, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint64_t*)keys, (const uint32_t*)vals, n,
                       (uint64_t*)tkeys, (uint32_t*)tvals, cap - 1);
    return (int)hipGetLastError();
}

int df_kg_probe(const void* epc0, const void* ip0, const void* epc1, const void* ip1,
                uint32_t n, const void* tkeys, const void* tvals, uint32_t cap,
                void* out, uint64_t stride, uint64_t base_row, uint64_t stream) {
    hipLaunchKernelGGL(k_kg_probe, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint32_t*)epc0, (const uint32_t*)ip0,
                       (const uint32_t*)epc1, (const uint32_t*)ip1, n,
                       (const uint64_t*)tkeys, (const uint32_t*)tvals, cap - 1,
                       (uint32_t*)out, stride, base_row);
    return (int)hipGetLastError();
}

int df_intern_many(const void* payload, const void* refs, const void* ref_rows,
                   const void* domains,
                   uint32_t C, uint32_t n, uint64_t ref_stride, uint64_t ref_base_row,
                   void* tkeys, uint32_t cap,
                   void* emit, void* emit_ctr, uint32_t emit_cap,
                   void* out_ids, uint64_t out_stride, uint64_t out_base_row,
                   uint64_t stream) {
    uint64_t total = (uint64_t)C * n;
    hipLaunchKernelGGL(k_intern_many, dim3(grid_for(total)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint8_t*)payload, (const uint64_t*)refs,
                       (const uint16_t*)ref_rows,
                       (const uint8_t*)domains, C, n, ref_stride, ref_base_row,
                       (uint64_t*)tkeys, cap - 1,
                       (uint64_t*)emit, (uint32_t*)emit_ctr, emit_cap,
                       (uint32_t*)out_ids, out_stride, out_base_row);
    return (int)hipGetLastError();
}

int df_intern_attrs(const void* payload, const void* attr_refs,
                    const void* attr_cnt, uint32_t n, uint64_t stride,
                    uint64_t base_row, uint64_t ref_stride,
                    void* tkeys, uint32_t cap,
                    void* emit, void* emit_ctr, uint32_t emit_cap,
                    const void* attr_start, void* attr_pool,
                    uint64_t stream) {
    hipLaunchKernelGGL(k_intern_attrs, dim3(grid_for(n)), dim3(BLOCK), 0,
                       STREAM(stream),
                       (const uint8_t*)payload, (const uint64_t*)attr_refs,
                       (const uint8_t*)attr_cnt, n, stride, base_row,
                       ref_stride, (uint64_t*)tkeys, cap - 1,
                       (uint64_t*)emit, (uint32_t*)emit_ctr, emit_cap,
                       (const uint32_t*)attr_start, (int32_t*)attr_pool);
    return (int)hipGetLastError();
}

int df_pool_lens(const void* strc, const void* pool_cols, uint32_t npc, uint32_t n,
                 uint64_t stride, uint64_t base_row, void* row_len, uint64_t stream) {
    hipLaunchKernelGGL(k_pool_lens, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint64_t*)strc, (const uint8_t*)pool_cols, npc, n,
                       stride, base_row, (uint32_t*)row_len);
    return (int)hipGetLastError();
}

int df_pool_gather(const void* payload, const void* strc, const void* pool_cols,
                   uint32_t npc, uint32_t n, uint64_t stride, uint64_t base_row,
                   const void* row_start, void* pool, uint64_t pool_base,
                   void* out_rowref, void* out_lens,
                   uint64_t out_stride, uint64_t out_base_row,
                   uint64_t stream) {
    hipLaunchKernelGGL(k_pool_gather, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       (const uint8_t*)payload, (const uint64_t*)strc,
                       (const uint8_t*)pool_cols, npc, n, stride, base_row,
                       (const uint64_t*)row_start, (uint8_t*)pool, pool_base,
                       (uint64_t*)out_rowref, (int16_t*)out_lens,
                       out_stride, out_base_row);
    return (int)hipGetLastError();
}

int df_rollup_l7(void* u64c, void* u32c, void* u8c, uint64_t stride,
                 uint64_t base_row, uint32_t n, uint64_t time_base_s,
                 const void* specs, uint32_t n_tables, const void* ptrs,
                 const void* caps, uint64_t stream) {
    L7Cols cols{(uint64_t*)u64c, (uint32_t*)u32c, (uint8_t*)u8c,
                nullptr, nullptr, nullptr, stride, base_row};
    RuMulti mu;
    RuTablePtrs tp;
    ru_multi_fill(specs, n_tables, (const uint64_t*)ptrs,
                  (const uint32_t*)caps, mu, tp);
    uint32_t blocks = grid_for(n);
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_rollup_l7, dim3(blocks), dim3(BLOCK), 0, STREAM(stream),
                       cols, n, time_base_s, mu, tp);
    return (int)hipGetLastError();
}

int df_query_agg(const void* u64c, const void* u32c, const void* u8c,
                 const void* didc,
                 const void* kg_tk, const void* kg_tv, uint32_t kg_cap,
                 const void* attr_pool,
                 const void* attr_start, const void* attr_cnt,
                 const void* str_rowref, const void* str_lens,
                 const void* pool,
                 uint64_t stride, uint64_t n_rows,
                 const void* spec,  // QuerySpec, host-built bytes
                 uint32_t n, uint64_t base_row,
                 void* gkeys, void* graw, void* gvals, uint32_t cap,
                 uint64_t stream) {
    SegView s{(const uint64_t*)u64c, (const uint32_t*)u32c, (const uint8_t*)u8c,
              (const uint32_t*)didc,
              (const uint64_t*)kg_tk, (const uint32_t*)kg_tv,
              kg_cap ? kg_cap - 1 : 0,
              (const int32_t*)attr_pool, (const uint32_t*)attr_start,
              (const uint8_t*)attr_cnt, (const uint64_t*)str_rowref,
              (const int16_t*)str_lens,
              (const uint8_t*)pool, stride, n_rows};
    QuerySpec q;
    __builtin_memcpy(&q, spec, sizeof(QuerySpec));
    // grid-stride kernel: bounded grid so each workgroup amortizes its
    // LDS group table over many rows (still >> 256 workgroups: 4096
    // covers all 8 XCDs with deep occupancy)
    uint32_t blocks = grid_for(n);
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_query_agg, dim3(blocks), dim3(BLOCK), 0, STREAM(stream),
                       s, q, n, base_row, (uint64_t*)gkeys, (uint64_t*)graw,
                       (unsigned long long*)gvals, cap - 1);
    return (int)hipGetLastError();
}

int df_qpart_agg(const void* u64c, const void* u32c, const void* u8c,
                 const void* didc,
                 const void* kg_tk, const void* kg_tv, uint32_t kg_cap,
                 const void* attr_pool,
                 const void* attr_start, const void* attr_cnt,
                 const void* str_rowref, const void* str_lens,
                 const void* pool,
                 uint64_t stride, uint64_t n_rows,
                 const void* spec, uint32_t n, uint64_t base_row,
                 void* counts, void* cursors, void* row_scratch,
                 void* gkeys, void* graw, void* gvals, uint32_t cap,
                 uint64_t stream) {
    // Three-pass radix-partitioned aggregation. counts/cursors are
    // QPART_NB u32 device buffers; the caller pre-fills cursors with the
    // exclusive prefix sums of counts AFTER the count pass via
    // df_qpart_prefix (all device-side; no host sync).
    SegView s{(const uint64_t*)u64c, (const uint32_t*)u32c, (const uint8_t*)u8c,
              (const uint32_t*)didc,
              (const uint64_t*)kg_tk, (const uint32_t*)kg_tv,
              kg_cap ? kg_cap - 1 : 0,
              (const int32_t*)attr_pool, (const uint32_t*)attr_start,
              (const uint8_t*)attr_cnt, (const uint64_t*)str_rowref,
              (const int16_t*)str_lens,
              (const uint8_t*)pool, stride, n_rows};
    QuerySpec q;
    __builtin_memcpy(&q, spec, sizeof(QuerySpec));
    uint32_t blocks = grid_for(n);
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_qpart_count, dim3(blocks), dim3(BLOCK), 0,
                       STREAM(stream), s, q, n, base_row,
                       (uint32_t*)counts);
    // device-side exclusive scan of 256 counters (one tiny block) —
    // writes cursors = starts and counts = ends (inclusive scan)
    hipLaunchKernelGGL(k_qpart_prefix, dim3(1), dim3(QPART_NB), 0,
                       STREAM(stream), (uint32_t*)counts,
                       (uint32_t*)cursors);
    hipLaunchKernelGGL(k_qpart_scatter, dim3(blocks), dim3(BLOCK), 0,
                       STREAM(stream), s, q, n, base_row,
                       (uint32_t*)cursors, (uint64_t*)row_scratch);
    // after scatter, cursors[b] == ends[b] (inclusive scan) and
    // starts[b] = ends[b] - count — k_qpart_agg derives lo from the
    // auxiliary starts written by k_qpart_prefix into counts
    constexpr uint32_t BPB = 16;
    hipLaunchKernelGGL(k_qpart_agg, dim3(QPART_NB * BPB), dim3(BLOCK), 0,
                       STREAM(stream), q, (const uint64_t*)row_scratch,
                       (const uint32_t*)counts, (const uint32_t*)cursors,
                       (uint64_t*)gkeys, (uint64_t*)graw,
                       (unsigned long long*)gvals, cap - 1, BPB);
    return (int)hipGetLastError();
}

int df_query_select(const void* u64c, const void* u32c, const void* u8c,
                    const void* didc,
                    const void* kg_tk, const void* kg_tv, uint32_t kg_cap,
                    const void* attr_pool,
                    const void* attr_start, const void* attr_cnt,
                    const void* str_rowref, const void* str_lens,
                    const void* pool,
                    uint64_t stride, uint64_t n_rows,
                    const void* spec, uint32_t n, uint64_t base_row,
                    void* out_rows, void* out_ctr, uint32_t out_cap,
                    uint64_t stream) {
    SegView s{(const uint64_t*)u64c, (const uint32_t*)u32c, (const uint8_t*)u8c,
              (const uint32_t*)didc,
              (const uint64_t*)kg_tk, (const uint32_t*)kg_tv,
              kg_cap ? kg_cap - 1 : 0,
              (const int32_t*)attr_pool, (const uint32_t*)attr_start,
              (const uint8_t*)attr_cnt, (const uint64_t*)str_rowref,
              (const int16_t*)str_lens,
              (const uint8_t*)pool, stride, n_rows};
    QuerySpec q;
    __builtin_memcpy(&q, spec, sizeof(QuerySpec));
    hipLaunchKernelGGL(k_query_select, dim3(grid_for(n)), dim3(BLOCK), 0, STREAM(stream),
                       s, q, n, base_row, (uint64_t*)out_rows, (uint32_t*)out_ctr,
                       out_cap);
    return (int)hipGetLastError();
}

// rocPRIM device radix sort of u64 keys (grouped-percentile path: the
// composite (group << 44 | value) sort replaces torch.argsort on the
// quantile gather — VERDICT r1 #8). Two-phase: *temp_bytes == 0 sizes
// the temp buffer; second call sorts in place (double-buffered).
int df_sort_u64(void* data, void* data_alt, uint32_t n, void* temp,
                uint64_t* temp_bytes, uint64_t stream) {
    size_t bytes = (size_t)*temp_bytes;
    rocprim::double_buffer<uint64_t> keys((uint64_t*)data,
                                          (uint64_t*)data_alt);
    hipError_t rc = rocprim::radix_sort_keys(
        temp == nullptr ? nullptr : temp, bytes, keys, n, 0, 64,
        STREAM(stream));
    if (rc != hipSuccess) return (int)rc;
    *temp_bytes = bytes;
    if (temp != nullptr && keys.current() != (uint64_t*)data) {
        // result landed in the alternate buffer: copy back
        (void)hipMemcpyAsync(data, data_alt, (size_t)n * 8,
                             hipMemcpyDeviceToDevice, STREAM(stream));
    }
    return 0;
}

int df_spec_sizes(uint32_t* qterm, uint32_t* qkey, uint32_t* qagg, uint32_t* qspec) {
    *qterm = sizeof(QTerm); *qkey = sizeof(QKey); *qagg = sizeof(QAgg);
    *qspec = sizeof(QuerySpec);
    return 0;
}

int df_pack_bits(const void* src, uint32_t n, uint32_t base, uint32_t bits,
                 void* out, uint32_t out_words, uint64_t stream) {
    hipLaunchKernelGGL(k_pack_bits, dim3(grid_for(out_words)), dim3(BLOCK),
                       0, STREAM(stream), (const uint32_t*)src, n, base,
                       bits, (uint32_t*)out, out_words);
    return (int)hipGetLastError();
}

int df_unpack_bits(const void* packed, uint32_t n, uint32_t base,
                   uint32_t bits, void* out, uint64_t stream) {
    hipLaunchKernelGGL(k_unpack_bits, dim3(grid_for(n)), dim3(BLOCK), 0,
                       STREAM(stream), (const uint32_t*)packed, n, base,
                       bits, (uint32_t*)out);
    return (int)hipGetLastError();
}

}  // extern "C"
