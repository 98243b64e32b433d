// HTTP/2 + HPACK + gRPC parsing for the agent core.
// Reference counterparts: agent/src/flow_generator/protocol_logs/http.rs
// (HTTP/2 path) and the go_http2 uprobe pipeline; this is a fresh
// RFC 7540/7541 implementation (static table + dynamic table + canonical
// Huffman; single-packet header blocks — CONTINUATION reassembly is a
// round-2 item).
#pragma once
#include <cstdint>
#include <cstring>
#include <string>
#include <utility>
#include <vector>

#include "hpack_huffman.h"

namespace h2 {

// ---- Huffman decode tree (built once) ----
struct HufNode { int16_t next[2]; int16_t sym; };

inline std::vector<HufNode>& huf_tree() {
    static std::vector<HufNode> tree = [] {
        std::vector<HufNode> t;
        t.push_back({{-1, -1}, -1});
        for (int s = 0; s < 256; s++) {
            uint32_t code = HPACK_HUF_CODE[s];
            int len = HPACK_HUF_LEN[s];
            int cur = 0;
            for (int b = len - 1; b >= 0; b--) {
                int bit = (code >> b) & 1;
                if (t[cur].next[bit] < 0) {
                    t[cur].next[bit] = (int16_t)t.size();
                    t.push_back({{-1, -1}, -1});
                }
                cur = t[cur].next[bit];
            }
            t[cur].sym = (int16_t)s;
        }
        return t;
    }();
    return tree;
}

inline bool huf_decode(const uint8_t* p, uint32_t n, std::string& out) {
    auto& t = huf_tree();
    int cur = 0;
    for (uint32_t i = 0; i < n; i++) {
        for (int b = 7; b >= 0; b--) {
            int bit = (p[i] >> b) & 1;
            int nxt = t[cur].next[bit];
            if (nxt < 0) return false;
            cur = nxt;
            if (t[cur].sym >= 0) {
                out.push_back((char)t[cur].sym);
                cur = 0;
            }
        }
    }
    return true;  // trailing 1-bits padding left mid-tree: fine
}

// ---- HPACK static table (RFC 7541 Appendix A) ----
struct StaticEntry { const char* name; const char* value; };
static const StaticEntry HPACK_STATIC[62] = {
    {"", ""},  // 1-based
    {":authority", ""}, {":method", "GET"}, {":method", "POST"},
    {":path", "/"}, {":path", "/index.html"}, {":scheme", "http"},
    {":scheme", "https"}, {":status", "200"}, {":status", "204"},
    {":status", "206"}, {":status", "304"}, {":status", "400"},
    {":status", "404"}, {":status", "500"}, {"accept-charset", ""},
    {"accept-encoding", "gzip, deflate"}, {"accept-language", ""},
    {"accept-ranges", ""}, {"accept", ""},
    {"access-control-allow-origin", ""}, {"age", ""}, {"allow", ""},
    {"authorization", ""}, {"cache-control", ""}, {"content-disposition", ""},
    {"content-encoding", ""}, {"content-language", ""}, {"content-length", ""},
    {"content-location", ""}, {"content-range", ""}, {"content-type", ""},
    {"cookie", ""}, {"date", ""}, {"etag", ""}, {"expect", ""},
    {"expires", ""}, {"from", ""}, {"host", ""}, {"if-match", ""},
    {"if-modified-since", ""}, {"if-none-match", ""}, {"if-range", ""},
    {"if-unmodified-since", ""}, {"last-modified", ""}, {"link", ""},
    {"location", ""}, {"max-forwards", ""}, {"proxy-authenticate", ""},
    {"proxy-authorization", ""}, {"range", ""}, {"referer", ""},
    {"refresh", ""}, {"retry-after", ""}, {"server", ""}, {"set-cookie", ""},
    {"strict-transport-security", ""}, {"transfer-encoding", ""},
    {"user-agent", ""}, {"vary", ""}, {"via", ""}, {"www-authenticate", ""},
};

using Header = std::pair<std::string, std::string>;
using DynTable = std::vector<Header>;  // newest first

inline bool hpack_int(const uint8_t* p, uint32_t n, uint32_t& pos,
                      int prefix, uint32_t& out) {
    if (pos >= n) return false;
    uint32_t mask = (1u << prefix) - 1;
    out = p[pos++] & mask;
    if (out < mask) return true;
    uint32_t m = 0;
    while (pos < n) {
        uint8_t b = p[pos++];
        out += (uint32_t)(b & 0x7F) << m;
        if (!(b & 0x80)) return true;
        m += 7;
        if (m > 28) return false;
    }
    return false;
}

inline bool hpack_string(const uint8_t* p, uint32_t n, uint32_t& pos,
                         std::string& out) {
    if (pos >= n) return false;
    bool huff = p[pos] & 0x80;
    uint32_t len;
    if (!hpack_int(p, n, pos, 7, len)) return false;
    if (pos + len > n) return false;
    out.clear();
    bool ok = true;
    if (huff) ok = huf_decode(p + pos, len, out);
    else out.assign((const char*)p + pos, len);
    pos += len;
    return ok;
}

inline bool lookup(const DynTable& dyn, uint32_t idx, Header& out) {
    if (idx == 0) return false;
    if (idx <= 61) {
        out = {HPACK_STATIC[idx].name, HPACK_STATIC[idx].value};
        return true;
    }
    uint32_t d = idx - 62;
    if (d >= dyn.size()) return false;
    out = dyn[d];
    return true;
}

// Decode one HPACK header block; appends to headers, updates dyn table.
inline bool hpack_decode(const uint8_t* p, uint32_t n, DynTable& dyn,
                         std::vector<Header>& headers) {
    uint32_t pos = 0;
    while (pos < n) {
        uint8_t b = p[pos];
        if (b & 0x80) {  // indexed
            uint32_t idx;
            if (!hpack_int(p, n, pos, 7, idx)) return false;
            Header h;
            if (!lookup(dyn, idx, h)) return false;
            headers.push_back(h);
        } else if (b & 0x40) {  // literal with incremental indexing
            uint32_t idx;
            if (!hpack_int(p, n, pos, 6, idx)) return false;
            Header h;
            if (idx) {
                if (!lookup(dyn, idx, h)) return false;
            } else if (!hpack_string(p, n, pos, h.first)) {
                return false;
            }
            if (!hpack_string(p, n, pos, h.second)) return false;
            dyn.insert(dyn.begin(), h);
            if (dyn.size() > 256) dyn.pop_back();
            headers.push_back(h);
        } else if ((b & 0xE0) == 0x20) {  // dynamic table size update
            uint32_t sz;
            if (!hpack_int(p, n, pos, 5, sz)) return false;
            if (sz == 0) dyn.clear();
        } else {  // literal without indexing / never indexed (prefix 4)
            uint32_t idx;
            if (!hpack_int(p, n, pos, 4, idx)) return false;
            Header h;
            if (idx) {
                if (!lookup(dyn, idx, h)) return false;
            } else if (!hpack_string(p, n, pos, h.first)) {
                return false;
            }
            if (!hpack_string(p, n, pos, h.second)) return false;
            headers.push_back(h);
        }
    }
    return true;
}

// ---- frame walk ----
enum { F_DATA = 0, F_HEADERS = 1, F_SETTINGS = 4, F_CONTINUATION = 9 };

struct FrameView {
    uint8_t type, flags;
    uint32_t stream_id, len;
    const uint8_t* payload;
};

inline bool next_frame(const uint8_t* p, uint32_t n, uint32_t& pos,
                       FrameView& f) {
    if (pos + 9 > n) return false;
    f.len = (p[pos] << 16) | (p[pos + 1] << 8) | p[pos + 2];
    f.type = p[pos + 3];
    f.flags = p[pos + 4];
    f.stream_id = ((p[pos + 5] & 0x7F) << 24) | (p[pos + 6] << 16) |
                  (p[pos + 7] << 8) | p[pos + 8];
    if (f.len > (1u << 24) - 1 || pos + 9 + f.len > n) return false;
    f.payload = p + pos + 9;
    pos += 9 + f.len;
    return true;
}

static const char PREFACE[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";

// Does this payload look like an HTTP/2 segment? (preface or a clean
// frame sequence covering most of the payload)
inline bool looks_like_http2(const uint8_t* p, uint32_t n) {
    if (n >= 24 && memcmp(p, PREFACE, 24) == 0) return true;
    if (n < 9) return false;
    uint32_t pos = 0;
    FrameView f;
    int frames = 0;
    uint8_t first_type = 0xFF;
    while (next_frame(p, n, pos, f)) {
        if (f.type > 9) return false;
        if (frames == 0) first_type = f.type;
        frames++;
        if (frames >= 2) break;
    }
    // one full HEADERS/SETTINGS frame also qualifies, covering the
    // payload exactly or followed by a truncated frame (TCP-segmented
    // captures end mid-frame; the tail must still look like a frame
    // header with a sane type). NB: next_frame fills f before its
    // length check, so use the remembered first_type.
    if (frames >= 2) return true;
    if (frames != 1 || (first_type != F_HEADERS && first_type != F_SETTINGS))
        return false;
    if (pos == n) return true;
    if (n - pos >= 9) return p[pos + 3] <= 9;
    return true;  // < 9 tail bytes: indeterminate, accept
}

}  // namespace h2
