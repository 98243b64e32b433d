// Native OTLP TracesData -> AppProtoLogsData converter.
//
// The reference converts OTel spans inside the Go ingester
// (otel_import.go); here the conversion is C++ (OpenMP over spans) so a
// 10M-span OTLP stream feeds the GPU pipeline at wire speed — the Python
// twin (deepflow_amd/ingest/otel.py span_to_l7) is the semantic oracle
// and stays authoritative for tests.
//
// Output framing matches framing.pack_records(): [u32 LE len][record]...
#include <cstdint>
#include <cstring>
#include <string>
#include <vector>
#ifdef _OPENMP
#include <omp.h>
#endif
#include "pbenc.h"

namespace {

struct Rd {
    const uint8_t* p;
    uint64_t n;
    uint64_t pos = 0;
    bool ok = true;
    uint64_t varint() {
        uint64_t v = 0;
        int shift = 0;
        while (pos < n && shift < 70) {
            uint8_t b = p[pos++];
            v |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
        }
        ok = false;
        return 0;
    }
    // returns field number, sets wire type; 0 = end/error
    uint32_t tag(uint32_t& wt) {
        if (pos >= n) return 0;
        uint64_t t = varint();
        wt = (uint32_t)(t & 7);
        return (uint32_t)(t >> 3);
    }
    uint64_t fixed64() {
        if (pos + 8 > n) { ok = false; return 0; }
        uint64_t v;
        memcpy(&v, p + pos, 8);
        pos += 8;
        return v;
    }
    Rd sub() {
        uint64_t ln = varint();
        if (pos + ln > n) { ok = false; ln = 0; }
        Rd r{p + pos, ln};
        pos += ln;
        return r;
    }
    void skip(uint32_t wt) {
        switch (wt) {
            case 0: varint(); break;
            case 1: pos += 8; break;
            case 2: { uint64_t ln = varint(); pos += ln; break; }
            case 5: pos += 4; break;
            default: ok = false;
        }
    }
};

struct StrView {
    const char* p = nullptr;
    uint32_t n = 0;
    bool eq(const char* s) const {
        size_t l = strlen(s);
        return n == l && memcmp(p, s, l) == 0;
    }
    bool empty() const { return n == 0; }
};

StrView rd_str(Rd& r) {
    uint64_t ln = r.varint();
    if (r.pos + ln > r.n) { r.ok = false; return {}; }
    StrView s{(const char*)r.p + r.pos, (uint32_t)ln};
    r.pos += ln;
    return s;
}

// AnyValue -> string view; non-string scalars render into `scratch`
// (no heap traffic on the hot path — matches _attr_val in otel.py)
StrView any_value_view(Rd r, char* scratch, size_t scratch_cap) {
    uint32_t wt;
    while (uint32_t f = r.tag(wt)) {
        if (f == 1 && wt == 2) {           // string_value
            return rd_str(r);
        } else if (f == 2 && wt == 0) {    // bool
            const char* b = r.varint() ? "true" : "false";
            return {b, (uint32_t)strlen(b)};
        } else if (f == 3 && wt == 0) {    // int
            int n = snprintf(scratch, scratch_cap, "%lld",
                             (long long)(int64_t)r.varint());
            return {scratch, (uint32_t)n};
        } else if (f == 4 && wt == 1) {    // double
            double d;
            uint64_t v = r.fixed64();
            memcpy(&d, &v, 8);
            int n = snprintf(scratch, scratch_cap, "%g", d);
            return {scratch, (uint32_t)n};
        } else {
            r.skip(wt);
        }
    }
    return {};
}

void any_value_str(Rd r, std::string& out) {
    char scratch[32];
    StrView v = any_value_view(r, scratch, sizeof scratch);
    out.assign(v.p ? v.p : "", v.n);
}

struct Attr {
    std::string key, val;
};

struct SpanView {
    const uint8_t* p;
    uint64_t n;
    std::string service;  // resource service.name
};

void hex_of(const uint8_t* b, uint32_t n, std::string& out) {
    static const char* H = "0123456789abcdef";
    out.resize(n * 2);
    for (uint32_t i = 0; i < n; i++) {
        out[2 * i] = H[b[i] >> 4];
        out[2 * i + 1] = H[b[i] & 0xF];
    }
}

bool well_known(const std::string& k) {
    static const char* W[] = {
        "http.method", "http.request.method", "http.target", "url.path",
        "http.url", "http.host", "server.address", "http.status_code",
        "http.response.status_code", "rpc.system", "rpc.method",
        "rpc.service", "rpc.grpc.status_code", "db.system", "db.operation",
        "db.name", "db.statement"};
    for (const char* w : W)
        if (k == w) return true;
    return false;
}

// convert one OTLP Span submessage -> AppProtoLogsData record bytes.
// Allocation-free per span: attrs and ids live in stack views/buffers.
constexpr uint32_t MAX_ATTRS = 48;

struct AttrSlot {
    StrView key, val;
    char scratch[32];
};

void convert_span(const SpanView& sv, std::vector<uint8_t>& out) {
    Rd r{sv.p, sv.n};
    char trace_hex[64], span_hex[64], parent_hex[64];
    uint32_t trace_n = 0, span_n = 0, parent_n = 0;
    StrView name;
    uint64_t kind = 0, t0 = 0, t1 = 0, status_code = 0;
    AttrSlot attrs[MAX_ATTRS];
    uint32_t n_attrs = 0;
    auto hex_into = [](StrView s, char* dst, uint32_t cap) -> uint32_t {
        static const char* H = "0123456789abcdef";
        uint32_t n = s.n * 2 > cap ? cap / 2 : s.n;
        for (uint32_t i = 0; i < n; i++) {
            dst[2 * i] = H[(uint8_t)s.p[i] >> 4];
            dst[2 * i + 1] = H[(uint8_t)s.p[i] & 0xF];
        }
        return n * 2;
    };
    uint32_t wt;
    while (uint32_t f = r.tag(wt)) {
        if (f == 1 && wt == 2) {
            trace_n = hex_into(rd_str(r), trace_hex, sizeof trace_hex);
        } else if (f == 2 && wt == 2) {
            span_n = hex_into(rd_str(r), span_hex, sizeof span_hex);
        } else if (f == 4 && wt == 2) {
            parent_n = hex_into(rd_str(r), parent_hex, sizeof parent_hex);
        } else if (f == 5 && wt == 2) {
            name = rd_str(r);
        } else if (f == 6 && wt == 0) {
            kind = r.varint();
        } else if (f == 7 && wt == 1) {
            t0 = r.fixed64();
        } else if (f == 8 && wt == 1) {
            t1 = r.fixed64();
        } else if (f == 9 && wt == 2) {  // KeyValue
            Rd kv = r.sub();
            if (n_attrs >= MAX_ATTRS) continue;
            AttrSlot& a = attrs[n_attrs];
            uint32_t wt2;
            bool got = false;
            while (uint32_t f2 = kv.tag(wt2)) {
                if (f2 == 1 && wt2 == 2) {
                    a.key = rd_str(kv);
                    got = true;
                } else if (f2 == 2 && wt2 == 2) {
                    a.val = any_value_view(kv.sub(), a.scratch,
                                           sizeof a.scratch);
                } else {
                    kv.skip(wt2);
                }
            }
            if (got) n_attrs++;
        } else if (f == 15 && wt == 2) {  // Status
            Rd st = r.sub();
            uint32_t wt2;
            while (uint32_t f2 = st.tag(wt2)) {
                if (f2 == 3 && wt2 == 0) status_code = st.varint();
                else st.skip(wt2);
            }
        } else {
            r.skip(wt);
        }
    }
    auto attr = [&](const char* k) -> const StrView* {
        for (uint32_t i = 0; i < n_attrs; i++)
            if (attrs[i].key.eq(k)) return &attrs[i].val;
        return nullptr;
    };
    auto to_i = [](const StrView* v) -> int64_t {
        if (!v || !v->n) return 0;
        char buf[24];
        uint32_t n = v->n < 23 ? v->n : 23;
        memcpy(buf, v->p, n);
        buf[n] = 0;
        return atoll(buf);
    };
    uint32_t tap_side = kind == 3 ? 1 : (kind == 2 ? 2 : 0);
    uint32_t proto = 0;
    StrView req_type, domain, resource, endpoint;
    int64_t code = 0;
    const StrView *v, *v2;
    if ((v = attr("http.method")) || (v = attr("http.request.method"))) {
        proto = 20;
        req_type = *v;
        if ((v2 = attr("http.target")) || (v2 = attr("url.path")) ||
            (v2 = attr("http.url")))
            resource = *v2;
        if ((v2 = attr("http.host")) || (v2 = attr("server.address")))
            domain = *v2;
        endpoint = name;
        if ((v2 = attr("http.status_code")) ||
            (v2 = attr("http.response.status_code")))
            code = to_i(v2);
    } else if (attr("rpc.system")) {
        proto = 41;
        if ((v2 = attr("rpc.method"))) req_type = *v2;
        if ((v2 = attr("rpc.service"))) domain = *v2;
        resource = name;
        endpoint = name;
        if ((v2 = attr("rpc.grpc.status_code"))) code = to_i(v2);
    } else if ((v = attr("db.system"))) {
        proto = v->eq("redis") ? 80 : 60;
        if ((v2 = attr("db.operation"))) req_type = *v2;
        if ((v2 = attr("db.name"))) domain = *v2;
        if ((v2 = attr("db.statement")) && v2->n) resource = *v2;
        else resource = name;
        endpoint = req_type;
    } else {
        resource = name;
        endpoint = name;
    }
    uint32_t status = status_code != 2 ? 0
                      : (code >= 400 && code < 500 ? 4 : 3);
    auto wk = [](StrView k) {
        static const char* W[] = {
            "http.method", "http.request.method", "http.target",
            "url.path", "http.url", "http.host", "server.address",
            "http.status_code", "http.response.status_code", "rpc.system",
            "rpc.method", "rpc.service", "rpc.grpc.status_code",
            "db.system", "db.operation", "db.name", "db.statement"};
        for (const char* w : W)
            if (k.eq(w)) return true;
        return false;
    };
    uint8_t buf[16384];
    dfpb::Buf b{buf, 0, sizeof buf};
    dfpb::f_m<512>(b, 1, [&](dfpb::Buf& s) {  // base
        dfpb::f_u(s, 1, t0);
        dfpb::f_u(s, 2, t1);
        dfpb::f_u(s, 8, tap_side);
        dfpb::f_m<64>(s, 9, [&](dfpb::Buf& h) {
            dfpb::f_u(h, 1, proto);
            dfpb::f_u(h, 2, 2);
            dfpb::f_u(h, 5, t1 > t0 ? (t1 - t0) / 1000 : 0);
        });
    });
    dfpb::f_m<4096>(b, 11, [&](dfpb::Buf& s) {  // req
        dfpb::f_s(s, 1, req_type.p, req_type.n);
        dfpb::f_s(s, 2, domain.p, domain.n);
        dfpb::f_s(s, 3, resource.p, resource.n);
        dfpb::f_s(s, 4, endpoint.p, endpoint.n);
    });
    dfpb::f_m<64>(b, 12, [&](dfpb::Buf& s) {  // resp
        dfpb::f_u(s, 1, status);
        dfpb::f_i(s, 2, code);
    });
    dfpb::f_m<256>(b, 14, [&](dfpb::Buf& s) {  // trace_info
        dfpb::f_s(s, 1, trace_hex, trace_n);
        dfpb::f_s(s, 2, span_hex, span_n);
        dfpb::f_s(s, 3, parent_hex, parent_n);
    });
    dfpb::f_m<8192>(b, 15, [&](dfpb::Buf& s) {  // ext_info
        dfpb::f_s(s, 1, sv.service.data(), sv.service.size());
        for (uint32_t i = 0; i < n_attrs; i++)
            if (!wk(attrs[i].key))
                dfpb::f_s(s, 16, attrs[i].key.p, attrs[i].key.n);
        for (uint32_t i = 0; i < n_attrs; i++)
            if (!wk(attrs[i].key))
                dfpb::f_s(s, 17, attrs[i].val.p, attrs[i].val.n);
    });
    uint32_t len = (uint32_t)b.len;
    size_t base = out.size();
    out.resize(base + 4 + len);
    memcpy(out.data() + base, &len, 4);
    memcpy(out.data() + base + 4, buf, len);
}

}  // namespace

extern "C" {

// OTLP TracesData bytes -> length-prefixed AppProtoLogsData payload.
// Returns bytes written (<=cap; computes full size regardless), -1 on
// malformed input.
int64_t df_otlp_to_l7(const uint8_t* src, uint64_t n, uint8_t* dst,
                      uint64_t cap) {
    // pass 1: collect span slices + per-resource service names
    std::vector<SpanView> spans;
    std::vector<std::string> services;
    Rd top{src, n};
    uint32_t wt;
    while (uint32_t f = top.tag(wt)) {
        if (f == 1 && wt == 2) {  // resource_spans
            Rd rs = top.sub();
            services.emplace_back();
            std::string& svc = services.back();
            size_t first_span = spans.size();
            uint32_t wt2;
            while (uint32_t f2 = rs.tag(wt2)) {
                if (f2 == 1 && wt2 == 2) {  // resource
                    Rd res = rs.sub();
                    uint32_t wt3;
                    while (uint32_t f3 = res.tag(wt3)) {
                        if (f3 == 1 && wt3 == 2) {  // attributes
                            Rd kv = res.sub();
                            std::string key, val;
                            uint32_t wt4;
                            while (uint32_t f4 = kv.tag(wt4)) {
                                if (f4 == 1 && wt4 == 2) {
                                    StrView s = rd_str(kv);
                                    key.assign(s.p, s.n);
                                } else if (f4 == 2 && wt4 == 2) {
                                    any_value_str(kv.sub(), val);
                                } else {
                                    kv.skip(wt4);
                                }
                            }
                            if (key == "service.name") svc = val;
                        } else {
                            res.skip(wt3);
                        }
                    }
                } else if (f2 == 2 && wt2 == 2) {  // scope_spans
                    Rd ss = rs.sub();
                    uint32_t wt3;
                    while (uint32_t f3 = ss.tag(wt3)) {
                        if (f3 == 2 && wt3 == 2) {  // span
                            Rd sp = ss.sub();
                            spans.push_back({sp.p, sp.n, {}});
                        } else {
                            ss.skip(wt3);
                        }
                    }
                } else {
                    rs.skip(wt2);
                }
            }
            for (size_t i = first_span; i < spans.size(); i++)
                spans[i].service = svc;
            if (!rs.ok) return -1;
        } else {
            top.skip(wt);
        }
    }
    if (!top.ok) return -1;
    // pass 2: convert spans in parallel
    int nt = 1;
#ifdef _OPENMP
    nt = omp_get_max_threads();
#endif
    std::vector<std::vector<uint8_t>> parts(nt);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < (int64_t)spans.size(); i++) {
        int t = 0;
#ifdef _OPENMP
        t = omp_get_thread_num();
#endif
        convert_span(spans[i], parts[t]);
    }
    uint64_t total = 0;
    for (auto& v : parts) total += v.size();
    uint64_t w = 0;
    for (auto& v : parts) {
        if (w + v.size() <= cap && dst)
            memcpy(dst + w, v.data(), v.size());
        w += v.size();
    }
    return (int64_t)total;
}

}  // extern "C"
