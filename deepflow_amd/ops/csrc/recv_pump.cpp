// Native receiver pump: one thread per agent connection doing
// recv -> trident deframe -> zstd decompress -> pinned byte ring.
//
// The Python receiver path (ingest/receiver.py) tops out around
// 0.6 GB/s per stream: every frame crosses the interpreter as bytes
// objects with slice copies.  The pump keeps the socket-to-pinned-memory
// path entirely native; Python only sees (cursor, length) pairs over a
// single-producer single-consumer ring and hands the pinned slices
// straight to hipMemcpyAsync.  Reference counterpart:
// server/libs/receiver/receiver.go (flow per-connection goroutines).
//
// Ring entry format (8-byte aligned):
//   [u64 payload_len][u64 meta][payload bytes][pad to 8]
//   meta = msg_type | (agent_id << 8) | (org_id << 24) | (team_id << 40)
// A u64 of ~0 at the first word is a wrap marker: skip to the next ring
// boundary.  Cursors are monotonic byte counts (position = cur % cap).

#include <atomic>
#include <cstdint>
#include <cstring>
#include <thread>
#include <vector>

#include <dlfcn.h>
#include <errno.h>
#include <sys/socket.h>
#include <unistd.h>

namespace {

constexpr uint64_t WRAP_MARK = ~0ull;
constexpr uint64_t MAX_FRAME = 64ull << 20;
constexpr int HEADER_LEN = 19;
constexpr int ENCODER_RAW = 0;
constexpr int ENCODER_ZSTD = 3;

typedef size_t (*zstd_fn4)(void*, size_t, const void*, size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
typedef unsigned long long (*zstd_size_fn)(const void*, size_t);

static void* zsym(const char* name) {
    static void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    return h ? dlsym(h, name) : nullptr;
}

struct Pump {
    int fd = -1;
    uint8_t* ring = nullptr;
    uint64_t cap = 0;
    std::atomic<uint64_t> head{0};
    std::atomic<uint64_t> tail{0};
    std::atomic<uint64_t> frames{0};
    std::atomic<uint64_t> wire_bytes{0};
    std::atomic<uint64_t> payload_bytes{0};
    std::atomic<uint64_t> bad_frames{0};
    std::atomic<int> stop{0};
    std::atomic<int> done{0};
    int accept_type = -1;  // msg_type filter; -1 = all
    std::thread th;
    std::vector<uint8_t> fbuf;  // one raw frame (after the 4B size)
    std::vector<uint8_t> dbuf;  // zstd scratch

    bool read_exact(uint8_t* dst, size_t n) {
        size_t got = 0;
        while (got < n && !stop.load(std::memory_order_relaxed)) {
            ssize_t r = recv(fd, dst + got, n - got, 0);
            if (r > 0) { got += (size_t)r; continue; }
            if (r == 0) return false;  // peer closed
            if (errno == EINTR || errno == EAGAIN || errno == EWOULDBLOCK)
                continue;
            return false;
        }
        return got == n;
    }

    // Reserve `need` contiguous bytes in the ring and return the write
    // position, or UINT64_MAX if stopping.  Blocks while the consumer
    // lags (bounded by ring capacity — backpressure, not drops, exactly
    // like a full TCP window).
    uint64_t reserve(uint64_t need) {
        for (;;) {
            uint64_t h = head.load(std::memory_order_relaxed);
            uint64_t pos = h % cap;
            uint64_t rem = cap - pos;
            uint64_t want = (rem < need) ? rem + need : need;
            while (cap - (h - tail.load(std::memory_order_acquire)) <
                   want) {
                if (stop.load(std::memory_order_relaxed))
                    return UINT64_MAX;
                usleep(50);
            }
            if (rem < need) {
                // not enough contiguous space before the boundary:
                // plant a wrap marker and skip to position 0
                memcpy(ring + pos, &WRAP_MARK, 8);
                head.store(h + rem, std::memory_order_release);
                continue;
            }
            return h;
        }
    }

    void publish(const uint8_t* payload, uint64_t n, uint64_t meta) {
        uint64_t need = 16 + ((n + 7) & ~7ull);
        if (need + 16 > cap) {  // larger than the ring: count, drop
            bad_frames.fetch_add(1, std::memory_order_relaxed);
            return;
        }
        uint64_t h = reserve(need);
        if (h == UINT64_MAX) return;
        uint64_t pos = h % cap;
        memcpy(ring + pos, &n, 8);
        memcpy(ring + pos + 8, &meta, 8);
        memcpy(ring + pos + 16, payload, n);
        head.store(h + need, std::memory_order_release);
        frames.fetch_add(1, std::memory_order_relaxed);
        payload_bytes.fetch_add(n, std::memory_order_relaxed);
    }

    // Reserve ring space for an entry of `n` payload bytes and write
    // its header; returns the payload write pointer (or null on stop).
    // The caller fills the payload then calls commit(h, n).
    uint8_t* begin_entry(uint64_t n, uint64_t meta, uint64_t& h) {
        uint64_t need = 16 + ((n + 7) & ~7ull);
        if (need + 16 > cap) {
            h = UINT64_MAX - 1;  // caller drains the payload instead
            return nullptr;
        }
        h = reserve(need);
        if (h == UINT64_MAX) return nullptr;
        uint64_t pos = h % cap;
        memcpy(ring + pos, &n, 8);
        memcpy(ring + pos + 8, &meta, 8);
        return ring + pos + 16;
    }

    void commit(uint64_t h, uint64_t n) {
        head.store(h + 16 + ((n + 7) & ~7ull), std::memory_order_release);
        frames.fetch_add(1, std::memory_order_relaxed);
        payload_bytes.fetch_add(n, std::memory_order_relaxed);
    }

    void run() {
        static zstd_fn4 zdec = (zstd_fn4)zsym("ZSTD_decompress");
        static zstd_iserr_fn ziserr = (zstd_iserr_fn)zsym("ZSTD_isError");
        static zstd_size_fn zsize =
            (zstd_size_fn)zsym("ZSTD_getFrameContentSize");
        uint8_t hdr[HEADER_LEN];
        while (!stop.load(std::memory_order_relaxed)) {
            if (!read_exact(hdr, HEADER_LEN)) break;
            uint64_t size = ((uint64_t)hdr[0] << 24) |
                            ((uint64_t)hdr[1] << 16) |
                            ((uint64_t)hdr[2] << 8) | hdr[3];
            if (size < (uint64_t)HEADER_LEN || size > MAX_FRAME) {
                bad_frames.fetch_add(1, std::memory_order_relaxed);
                break;  // stream is desynced; drop the connection
            }
            wire_bytes.fetch_add(size, std::memory_order_relaxed);
            int msg_type = hdr[4];
            int encoder = hdr[7];
            uint32_t team = (uint32_t)hdr[8] | ((uint32_t)hdr[9] << 8) |
                            ((uint32_t)hdr[10] << 16) |
                            ((uint32_t)hdr[11] << 24);
            uint16_t org = (uint16_t)(hdr[12] | (hdr[13] << 8));
            uint16_t agent = (uint16_t)(hdr[16] | (hdr[17] << 8));
            uint64_t meta = (uint64_t)(uint8_t)msg_type |
                            ((uint64_t)agent << 8) |
                            ((uint64_t)org << 24) |
                            ((uint64_t)team << 40);
            uint64_t pn = size - HEADER_LEN;
            bool want_it = accept_type < 0 || msg_type == accept_type;
            if (!want_it) {
                // skip the payload without publishing
                uint64_t left = pn;
                uint8_t sink[4096];
                bool ok = true;
                while (left) {
                    uint64_t c = left < sizeof(sink) ? left : sizeof(sink);
                    if (!read_exact(sink, c)) { ok = false; break; }
                    left -= c;
                }
                if (!ok) break;
                continue;
            }
            if (encoder == ENCODER_RAW) {
                // recv STRAIGHT into the reserved ring entry: the only
                // copy between the socket and pinned memory
                uint64_t h;
                uint8_t* dst = begin_entry(pn, meta, h);
                if (dst == nullptr) {
                    if (h != UINT64_MAX - 1) break;  // stopping
                    // frame larger than the ring: drain + count
                    bad_frames.fetch_add(1, std::memory_order_relaxed);
                    uint64_t left = pn;
                    uint8_t sink[4096];
                    bool ok = true;
                    while (left) {
                        uint64_t c =
                            left < sizeof(sink) ? left : sizeof(sink);
                        if (!read_exact(sink, c)) { ok = false; break; }
                        left -= c;
                    }
                    if (!ok) break;
                    continue;
                }
                if (!read_exact(dst, pn)) break;
                commit(h, pn);
            } else if (encoder == ENCODER_ZSTD) {
                if (fbuf.size() < pn) fbuf.resize(pn);
                if (!read_exact(fbuf.data(), pn)) break;
                if (!zdec || !ziserr) {
                    bad_frames.fetch_add(1, std::memory_order_relaxed);
                    continue;
                }
                unsigned long long want =
                    zsize ? zsize(fbuf.data(), pn)
                          : (unsigned long long)-1;
                if (want != (unsigned long long)-1 &&
                    want != (unsigned long long)-2 && want &&
                    want + 64 < cap / 2) {
                    // known content size: decompress straight into the
                    // reserved ring entry
                    uint64_t h;
                    uint8_t* dst = begin_entry(want, meta, h);
                    if (dst == nullptr) {
                        if (h != UINT64_MAX - 1) break;  // stopping
                        bad_frames.fetch_add(1,
                                             std::memory_order_relaxed);
                        continue;  // decompressed size exceeds the ring
                    }
                    size_t r = zdec(dst, want, fbuf.data(), pn);
                    if (ziserr(r)) {
                        // commit a zero-length entry to keep cursors
                        // consistent, count the failure
                        commit(h, 0);
                        bad_frames.fetch_add(1,
                                             std::memory_order_relaxed);
                    } else {
                        commit(h, r);
                    }
                } else {
                    uint64_t capg = pn * 20 + (1u << 20);
                    for (;;) {
                        if (dbuf.size() < capg) dbuf.resize(capg);
                        size_t r = zdec(dbuf.data(), capg, fbuf.data(),
                                        pn);
                        if (!ziserr(r)) {
                            publish(dbuf.data(), r, meta);
                            break;
                        }
                        capg *= 4;
                        if (capg > MAX_FRAME * 64) {
                            bad_frames.fetch_add(
                                1, std::memory_order_relaxed);
                            break;
                        }
                    }
                }
            } else {
                bad_frames.fetch_add(1, std::memory_order_relaxed);
            }
        }
        close(fd);
        fd = -1;
        done.store(1, std::memory_order_release);
    }
};

}  // namespace

extern "C" {

// ring must stay alive (and, for the GPU path, pinned) until
// df_pump_free.  fd ownership transfers to the pump.
void* df_pump_start(int fd, uint8_t* ring, uint64_t cap,
                    int accept_type) {
    Pump* p = new Pump();
    p->fd = fd;
    p->ring = ring;
    p->cap = cap;
    p->accept_type = accept_type;
    p->th = std::thread([p] { p->run(); });
    return p;
}

uint64_t df_pump_head(void* h) {
    return ((Pump*)h)->head.load(std::memory_order_acquire);
}

uint64_t df_pump_tail(void* h) {
    return ((Pump*)h)->tail.load(std::memory_order_relaxed);
}

void df_pump_set_tail(void* h, uint64_t t) {
    ((Pump*)h)->tail.store(t, std::memory_order_release);
}

int df_pump_done(void* h) {
    return ((Pump*)h)->done.load(std::memory_order_acquire);
}

void df_pump_stats(void* h, uint64_t* frames, uint64_t* wire_bytes,
                   uint64_t* payload_bytes, uint64_t* bad_frames) {
    Pump* p = (Pump*)h;
    *frames = p->frames.load(std::memory_order_relaxed);
    *wire_bytes = p->wire_bytes.load(std::memory_order_relaxed);
    *payload_bytes = p->payload_bytes.load(std::memory_order_relaxed);
    *bad_frames = p->bad_frames.load(std::memory_order_relaxed);
}

void df_pump_free(void* h) {
    Pump* p = (Pump*)h;
    p->stop.store(1, std::memory_order_release);
    if (p->fd >= 0) shutdown(p->fd, SHUT_RDWR);
    if (p->th.joinable()) p->th.join();
    delete p;
}

// Benchmark sender: push `blob` (pre-framed wire bytes) `reps` times
// through a connected socket. Lives here so e2e throughput tests are
// not capped by a Python send loop (ctypes releases the GIL for the
// whole call).
int64_t df_tcp_blast(int fd, const uint8_t* blob, uint64_t n,
                     uint32_t reps) {
    int64_t total = 0;
    for (uint32_t r = 0; r < reps; r++) {
        uint64_t off = 0;
        while (off < n) {
            ssize_t w = send(fd, blob + off, n - off, 0);
            if (w > 0) { off += (uint64_t)w; total += w; continue; }
            if (errno == EINTR) continue;
            return -1;
        }
    }
    return total;
}

}  // extern "C"
