// dfprof — in-process rocprofiler-sdk subscriber for the continuous GPU
// profiler (BASELINE config #5). Round 1 wrapped torch.profiler/kineto,
// paying ~0.7 s per capture window; here a window is
// rocprofiler_start/stop_context (microseconds), with kernel dispatch
// records aggregated natively, so 1-window/60 s continuous mode costs
// <<0.5%. No kineto, no CUPTI, no torch.
//
// rocprofiler-sdk requires tool registration BEFORE the HIP runtime
// initializes: dfp_register() (python: native_profiler.ensure_early())
// must run before the first torch/HIP GPU touch — the server and bench
// call it at process start. Two contexts:
//   - code-object context: always on, maps kernel_id -> kernel name
//   - dispatch context:    windowed, buffers kernel begin/end timestamps
//
// Build: hipcc -shared -fPIC gpuprof.cpp -lrocprofiler-sdk
#include <cstdint>
#include <cstring>
#include <map>
#include <mutex>
#include <string>

#include <rocprofiler-sdk/buffer.h>
#include <rocprofiler-sdk/buffer_tracing.h>
#include <rocprofiler-sdk/callback_tracing.h>
#include <rocprofiler-sdk/fwd.h>
#include <rocprofiler-sdk/registration.h>
#include <rocprofiler-sdk/rocprofiler.h>

namespace {

struct Agg {
    uint64_t count = 0;
    uint64_t total_ns = 0;
    uint64_t max_ns = 0;
};

std::mutex g_mu;
std::map<std::string, Agg> g_table;
std::map<uint64_t, std::string> g_kernel_names;
uint64_t g_records = 0;
rocprofiler_context_id_t g_ctx_code = {0};
rocprofiler_context_id_t g_ctx_disp = {0};
rocprofiler_buffer_id_t g_buf = {0};
int g_state = 0;  // 0 = unregistered, 1 = registered, 2 = initialized
int g_err = 0;    // first failing step in tool_init (diagnostics)

void code_object_cb(rocprofiler_callback_tracing_record_t record,
                    rocprofiler_user_data_t*, void*) {
    if (record.kind != ROCPROFILER_CALLBACK_TRACING_CODE_OBJECT ||
        record.operation !=
            ROCPROFILER_CODE_OBJECT_DEVICE_KERNEL_SYMBOL_REGISTER ||
        record.phase != ROCPROFILER_CALLBACK_PHASE_LOAD)
        return;
    auto* sym = (rocprofiler_callback_tracing_code_object_kernel_symbol_register_data_t*)
        record.payload;
    if (sym && sym->kernel_name) {
        std::lock_guard<std::mutex> lk(g_mu);
        g_kernel_names[sym->kernel_id] = sym->kernel_name;
    }
}

void buffer_cb(rocprofiler_context_id_t, rocprofiler_buffer_id_t,
               rocprofiler_record_header_t** headers, size_t n,
               void*, uint64_t) {
    std::lock_guard<std::mutex> lk(g_mu);
    for (size_t i = 0; i < n; i++) {
        rocprofiler_record_header_t* h = headers[i];
        if (h->category != ROCPROFILER_BUFFER_CATEGORY_TRACING ||
            h->kind != ROCPROFILER_BUFFER_TRACING_KERNEL_DISPATCH)
            continue;
        auto* rec =
            (rocprofiler_buffer_tracing_kernel_dispatch_record_t*)h->payload;
        uint64_t dur = rec->end_timestamp > rec->start_timestamp
                           ? rec->end_timestamp - rec->start_timestamp
                           : 0;
        if (dur == 0) continue;
        auto it = g_kernel_names.find(rec->dispatch_info.kernel_id);
        const std::string& nm =
            it != g_kernel_names.end()
                ? it->second
                : (g_kernel_names[rec->dispatch_info.kernel_id] =
                       "kernel_" +
                       std::to_string(rec->dispatch_info.kernel_id));
        Agg& a = g_table[nm];
        a.count++;
        a.total_ns += dur;
        if (dur > a.max_ns) a.max_ns = dur;
        g_records++;
    }
}

int tool_init(rocprofiler_client_finalize_t, void*) {
    rocprofiler_status_t rc;
    rc = rocprofiler_create_context(&g_ctx_code);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 100 + (int)rc; return -1; }
    rc = rocprofiler_create_context(&g_ctx_disp);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 200 + (int)rc; return -1; }
    rc = rocprofiler_configure_callback_tracing_service(
            g_ctx_code, ROCPROFILER_CALLBACK_TRACING_CODE_OBJECT, nullptr, 0,
            code_object_cb, nullptr);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 300 + (int)rc; return -1; }
    rc = rocprofiler_create_buffer(g_ctx_disp, 1 << 22, 3 << 20,
                                   ROCPROFILER_BUFFER_POLICY_LOSSLESS,
                                   buffer_cb, nullptr, &g_buf);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 400 + (int)rc; return -1; }
    rc = rocprofiler_configure_buffer_tracing_service(
            g_ctx_disp, ROCPROFILER_BUFFER_TRACING_KERNEL_DISPATCH, nullptr,
            0, g_buf);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 500 + (int)rc; return -1; }
    // names must be known for kernels loaded at any time
    rc = rocprofiler_start_context(g_ctx_code);
    if (rc != ROCPROFILER_STATUS_SUCCESS) { g_err = 600 + (int)rc; return -1; }
    g_state = 2;
    return 0;
}

void tool_fini(void*) {}

}  // namespace

extern "C" {

// canonical tool entry: rocprofiler dlopens the libraries listed in
// ROCP_TOOL_LIBRARIES at HIP runtime init and calls this export
// (ensure_early() sets the env var before torch loads the runtime)
rocprofiler_tool_configure_result_t* rocprofiler_configure(
    uint32_t /*version*/, const char* /*runtime_version*/,
    uint32_t /*priority*/, rocprofiler_client_id_t* id) {
    id->name = "dfprof";
    if (g_state == 2) return nullptr;  // single client
    g_state = 1;
    static rocprofiler_tool_configure_result_t cfg{
        sizeof(rocprofiler_tool_configure_result_t), &tool_init, &tool_fini,
        nullptr};
    return &cfg;
}

// best-effort late registration (processes that skipped ensure_early)
int dfp_register() {
    if (g_state >= 1) return 0;
    if (rocprofiler_force_configure(&rocprofiler_configure) !=
        ROCPROFILER_STATUS_SUCCESS)
        return -1;
    if (g_state == 0) g_state = 1;
    return 0;
}

int dfp_ready() { return g_state; }

int dfp_err() { return g_err; }

int dfp_start() {
    if (g_state != 2) return -10;  // not registered early enough
    return rocprofiler_start_context(g_ctx_disp) ==
                   ROCPROFILER_STATUS_SUCCESS
               ? 0
               : -1;
}

int dfp_stop() {
    if (g_state != 2) return -10;
    rocprofiler_stop_context(g_ctx_disp);
    rocprofiler_flush_buffer(g_buf);
    return 0;
}

// [u32 name_len][name][u64 count][u64 total_ns][u64 max_ns] ...
// returns bytes written (drains the table); 0-cap call sizes the buffer
uint64_t dfp_drain(uint8_t* out, uint64_t cap) {
    std::lock_guard<std::mutex> lk(g_mu);
    uint64_t need = 0;
    for (const auto& kv : g_table) need += 4 + kv.first.size() + 24;
    if (out == nullptr || cap < need) return need;
    uint64_t pos = 0;
    for (const auto& kv : g_table) {
        uint32_t nl = (uint32_t)kv.first.size();
        memcpy(out + pos, &nl, 4);
        pos += 4;
        memcpy(out + pos, kv.first.data(), nl);
        pos += nl;
        memcpy(out + pos, &kv.second.count, 8);
        memcpy(out + pos + 8, &kv.second.total_ns, 8);
        memcpy(out + pos + 16, &kv.second.max_ns, 8);
        pos += 24;
    }
    g_table.clear();
    return pos;
}

uint64_t dfp_record_count() { return g_records; }

}  // extern "C"
