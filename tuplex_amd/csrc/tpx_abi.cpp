// tpx_abi.cpp — C-ABI host layer of the MI355X-native TransformStage executor.
//
// Replaces the reference's execution backend for the normal-case path:
//  - stage compile: TransformStage.cc:763 compile (LLVM ORC JIT) -> hipRTC with an
//    hsaco disk cache keyed by source hash;
//  - stage execute: LocalBackend.cc:815 executeTransformStage +
//    TransformTask.cc:382 execute / :682 processMemorySource -> H2D staging, fused
//    stage kernel, filter-compaction prefix sums, row-serialize kernel, D2H;
//  - exception buffers: device (row,ec,opid) marks are materialised host-side into
//    the reference record format [row,ec,opID,size,data]
//    (core/include/physical/IExceptionableTask.h:20) with the input row bytes as
//    payload, so the host resolve path (ResolveTask semantics) is unchanged.
//
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC tpx_abi.cpp -lhiprtc
//        -o ../libtpx_gpu.so   (see build.py)

#include "../../include/tpx_abi.h"

#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>

#include <algorithm>
#include <cstdio>
#include <unistd.h>
#include <functional>
#include <cstring>
#include <fstream>
#include <map>
#include <mutex>
#include <sstream>
#include <string>
#include <vector>

// ---------------------------------------------------------------------------------

static thread_local std::string g_err;

static void set_err(const std::string& e) { g_err = e; }

extern "C" const char* tpx_last_error(void) { return g_err.c_str(); }

extern "C" int64_t tpx_version(void) { return (1 << 16) | 0; }

extern "C" int64_t tpx_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

extern "C" int64_t tpx_set_device(int64_t device) {
    hipError_t e = hipSetDevice((int)device);
    if (e != hipSuccess) { set_err(hipGetErrorString(e)); return (int64_t)e; }
    return 0;
}

#define HIP_CHECK(x)                                                         \
    do {                                                                     \
        hipError_t _e = (x);                                                 \
        if (_e != hipSuccess) {                                              \
            set_err(std::string(#x) + ": " + hipGetErrorString(_e));         \
            return -1;                                                       \
        }                                                                    \
    } while (0)

// ---------------------------------------------------------------------------------
// stage descriptor

enum ColKind { K_I64, K_F64, K_BOOL, K_STR };

struct ColDesc {
    ColKind kind;
    bool opt;
};

struct StageDesc {
    std::string source;  // mem | csv
    std::string sink;    // mem | csv
    std::string agg;     // "" | i64 | f64 — GPU-reducible aggregate fold
    std::string aggby;   // "" | i64 | f64 — by-key hash-reduce (key col 0)
    int aggkeystr = 0;   // by-key key column is a string
    int textmode = 0;    // text() source: rows split on every newline (no quotes)
    int split = 0;       // split parse/UDF kernels (csv): k_parse fills the
                         // cell park, k_main runs grid-stride over it
    int parkcopy = 0;    // 1 = dense byte park (default: zero-copy views)
    std::vector<int> used;  // park columns (projection pushdown)
    std::vector<ColDesc> in_cols, out_cols;
};

static bool parse_col(const std::string& v, ColDesc* c) {
    std::string s = v;
    c->opt = false;
    if (s.rfind("opt,", 0) == 0) { c->opt = true; s = s.substr(4); }
    if (s == "i64") c->kind = K_I64;
    else if (s == "f64") c->kind = K_F64;
    else if (s == "bool") c->kind = K_BOOL;
    else if (s == "str") c->kind = K_STR;
    else return false;
    return true;
}

static bool parse_desc(const char* text, StageDesc* d) {
    std::istringstream is(text);
    std::string line;
    std::map<std::string, std::string> kv;
    while (std::getline(is, line)) {
        auto p = line.find('=');
        if (p == std::string::npos) continue;
        kv[line.substr(0, p)] = line.substr(p + 1);
    }
    d->source = kv.count("source") ? kv["source"] : "mem";
    d->sink = kv.count("sink") ? kv["sink"] : "mem";
    d->agg = kv.count("agg") ? kv["agg"] : "";
    d->aggby = kv.count("aggby") ? kv["aggby"] : "";
    d->aggkeystr = kv.count("aggkeystr") ? atoi(kv["aggkeystr"].c_str()) : 0;
    d->textmode = kv.count("textmode") ? atoi(kv["textmode"].c_str()) : 0;
    d->split = kv.count("split") ? atoi(kv["split"].c_str()) : 0;
    d->parkcopy = kv.count("parkcopy") ? atoi(kv["parkcopy"].c_str()) : 0;
    if (kv.count("used")) {
        std::istringstream us(kv["used"]);
        std::string tok;
        while (std::getline(us, tok, ','))
            if (!tok.empty()) d->used.push_back(atoi(tok.c_str()));
    }
    int nin = atoi(kv["nin"].c_str());
    int nout = atoi(kv["nout"].c_str());
    for (int i = 0; i < nin; ++i) {
        ColDesc c;
        char key[16];
        snprintf(key, sizeof key, "in%d", i);
        if (!kv.count(key) || !parse_col(kv[key], &c)) return false;
        d->in_cols.push_back(c);
    }
    for (int i = 0; i < nout; ++i) {
        ColDesc c;
        char key[16];
        snprintf(key, sizeof key, "out%d", i);
        if (!kv.count(key) || !parse_col(kv[key], &c)) return false;
        d->out_cols.push_back(c);
    }
    return true;
}

// ---------------------------------------------------------------------------------
// stage handle

struct tpx_stage {
    std::string source_text;
    StageDesc desc;
    std::vector<char> code;  // hsaco
    hipModule_t module = nullptr;
    hipFunction_t k_main = nullptr, k_write = nullptr;
    hipFunction_t k_parse = nullptr;  // split mode (desc.split)
    hipFunction_t k_scan_block = nullptr, k_scan_add = nullptr;
    hipFunction_t k_pair_total = nullptr;
    hipFunction_t k_csv_chunk = nullptr, k_csv_sel = nullptr, k_csv_rows = nullptr;
    hipFunction_t k_red_f64 = nullptr, k_red_f64_fin = nullptr;
    hipFunction_t k_red_i64 = nullptr, k_red_i64_fin = nullptr;
    hipFunction_t k_hk_f64 = nullptr, k_hk_i64 = nullptr, k_hk_emit = nullptr;
    hipFunction_t k_hk_sf64 = nullptr, k_hk_si64 = nullptr,
                  k_hk_semit = nullptr;
    bool loaded = false;
};

static uint64_t fnv1a(const char* s, size_t n) {
    uint64_t h = 1469598103934665603ULL;
    for (size_t i = 0; i < n; ++i) {
        h ^= (unsigned char)s[i];
        h *= 1099511628211ULL;
    }
    return h;
}

static bool compile_hiprtc(const char* src, std::vector<char>* code, std::string* log) {
    hiprtcProgram prog;
    if (hiprtcCreateProgram(&prog, src, "tpx_stage.hip", 0, nullptr, nullptr) !=
        HIPRTC_SUCCESS) {
        *log = "hiprtcCreateProgram failed";
        return false;
    }
    // -ffp-contract=off: the reference's compiled path (x86 SSE2, no FMA)
    // and the oracle never contract a*b+c; fused multiply-add in fast_atod's
    // digit accumulation shifts parsed doubles by an ulp and breaks bit parity
    const char* opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17",
                          "-ffp-contract=off"};
    hiprtcResult rc = hiprtcCompileProgram(prog, 4, opts);
    size_t log_size = 0;
    hiprtcGetProgramLogSize(prog, &log_size);
    if (log_size > 1) {
        std::vector<char> lg(log_size);
        hiprtcGetProgramLog(prog, lg.data());
        log->assign(lg.data(), log_size - 1);
    }
    if (rc != HIPRTC_SUCCESS) {
        hiprtcDestroyProgram(&prog);
        return false;
    }
    size_t code_size = 0;
    hiprtcGetCodeSize(prog, &code_size);
    code->resize(code_size);
    hiprtcGetCode(prog, code->data());
    hiprtcDestroyProgram(&prog);
    return true;
}

extern "C" tpx_stage* tpx_stage_compile(const char* hip_source,
                                        const char* stage_desc,
                                        const char* cache_dir, int64_t flags) {
    auto* st = new tpx_stage();
    st->source_text = hip_source;
    if (!parse_desc(stage_desc, &st->desc)) {
        set_err("bad stage_desc");
        delete st;
        return nullptr;
    }
    // cache lookup
    std::string cache_path;
    if (cache_dir && *cache_dir) {
        char name[64];
        // key = source hash + length + COMPILE-OPTION GENERATION (bump when
        // hipRTC options change, or stale hsacos built under old options load)
        snprintf(name, sizeof name, "tpx_%016llx_%zu_g2.hsaco",
                 (unsigned long long)fnv1a(hip_source, strlen(hip_source)),
                 strlen(hip_source));
        cache_path = std::string(cache_dir) + "/" + name;
        std::ifstream f(cache_path, std::ios::binary);
        if (f) {
            st->code.assign(std::istreambuf_iterator<char>(f),
                            std::istreambuf_iterator<char>());
        }
    }
    if (st->code.empty()) {
        std::string log;
        if (!compile_hiprtc(hip_source, &st->code, &log)) {
            set_err("hipRTC compile failed:\n" + log);
            delete st;
            return nullptr;
        }
        if (!cache_path.empty()) {
            // atomic publish: concurrent ranks compile the same stage (the
            // multi-process engine path) — a torn cache file must never be
            // observable, so write to a pid-suffixed temp and rename
            std::string tmp = cache_path + "." + std::to_string(getpid());
            {
                std::ofstream f(tmp, std::ios::binary);
                f.write(st->code.data(), (std::streamsize)st->code.size());
            }
            if (rename(tmp.c_str(), cache_path.c_str()) != 0)
                unlink(tmp.c_str());
        }
    }
    if (flags & 1) return st;  // compile-only (no GPU present)

    if (hipModuleLoadData(&st->module, st->code.data()) != hipSuccess) {
        set_err("hipModuleLoadData failed (no GPU?)");
        delete st;
        return nullptr;
    }
    struct { const char* name; hipFunction_t* fn; bool required; } lut[] = {
        {"tpx_stage_main", &st->k_main, true},
        {"tpx_stage_parse", &st->k_parse, false},
        {"tpx_stage_write", &st->k_write, true},
        {"tpx_scan_block", &st->k_scan_block, true},
        {"tpx_scan_add", &st->k_scan_add, true},
        {"tpx_pair_total", &st->k_pair_total, true},
        {"tpx_csv_chunk_stats", &st->k_csv_chunk, false},
        {"tpx_csv_select_counts", &st->k_csv_sel, false},
        {"tpx_csv_emit_rows", &st->k_csv_rows, false},
        {"tpx_reduce_f64", &st->k_red_f64, false},
        {"tpx_reduce_f64_final", &st->k_red_f64_fin, false},
        {"tpx_reduce_i64", &st->k_red_i64, false},
        {"tpx_reduce_i64_final", &st->k_red_i64_fin, false},
        {"tpx_hashagg_f64", &st->k_hk_f64, false},
        {"tpx_hashagg_i64", &st->k_hk_i64, false},
        {"tpx_hashagg_emit", &st->k_hk_emit, false},
        {"tpx_hashagg_str_f64", &st->k_hk_sf64, false},
        {"tpx_hashagg_str_i64", &st->k_hk_si64, false},
        {"tpx_hashagg_str_emit", &st->k_hk_semit, false},
    };
    for (auto& e : lut) {
        hipError_t r = hipModuleGetFunction(e.fn, st->module, e.name);
        if (r != hipSuccess && e.required) {
            set_err(std::string("kernel not found: ") + e.name);
            delete st;
            return nullptr;
        }
        if (getenv("TPX_TRACE") && r == hipSuccess)
            fprintf(stderr, "[tpx] kernel %s = %p\n", e.name, (void*)*e.fn);
    }
    st->loaded = true;
    return st;
}

extern "C" void tpx_stage_free(tpx_stage* stage) {
    if (!stage) return;
    if (stage->module) hipModuleUnload(stage->module);
    delete stage;
}

extern "C" const char* tpx_stage_source(const tpx_stage* stage) {
    return stage->source_text.c_str();
}

// ---------------------------------------------------------------------------------
// execution helpers

struct DevBuf {
    void* p = nullptr;
    ~DevBuf() { if (p) hipFree(p); }
    hipError_t alloc(size_t n) { return hipMalloc(&p, n ? n : 8); }
};

// Per-device workspace arena, reused across executes (the device partition
// manager's scratch — replaces per-call hipMalloc churn, which measured ~70 ms
// of a 110 ms step). Single-threaded per device, like the reference's
// one-task-per-thread discipline. Pointers stay valid until the NEXT execute.
struct DevArena {
    std::vector<std::pair<void*, size_t>> blocks;
    size_t cur = 0, off = 0, used = 0;
    void* take(size_t n) {
        n = (n + 255) & ~(size_t)255;
        if (!n) n = 256;
        used += n;
        while (cur < blocks.size()) {
            if (off + n <= blocks[cur].second) {
                void* p = (char*)blocks[cur].first + off;
                off += n;
                return p;
            }
            ++cur;
            off = 0;
        }
        size_t bs = std::max(n, (size_t)256 << 20);
        void* p = nullptr;
        if (hipMalloc(&p, bs) != hipSuccess) {
            set_err("arena hipMalloc failed");
            return nullptr;
        }
        blocks.emplace_back(p, bs);
        cur = blocks.size() - 1;
        off = n;
        return p;
    }
    void begin() {
        if (blocks.size() > 1) {  // consolidate to one block at the high-water mark
            size_t want = used + (used >> 2);
            (void)hipDeviceSynchronize();
            for (auto& b : blocks) (void)hipFree(b.first);
            blocks.clear();
            void* p = nullptr;
            if (hipMalloc(&p, want) == hipSuccess) blocks.emplace_back(p, want);
        }
        cur = 0;
        off = 0;
        used = 0;
    }
};
static DevArena g_arena[64];

// persistent per-device string heap / exception buffer / counters
struct Persist {
    void* heap = nullptr;
    size_t heap_cap = 0;
    void* exc = nullptr;
    size_t exc_cap_recs = 0;
    void* counters = nullptr;  // [cursor_c at 16*c (c<4) ... exc_count at 80]
};
static Persist g_persist[64];

// two extra per-device streams for chunk pipelining (scans/writes of chunk c
// overlap the main kernel of chunk c+1)
static hipStream_t g_cstream[64][2];
// small pinned staging area per device for tiny D2H reads (scan totals,
// counters): pageable 8B copies cost a hidden staging hop each
static void* g_pinned[64];
static void* pinned(int dev) {
    if (!g_pinned[dev]) (void)hipHostMalloc(&g_pinned[dev], 1024);
    return g_pinned[dev];
}

// per-device event pool: hipEventCreate/Destroy cost ~10 us each and run_core
// uses ~28 events per step — pooled, that overhead disappears
struct EventPool {
    std::vector<hipEvent_t> evs;
    size_t next = 0;
    hipEvent_t get() {
        if (next == evs.size()) {
            hipEvent_t e = nullptr;
            (void)hipEventCreate(&e);
            evs.push_back(e);
        }
        return evs[next++];
    }
    void reset() { next = 0; }
};
static EventPool g_events[64];
static hipStream_t cstream(int dev, int i) {
    if (!g_cstream[dev][i])
        (void)hipStreamCreateWithFlags(&g_cstream[dev][i], hipStreamNonBlocking);
    return g_cstream[dev][i];
}

static int cur_device() {
    int d = 0;
    (void)hipGetDevice(&d);
    return d;
}

#define ARENA_TAKE(var, n)                                                   \
    void* var = g_arena[dev].take((size_t)(n));                              \
    if (!var) return -1;


static int launch(hipFunction_t f, unsigned grid, unsigned block, hipStream_t s,
                  void** args) {
    static int trace = getenv("TPX_TRACE") ? 1 : 0;
    if (trace) {
        fprintf(stderr, "[tpx] launch f=%p grid=%u block=%u\n", (void*)f,
                grid, block);
        (void)hipDeviceSynchronize();
    }
    hipError_t e = hipModuleLaunchKernel(f, grid, 1, 1, block, 1, 1, 0, s, args,
                                         nullptr);
    if (trace) {
        hipError_t e2 = hipDeviceSynchronize();
        fprintf(stderr, "[tpx] done  f=%p rc=%d sync=%d\n", (void*)f, (int)e,
                (int)e2);
    }
    if (e != hipSuccess) { set_err(hipGetErrorString(e)); return -1; }
    return 0;
}

// exclusive scan of device i64 array (n elements), in -> out; returns total via
// host copy of (out[n-1] + in[n-1]).
static size_t col_slot_bytes(const ColDesc& c, long long n, int slot) {
    if (slot == 0) return (size_t)n * 8;                       // value / ptr
    if (slot == 1) return c.kind == K_STR ? (size_t)n * 4 : 0; // len
    return c.opt ? (size_t)n : 0;                              // null mask
}

struct ExcRec {
    long long row, ec, opid, off_start, off_end;  // offsets into the input bytes
};

// exclusive scan of device i64 array (n elements), in -> out; total = out[n-1] +
// in[n-1]. Temporaries come from the workspace arena.
static int dev_scan(tpx_stage* st, hipStream_t stream, long long* d_in,
                    long long* d_out, long long n, long long* total) {
    int dev = cur_device();
    const long long BLOCK = 2048;  // TPX_SCAN_BLOCK
    long long nblocks = (n + BLOCK - 1) / BLOCK;
    if (nblocks == 0) { if (total) *total = 0; return 0; }
    ARENA_TAKE(sums, (size_t)nblocks * 8);
    void* in_p = d_in;
    void* out_p = d_out;
    void* a1[] = {&in_p, &out_p, &sums, &n};
    if (launch(st->k_scan_block, (unsigned)nblocks, 256, stream, a1)) return -1;
    if (nblocks > 1) {
        ARENA_TAKE(sums_scan, (size_t)nblocks * 8);
        long long dummy;
        if (dev_scan(st, stream, (long long*)sums, (long long*)sums_scan,
                     nblocks, &dummy))
            return -1;
        void* a2[] = {&out_p, &sums_scan, &n};
        if (launch(st->k_scan_add, (unsigned)nblocks, 256, stream, a2)) return -1;
    }
    if (!total) return 0;  // caller fetches totals itself (tpx_pair_total)
    long long last_out = 0, last_in = 0;
    HIP_CHECK(hipMemcpyAsync(&last_out, d_out + (n - 1), 8,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipMemcpyAsync(&last_in, d_in + (n - 1), 8, hipMemcpyDeviceToHost,
                             stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    *total = last_out + last_in;
    return 0;
}

// ---------------------------------------------------------------------------------
// shared stage-execution core (mem + csv sources share the fixed kernel signature:
// tpx_stage_main(in_data, in_offs, n, row0, heap..., keep..., exc..., outv))

struct PayloadSrc {
    const uint8_t* bytes;         // flat input bytes (csv) — off_start/off_end index here
    const tpx_partition* parts;   // or: partition list (mem)
    int64_t n_parts;
};

// pre_chunk(row_lo, row_hi, stream): optional hook launched on a chunk's
// stream BEFORE its main kernel — the csv path uses it to emit that chunk's
// row offsets just-in-time, overlapping the previous chunk's main kernel.
using PreChunkFn = std::function<int(long long, long long, hipStream_t)>;

static int64_t run_core(tpx_stage* st, void* d_in, void* d_offs, long long n,
                        long long row0, long long in_bytes, tpx_result* res,
                        const PayloadSrc& psrc, hipStream_t stream,
                        int64_t flags = 0, const PreChunkFn* pre_chunk = nullptr,
                        const std::vector<void*>* in_tab = nullptr) {
    const StageDesc& D = st->desc;
    int dev = cur_device();
    Persist& P = g_persist[dev];
    res->in_num_rows = n;
    EventPool& EP = g_events[dev];
    EP.reset();  // previous call's events are fully consumed by its return
    hipEvent_t ev1 = EP.get(), ev2 = EP.get(), ev3 = EP.get();
    hipEvent_t evm0 = EP.get(), evm1 = EP.get(), evs1 = EP.get();
    (void)evm1;
    hipEventRecord(ev1, stream);

    // output columnar buffers (3 slots per column) from the arena
    int nout = (int)D.out_cols.size();
    std::vector<void*> outv((size_t)nout * 3, nullptr);
    for (int k = 0; k < nout; ++k)
        for (int s = 0; s < 3; ++s) {
            size_t b = col_slot_bytes(D.out_cols[(size_t)k], n, s);
            if (b) {
                outv[(size_t)k * 3 + s] = g_arena[dev].take(b);
                if (!outv[(size_t)k * 3 + s]) return -1;
            }
        }
    ARENA_TAKE(d_outv, outv.size() * sizeof(void*) + 8);
    HIP_CHECK(hipMemcpyAsync(d_outv, outv.data(), outv.size() * sizeof(void*),
                             hipMemcpyHostToDevice, stream));

    ARENA_TAKE(d_keep, (size_t)n);
    ARENA_TAKE(d_keep01, (size_t)n * 8);
    ARENA_TAKE(d_sizes, (size_t)n * 8);

    // persistent heap / exception buffer / counters
    if (!P.counters)
        HIP_CHECK(hipMalloc(&P.counters, 160));
    unsigned long long exc_cap =
        (unsigned long long)std::min<long long>(std::max<long long>(n, 1024),
                                                1 << 20);
    if (P.exc_cap_recs < exc_cap) {
        if (P.exc) (void)hipFree(P.exc);
        HIP_CHECK(hipMalloc(&P.exc, (size_t)exc_cap * sizeof(ExcRec)));
        P.exc_cap_recs = exc_cap;
    }
    exc_cap = P.exc_cap_recs;

    // chunk pipelining: split the rows into C ranges; each chunk's compaction
    // scans + write kernel run on a side stream and overlap the NEXT chunk's
    // main kernel (the step is otherwise strictly sequential; main dominates).
    // Aggregate sinks consume the full keep/outv arrays at once -> C=1.
    bool mem_sink = D.sink == "mem";
    int C = 1;
    if (D.agg.empty() && D.aggby.empty() && n >= (1 << 20) &&
        (d_offs || in_tab)) {
        static int env_c = [] {
            const char* e = getenv("TPX_CHUNKS");
            int v = e ? atoi(e) : 4;
            return v < 1 ? 1 : (v > 8 ? 8 : v);
        }();
        C = env_c;
    }
    hipStream_t S[2] = {stream, stream};
    if (C > 1) { S[0] = cstream(dev, 0); S[1] = cstream(dev, 1); }
    // chunk split: sizes proportional to ratio^c (ratio<1 -> later chunks
    // smaller, shrinking the non-overlapped scan+write tail of the last chunk)
    static double env_ratio = [] {
        const char* e = getenv("TPX_CHUNK_RATIO");
        double v = e ? atof(e) : 1.0;  // measured: equal split beats 0.75
        return (v > 0.1 && v <= 1.0) ? v : 1.0;
    }();
    // tail split: equal chunks, but the LAST one halved into two. Measured
    // NEGATIVE on Zillow Z1 (772 vs 804 M rows/s: the extra launch + lower
    // wave efficiency at 5 chunks cost more than the halved tail saved) —
    // default off, kept for other shapes (TPX_TAIL_SPLIT=1).
    static int env_tail = [] {
        const char* e = getenv("TPX_TAIL_SPLIT");
        return e ? atoi(e) : 0;
    }();
    if (env_tail && C > 1 && C < 8) ++C;
    long long cstart[9] = {0};
    long long ccnt[8] = {0};
    {
        int CW = (env_tail && C > 2) ? C - 1 : C;  // weight units: last 2 = half
        double wsum = 0, w = 1;
        for (int c = 0; c < CW; ++c) { wsum += w; w *= env_ratio; }
        double acc = 0;
        w = 1;
        for (int c = 0; c < C; ++c) {
            bool half = env_tail && C > 2 && c >= C - 2;
            acc += half ? w * 0.5 : w;
            cstart[c + 1] = c + 1 == C ? n : (long long)(n * acc / wsum);
            if (!half) w *= env_ratio;
        }
        for (int c = 0; c < C; ++c) ccnt[c] = cstart[c + 1] - cstart[c];
    }
    // per-chunk biased outv pointer tables (chunk rows index columnar slot i
    // locally); chunk 0 table == the unbiased one
    void* d_outv_c[8] = {d_outv};
    if (C > 1) {
        std::vector<void*> ovall(outv.size() * (size_t)(C - 1));
        for (int c = 1; c < C; ++c) {
            void** ov = ovall.data() + outv.size() * (size_t)(c - 1);
            for (int k = 0; k < nout; ++k) {
                const ColDesc& col = D.out_cols[(size_t)k];
                ov[(size_t)k * 3 + 0] =
                    (char*)outv[(size_t)k * 3 + 0] + cstart[c] * 8;
                ov[(size_t)k * 3 + 1] = col.kind == K_STR
                    ? (char*)outv[(size_t)k * 3 + 1] + cstart[c] * 4 : nullptr;
                ov[(size_t)k * 3 + 2] = col.opt
                    ? (char*)outv[(size_t)k * 3 + 2] + cstart[c] : nullptr;
            }
        }
        void* blk = g_arena[dev].take(ovall.size() * sizeof(void*) + 8);
        if (!blk) return -1;
        HIP_CHECK(hipMemcpyAsync(blk, ovall.data(), ovall.size() * sizeof(void*),
                                 hipMemcpyHostToDevice, stream));
        for (int c = 1; c < C; ++c)
            d_outv_c[c] = (char*)blk + outv.size() * sizeof(void*) * (size_t)(c - 1);
    }
    // columnar source: per-chunk biased INPUT tables (same 3-slot layout as
    // outputs; strings keep their unbiased data base, offsets/masks shift)
    void* d_in_c[8] = {d_in};
    if (C > 1 && in_tab) {
        size_t ns = in_tab->size();
        std::vector<void*> iall(ns * (size_t)(C - 1));
        for (int c = 1; c < C; ++c) {
            void** tv = iall.data() + ns * (size_t)(c - 1);
            for (size_t k3 = 0; k3 < ns; k3 += 3) {
                const ColDesc& col = D.in_cols[k3 / 3];
                char* v0 = (char*)(*in_tab)[k3];
                char* v1 = (char*)(*in_tab)[k3 + 1];
                char* v2 = (char*)(*in_tab)[k3 + 2];
                long long stride0 = col.kind == K_BOOL ? 1 : 8;
                tv[k3] = v0 ? v0 + cstart[c] * stride0 : nullptr;
                tv[k3 + 1] = v1;  // string data base: unbiased
                tv[k3 + 2] = v2 ? v2 + cstart[c] : nullptr;
            }
        }
        void* blk = g_arena[dev].take(iall.size() * sizeof(void*) + 8);
        if (!blk) return -1;
        HIP_CHECK(hipMemcpyAsync(blk, iall.data(), iall.size() * sizeof(void*),
                                 hipMemcpyHostToDevice, stream));
        for (int c = 1; c < C; ++c)
            d_in_c[c] = (char*)blk + ns * sizeof(void*) * (size_t)(c - 1);
    }

    // split mode (D.split): allocate the cell park — typed value arrays per
    // used input column, a string byte park with one bump cursor, and per-row
    // prc/dirty arrays; the k_main table gets them as slots [3*nin, 3*nin+1].
    bool split = D.split && d_offs && st->k_parse != nullptr;
    void* d_park_c[8] = {nullptr};
    void* d_strbuf = nullptr;
    void* d_strcur = nullptr;
    void* d_prc = nullptr;
    void* d_dirty = nullptr;
    if (split) {
        int nin = (int)D.in_cols.size();
        std::vector<void*> parkv((size_t)nin * 3 + 2, nullptr);
        for (int k : D.used) {
            const ColDesc& col = D.in_cols[(size_t)k];
            parkv[(size_t)k * 3] = g_arena[dev].take((size_t)n * 8);
            if (!parkv[(size_t)k * 3]) return -1;
            if (col.kind == K_STR) {
                parkv[(size_t)k * 3 + 1] = g_arena[dev].take((size_t)n * 4);
                if (!parkv[(size_t)k * 3 + 1]) return -1;
            }
            if (col.opt) {
                parkv[(size_t)k * 3 + 2] = g_arena[dev].take((size_t)n);
                if (!parkv[(size_t)k * 3 + 2]) return -1;
            }
        }
        d_prc = g_arena[dev].take((size_t)n * 8);
        d_dirty = g_arena[dev].take((size_t)n * 8);
        if (!d_prc || !d_dirty) return -1;
        if (D.parkcopy) {
            // park cells <= raw row bytes; +8/row pad + SWAR overread slack
            d_strbuf = g_arena[dev].take((size_t)in_bytes + (size_t)n * 8 +
                                         4096);
            d_strcur = g_arena[dev].take(256);
            if (!d_strbuf || !d_strcur) return -1;
            HIP_CHECK(hipMemsetAsync(d_strcur, 0, 8, stream));
        }
        parkv[(size_t)nin * 3] = d_prc;
        parkv[(size_t)nin * 3 + 1] = d_dirty;
        std::vector<void*> pall(parkv.size() * (size_t)C);
        for (int c = 0; c < C; ++c) {
            void** tv = pall.data() + parkv.size() * (size_t)c;
            for (int k = 0; k < nin; ++k) {
                char* v0 = (char*)parkv[(size_t)k * 3];
                char* v1 = (char*)parkv[(size_t)k * 3 + 1];
                char* v2 = (char*)parkv[(size_t)k * 3 + 2];
                tv[k * 3] = v0 ? v0 + cstart[c] * 8 : nullptr;
                tv[k * 3 + 1] = v1 ? v1 + cstart[c] * 4 : nullptr;
                tv[k * 3 + 2] = v2 ? v2 + cstart[c] : nullptr;
            }
            tv[nin * 3] = (char*)d_prc + cstart[c] * 8;
            tv[nin * 3 + 1] = (char*)d_dirty + cstart[c] * 8;
        }
        void* blk = g_arena[dev].take(pall.size() * sizeof(void*) + 8);
        if (!blk) return -1;
        HIP_CHECK(hipMemcpyAsync(blk, pall.data(),
                                 pall.size() * sizeof(void*),
                                 hipMemcpyHostToDevice, stream));
        for (int c = 0; c < C; ++c)
            d_park_c[c] = (char*)blk +
                          parkv.size() * sizeof(void*) * (size_t)c;
    }

    // +thread-chunk slack per LAUNCH: <=2048 blocks x 256 threads x 256 B
    unsigned long long heap_cap =
        (unsigned long long)std::max<long long>(in_bytes + (in_bytes >> 1) +
                                                C * (160ll << 20), 1 << 20);
    if (P.heap_cap < heap_cap) {
        if (P.heap) (void)hipFree(P.heap);
        HIP_CHECK(hipMalloc(&P.heap, heap_cap + 16));  // tpx_memcpy over-read pad
        P.heap_cap = heap_cap;
    }
    heap_cap = P.heap_cap;

    void* d_exc_count = (char*)P.counters + 128;
    unsigned long long exc_count = 0;
    // [m1, s0, s1, w0, w1] per chunk + 2 join events
    std::vector<hipEvent_t> cev((size_t)C * 5 + 2);
    for (auto& e : cev) e = EP.get();
    long long total_rows = 0, total_bytes = 0;
    long long chunk_rows[8] = {0}, chunk_bytes[8] = {0};
    void* chunk_out[8] = {nullptr};
    void* d_out = nullptr;
    void* d_out_offs = nullptr;
    void* d_out_rowidx = nullptr;
    bool sink_done = false;
    for (int attempt = 0;; ++attempt) {
        HIP_CHECK(hipMemsetAsync(P.counters, 0, 160, stream));
        hipEventRecord(evm0, stream);
        unsigned long long hs = heap_cap / (unsigned long long)C & ~255ull;
        for (int c = 0; c < C; ++c) {
            hipStream_t sc = S[c & 1];
            if (C > 1) hipStreamWaitEvent(sc, evm0, 0);
            if (pre_chunk && attempt == 0 &&
                (*pre_chunk)(cstart[c], cstart[c + 1], sc))
                return -1;
            // main runs 128-thread blocks (2 waves x 16 KiB LDS staging =
            // 32 KiB -> 5 blocks/CU = 10 waves/CU); each wave owns 64 rows
            long long nc = ccnt[c];
            long long waves = (nc + 63) / 64;
            unsigned grid = (unsigned)std::min<long long>((waves + 1) / 2, 4096);
            void* in_offs_c = d_offs ? (char*)d_offs + cstart[c] * 8 : nullptr;
            void* d_in_use = in_tab ? d_in_c[c] : d_in;
            long long row0_c = row0 + cstart[c];
            void* heap_c = (char*)P.heap + (unsigned long long)c * hs;
            void* cursor_c = (char*)P.counters + 16 * c;
            void* keep_c = (char*)d_keep + cstart[c];
            void* keep01_c = (char*)d_keep01 + cstart[c] * 8;
            void* sizes_c = (char*)d_sizes + cstart[c] * 8;
            if (split) {
                if (attempt == 0) {  // park is attempt-invariant
                    void* prc_c = (char*)d_prc + cstart[c] * 8;
                    void* dirty_c = (char*)d_dirty + cstart[c] * 8;
                    void* pargs[] = {&d_in, &in_offs_c, &nc, &row0_c,
                                     &d_strbuf, &d_strcur, &d_park_c[c],
                                     &prc_c, &dirty_c};
                    if (launch(st->k_parse, grid, 128, sc, pargs)) return -1;
                }
                d_in_use = d_park_c[c];
            }
            void* args[] = {&d_in_use, &in_offs_c, &nc, &row0_c, &heap_c,
                            &cursor_c, &hs, &keep_c, &keep01_c, &sizes_c,
                            &P.exc, &d_exc_count, &exc_cap, &d_outv_c[c]};
            if (split) {
                unsigned g2 = (unsigned)std::min<long long>(
                    (nc + 255) / 256, 4096);
                if (g2 < 1) g2 = 1;
                if (launch(st->k_main, g2, 256, sc, args)) return -1;
            } else {
                if (launch(st->k_main, grid, 128, sc, args)) return -1;
            }
            hipEventRecord(cev[(size_t)c * 5], sc);
        }
        // C>1 (non-agg sink): pipelined per-chunk scans + writes launched
        // OPTIMISTICALLY before the overflow check (overflowed rows are marked
        // exceptions with keep=false, so the write reads no garbage; on the
        // rare overflow the attempt is simply redone)
        if (C > 1) {
            total_rows = total_bytes = 0;
            d_out_offs = g_arena[dev].take(((size_t)n + 1) * 8);
            d_out_rowidx = g_arena[dev].take(((size_t)n + 1) * 8);
            void* d_keep_scan = g_arena[dev].take((size_t)n * 8);
            void* d_size_scan = g_arena[dev].take((size_t)n * 8);
            if (!d_out_offs || !d_out_rowidx || !d_keep_scan || !d_size_scan)
                return -1;
            for (int c = 0; c < C; ++c) {
                hipStream_t sc = S[c & 1];
                void* keep01_c = (char*)d_keep01 + cstart[c] * 8;
                void* sizes_c = (char*)d_sizes + cstart[c] * 8;
                void* kscan_c = (char*)d_keep_scan + cstart[c] * 8;
                void* sscan_c = (char*)d_size_scan + cstart[c] * 8;
                long long rows_c = 0, bytes_c = 0;
                hipEventRecord(cev[(size_t)c * 5 + 1], sc);
                if (dev_scan(st, sc, (long long*)keep01_c,
                             (long long*)kscan_c, ccnt[c], nullptr))
                    return -1;
                if (dev_scan(st, sc, (long long*)sizes_c, (long long*)sscan_c,
                             ccnt[c], nullptr))
                    return -1;
                {   // both totals with one tiny kernel + one pinned D2H
                    void* slot = g_arena[dev].take(16);
                    if (!slot) return -1;
                    void* ta[] = {&keep01_c, &kscan_c, &ccnt[c], &sizes_c,
                                  &sscan_c, &ccnt[c], &slot};
                    if (launch(st->k_pair_total, 1, 64, sc, ta)) return -1;
                    long long* host = (long long*)pinned(dev);
                    HIP_CHECK(hipMemcpyAsync(host, slot, 16,
                                             hipMemcpyDeviceToHost, sc));
                    HIP_CHECK(hipStreamSynchronize(sc));
                    rows_c = host[0];
                    bytes_c = host[1];
                }
                hipEventRecord(cev[(size_t)c * 5 + 2], sc);
                chunk_out[c] = g_arena[dev].take((size_t)bytes_c + 16);
                if (!chunk_out[c]) return -1;
                long long out_byte0 = (mem_sink ? 8 : 0) + total_bytes;
                void* keep_c = (char*)d_keep + cstart[c];
                void* offs_c = (char*)d_out_offs + total_rows * 8;
                void* ridx_c = (char*)d_out_rowidx + total_rows * 8;
                long long row0_c = row0 + cstart[c];
                unsigned grid = (unsigned)std::min<long long>(
                    (ccnt[c] + 127) / 128, 8192);
                if (grid < 1) grid = 1;
                void* args[] = {&keep_c, &kscan_c, &sscan_c, &ccnt[c], &row0_c,
                                &d_outv_c[c], &chunk_out[c], &offs_c, &ridx_c,
                                &rows_c, &bytes_c, &out_byte0};
                hipEventRecord(cev[(size_t)c * 5 + 3], sc);
                if (launch(st->k_write, grid, 128, sc, args)) return -1;
                hipEventRecord(cev[(size_t)c * 5 + 4], sc);
                chunk_rows[c] = rows_c;
                chunk_bytes[c] = bytes_c;
                total_rows += rows_c;
                total_bytes += bytes_c;
            }
        }
        // overflow check (all mains have completed: each chunk's scans synced
        // its stream for C>1; explicit sync below covers C==1)
        unsigned long long* counters =
            (unsigned long long*)((char*)pinned(dev) + 256);
        HIP_CHECK(hipMemcpyAsync(counters, P.counters, 160,
                                 hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        if (C == 1) {
            // the copy above raced the main kernel only through stream order
            // (same stream) -> already correct
        }
        unsigned long long max_used = 0;
        for (int c = 0; c < C; ++c)
            max_used = std::max(max_used, counters[c * 2]);
        exc_count = counters[16];
        bool heap_overflow = max_used > hs;
        bool exc_overflow = exc_count > exc_cap;
        if (!heap_overflow && !exc_overflow) {
            sink_done = C > 1;
            break;
        }
        if (attempt >= 3) { set_err("retry limit (heap/exc overflow)"); return -1; }
        if (heap_overflow) {
            (void)hipFree(P.heap);
            P.heap = nullptr;
            P.heap_cap = 0;
            heap_cap = (max_used + (max_used >> 1) + (16 << 20)) *
                       (unsigned long long)C;
            HIP_CHECK(hipMalloc(&P.heap, heap_cap + 16));
            P.heap_cap = heap_cap;
        }
        if (exc_overflow) {
            (void)hipFree(P.exc);
            P.exc = nullptr;
            P.exc_cap_recs = 0;
            exc_cap = exc_count + 1024;
            HIP_CHECK(hipMalloc(&P.exc, (size_t)exc_cap * sizeof(ExcRec)));
            P.exc_cap_recs = exc_cap;
        }
    }

    if (!D.aggby.empty() && D.aggkeystr) {
        // STRING-key hash-reduce: slots claim the first row's (ptr,len); a
        // not-yet-visible len makes the prober move on, so the same string
        // may land in several slots — the HOST merges the emitted pairs by
        // key bytes (sum is commutative). Key bytes are fetched from device
        // memory (views into in_data / the string heap, both live here).
        bool is_f64 = D.aggby == "f64";
        hipFunction_t kfill = is_f64 ? st->k_hk_sf64 : st->k_hk_si64;
        if (!kfill || !st->k_hk_semit) {
            set_err("string hashagg kernels missing");
            return -1;
        }
        unsigned long long tsize = 1ull << 20;
        unsigned long long used = 0;
        void* tkeys = nullptr;
        void* tlens = nullptr;
        void* tvals = nullptr;
        ARENA_TAKE(d_hk_state, 40);
        for (int attempt = 0;; ++attempt) {
            tkeys = g_arena[dev].take(tsize * 8);
            tlens = g_arena[dev].take(tsize * 4);
            tvals = g_arena[dev].take(tsize * 8);
            if (!tkeys || !tlens || !tvals) return -1;
            HIP_CHECK(hipMemsetAsync(tkeys, 0, tsize * 8, stream));
            HIP_CHECK(hipMemsetAsync(tlens, 0, tsize * 4, stream));
            HIP_CHECK(hipMemsetAsync(tvals, 0, tsize * 8, stream));
            HIP_CHECK(hipMemsetAsync(d_hk_state, 0, 40, stream));
            unsigned long long tmask = tsize - 1;
            void* kptr = outv[0];
            void* klen = outv[1];
            void* vals = outv[3];
            void* d_used = d_hk_state;
            void* d_ovf = (char*)d_hk_state + 8;
            unsigned grid2 = (unsigned)std::min<long long>((n + 255) / 256,
                                                           2048);
            void* a1[] = {&kptr, &klen, &vals, &d_keep, &n, &tkeys, &tlens,
                          &tvals, &tmask, &d_used, &d_ovf};
            if (launch(kfill, grid2, 256, stream, a1)) return -1;
            unsigned long long st8[2] = {0, 0};
            HIP_CHECK(hipMemcpyAsync(st8, d_hk_state, 16,
                                     hipMemcpyDeviceToHost, stream));
            HIP_CHECK(hipStreamSynchronize(stream));
            used = st8[0];
            bool ovf = st8[1] != 0 || used * 2 > tsize;
            if (!ovf) break;
            if (attempt >= 3 || tsize >= (1ull << 27)) {
                set_err("aggregateByKey table overflow");
                return -1;
            }
            tsize *= 8;
        }
        hipEventRecord(evs1, stream);
        long long cap = (long long)used + 1;
        ARENA_TAKE(d_tri, (size_t)cap * 24 + 16);
        void* d_oidx = (char*)d_hk_state + 32;
        {
            long long ts_ll = (long long)tsize;
            unsigned grid2 = (unsigned)std::min<long long>(
                ((long long)tsize + 255) / 256, 4096);
            void* a2[] = {&tkeys, &tlens, &tvals, &ts_ll, &d_tri, &d_oidx};
            if (launch(st->k_hk_semit, grid2, 256, stream, a2)) return -1;
        }
        hipEventRecord(ev2, stream);
        unsigned long long nout = 0;
        HIP_CHECK(hipMemcpyAsync(&nout, d_oidx, 8, hipMemcpyDeviceToHost,
                                 stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        std::vector<long long> tri((size_t)nout * 3);
        if (nout)
            HIP_CHECK(hipMemcpy(tri.data(), d_tri, (size_t)nout * 24,
                                hipMemcpyDeviceToHost));
        // fetch key bytes + merge duplicate slots by content
        std::map<std::string, long long> merged;
        for (unsigned long long e = 0; e < nout; ++e) {
            long long len = tri[e * 3 + 1];
            std::string key((size_t)len, 0);
            if (len)
                HIP_CHECK(hipMemcpy(&key[0], (void*)(uintptr_t)tri[e * 3],
                                    (size_t)len, hipMemcpyDeviceToHost));
            long long vb = tri[e * 3 + 2];
            auto it = merged.find(key);
            if (it == merged.end()) {
                merged[key] = vb;
            } else if (is_f64) {
                double acc;
                memcpy(&acc, &it->second, 8);
                double add;
                memcpy(&add, &vb, 8);
                acc += add;
                memcpy(&it->second, &acc, 8);
            } else {
                it->second += vb;
            }
        }
        // build the (str, val) output partition (Serializer.cc layout:
        // 2 slots + varlen_total + bytes, str slot = off|size<<32, off from
        // the slot's own address)
        size_t total_bytes2 = 8;
        for (auto& kvp : merged)
            total_bytes2 += 24 + kvp.first.size() + 1;
        res->out_size = (int64_t)total_bytes2;
        res->out_num_rows = (int64_t)merged.size();
        res->out_data = (uint8_t*)malloc(total_bytes2);
        ((int64_t*)res->out_data)[0] = (int64_t)merged.size();
        res->out_row_offsets = (int64_t*)malloc((merged.size() + 1) * 8);
        res->out_row_indices = (int64_t*)malloc((merged.size() + 1) * 8);
        {
            uint8_t* w = res->out_data + 8;
            long long ri = 0;
            for (auto& kvp : merged) {
                res->out_row_offsets[ri] = (int64_t)(w - res->out_data);
                res->out_row_indices[ri] = ri;
                long long len = (long long)kvp.first.size();
                ((int64_t*)w)[0] = 24 | ((len + 1) << 32);
                ((int64_t*)w)[1] = kvp.second;
                ((int64_t*)w)[2] = len + 1;
                memcpy(w + 24, kvp.first.data(), (size_t)len);
                w[24 + len] = 0;
                w += 24 + len + 1;
                ++ri;
            }
            res->out_row_offsets[ri] = (int64_t)total_bytes2;
        }
        res->bytes_out = res->out_size;
    } else if (!D.aggby.empty()) {
        // by-key hash-reduce (hashmap.cc analog); output order unpinned
        hipFunction_t kfill = D.aggby == "f64" ? st->k_hk_f64 : st->k_hk_i64;
        if (!kfill || !st->k_hk_emit) { set_err("hashagg kernels missing"); return -1; }
        unsigned long long tsize = 1ull << 20;
        unsigned long long used = 0;
        void* tkeys = nullptr;
        void* tvals = nullptr;
        ARENA_TAKE(d_hk_state, 40);  // used, overflow, special_cnt, special_val, out_idx
        for (int attempt = 0;; ++attempt) {
            tkeys = g_arena[dev].take(tsize * 8);
            tvals = g_arena[dev].take(tsize * 8);
            if (!tkeys || !tvals) return -1;
            HIP_CHECK(hipMemsetAsync(tkeys, 0xFF, tsize * 8, stream));
            HIP_CHECK(hipMemsetAsync(tvals, 0, tsize * 8, stream));
            HIP_CHECK(hipMemsetAsync(d_hk_state, 0, 40, stream));
            unsigned long long tmask = tsize - 1;
            void* keys = outv[0];
            void* vals = outv[3];
            void* d_used = d_hk_state;
            void* d_ovf = (char*)d_hk_state + 8;
            void* d_scnt = (char*)d_hk_state + 16;
            void* d_sval = (char*)d_hk_state + 24;
            unsigned grid2 = (unsigned)std::min<long long>((n + 255) / 256, 2048);
            void* a1[] = {&keys, &vals, &d_keep, &n, &tkeys, &tvals, &tmask,
                          &d_used, &d_ovf, &d_scnt, &d_sval};
            if (launch(kfill, grid2, 256, stream, a1)) return -1;
            unsigned long long st8[2] = {0, 0};
            HIP_CHECK(hipMemcpyAsync(st8, d_hk_state, 16, hipMemcpyDeviceToHost,
                                     stream));
            HIP_CHECK(hipStreamSynchronize(stream));
            used = st8[0];
            bool ovf = st8[1] != 0 || used * 2 > tsize;
            if (!ovf) break;
            if (attempt >= 3 || tsize >= (1ull << 27)) {
                set_err("aggregateByKey table overflow");
                return -1;
            }
            tsize *= 8;
        }
        hipEventRecord(evs1, stream);
        long long out_rows_cap = (long long)used + 1;
        long long out_total = 8 + out_rows_cap * 16;
        ARENA_TAKE(d_out, (size_t)out_total + 16);
        void* d_oidx = (char*)d_hk_state + 32;
        {
            long long ts_ll = (long long)tsize;
            void* d_scnt = (char*)d_hk_state + 16;
            void* d_sval = (char*)d_hk_state + 24;
            unsigned grid2 = (unsigned)std::min<long long>(((long long)tsize + 255) / 256, 4096);
            void* a2[] = {&tkeys, &tvals, &ts_ll, &d_scnt, &d_sval, &d_out,
                          &d_oidx};
            if (launch(st->k_hk_emit, grid2, 256, stream, a2)) return -1;
        }
        hipEventRecord(ev2, stream);
        unsigned long long nout_rows = 0;
        HIP_CHECK(hipMemcpyAsync(&nout_rows, d_oidx, 8, hipMemcpyDeviceToHost,
                                 stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        long long total = (long long)nout_rows;
        res->out_size = 8 + total * 16;
        res->out_num_rows = total;
        res->out_data = (uint8_t*)malloc((size_t)res->out_size);
        ((int64_t*)res->out_data)[0] = total;
        if (total)
            HIP_CHECK(hipMemcpy(res->out_data + 8, (char*)d_out + 8,
                                (size_t)total * 16, hipMemcpyDeviceToHost));
        res->out_row_offsets = (int64_t*)malloc(((size_t)total + 1) * 8);
        res->out_row_indices = (int64_t*)malloc(((size_t)total + 1) * 8);
        for (long long i = 0; i <= total; ++i)
            res->out_row_offsets[i] = 8 + i * 16;
        for (long long i = 0; i < total; ++i) res->out_row_indices[i] = i;
        res->bytes_out = res->out_size;
    } else if (!D.agg.empty()) {
        // GPU aggregate fold (Q6/count/sum): deterministic masked reduce of the
        // per-row expr column; result = a 1-row partition [numRows=1][value]
        bool is_f64 = D.agg == "f64";
        hipFunction_t kr = is_f64 ? st->k_red_f64 : st->k_red_i64;
        hipFunction_t krf = is_f64 ? st->k_red_f64_fin : st->k_red_i64_fin;
        if (!kr || !krf) { set_err("reduce kernels missing"); return -1; }
        long long nb = std::min<long long>((n + 2047) / 2048, 1024);
        if (nb < 1) nb = 1;
        ARENA_TAKE(d_partials, (size_t)nb * 8);
        ARENA_TAKE(d_res8, 8);
        void* vals = outv[0];
        void* a1[] = {&vals, &d_keep, &n, &d_partials};
        if (launch(kr, (unsigned)nb, 256, stream, a1)) return -1;
        void* a2[] = {&d_partials, &nb, &d_res8};
        if (launch(krf, 1, 256, stream, a2)) return -1;
        hipEventRecord(evs1, stream);
        hipEventRecord(ev2, stream);
        long long bits = 0;
        HIP_CHECK(hipMemcpyAsync(&bits, d_res8, 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        res->out_size = 16;
        res->out_num_rows = 1;
        res->out_data = (uint8_t*)malloc(16);
        ((int64_t*)res->out_data)[0] = 1;
        ((int64_t*)res->out_data)[1] = bits;
        res->out_row_offsets = (int64_t*)malloc(16);
        res->out_row_offsets[0] = 8;
        res->out_row_offsets[1] = 16;
        res->out_row_indices = (int64_t*)malloc(8);
        res->out_row_indices[0] = 0;
    } else {
    long long out_total;
    if (sink_done) {
        // C>1: per-chunk writes already done; gather the chunk buffers into
        // the final contiguous output
        out_total = (mem_sink ? 8 : 0) + total_bytes;
        d_out = g_arena[dev].take((size_t)out_total + 16);
        if (!d_out) return -1;
        long long acc = mem_sink ? 8 : 0;
        for (int c = 0; c < C; ++c) {
            if (chunk_bytes[c])
                HIP_CHECK(hipMemcpyAsync((char*)d_out + acc, chunk_out[c],
                                         (size_t)chunk_bytes[c],
                                         hipMemcpyDeviceToDevice, S[c & 1]));
            acc += chunk_bytes[c];
        }
        // join the side streams back into `stream`
        hipEventRecord(cev[(size_t)C * 5], S[0]);
        hipEventRecord(cev[(size_t)C * 5 + 1], S[1]);
        hipStreamWaitEvent(stream, cev[(size_t)C * 5], 0);
        hipStreamWaitEvent(stream, cev[(size_t)C * 5 + 1], 0);
        hipEventRecord(evs1, stream);
    } else {
        // C==1: compaction scans + one write into the final buffer
        ARENA_TAKE(d_keep_scan, (size_t)n * 8);
        ARENA_TAKE(d_size_scan, (size_t)n * 8);
        if (dev_scan(st, stream, (long long*)d_keep01, (long long*)d_keep_scan,
                     n, &total_rows))
            return -1;
        if (dev_scan(st, stream, (long long*)d_sizes, (long long*)d_size_scan,
                     n, &total_bytes))
            return -1;
        hipEventRecord(evs1, stream);
        out_total = (mem_sink ? 8 : 0) + total_bytes;
        d_out = g_arena[dev].take((size_t)out_total + 16);
        d_out_offs = g_arena[dev].take(((size_t)total_rows + 1) * 8);
        d_out_rowidx = g_arena[dev].take(((size_t)total_rows + 1) * 8);
        if (!d_out || !d_out_offs || !d_out_rowidx) return -1;
        unsigned grid = (unsigned)std::min<long long>((n + 127) / 128, 8192);
        if (grid < 1) grid = 1;
        void* out_base = (char*)d_out + (mem_sink ? 8 : 0);
        long long out_byte0 = mem_sink ? 8 : 0;
        void* args[] = {&d_keep, &d_keep_scan, &d_size_scan, &n, &row0,
                        &d_outv, &out_base, &d_out_offs, &d_out_rowidx,
                        &total_rows, &total_bytes, &out_byte0};
        if (launch(st->k_write, grid, 128, stream, args)) return -1;
    }
    // header + sentinel are host-written (the kernels write only row entries)
    long long sentinel = out_total;
    HIP_CHECK(hipMemcpyAsync((char*)d_out_offs + total_rows * 8, &sentinel, 8,
                             hipMemcpyHostToDevice, stream));
    if (mem_sink)
        HIP_CHECK(hipMemcpyAsync(d_out, &total_rows, 8, hipMemcpyHostToDevice,
                                 stream));
    hipEventRecord(ev2, stream);

    // D2H (skipped when the caller keeps outputs device-resident, flags bit1)
    res->out_size = out_total;
    res->out_num_rows = total_rows;
    if (!(flags & 2)) {
        res->out_data = (uint8_t*)malloc((size_t)out_total);
        HIP_CHECK(hipMemcpyAsync(res->out_data, d_out, (size_t)out_total,
                                 hipMemcpyDeviceToHost, stream));
        res->out_row_offsets = (int64_t*)malloc(((size_t)total_rows + 1) * 8);
        HIP_CHECK(hipMemcpyAsync(res->out_row_offsets, d_out_offs,
                                 ((size_t)total_rows + 1) * 8,
                                 hipMemcpyDeviceToHost, stream));
        res->out_row_indices = (int64_t*)malloc(((size_t)total_rows + 1) * 8);
        if (total_rows)
            HIP_CHECK(hipMemcpyAsync(res->out_row_indices, d_out_rowidx,
                                     (size_t)total_rows * 8,
                                     hipMemcpyDeviceToHost, stream));
    }
    res->bytes_out = out_total;
    }
    std::vector<ExcRec> excs((size_t)exc_count);
    if (exc_count)
        HIP_CHECK(hipMemcpyAsync(excs.data(), P.exc,
                                 (size_t)exc_count * sizeof(ExcRec),
                                 hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    hipEventRecord(ev3, stream);
    HIP_CHECK(hipEventSynchronize(ev3));

    // exception buffer in the reference record format (IExceptionableTask.h:20)
    if (exc_count) {
        std::sort(excs.begin(), excs.end(),
                  [](const ExcRec& a, const ExcRec& b) { return a.row < b.row; });
        std::vector<long long> row_base;
        if (psrc.parts) {
            row_base.resize((size_t)psrc.n_parts + 1);
            row_base[0] = 0;
            for (int64_t pp = 0; pp < psrc.n_parts; ++pp)
                row_base[(size_t)pp + 1] =
                    row_base[(size_t)pp] + psrc.parts[pp].num_rows;
        }
        size_t total = 0;
        std::vector<std::vector<uint8_t>> fetched;
        std::vector<std::pair<const uint8_t*, long long>> payloads(excs.size());
        for (size_t i = 0; i < excs.size(); ++i) {
            if (psrc.bytes) {
                payloads[i] = {psrc.bytes + excs[i].off_start,
                               excs[i].off_end - excs[i].off_start};
            } else if (psrc.parts) {
                long long r = excs[i].row - row0;
                int64_t pp = (int64_t)(std::upper_bound(row_base.begin(),
                                                        row_base.end(), r) -
                                       row_base.begin()) - 1;
                long long lr = r - row_base[(size_t)pp];
                const tpx_partition& PP = psrc.parts[pp];
                payloads[i] = {PP.data + PP.row_offsets[lr],
                               PP.row_offsets[lr + 1] - PP.row_offsets[lr]};
            } else {
                // device-only input: copy the raw line back (rare path)
                long long len = excs[i].off_end - excs[i].off_start;
                fetched.emplace_back((size_t)len);
                HIP_CHECK(hipMemcpy(fetched.back().data(),
                                    (const char*)d_in + excs[i].off_start,
                                    (size_t)len, hipMemcpyDeviceToHost));
                payloads[i] = {fetched.back().data(), len};
            }
            total += 32 + (size_t)payloads[i].second;
        }
        res->exc_data = (uint8_t*)malloc(total);
        uint8_t* w = res->exc_data;
        for (size_t i = 0; i < excs.size(); ++i) {
            int64_t* ib = (int64_t*)w;
            ib[0] = excs[i].row;
            ib[1] = excs[i].ec;
            ib[2] = excs[i].opid;
            ib[3] = payloads[i].second;
            memcpy(w + 32, payloads[i].first, (size_t)payloads[i].second);
            w += 32 + payloads[i].second;
        }
        res->exc_size = (int64_t)total;
        res->exc_num_rows = (int64_t)exc_count;
    }
    if (!D.agg.empty() && D.aggby.empty()) res->bytes_out = 16;
    float ms = 0;
    hipEventElapsedTime(&ms, ev1, ev2); res->t_kernel_ms = ms;
    hipEventElapsedTime(&ms, ev2, ev3); res->t_d2h_ms = ms;
    // main: shared start -> latest chunk-main end (spans both side streams)
    float tmain = 0;
    for (int c = 0; c < C; ++c) {
        hipEventElapsedTime(&ms, evm0, cev[(size_t)c * 5]);
        tmain = std::max(tmain, ms);
    }
    res->t_main_ms = tmain;
    if (sink_done) {
        float tc = 0, tw = 0;
        for (int c = 0; c < C; ++c) {
            hipEventElapsedTime(&ms, cev[(size_t)c * 5 + 1],
                                cev[(size_t)c * 5 + 2]);
            tc += ms;
            hipEventElapsedTime(&ms, cev[(size_t)c * 5 + 3],
                                cev[(size_t)c * 5 + 4]);
            tw += ms;
        }
        res->t_compact_ms = tc;
        res->t_write_ms = tw;
    } else {
        hipEventElapsedTime(&ms, cev[0], evs1); res->t_compact_ms = ms;
        hipEventElapsedTime(&ms, evs1, ev2); res->t_write_ms = ms;
    }
    return 0;
}

static void empty_result(tpx_result* res, bool mem_sink) {
    if (mem_sink) {
        res->out_data = (uint8_t*)malloc(8);
        memset(res->out_data, 0, 8);
        res->out_size = 8;
    } else {
        res->out_data = (uint8_t*)malloc(1);
        res->out_size = 0;
    }
    res->out_row_offsets = (int64_t*)malloc(8);
    res->out_row_offsets[0] = mem_sink ? 8 : 0;
    res->out_row_indices = (int64_t*)malloc(8);
}

// ---------------------------------------------------------------------------------
// columnar source (ORC/Arrow ingest — io OrcReader analog): slots[] holds 3
// DEVICE pointers per input column ([values-or-offsets, string-data,
// null-mask]; unused/pushed-down columns null). No row bytes exist, so
// exception payloads are empty and the host replays from the original table.

extern "C" int64_t tpx_stage_execute_col(tpx_stage* st, void* const* slots,
                                         int64_t n_slots, int64_t n_rows,
                                         int64_t in_bytes, int64_t first_row,
                                         int64_t flags, tpx_result* res) {
    memset(res, 0, sizeof(*res));
    if (!st->loaded) { set_err("stage not loaded on a GPU"); return -1; }
    hipStream_t stream = nullptr;
    int dev = cur_device();
    g_arena[dev].begin();
    res->bytes_in = in_bytes;
    if (n_rows == 0) { empty_result(res, st->desc.sink == "mem"); return 0; }
    ARENA_TAKE(d_tab, (size_t)n_slots * sizeof(void*) + 8);
    HIP_CHECK(hipMemcpyAsync(d_tab, slots, (size_t)n_slots * sizeof(void*),
                             hipMemcpyHostToDevice, stream));
    PayloadSrc psrc{nullptr, nullptr, 0};
    std::vector<void*> ht(slots, slots + n_slots);
    return run_core(st, d_tab, nullptr, n_rows, first_row, in_bytes, res, psrc,
                    stream, flags, nullptr, &ht);
}

// ---------------------------------------------------------------------------------
// mem source (TransformTask.cc:682 processMemorySource analog)

extern "C" int64_t tpx_stage_execute(tpx_stage* st, const tpx_partition* parts,
                                     int64_t n_parts, tpx_result* res) {
    memset(res, 0, sizeof(*res));
    if (!st->loaded) { set_err("stage not loaded on a GPU"); return -1; }
    hipStream_t stream = nullptr;
    int dev = cur_device();
    g_arena[dev].begin();

    long long n = 0, in_bytes = 0;
    for (int64_t p = 0; p < n_parts; ++p) {
        n += parts[p].num_rows;
        in_bytes += parts[p].size;
    }
    res->bytes_in = in_bytes;
    if (n == 0) { empty_result(res, st->desc.sink == "mem"); return 0; }

    hipEvent_t ev0, ev1;
    hipEventCreate(&ev0); hipEventCreate(&ev1);
    hipEventRecord(ev0, stream);
    ARENA_TAKE(d_in, (size_t)in_bytes + 80);  // scan-window/group overread slack
    ARENA_TAKE(d_offs, (size_t)(n + 1) * 8);
    {
        std::vector<long long> offs((size_t)n + 1);
        long long byte_base = 0, row_base = 0;
        for (int64_t p = 0; p < n_parts; ++p) {
            const tpx_partition& P = parts[p];
            HIP_CHECK(hipMemcpyAsync((char*)d_in + byte_base, P.data,
                                     (size_t)P.size, hipMemcpyHostToDevice, stream));
            for (long long r = 0; r < P.num_rows; ++r)
                offs[(size_t)(row_base + r)] = byte_base + P.row_offsets[r];
            byte_base += P.size;
            row_base += P.num_rows;
        }
        offs[(size_t)n] = byte_base;
        HIP_CHECK(hipMemcpyAsync(d_offs, offs.data(), ((size_t)n + 1) * 8,
                                 hipMemcpyHostToDevice, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
    }
    hipEventRecord(ev1, stream);
    HIP_CHECK(hipEventSynchronize(ev1));
    float ms = 0;
    hipEventElapsedTime(&ms, ev0, ev1);
    res->t_h2d_ms = ms;
    (void)hipEventDestroy(ev0); (void)hipEventDestroy(ev1);

    PayloadSrc psrc{nullptr, parts, n_parts};
    return run_core(st, d_in, d_offs, n, 0, in_bytes, res, psrc, stream);
}

// ---------------------------------------------------------------------------------
// csv source (TransformTask.cc:724 processFileSource + CSVReader.cc:390 analog;
// row boundaries found on-device by the quote-parity scan kernels)

static int64_t csv_boundary_and_core(tpx_stage* st, void* d_in_p, long long size,
                                     const uint8_t* host_bytes,
                                     int64_t first_global_row, int64_t flags,
                                     tpx_result* res) {
    hipStream_t stream = nullptr;
    int dev = cur_device();
    hipEvent_t eb0, eb1;
    hipEventCreate(&eb0); hipEventCreate(&eb1);
    hipEventRecord(eb0, stream);
    long long nchunks = (size + 4095) / 4096;  // TPX_CSV_CHUNK
    ARENA_TAKE(d_q, (size_t)nchunks * 8);
    ARENA_TAKE(d_c0, (size_t)nchunks * 8);
    ARENA_TAKE(d_c1, (size_t)nchunks * 8);
    ARENA_TAKE(d_qs, (size_t)nchunks * 8);
    ARENA_TAKE(d_rc, (size_t)nchunks * 8);
    ARENA_TAKE(d_base, (size_t)nchunks * 8);
    // wave-cooperative kernels: one wave per chunk, 4 waves per 256-thread block
    unsigned grid_w = (unsigned)std::min<long long>((nchunks + 3) / 4, 8192);
    unsigned grid_t = (unsigned)std::min<long long>((nchunks + 255) / 256, 8192);
    int quotes_on = st->desc.textmode ? 0 : 1;
    {
        void* args[] = {&d_in_p, (void*)&size, &nchunks, &d_q, &d_c0, &d_c1,
                        &quotes_on};
        if (launch(st->k_csv_chunk, grid_w, 256, stream, args)) return -1;
    }
    long long qtotal = 0;
    if (dev_scan(st, stream, (long long*)d_q, (long long*)d_qs, nchunks, &qtotal))
        return -1;
    {
        void* args[] = {&d_qs, &d_c0, &d_c1, &d_rc, &nchunks};
        if (launch(st->k_csv_sel, grid_t, 256, stream, args)) return -1;
    }
    long long nrows = 0;
    if (dev_scan(st, stream, (long long*)d_rc, (long long*)d_base, nchunks,
                 &nrows))
        return -1;
    if (nrows == 0) { empty_result(res, st->desc.sink == "mem"); return 0; }
    ARENA_TAKE(d_offs, ((size_t)nrows + 1) * 8);
    hipEventRecord(eb1, stream);
    HIP_CHECK(hipEventSynchronize(eb1));
    float ms = 0;
    hipEventElapsedTime(&ms, eb0, eb1);
    res->t_boundary_ms = ms;  // stats+scans+select; per-range emits overlap main
    (void)hipEventDestroy(eb0); (void)hipEventDestroy(eb1);

    // row offsets are emitted per main-kernel chunk, on that chunk's stream,
    // so emit(range c) overlaps main(chunk c-1) (ranged tpx_csv_emit_rows)
    PreChunkFn emit_range = [&](long long lo, long long hi,
                                hipStream_t sc) -> int {
        void* args[] = {&d_in_p, (void*)&size, &nchunks, &d_qs, &d_base,
                        &d_offs, &quotes_on, &lo, &hi};
        return launch(st->k_csv_rows, grid_w, 256, sc, args);
    };

    PayloadSrc psrc{host_bytes, nullptr, 0};
    return run_core(st, d_in_p, d_offs, nrows, first_global_row, size, res, psrc,
                    stream, flags, &emit_range);
}

extern "C" int64_t tpx_stage_execute_csv(tpx_stage* st, const uint8_t* csv_bytes,
                                         int64_t size, int64_t first_global_row,
                                         tpx_result* res) {
    memset(res, 0, sizeof(*res));
    if (!st->loaded) { set_err("stage not loaded on a GPU"); return -1; }
    if (!st->k_csv_chunk || !st->k_csv_sel || !st->k_csv_rows) {
        set_err("stage compiled without csv kernels");
        return -1;
    }
    hipStream_t stream = nullptr;
    int dev = cur_device();
    g_arena[dev].begin();
    res->bytes_in = size;
    if (size == 0) { empty_result(res, st->desc.sink == "mem"); return 0; }

    hipEvent_t ev0, ev1;
    hipEventCreate(&ev0); hipEventCreate(&ev1);
    hipEventRecord(ev0, stream);
    ARENA_TAKE(d_in, (size_t)size + 80);  // mask-walk 64-B group overread slack
    HIP_CHECK(hipMemcpyAsync(d_in, csv_bytes, (size_t)size,
                             hipMemcpyHostToDevice, stream));
    hipEventRecord(ev1, stream);
    HIP_CHECK(hipEventSynchronize(ev1));
    float ms = 0;
    hipEventElapsedTime(&ms, ev0, ev1);
    res->t_h2d_ms = ms;
    (void)hipEventDestroy(ev0); (void)hipEventDestroy(ev1);

    return csv_boundary_and_core(st, d_in, size, csv_bytes, first_global_row, 0,
                                 res);
}

// ---------------------------------------------------------------------------------
// device-resident input helpers (bench / cache())

extern "C" uint64_t tpx_dev_alloc(int64_t size) {
    void* p = nullptr;
    if (hipMalloc(&p, (size_t)size + 128) != hipSuccess) return 0;  // memcpy + mask-walk group overread pad
    return (uint64_t)(uintptr_t)p;
}

extern "C" int64_t tpx_dev_upload(uint64_t dst, const void* src, int64_t size) {
    HIP_CHECK(hipMemcpy((void*)(uintptr_t)dst, src, (size_t)size,
                        hipMemcpyHostToDevice));
    return 0;
}

extern "C" void tpx_dev_free(uint64_t ptr) {
    if (ptr) (void)hipFree((void*)(uintptr_t)ptr);
}

// pinned host staging for streamed ingestion (file -> pinned ring -> DMA):
// pageable hipMemcpy runs ~6 GB/s, pinned ~20+ GB/s, and reads can overlap
extern "C" uint64_t tpx_pinned_alloc(int64_t size) {
    void* p = nullptr;
    if (hipHostMalloc(&p, (size_t)size) != hipSuccess) return 0;
    return (uint64_t)(uintptr_t)p;
}

extern "C" void tpx_pinned_free(uint64_t ptr) {
    if (ptr) (void)hipHostFree((void*)(uintptr_t)ptr);
}

extern "C" int64_t tpx_stage_execute_csv_dev(tpx_stage* st, uint64_t dev_bytes,
                                             int64_t size,
                                             int64_t first_global_row,
                                             int64_t flags, tpx_result* res) {
    memset(res, 0, sizeof(*res));
    if (!st->loaded) { set_err("stage not loaded on a GPU"); return -1; }
    res->bytes_in = size;
    int dev = cur_device();
    g_arena[dev].begin();
    if (size == 0) { empty_result(res, st->desc.sink == "mem"); return 0; }
    return csv_boundary_and_core(st, (void*)(uintptr_t)dev_bytes, size, nullptr,
                                 first_global_row, flags, res);
}

extern "C" void tpx_result_free(tpx_result* res) {
    if (!res) return;
    free(res->out_data);
    free(res->out_row_offsets);
    free(res->out_row_indices);
    free(res->exc_data);
    memset(res, 0, sizeof(*res));
}
