// tpx_rt.hip.h — device runtime for generated TransformStage kernels (gfx950).
//
// This is the MI355X replacement for the reference's C runtime the JIT'd code links
// against (tuplex/runtime/src/Runtime.cc: rtmalloc arena :186, fast_atoi64/atod
// wrappers :319-383, string functions StringFunctions.cc) plus the per-stage LLVM
// helpers. Parse semantics restate tuplex/utils/src/StringUtils.cc:22 fast_atoi64 /
// :71 fast_atod verbatim (including quirks: "-" -> 0, digit-by-digit f64
// accumulation, exponent clamp 308) so GPU results match the reference's compiled
// path bit-for-bit.
//
// Strings are byte views (tstr). Char-index-sensitive ops (len/find/rfind/getitem/
// slice/case) verify ASCII-ness while scanning and divert non-ASCII rows with
// EC_NCV (NORMALCASEVIOLATION, ExceptionCodes.h:107) -> the host interpreter replays
// them with full CPython unicode semantics — the reference's own dual-mode design.
//
// Prepended (with this header) to every generated stage source and compiled by
// hipRTC; see tuplex_amd/codegen.py and csrc/tpx_abi.cpp.
#pragma once
#if !defined(__HIPCC_RTC__) && !defined(__HIPRTC__)
#include <hip/hip_runtime.h>  // hipRTC injects the builtins itself
#endif

#define TPX_WAVE 64

// ExceptionCodes.h:26 values (subset used on device)
#define EC_OK 0
#define EC_NCV 7            // NORMALCASEVIOLATION -> silent host replay
#define EC_CSV_UNDERRUN 20
#define EC_CSV_OVERRUN 21
#define EC_NULLERROR 50
#define EC_I64PARSE 52
#define EC_F64PARSE 53
#define EC_DOUBLEQUOTE 55
#define EC_BADPARSE 70      // BADPARSE_STRING_INPUT
#define EC_INDEXERROR 111
#define EC_MEMORYERROR 114
#define EC_OVERFLOW 118
#define EC_TYPEERROR 129
#define EC_VALUEERROR 135
#define EC_ZERODIV 136

struct tstr {
    const char* p;
    long long n;
};

// String heap: per-thread chunked bump allocation (one global atomic per 256-B
// chunk per thread instead of one per allocation — a single global cursor word
// saturates at ~88 atomics/µs, MI355X_MICROARCH.md §dequeue). The rtmalloc-arena
// analog (Runtime.cc:186), device-side.
#define TPX_HEAP_CHUNK 256

struct TpxHeap {
    char* base;
    unsigned long long* cursor;  // device-global bump cursor
    unsigned long long cap;
    char* tl_cur;                // per-thread chunk (registers)
    char* tl_end;
};

__device__ __forceinline__ char* tpx_alloc(TpxHeap& h, long long n) {
    if (n <= 0) return h.base;  // zero-size alloc: any valid pointer
    if (n > TPX_HEAP_CHUNK) {
        unsigned long long off = atomicAdd(h.cursor, (unsigned long long)n);
        if (off + (unsigned long long)n > h.cap) return nullptr;
        return h.base + off;
    }
    if (h.tl_cur + n > h.tl_end) {
        unsigned long long off =
            atomicAdd(h.cursor, (unsigned long long)TPX_HEAP_CHUNK);
        if (off + TPX_HEAP_CHUNK > h.cap) return nullptr;
        h.tl_cur = h.base + off;
        h.tl_end = h.tl_cur + TPX_HEAP_CHUNK;
    }
    char* p = h.tl_cur;
    h.tl_cur += n;
    return p;
}

__device__ __forceinline__ bool tpx_is_ascii_byte(unsigned char c) { return c < 0x80; }

// ---- SWAR 8-byte scanning (the scanning loops dominate this HBM/LDS-bound
// path; byte-at-a-time dependent loads are issue/latency-bound) -----------------

#define TPX_SWAR_ONE 0x0101010101010101ULL
#define TPX_SWAR_HIGH 0x8080808080808080ULL

// zero-byte detector: nonzero bits at 0x80 positions of bytes equal to zero
__device__ __forceinline__ unsigned long long tpx_swar_zero(unsigned long long x) {
    // EXACT per-byte zero detect: (x|HIGH) - ONE never borrows across bytes,
    // so every byte's high bit is independent. The classic
    // (x-ONE) & ~x & HIGH is only first-match exact: a borrow from a true
    // zero byte falsely flags a following 0x01 byte (ex: scanning for ','
    // flags the '-' of ",-471.04" — cell walk and quote-parity masks use
    // these bits POSITIONALLY, so they must be exact).
    return ~(x | ((x | TPX_SWAR_HIGH) - TPX_SWAR_ONE)) & TPX_SWAR_HIGH;
}

// first index of byte c in p[0..n), or -1. Head byte loop to 8-alignment,
// then 32-B iterations with FOUR INDEPENDENT 8-B loads (the scans run over
// LDS/L1 where a dependent 8-B chain pays ~50 cycles per step; 4 loads in
// flight cut the chain depth 4x), then an 8-B + byte tail. Semantics exactly
// match the byte loop.
__device__ __forceinline__ long long tpx_memchr(const char* p, long long n,
                                                char c) {
    unsigned long long pat = TPX_SWAR_ONE * (unsigned char)c;
    long long i = 0;
    while (i < n && (((unsigned long long)(p + i)) & 7)) {
        if (p[i] == c) return i;
        ++i;
    }
    for (; i + 32 <= n; i += 32) {
        unsigned long long v0 = *(const unsigned long long*)(p + i);
        unsigned long long v1 = *(const unsigned long long*)(p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(p + i + 24);
        unsigned long long h0 = tpx_swar_zero(v0 ^ pat);
        unsigned long long h1 = tpx_swar_zero(v1 ^ pat);
        unsigned long long h2 = tpx_swar_zero(v2 ^ pat);
        unsigned long long h3 = tpx_swar_zero(v3 ^ pat);
        if (h0 | h1 | h2 | h3) {
            if (h0) return i + (__ffsll((long long)h0) - 1) / 8;
            if (h1) return i + 8 + (__ffsll((long long)h1) - 1) / 8;
            if (h2) return i + 16 + (__ffsll((long long)h2) - 1) / 8;
            return i + 24 + (__ffsll((long long)h3) - 1) / 8;
        }
    }
    for (; i + 8 <= n; i += 8) {
        unsigned long long v = *(const unsigned long long*)(p + i);
        unsigned long long hit = tpx_swar_zero(v ^ pat);
        if (hit) return i + (__ffsll((long long)hit) - 1) / 8;
    }
    for (; i < n; ++i)
        if (p[i] == c) return i;
    return -1;
}

// tpx_memchr that also ORs the high (non-ASCII) bits of every byte it scans
// into *hi — lets the CSV cell walk double as the per-row ASCII gate (the walk
// already touches every row byte; a separate tpx_ascii pass re-reads the row)
__device__ __forceinline__ long long tpx_memchr_hi(const char* p, long long n,
                                                   char c,
                                                   unsigned long long* hi) {
    unsigned long long pat = TPX_SWAR_ONE * (unsigned char)c;
    long long i = 0;
    while (i < n && (((unsigned long long)(p + i)) & 7)) {
        *hi |= (unsigned char)p[i] & 0x80u;
        if (p[i] == c) return i;
        ++i;
    }
    for (; i + 32 <= n; i += 32) {
        unsigned long long v0 = *(const unsigned long long*)(p + i);
        unsigned long long v1 = *(const unsigned long long*)(p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(p + i + 24);
        unsigned long long h0 = tpx_swar_zero(v0 ^ pat);
        unsigned long long h1 = tpx_swar_zero(v1 ^ pat);
        unsigned long long h2 = tpx_swar_zero(v2 ^ pat);
        unsigned long long h3 = tpx_swar_zero(v3 ^ pat);
        if (h0 | h1 | h2 | h3) {
            // hi must cover exactly the bytes scanned up to the hit
            if (h0) { *hi |= v0 & TPX_SWAR_HIGH;
                      return i + (__ffsll((long long)h0) - 1) / 8; }
            *hi |= v0 & TPX_SWAR_HIGH;
            if (h1) { *hi |= v1 & TPX_SWAR_HIGH;
                      return i + 8 + (__ffsll((long long)h1) - 1) / 8; }
            *hi |= v1 & TPX_SWAR_HIGH;
            if (h2) { *hi |= v2 & TPX_SWAR_HIGH;
                      return i + 16 + (__ffsll((long long)h2) - 1) / 8; }
            *hi |= (v2 | v3) & TPX_SWAR_HIGH;
            return i + 24 + (__ffsll((long long)h3) - 1) / 8;
        }
        *hi |= (v0 | v1 | v2 | v3) & TPX_SWAR_HIGH;
    }
    for (; i + 8 <= n; i += 8) {
        unsigned long long v = *(const unsigned long long*)(p + i);
        *hi |= v & TPX_SWAR_HIGH;
        unsigned long long hit = tpx_swar_zero(v ^ pat);
        if (hit) return i + (__ffsll((long long)hit) - 1) / 8;
    }
    for (; i < n; ++i) {
        *hi |= (unsigned char)p[i] & 0x80u;
        if (p[i] == c) return i;
    }
    return -1;
}

// first index where p[i]==c1 or p[i]==c2, or -1
__device__ __forceinline__ long long tpx_memchr2(const char* p, long long n,
                                                 char c1, char c2) {
    unsigned long long pat1 = TPX_SWAR_ONE * (unsigned char)c1;
    unsigned long long pat2 = TPX_SWAR_ONE * (unsigned char)c2;
    long long i = 0;
    while (i < n && (((unsigned long long)(p + i)) & 7)) {
        if (p[i] == c1 || p[i] == c2) return i;
        ++i;
    }
    for (; i + 8 <= n; i += 8) {
        unsigned long long v = *(const unsigned long long*)(p + i);
        unsigned long long hit = tpx_swar_zero(v ^ pat1) | tpx_swar_zero(v ^ pat2);
        if (hit) return i + (__ffsll((long long)hit) - 1) / 8;
    }
    for (; i < n; ++i)
        if (p[i] == c1 || p[i] == c2) return i;
    return -1;
}

// last index of byte c in p[0..n), or -1 (backward SWAR; 32-B iterations with
// four independent loads, same rationale as tpx_memchr)
__device__ __forceinline__ long long tpx_memrchr(const char* p, long long n,
                                                 char c) {
    unsigned long long pat = TPX_SWAR_ONE * (unsigned char)c;
    long long i = n;
    while (i > 0 && (((unsigned long long)(p + i)) & 7)) {
        --i;
        if (p[i] == c) return i;
    }
    while (i >= 32) {
        i -= 32;
        unsigned long long v0 = *(const unsigned long long*)(p + i);
        unsigned long long v1 = *(const unsigned long long*)(p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(p + i + 24);
        unsigned long long h0 = tpx_swar_zero(v0 ^ pat);
        unsigned long long h1 = tpx_swar_zero(v1 ^ pat);
        unsigned long long h2 = tpx_swar_zero(v2 ^ pat);
        unsigned long long h3 = tpx_swar_zero(v3 ^ pat);
        if (h0 | h1 | h2 | h3) {
            if (h3) return i + 24 + (63 - __builtin_clzll(h3)) / 8;
            if (h2) return i + 16 + (63 - __builtin_clzll(h2)) / 8;
            if (h1) return i + 8 + (63 - __builtin_clzll(h1)) / 8;
            return i + (63 - __builtin_clzll(h0)) / 8;
        }
    }
    while (i >= 8) {
        i -= 8;
        unsigned long long v = *(const unsigned long long*)(p + i);
        unsigned long long hit = tpx_swar_zero(v ^ pat);
        if (hit) return i + (63 - __builtin_clzll(hit)) / 8;
    }
    while (i > 0) {
        --i;
        if (p[i] == c) return i;
    }
    return -1;
}

// scan for non-ASCII; true if all ASCII (SWAR high-bit sweep)
__device__ __forceinline__ bool tpx_ascii(const tstr s) {
    const char* p = s.p;
    long long n = s.n;
    long long i = 0;
    while (i < n && (((unsigned long long)(p + i)) & 7)) {
        if ((unsigned char)p[i] >= 0x80) return false;
        ++i;
    }
    for (; i + 32 <= n; i += 32) {
        unsigned long long v0 = *(const unsigned long long*)(p + i);
        unsigned long long v1 = *(const unsigned long long*)(p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(p + i + 24);
        if ((v0 | v1 | v2 | v3) & TPX_SWAR_HIGH) return false;
    }
    for (; i + 8 <= n; i += 8)
        if (*(const unsigned long long*)(p + i) & TPX_SWAR_HIGH) return false;
    for (; i < n; ++i)
        if ((unsigned char)p[i] >= 0x80) return false;
    return true;
}

// vectorized copy: align dest, funnel-shift unaligned source words. May read up
// to 7 bytes past s+n (every device buffer carries >=16 B tail padding).
__device__ __forceinline__ void tpx_memcpy(char* d, const char* s, long long n) {
    if (n < 16) {
        for (long long i = 0; i < n; ++i) d[i] = s[i];
        return;
    }
    long long i = 0;
    while (((unsigned long long)(d + i)) & 7) { d[i] = s[i]; ++i; }
    long long k = ((unsigned long long)(s + i)) & 7;
    const unsigned long long* sw =
        (const unsigned long long*)(s + i - k);
    if (k == 0) {
        for (; i + 8 <= n; i += 8) *(unsigned long long*)(d + i) = *sw++;
    } else {
        unsigned long long prev = *sw++;
        int sh = 8 * (int)k;
        for (; i + 8 <= n; i += 8) {
            unsigned long long cur = *sw++;
            *(unsigned long long*)(d + i) = (prev >> sh) | (cur << (64 - sh));
            prev = cur;
        }
    }
    for (; i < n; ++i) d[i] = s[i];
}

// translate an LDS-staged string view back to its global-memory address before it
// escapes the kernel (columnar string cells must outlive the LDS staging window).
// (p - lds_lo) + span_start is the byte offset in the input. The LDS test uses
// the HARDWARE aperture check (__builtin_amdgcn_is_shared), NOT a pointer
// range compare: relational comparison of pointers with different provenance
// is unspecified and the optimizer folds it once the row pointer is a
// staged-or-global generic select (observed: untranslated LDS-aperture
// pointers escaping to the write kernel -> APERTURE_VIOLATION).
__device__ __forceinline__ tstr tpx_to_global(tstr s, const char* lds_lo,
                                              const char* lds_hi,
                                              const unsigned char* gbase,
                                              long long span_start) {
#ifdef TPX_HOST_TEST
    bool in_lds = s.p >= lds_lo && s.p < lds_hi;
#else
    bool in_lds = __builtin_amdgcn_is_shared((const void*)s.p);
#endif
    if (in_lds)
        return tstr{(const char*)gbase + span_start +
                        ((unsigned long long)s.p - (unsigned long long)lds_lo),
                    s.n};
    return s;
}

// ---- python-semantics helpers ---------------------------------------------------

__device__ __forceinline__ long long tpx_floordiv_i64(long long a, long long b, int* ec) {
    if (b == 0) { *ec = EC_ZERODIV; return 0; }
    long long q = a / b;
    if ((a % b != 0) && ((a < 0) != (b < 0))) q -= 1;  // python floor semantics
    return q;
}

__device__ __forceinline__ long long tpx_mod_i64(long long a, long long b, int* ec) {
    if (b == 0) { *ec = EC_ZERODIV; return 0; }
    long long r = a % b;
    if (r != 0 && ((r < 0) != (b < 0))) r += b;
    return r;
}

__device__ __forceinline__ double tpx_truediv(double a, double b, int* ec) {
    if (b == 0.0) { *ec = EC_ZERODIV; return 0.0; }
    return a / b;
}

__device__ __forceinline__ double tpx_floordiv_f64(double a, double b, int* ec) {
    if (b == 0.0) { *ec = EC_ZERODIV; return 0.0; }
    return floor(a / b);
}

__device__ __forceinline__ double tpx_mod_f64(double a, double b, int* ec) {
    if (b == 0.0) { *ec = EC_ZERODIV; return 0.0; }
    double r = fmod(a, b);
    if (r != 0.0 && ((r < 0.0) != (b < 0.0))) r += b;
    return r;
}

// ---- string ops -----------------------------------------------------------------

// len: python counts CHARS; byte len == char len only for ASCII (divert otherwise)
__device__ __forceinline__ long long tpx_len(const tstr s, int* ec) {
    return s.n;  // rows are ASCII-gated once at load (EC_NCV divert)
}

// find/rfind return python CHAR index; ASCII-gate the prefix we index over
__device__ __forceinline__ long long tpx_find(const tstr s, const tstr needle, int* ec) {
    if (needle.n == 0) return 0;
    long long i = 0;
    long long last = s.n - needle.n;
    while (i <= last) {
        long long k = tpx_memchr(s.p + i, last - i + 1, needle.p[0]);
        if (k < 0) return -1;
        i += k;
        bool m = true;
        for (long long j = 1; j < needle.n; ++j)
            if (s.p[i + j] != needle.p[j]) { m = false; break; }
        if (m) return i;
        ++i;
    }
    return -1;
}

__device__ __forceinline__ long long tpx_rfind(const tstr s, const tstr needle, int* ec) {
    if (needle.n == 0) return s.n;
    long long hi = s.n - needle.n;  // highest candidate start
    while (hi >= 0) {
        long long k = tpx_memrchr(s.p, hi + 1, needle.p[0]);
        if (k < 0) return -1;
        bool m = true;
        for (long long j = 1; j < needle.n; ++j)
            if (s.p[k + j] != needle.p[j]) { m = false; break; }
        if (m) return k;
        hi = k - 1;
    }
    return -1;
}

__device__ __forceinline__ bool tpx_contains(const tstr s, const tstr needle) {
    if (needle.n == 0) return true;
    long long i = 0;
    long long last = s.n - needle.n;
    while (i <= last) {
        long long k = tpx_memchr(s.p + i, last - i + 1, needle.p[0]);
        if (k < 0) return false;
        i += k;
        bool m = true;
        for (long long j = 1; j < needle.n; ++j)
            if (s.p[i + j] != needle.p[j]) { m = false; break; }
        if (m) return true;
        ++i;
    }
    return false;
}

__device__ __forceinline__ bool tpx_startswith(const tstr s, const tstr p) {
    if (p.n > s.n) return false;
    for (long long j = 0; j < p.n; ++j)
        if (s.p[j] != p.p[j]) return false;
    return true;
}

__device__ __forceinline__ bool tpx_endswith(const tstr s, const tstr p) {
    if (p.n > s.n) return false;
    for (long long j = 0; j < p.n; ++j)
        if (s.p[s.n - p.n + j] != p.p[j]) return false;
    return true;
}

__device__ __forceinline__ bool tpx_streq(const tstr a, const tstr b) {
    if (a.n != b.n) return false;
    for (long long i = 0; i < a.n; ++i)
        if (a.p[i] != b.p[i]) return false;
    return true;
}

// UTF-8 byte-lexicographic order == code-point order, so byte compare is exact
__device__ __forceinline__ int tpx_strcmp(const tstr a, const tstr b) {
    long long n = a.n < b.n ? a.n : b.n;
    for (long long i = 0; i < n; ++i) {
        unsigned char x = a.p[i], y = b.p[i];
        if (x != y) return x < y ? -1 : 1;
    }
    return a.n == b.n ? 0 : (a.n < b.n ? -1 : 1);
}

// x[i]: python char access with negative wrap; view, no alloc
__device__ __forceinline__ tstr tpx_getitem(const tstr s, long long i, int* ec) {
    long long idx = i < 0 ? i + s.n : i;
    if (idx < 0 || idx >= s.n) { *ec = EC_INDEXERROR; return tstr{s.p, 0}; }
    return tstr{s.p + idx, 1};
}

// s[lo:hi]: python clamp semantics; view
__device__ __forceinline__ tstr tpx_slice(const tstr s, long long lo, bool has_lo,
                                          long long hi, bool has_hi, int* ec) {
    long long a = has_lo ? lo : 0;
    long long b = has_hi ? hi : s.n;
    if (a < 0) a += s.n;
    if (b < 0) b += s.n;
    a = a < 0 ? 0 : (a > s.n ? s.n : a);
    b = b < 0 ? 0 : (b > s.n ? s.n : b);
    if (b < a) b = a;
    return tstr{s.p + a, b - a};
}

// s.center(width, fill) — CPython's left-bias quirk
// (left = marg/2 + (marg & width & 1)); fill must be exactly one char
// (TypeError otherwise, like CPython / strCenter StringFunctions.cc:224)
__device__ __forceinline__ tstr tpx_center(TpxHeap& h, const tstr s,
                                           long long width, const tstr fill,
                                           int* ec) {
    if (fill.n != 1) { *ec = EC_TYPEERROR; return tstr{s.p, 0}; }
    if (s.n >= width) return s;
    long long marg = width - s.n;
    long long left = marg / 2 + (marg & width & 1);
    char* d = tpx_alloc(h, width);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    char fc = fill.p[0];
    for (long long i = 0; i < left; ++i) d[i] = fc;
    for (long long i = 0; i < s.n; ++i) d[left + i] = s.p[i];
    for (long long i = left + s.n; i < width; ++i) d[i] = fc;
    return tstr{d, width};
}

// str * int repetition (python sequence semantics: n <= 0 -> "")
__device__ __forceinline__ tstr tpx_strmul(TpxHeap& h, const tstr s,
                                           long long n, int* ec) {
    if (n <= 0 || s.n == 0) return tstr{s.p, 0};
    long long total = s.n * n;
    char* d = tpx_alloc(h, total);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    for (long long r = 0; r < n; ++r)
        for (long long i = 0; i < s.n; ++i)
            d[r * s.n + i] = s.p[i];
    return tstr{d, total};
}

__device__ __forceinline__ tstr tpx_lower(TpxHeap& h, const tstr s, int* ec) {
    char* d = tpx_alloc(h, s.n);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    for (long long i = 0; i < s.n; ++i) {
        unsigned char c = s.p[i];
        d[i] = (c >= 'A' && c <= 'Z') ? c + 32 : c;
    }
    return tstr{d, s.n};
}

__device__ __forceinline__ tstr tpx_upper(TpxHeap& h, const tstr s, int* ec) {
    char* d = tpx_alloc(h, s.n);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    for (long long i = 0; i < s.n; ++i) {
        unsigned char c = s.p[i];
        d[i] = (c >= 'a' && c <= 'z') ? c - 32 : c;
    }
    return tstr{d, s.n};
}

// x[0].upper() + x[1:].lower() fused (codegen peephole): one alloc, one pass
// instead of getitem+upper+slice+lower+concat (3 allocs, 3 passes). Empty x
// raises EC_INDEXERROR exactly like the unfused x[0] (tpx_getitem above).
__device__ __forceinline__ tstr tpx_capitalize_ix(TpxHeap& h, const tstr s, int* ec) {
    if (s.n <= 0) { *ec = EC_INDEXERROR; return tstr{s.p, 0}; }
    char* d = tpx_alloc(h, s.n);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    unsigned char c0 = s.p[0];
    d[0] = (c0 >= 'a' && c0 <= 'z') ? c0 - 32 : c0;
    for (long long i = 1; i < s.n; ++i) {
        unsigned char c = s.p[i];
        d[i] = (c >= 'A' && c <= 'Z') ? c + 32 : c;
    }
    return tstr{d, s.n};
}

// int(s.replace(ch, '')) fused (codegen peephole): for short cells the dropped
// copy goes through a stack buffer (no heap alloc, no extra pass); the >31B
// tail falls back to replace+parse. Exact by construction: the same bytes that
// replace() would produce feed the same tpx_int_str.
__device__ __forceinline__ tstr tpx_replace(TpxHeap& h, const tstr s, const tstr a,
                                            const tstr b, int* ec);
__device__ __forceinline__ long long tpx_int_str(const tstr s, int* ec);
__device__ __forceinline__ long long tpx_int_drop(TpxHeap& h, const tstr s,
                                                  char drop, const tstr drop_s,
                                                  int* ec) {
    if (s.n <= 31) {
        char buf[32];
        long long m = 0;
        for (long long i = 0; i < s.n; ++i) {
            char c = s.p[i];
            if (c != drop) buf[m++] = c;
        }
        return tpx_int_str(tstr{buf, m}, ec);
    }
    tstr r = tpx_replace(h, s, drop_s, tstr{s.p, 0}, ec);
    if (*ec) return 0;
    return tpx_int_str(r, ec);
}

__device__ __forceinline__ tstr tpx_swapcase(TpxHeap& h, const tstr s, int* ec) {
    char* d = tpx_alloc(h, s.n);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    for (long long i = 0; i < s.n; ++i) {
        unsigned char c = s.p[i];
        d[i] = (c >= 'a' && c <= 'z') ? c - 32 : ((c >= 'A' && c <= 'Z') ? c + 32 : c);
    }
    return tstr{d, s.n};
}

__device__ __forceinline__ bool tpx_is_pyws(unsigned char c) {
    return c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\x0b' || c == '\x0c';
}

// str.strip(): python strips unicode whitespace; edge bytes >=0x80 could be
// multi-byte whitespace -> divert; interior bytes don't matter
__device__ __forceinline__ tstr tpx_strip(const tstr s, int* ec) {
    long long a = 0, b = s.n;
    while (a < b && tpx_is_pyws(s.p[a])) ++a;
    while (b > a && tpx_is_pyws(s.p[b - 1])) --b;
    return tstr{s.p + a, b - a};
}

__device__ __forceinline__ tstr tpx_concat(TpxHeap& h, const tstr a, const tstr b, int* ec) {
    char* d = tpx_alloc(h, a.n + b.n);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{a.p, 0}; }
    tpx_memcpy(d, a.p, a.n);
    tpx_memcpy(d + a.n, b.p, b.n);
    return tstr{d, a.n + b.n};
}

__device__ __forceinline__ tstr tpx_replace(TpxHeap& h, const tstr s, const tstr a,
                                            const tstr b, int* ec) {
    if (a.n == 0) {
        // python: '' needle inserts b between every CHAR (rows ASCII-gated)
        long long outn = s.n + (s.n + 1) * b.n;
        char* d = tpx_alloc(h, outn);
        if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
        long long w = 0;
        for (long long i = 0; i <= s.n; ++i) {
            for (long long j = 0; j < b.n; ++j) d[w++] = b.p[j];
            if (i < s.n) d[w++] = s.p[i];
        }
        return tstr{d, outn};
    }
    // count matches (byte-exact for UTF-8)
    long long cnt = 0;
    for (long long i = 0; i + a.n <= s.n;) {
        bool m = true;
        for (long long j = 0; j < a.n; ++j)
            if (s.p[i + j] != a.p[j]) { m = false; break; }
        if (m) { ++cnt; i += a.n; } else { ++i; }
    }
    if (cnt == 0) return s;
    long long outn = s.n + cnt * (b.n - a.n);
    char* d = tpx_alloc(h, outn);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{s.p, 0}; }
    long long w = 0;
    for (long long i = 0; i < s.n;) {
        bool m = (i + a.n <= s.n);
        if (m) for (long long j = 0; j < a.n; ++j)
            if (s.p[i + j] != a.p[j]) { m = false; break; }
        if (m) {
            for (long long j = 0; j < b.n; ++j) d[w++] = b.p[j];
            i += a.n;
        } else {
            d[w++] = s.p[i++];
        }
    }
    return tstr{d, outn};
}

// s.split(sep)[idx] fused — python list-index semantics (negative wraps over the
// PART count; IndexError when out of range; ValueError on empty separator)
__device__ __forceinline__ tstr tpx_splitget(const tstr s, const tstr sep,
                                             long long idx, int* ec) {
    if (sep.n == 0) { *ec = EC_VALUEERROR; return tstr{s.p, 0}; }
    if (idx < 0) {
        long long parts = 1;
        for (long long i = 0; i + sep.n <= s.n;) {
            bool m = true;
            for (long long j = 0; j < sep.n; ++j)
                if (s.p[i + j] != sep.p[j]) { m = false; break; }
            if (m) { ++parts; i += sep.n; } else { ++i; }
        }
        idx += parts;
        if (idx < 0) { *ec = EC_INDEXERROR; return tstr{s.p, 0}; }
    }
    long long k = 0, start = 0;
    for (long long i = 0; i + sep.n <= s.n;) {
        bool m = true;
        for (long long j = 0; j < sep.n; ++j)
            if (s.p[i + j] != sep.p[j]) { m = false; break; }
        if (m) {
            if (k == idx) return tstr{s.p + start, i - start};
            ++k;
            i += sep.n;
            start = i;
        } else {
            ++i;
        }
    }
    if (k == idx) return tstr{s.p + start, s.n - start};
    *ec = EC_INDEXERROR;
    return tstr{s.p, 0};
}

// ---- parse (StringUtils.cc:22 fast_atoi64 / :71 fast_atod, restated) ------------

__device__ __forceinline__ int tpx_fast_atoi64(const char* start, const char* end,
                                               long long* out) {
    if (start == end) return EC_NULLERROR;
    long long x = 0;
    const char* p = start;
    bool neg = false;
    if (*p == '-') { neg = true; ++p; }
    while (p < end && *p >= '0' && *p <= '9') {
        x = x * 10 + (*p - '0');  // i64 wrap like the compiled path
        ++p;
    }
    if (p != end) return EC_I64PARSE;
    *out = neg ? -x : x;
    return EC_OK;
}

__device__ __forceinline__ int tpx_fast_atod(const char* start, const char* end,
                                             double* out) {
    if (start == end) return EC_NULLERROR;
    const char* p = start;
    double sign = 1.0;
    if (p < end && *p == '-') { sign = -1.0; ++p; }
    else if (p < end && *p == '+') ++p;
    double value = 0.0;
    while (p < end && *p >= '0' && *p <= '9') { value = 10.0 * value + (*p - '0'); ++p; }
    if (p < end && *p == '.') {
        double pow10 = 10.0;
        ++p;
        while (p < end && *p >= '0' && *p <= '9') {
            value += (*p - '0') / pow10;
            pow10 *= 10.0;
            ++p;
        }
    }
    int frac = 0;
    double scale = 1.0;
    if (p < end && (*p == 'e' || *p == 'E')) {
        ++p;
        if (p < end && *p == '-') { frac = 1; ++p; }
        else if (p < end && *p == '+') ++p;
        unsigned int exponent = 0;
        while (p < end && *p >= '0' && *p <= '9') { exponent = exponent * 10 + (*p - '0'); ++p; }
        if (exponent > 308) exponent = 308;
        while (exponent >= 50) { scale *= 1e50; exponent -= 50; }
        while (exponent >= 8)  { scale *= 1e8;  exponent -= 8; }
        while (exponent > 0)   { scale *= 10.0; exponent -= 1; }
    }
    int nanmatch = 0, infmatch = 0;
    if (p == start) {
        const char* t = "nan";
        while (nanmatch < 3 && p < end &&
               (*p == t[nanmatch] || *p == t[nanmatch] - 32)) { ++p; ++nanmatch; }
    }
    if (p == start) {
        const char* t = "infinity";
        while (infmatch < 8 && p < end &&
               (*p == t[infmatch] || *p == t[infmatch] - 32)) { ++p; ++infmatch; }
    }
    if (p != end) return EC_F64PARSE;
    if (nanmatch == 3) { *out = __builtin_nan(""); return EC_OK; }
    if (infmatch == 3 || infmatch == 8) { *out = sign * __builtin_inf(); return EC_OK; }
    *out = sign * (frac ? (value / scale) : (value * scale));
    return EC_OK;
}

// Runtime.cc:319 wrappers: trim python whitespace both ends, then parse;
// parse error -> VALUEERROR
__device__ __forceinline__ long long tpx_int_str(const tstr s, int* ec) {
    const char* a = s.p;
    const char* b = s.p + s.n;
    while (a < b && tpx_is_pyws(*a)) ++a;
    while (b > a && tpx_is_pyws(*(b - 1))) --b;
    long long v = 0;
    int r = tpx_fast_atoi64(a, b, &v);
    if (r != EC_OK) *ec = EC_VALUEERROR;
    return v;
}

__device__ __forceinline__ double tpx_float_str(const tstr s, int* ec) {
    const char* a = s.p;
    const char* b = s.p + s.n;
    while (a < b && tpx_is_pyws(*a)) ++a;
    while (b > a && tpx_is_pyws(*(b - 1))) --b;
    double v = 0.0;
    int r = tpx_fast_atod(a, b, &v);
    if (r != EC_OK) *ec = EC_VALUEERROR;
    return v;
}

// int(float): python truncates toward zero; OverflowError outside i64
__device__ __forceinline__ long long tpx_int_f64(double x, int* ec) {
    if (!(x > -9.223372036854776e18 && x < 9.223372036854776e18)) {
        *ec = x == x ? EC_OVERFLOW : EC_VALUEERROR;  // NaN -> ValueError
        return 0;
    }
    return (long long)x;
}

// ---- int -> str -----------------------------------------------------------------

__device__ __forceinline__ long long tpx_i64_digits(long long v) {
    unsigned long long x = v < 0 ? (unsigned long long)(-(v + 1)) + 1 : (unsigned long long)v;
    long long d = 1;
    while (x >= 10) { x /= 10; ++d; }
    return d + (v < 0 ? 1 : 0);
}

__device__ __forceinline__ void tpx_i64_write(char* dst, long long v, long long len) {
    unsigned long long x = v < 0 ? (unsigned long long)(-(v + 1)) + 1 : (unsigned long long)v;
    long long i = len;
    do { dst[--i] = '0' + (char)(x % 10); x /= 10; } while (x);
    if (v < 0) dst[0] = '-';
}

__device__ __forceinline__ tstr tpx_str_i64(TpxHeap& h, long long v, int* ec) {
    long long len = tpx_i64_digits(v);
    char* d = tpx_alloc(h, len);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{nullptr, 0}; }
    tpx_i64_write(d, v, len);
    return tstr{d, len};
}

// '%0Nd' % v (python semantics: zero-pad to width, sign before zeros)
__device__ __forceinline__ tstr tpx_fmt0d(TpxHeap& h, long long width, long long v, int* ec) {
    long long dl = tpx_i64_digits(v);
    long long len = dl < width ? width : dl;
    char* d = tpx_alloc(h, len);
    if (!d) { *ec = EC_MEMORYERROR; return tstr{nullptr, 0}; }
    long long pad = len - dl;
    for (long long i = 0; i < pad; ++i) d[i] = '0';
    tpx_i64_write(d + pad, v, dl);
    if (v < 0 && pad > 0) {  // sign moves to front: '-007'
        d[pad] = '0';
        d[0] = '-';
    }
    return tstr{d, len};
}

__device__ __forceinline__ tstr tpx_str_bool(bool b) {
    return b ? tstr{"True", 4} : tstr{"False", 5};
}

__device__ __forceinline__ tstr tpx_str_none() { return tstr{"None", 4}; }

// ---- CSV output cell (RFC-4180: quote iff cell contains delim/quote/CR/LF;
//      '"' doubled) ---------------------------------------------------------------

// SWAR: 8 bytes per step (called twice per output string — size pass in the
// main kernel, then the write kernel — so the byte loop was ~2x70B of serial
// dependent loads per kept row). Head/tail stay byte-wise: no over-read.
__device__ __forceinline__ bool tpx_csv_needs_quote(const tstr s, long long* nquotes) {
    bool need = false;
    long long q = 0;
    long long i = 0;
    while (i < s.n && (((unsigned long long)(s.p + i)) & 7)) {
        char c = s.p[i];
        if (c == '"') { ++q; need = true; }
        else if (c == ',' || c == '\n' || c == '\r') need = true;
        ++i;
    }
    for (; i + 32 <= s.n; i += 32) {  // 4 independent loads per iteration
        unsigned long long v0 = *(const unsigned long long*)(s.p + i);
        unsigned long long v1 = *(const unsigned long long*)(s.p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(s.p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(s.p + i + 24);
        unsigned long long hq =
            tpx_swar_zero(v0 ^ (TPX_SWAR_ONE * (unsigned long long)'"')) |
            tpx_swar_zero(v1 ^ (TPX_SWAR_ONE * (unsigned long long)'"')) >> 1 |
            tpx_swar_zero(v2 ^ (TPX_SWAR_ONE * (unsigned long long)'"')) >> 2 |
            tpx_swar_zero(v3 ^ (TPX_SWAR_ONE * (unsigned long long)'"')) >> 3;
        unsigned long long ho = 0;
        ho |= tpx_swar_zero(v0 ^ (TPX_SWAR_ONE * (unsigned long long)',')) |
              tpx_swar_zero(v0 ^ (TPX_SWAR_ONE * (unsigned long long)'\n')) |
              tpx_swar_zero(v0 ^ (TPX_SWAR_ONE * (unsigned long long)'\r'));
        ho |= tpx_swar_zero(v1 ^ (TPX_SWAR_ONE * (unsigned long long)',')) |
              tpx_swar_zero(v1 ^ (TPX_SWAR_ONE * (unsigned long long)'\n')) |
              tpx_swar_zero(v1 ^ (TPX_SWAR_ONE * (unsigned long long)'\r'));
        ho |= tpx_swar_zero(v2 ^ (TPX_SWAR_ONE * (unsigned long long)',')) |
              tpx_swar_zero(v2 ^ (TPX_SWAR_ONE * (unsigned long long)'\n')) |
              tpx_swar_zero(v2 ^ (TPX_SWAR_ONE * (unsigned long long)'\r'));
        ho |= tpx_swar_zero(v3 ^ (TPX_SWAR_ONE * (unsigned long long)',')) |
              tpx_swar_zero(v3 ^ (TPX_SWAR_ONE * (unsigned long long)'\n')) |
              tpx_swar_zero(v3 ^ (TPX_SWAR_ONE * (unsigned long long)'\r'));
        if (hq) { q += __builtin_popcountll(hq); need = true; }
        if (ho) need = true;
    }
    for (; i + 8 <= s.n; i += 8) {
        unsigned long long v = *(const unsigned long long*)(s.p + i);
        unsigned long long hq =
            tpx_swar_zero(v ^ (TPX_SWAR_ONE * (unsigned long long)'"'));
        unsigned long long ho =
            tpx_swar_zero(v ^ (TPX_SWAR_ONE * (unsigned long long)',')) |
            tpx_swar_zero(v ^ (TPX_SWAR_ONE * (unsigned long long)'\n')) |
            tpx_swar_zero(v ^ (TPX_SWAR_ONE * (unsigned long long)'\r'));
        if (hq) { q += __builtin_popcountll(hq); need = true; }
        if (ho) need = true;
    }
    for (; i < s.n; ++i) {
        char c = s.p[i];
        if (c == '"') { ++q; need = true; }
        else if (c == ',' || c == '\n' || c == '\r') need = true;
    }
    *nquotes = q;
    return need;
}

__device__ __forceinline__ long long tpx_csv_cell_len(const tstr s) {
    long long q;
    return tpx_csv_needs_quote(s, &q) ? s.n + q + 2 : s.n;
}

__device__ __forceinline__ char* tpx_csv_cell_write(char* w, const tstr s) {
    long long q;
    if (tpx_csv_needs_quote(s, &q)) {
        *w++ = '"';
        for (long long i = 0; i < s.n; ++i) {
            char c = s.p[i];
            *w++ = c;
            if (c == '"') *w++ = '"';
        }
        *w++ = '"';
    } else {
        tpx_memcpy(w, s.p, s.n);
        w += s.n;
    }
    return w;
}

// ---- %f output for csv f64 cells (PipelineBuilder.cc:1413 formats doubles
// with "%f"): EXACT round-half-even of v*10^6 on the binary value via 128-bit
// fixed point (bit-identical to CPython's correctly-rounded "%f", validated
// against 500k fuzz doubles). Values whose 6-decimal representation exceeds
// 63 bits of digits (|v| >~ 9.2e12), nan and inf divert the row (NCV-style:
// the host formatter produces the identical text on replay). -----------------

__device__ __forceinline__ int tpx_f64_csv_n(double v, unsigned long long* N,
                                             bool* neg) {
    unsigned long long bits = (unsigned long long)__double_as_longlong(v);
    *neg = (bits >> 63) != 0;
    int exp = (int)((bits >> 52) & 0x7FF);
    unsigned long long man = bits & ((1ULL << 52) - 1);
    if (exp == 0x7FF) return 7;  // nan/inf -> host replay
    unsigned long long m;
    int e;
    if (exp == 0) { m = man; e = -1074; }
    else { m = man | (1ULL << 52); e = exp - 1075; }
    unsigned __int128 M = (unsigned __int128)m * 15625u;  // * 5^6
    int k = e + 6;
    if (k >= 0) {
        if (k >= 62) return 7;
        unsigned __int128 S = M << k;
        if ((unsigned long long)(S >> 63)) return 7;
        *N = (unsigned long long)S;
        return 0;
    }
    k = -k;
    if (k >= 69) { *N = 0; return 0; }
    unsigned __int128 Nw = M >> k;
    unsigned __int128 rem = M & (((unsigned __int128)1 << k) - 1);
    unsigned __int128 half = (unsigned __int128)1 << (k - 1);
    if (rem > half || (rem == half && ((unsigned long long)Nw & 1))) ++Nw;
    if ((unsigned long long)(Nw >> 63)) return 7;
    *N = (unsigned long long)Nw;
    return 0;
}

__device__ __forceinline__ long long tpx_f64_csv_len(unsigned long long N,
                                                     bool neg) {
    int d = 1;
    unsigned long long t = N;
    while (t >= 10) { t /= 10; ++d; }
    if (d < 7) d = 7;
    return (neg ? 1 : 0) + d + 1;
}

__device__ __forceinline__ char* tpx_f64_csv_write(char* w,
                                                   unsigned long long N,
                                                   bool neg) {
    char buf[21];
    int nd = 0;
    unsigned long long t = N;
    do { buf[nd++] = (char)('0' + (t % 10)); t /= 10; } while (t);
    while (nd < 7) buf[nd++] = '0';
    if (neg) *w++ = '-';
    for (int i = nd - 1; i >= 6; --i) *w++ = buf[i];
    *w++ = '.';
    for (int i = 5; i >= 0; --i) *w++ = buf[i];
    return w;
}

// ---- CSV input: row-boundary detection + cell split ------------------------------
//
// Replaces the reference's chunked CSV reading (CSVReader.cc:390; chunk-boundary
// row-start detection utils/src/CSVUtils.cc:1494 findLineStart) and the generated
// row parser's scan (CSVParseRowGenerator.cc SSE4.2 spanner :355). RFC-4180 rule:
// a '\n' is a row boundary iff the number of '"' before it is even (every quote
// toggles in/out of a quoted cell; an escaped "" toggles twice). Per-chunk quote
// counts + an exclusive scan give each chunk's start parity exactly — a
// parallel-exact restatement of the reference's sequential state machine.

#define TPX_CSV_CHUNK 4096
#define TPX_CSV_LANE_BYTES 64  /* 4096 / 64 lanes */

#ifndef TPX_HOST_TEST  // device-only: wave helpers + boundary-scan kernels
// wave helpers: 64-lane exclusive scan / reduction via shfl
__device__ __forceinline__ long long tpx_wave_exscan(long long v) {
    long long x = v;
    for (int o = 1; o < 64; o <<= 1) {
        long long y = __shfl_up((long long)x, o);
        if ((threadIdx.x & 63) >= o) x += y;
    }
    return x - v;  // exclusive
}

__device__ __forceinline__ long long tpx_wave_sum(long long v) {
    for (int o = 32; o; o >>= 1) v += __shfl_down((long long)v, o);
    return __shfl((long long)v, 0);
}

// per-lane scan of its 64-byte slice: counts quotes and newlines at even/odd
// LOCAL quote parity. Coalesced uint4 loads when the slice is full.
__device__ __forceinline__ void tpx_scan64(const char* __restrict__ data,
                                           long long a, long long b,
                                           int quotes_on,
                                           long long* lq, long long* l0,
                                           long long* l1) {
    long long qq = 0, e0 = 0, e1 = 0;
    if (b - a == TPX_CSV_LANE_BYTES && ((a & 15) == 0)) {
        #pragma unroll
        for (int v = 0; v < 4; ++v) {
            uint4 w = *(const uint4*)(data + a + v * 16);
            unsigned words[4] = {w.x, w.y, w.z, w.w};
            #pragma unroll
            for (int k = 0; k < 4; ++k)
                #pragma unroll
                for (int s = 0; s < 4; ++s) {
                    unsigned ch = (words[k] >> (8 * s)) & 0xFF;
                    if (quotes_on && ch == '"') ++qq;
                    else if (ch == '\n') { if (qq & 1) ++e1; else ++e0; }
                }
        }
    } else {
        for (long long i = a; i < b; ++i) {
            char ch = data[i];
            if (quotes_on && ch == '"') ++qq;
            else if (ch == '\n') { if (qq & 1) ++e1; else ++e0; }
        }
    }
    *lq = qq; *l0 = e0; *l1 = e1;
}

// wave-cooperative chunk stats: one wave per 4 KB chunk, lane i owns bytes
// [i*64, i*64+64); lane parity composed by a wave scan — parallel-exact
// restatement of the sequential quote-parity state machine.
extern "C" __global__ void tpx_csv_chunk_stats(const char* __restrict__ data,
                                               long long size, long long nchunks,
                                               long long* __restrict__ q,
                                               long long* __restrict__ c0,
                                               long long* __restrict__ c1,
                                               int quotes_on) {
    int lane = threadIdx.x & 63;
    int wid = threadIdx.x >> 6;
    int wpb = blockDim.x >> 6;
    long long wstride = (long long)gridDim.x * wpb;
    for (long long c = (long long)blockIdx.x * wpb + wid; c < nchunks;
         c += wstride) {
        long long a = c * TPX_CSV_CHUNK + (long long)lane * TPX_CSV_LANE_BYTES;
        long long b = a + TPX_CSV_LANE_BYTES;
        if (a > size) a = size;
        if (b > size) b = size;
        long long lq, l0, l1;
        tpx_scan64(data, a, b, quotes_on, &lq, &l0, &l1);
        long long pref = tpx_wave_exscan(lq);
        bool odd = (pref & 1) != 0;
        long long tot_q = tpx_wave_sum(lq);
        long long tot0 = tpx_wave_sum(odd ? l1 : l0);
        long long tot1 = tpx_wave_sum(odd ? l0 : l1);
        if (lane == 0) { q[c] = tot_q; c0[c] = tot0; c1[c] = tot1; }
    }
}

extern "C" __global__ void tpx_csv_select_counts(const long long* __restrict__ qscan,
                                                 const long long* __restrict__ c0,
                                                 const long long* __restrict__ c1,
                                                 long long* __restrict__ rc,
                                                 long long nchunks) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long c = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         c < nchunks; c += stride)
        rc[c] = (qscan[c] & 1) ? c1[c] : c0[c];
}

// row_lo/row_hi restrict emission to byte-chunks owning rows in [row_lo,row_hi)
// so the offsets for one main-kernel chunk can be emitted just-in-time on that
// chunk's stream (overlapping the previous chunk's main); chunks straddling a
// range edge are re-emitted by the neighbouring call — idempotent writes.
extern "C" __global__ void tpx_csv_emit_rows(const char* __restrict__ data,
                                             long long size, long long nchunks,
                                             const long long* __restrict__ qscan,
                                             const long long* __restrict__ base,
                                             long long* __restrict__ row_offs,
                                             int quotes_on, long long row_lo,
                                             long long row_hi) {
    int lane = threadIdx.x & 63;
    int wid = threadIdx.x >> 6;
    int wpb = blockDim.x >> 6;
    long long wstride = (long long)gridDim.x * wpb;
    if (blockIdx.x == 0 && threadIdx.x == 0 && row_lo == 0) row_offs[0] = 0;
    for (long long c = (long long)blockIdx.x * wpb + wid; c < nchunks;
         c += wstride) {
        if (base[c] >= row_hi) continue;
        if (c + 1 < nchunks && base[c + 1] <= row_lo) continue;
        long long a = c * TPX_CSV_CHUNK + (long long)lane * TPX_CSV_LANE_BYTES;
        long long b = a + TPX_CSV_LANE_BYTES;
        if (a > size) a = size;
        if (b > size) b = size;
        long long lq, l0, l1;
        tpx_scan64(data, a, b, quotes_on, &lq, &l0, &l1);
        long long pref_q = tpx_wave_exscan(lq);
        long long start_par = (qscan[c] + pref_q) & 1;
        long long sel = start_par ? l1 : l0;   // valid newlines in this lane
        long long lane_base = base[c] + tpx_wave_exscan(sel);
        // second pass: emit offsets for valid newlines (vector loads when full)
        long long qq = 0, idx = lane_base;
        if (b - a == TPX_CSV_LANE_BYTES && ((a & 15) == 0)) {
            #pragma unroll
            for (int v = 0; v < 4; ++v) {
                uint4 w = *(const uint4*)(data + a + v * 16);
                unsigned words[4] = {w.x, w.y, w.z, w.w};
                #pragma unroll
                for (int k = 0; k < 4; ++k)
                    #pragma unroll
                    for (int s2 = 0; s2 < 4; ++s2) {
                        unsigned ch = (words[k] >> (8 * s2)) & 0xFF;
                        if (quotes_on && ch == '"') ++qq;
                        else if (ch == '\n' && (((qq + start_par) & 1) == 0))
                            row_offs[1 + idx++] = a + v * 16 + k * 4 + s2 + 1;
                    }
            }
        } else {
            for (long long i = a; i < b; ++i) {
                char ch = data[i];
                if (quotes_on && ch == '"') ++qq;
                else if (ch == '\n' && (((qq + start_par) & 1) == 0))
                    row_offs[1 + idx++] = i + 1;
            }
        }
    }
}

#endif  // TPX_HOST_TEST

// one CSV cell starting at p; returns pointer to next cell start. *more = a
// delimiter was consumed (another cell follows). flags: 1 quoted, 2 contains ""
// escapes (diverted to host), 4 structurally bad (unterminated quote / junk after
// closing quote — reference DOUBLEQUOTEERROR class), 8 content may need CSV-
// OUTPUT quoting (quoted cells always; unquoted cells iff a ','/'"'/'\r' byte
// was seen in the content — lets the csv-sink size pass skip its quote scan
// for values provably derived from clean input cells, codegen._thread_qfree)
struct tpx_cell { const char* p; long long n; int flags; };

// tpx_memchr_hi that additionally reports whether any output-special byte
// ('"', '\r', and ',' when chk_comma — needed only when the source delimiter
// is not ',') occurs among the scanned bytes BEFORE the hit. Word-granular:
// in the hit word only bytes below the hit position count.
__device__ __forceinline__ long long tpx_memchr_hi_spec(
        const char* p, long long n, char c, unsigned long long* hi,
        int chk_comma, int* spec) {
    unsigned long long pat = TPX_SWAR_ONE * (unsigned char)c;
    unsigned long long sp_acc = 0;
    long long i = 0;
    #define TPX_SPECW(v) \
        (tpx_swar_zero((v) ^ (TPX_SWAR_ONE * (unsigned long long)'"')) | \
         tpx_swar_zero((v) ^ (TPX_SWAR_ONE * (unsigned long long)'\r')) | \
         (chk_comma ? tpx_swar_zero((v) ^ (TPX_SWAR_ONE * (unsigned long long)',')) : 0ULL))
    #define TPX_SPECB(ch) \
        ((ch) == '"' || (ch) == '\r' || (chk_comma && (ch) == ','))
    while (i < n && (((unsigned long long)(p + i)) & 7)) {
        unsigned char ch = p[i];
        *hi |= ch & 0x80u;
        if ((char)ch == c) { if (sp_acc) *spec = 1; return i; }
        if (TPX_SPECB(ch)) sp_acc = 1;
        ++i;
    }
    for (; i + 32 <= n; i += 32) {
        unsigned long long v0 = *(const unsigned long long*)(p + i);
        unsigned long long v1 = *(const unsigned long long*)(p + i + 8);
        unsigned long long v2 = *(const unsigned long long*)(p + i + 16);
        unsigned long long v3 = *(const unsigned long long*)(p + i + 24);
        unsigned long long h0 = tpx_swar_zero(v0 ^ pat);
        unsigned long long h1 = tpx_swar_zero(v1 ^ pat);
        unsigned long long h2 = tpx_swar_zero(v2 ^ pat);
        unsigned long long h3 = tpx_swar_zero(v3 ^ pat);
        if (h0 | h1 | h2 | h3) {
            unsigned long long hw, sw;
            long long base;
            if (h0) { hw = h0; sw = TPX_SPECW(v0); base = i;
                      *hi |= v0 & TPX_SWAR_HIGH; }
            else if (h1) { hw = h1; sw = TPX_SPECW(v1); base = i + 8;
                           sp_acc |= TPX_SPECW(v0);
                           *hi |= (v0 | v1) & TPX_SWAR_HIGH; }
            else if (h2) { hw = h2; sw = TPX_SPECW(v2); base = i + 16;
                           sp_acc |= TPX_SPECW(v0) | TPX_SPECW(v1);
                           *hi |= (v0 | v1 | v2) & TPX_SWAR_HIGH; }
            else { hw = h3; sw = TPX_SPECW(v3); base = i + 24;
                   sp_acc |= TPX_SPECW(v0) | TPX_SPECW(v1) | TPX_SPECW(v2);
                   *hi |= (v0 | v1 | v2 | v3) & TPX_SWAR_HIGH; }
            long long b = (__ffsll((long long)hw) - 1) / 8;
            sp_acc |= sw & ((b ? (1ULL << (8 * b)) : 1ULL) - 1);
            if (sp_acc) *spec = 1;
            return base + b;
        }
        sp_acc |= TPX_SPECW(v0) | TPX_SPECW(v1) | TPX_SPECW(v2) | TPX_SPECW(v3);
        *hi |= (v0 | v1 | v2 | v3) & TPX_SWAR_HIGH;
    }
    for (; i + 8 <= n; i += 8) {
        unsigned long long v = *(const unsigned long long*)(p + i);
        *hi |= v & TPX_SWAR_HIGH;
        unsigned long long hit = tpx_swar_zero(v ^ pat);
        if (hit) {
            long long b = (__ffsll((long long)hit) - 1) / 8;
            sp_acc |= TPX_SPECW(v) & ((b ? (1ULL << (8 * b)) : 1ULL) - 1);
            if (sp_acc) *spec = 1;
            return i + b;
        }
        sp_acc |= TPX_SPECW(v);
    }
    for (; i < n; ++i) {
        unsigned char ch = p[i];
        *hi |= ch & 0x80u;
        if ((char)ch == c) { if (sp_acc) *spec = 1; return i; }
        if (TPX_SPECB(ch)) sp_acc = 1;
    }
    if (sp_acc) *spec = 1;
    return -1;
    #undef TPX_SPECW
    #undef TPX_SPECB
}

// hi: accumulates non-ASCII high bits of every scanned byte (row ASCII gate
// fused into the walk — delimiters/quotes themselves are always ASCII)
__device__ __forceinline__ const char* tpx_csv_next_cell(const char* p,
                                                         const char* end,
                                                         tpx_cell* c, bool* more,
                                                         char delim,
                                                         unsigned long long* hi) {
    c->flags = 0;
    *more = false;
    if (p < end && *p == '"') {
        const char* s = p + 1;
        const char* q = s;
        bool esc = false;
        while (q < end) {
            long long k = tpx_memchr_hi(q, end - q, '"', hi);
            if (k < 0) { q = end; break; }
            q += k;
            if (q + 1 < end && q[1] == '"') { esc = true; q += 2; continue; }
            break;
        }
        if (q >= end) { c->p = p; c->n = end - p; c->flags = 4; return end; }
        // bit 8: quoted content was quoted for a reason (or we don't track it)
        c->p = s; c->n = q - s; c->flags = 1 | (esc ? 2 : 0) | 8;
        ++q;
        if (q < end && *q != delim) c->flags |= 4;
        long long kd = tpx_memchr_hi(q, end - q, delim, hi);
        q = kd < 0 ? end : q + kd;
        if (q < end) { *more = true; ++q; }
        return q;
    }
    int spec = 0;
    long long kd = tpx_memchr_hi_spec(p, end - p, delim, hi,
                                      delim != ',', &spec);
    const char* q = kd < 0 ? end : p + kd;
    c->p = p; c->n = q - p;
    if (spec) c->flags |= 8;
    if (q < end) { *more = true; ++q; }
    return q;
}

// ---- mask-based cell walk -------------------------------------------------------
// The per-cell memchr walk is a serial dependent-load chain (~10 sequential
// cells x ~50-cycle LDS loads per SWAR step). This walk instead classifies the
// row's bytes in 64-BYTE GROUPS — 8 independent 8-B loads, then pure-ALU
// bitmask derivation (movemask trick) — and consumes cell boundaries from the
// group bitmasks. Groups load lazily and monotonically (each at most once per
// row); semantics are EXACTLY tpx_csv_next_cell's, including flag bits and the
// fused ASCII gate. Mirrors the reference CSVParseRowGenerator's SSE spanner
// idea (:355) at CDNA4 width.

struct tpx_grp { unsigned long long q, d, s; };

struct tpx_mwalk {
    const char* ab;   // 8-aligned base; bit i of a group mask = byte ab[64g+i]
    int off, endb;    // row occupies bytes [off, endb) relative to ab
    int g;            // loaded group index (-1 = none)
    int pos;          // next cell start
    bool more;
    tpx_grp G;
    unsigned long long hib;  // non-ASCII bits seen in loaded groups (row-ranged)
};

// gather the 0x80-position hit bits of one SWAR word into 8 contiguous bits
// (byte j -> bit j); all 64 partial products land on distinct bit positions,
// so the multiply is carry-free and exact
__device__ __forceinline__ unsigned long long tpx_mm8(unsigned long long h) {
    return ((h >> 7) * 0x0102040810204080ULL) >> 56;
}

__device__ __forceinline__ void tpx_mw_group(tpx_mwalk& S, int g, char delim,
                                             int chk_comma) {
    const unsigned long long* w =
        (const unsigned long long*)(S.ab + ((long long)g << 6));
    unsigned long long pq = TPX_SWAR_ONE * (unsigned long long)'"';
    unsigned long long pd = TPX_SWAR_ONE * (unsigned char)delim;
    unsigned long long pr = TPX_SWAR_ONE * (unsigned long long)'\r';
    unsigned long long pc = TPX_SWAR_ONE * (unsigned long long)',';
    unsigned long long q = 0, d = 0, s = 0, h = 0;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
        unsigned long long v = w[k];
        q |= tpx_mm8(tpx_swar_zero(v ^ pq)) << (8 * k);
        d |= tpx_mm8(tpx_swar_zero(v ^ pd)) << (8 * k);
        unsigned long long sv = tpx_swar_zero(v ^ pr);
        if (chk_comma) sv |= tpx_swar_zero(v ^ pc);
        s |= tpx_mm8(sv) << (8 * k);
        h |= tpx_mm8(v & TPX_SWAR_HIGH) << (8 * k);
    }
    int base = g << 6;
    int lo = S.off - base; if (lo < 0) lo = 0;          // lo in [0, 63]
    int hi = S.endb - base; if (hi > 64) hi = 64;
    unsigned long long rm = (hi <= lo) ? 0ULL :
        ((hi >= 64 ? ~0ULL : ((1ULL << hi) - 1)) & ~((1ULL << lo) - 1));
    S.G.q = q & rm;
    S.G.d = d & rm;
    S.G.s = (s | q) & rm;  // a '"' byte in unquoted content also forces quoting
    S.hib |= h & rm;
    S.g = g;
}

__device__ __forceinline__ void tpx_mw_init(tpx_mwalk& S, const char* rp,
                                            const char* rend) {
    S.ab = (const char*)((unsigned long long)rp & ~7ULL);
    S.off = (int)(rp - S.ab);
    S.endb = S.off + (int)(rend - rp);
    S.g = -1;
    S.pos = S.off;
    S.more = true;
    S.hib = 0;
}

__device__ __forceinline__ int tpx_mw_test_q(tpx_mwalk& S, int k, char delim,
                                             int chk) {
    if (k >= S.endb) return 0;
    if ((k >> 6) != S.g) tpx_mw_group(S, k >> 6, delim, chk);
    return (int)((S.G.q >> (k & 63)) & 1);
}

__device__ __forceinline__ int tpx_mw_test_d(tpx_mwalk& S, int k, char delim,
                                             int chk) {
    if (k >= S.endb) return 0;
    if ((k >> 6) != S.g) tpx_mw_group(S, k >> 6, delim, chk);
    return (int)((S.G.d >> (k & 63)) & 1);
}

// next delim bit >= k (or endb); ORs output-special bits in [k, result) into
// *spec — the word-granular analog of tpx_memchr_hi_spec
__device__ __forceinline__ int tpx_mw_next_d(tpx_mwalk& S, int k, char delim,
                                             int chk, int* spec) {
    unsigned long long sp_acc = 0;
    for (;;) {
        if (k >= S.endb) { if (sp_acc) *spec = 1; return S.endb; }
        if ((k >> 6) != S.g) tpx_mw_group(S, k >> 6, delim, chk);
        int local = k & 63;
        unsigned long long ge = local ? (~0ULL << local) : ~0ULL;
        unsigned long long d = S.G.d & ge;
        unsigned long long s = S.G.s & ge;
        if (d) {
            int b = __ffsll((long long)d) - 1;
            sp_acc |= s & ((b ? (1ULL << b) : 1ULL) - 1);
            if (sp_acc) *spec = 1;
            return (S.g << 6) + b;
        }
        sp_acc |= s;
        k = (S.g + 1) << 6;
    }
}

__device__ __forceinline__ int tpx_mw_next_q(tpx_mwalk& S, int k, char delim,
                                             int chk) {
    for (;;) {
        if (k >= S.endb) return S.endb;
        if ((k >> 6) != S.g) tpx_mw_group(S, k >> 6, delim, chk);
        int local = k & 63;
        unsigned long long ge = local ? (~0ULL << local) : ~0ULL;
        unsigned long long q = S.G.q & ge;
        if (q) return (S.g << 6) + (__ffsll((long long)q) - 1);
        k = (S.g + 1) << 6;
    }
}

// one cell via the group masks — flag/position semantics identical to
// tpx_csv_next_cell (incl. bit 8 output-special tracking)
__device__ __forceinline__ void tpx_mw_cell(tpx_mwalk& S, tpx_cell* c,
                                            char delim, int chk) {
    c->flags = 0;
    S.more = false;
    int pos = S.pos;
    if (pos < S.endb && tpx_mw_test_q(S, pos, delim, chk)) {  // quoted cell
        int q = pos + 1;
        bool esc = false;
        for (;;) {
            int cq = tpx_mw_next_q(S, q, delim, chk);
            if (cq >= S.endb) {  // unterminated
                c->p = S.ab + pos; c->n = S.endb - pos; c->flags = 4;
                S.pos = S.endb;
                return;
            }
            if (cq + 1 < S.endb && tpx_mw_test_q(S, cq + 1, delim, chk)) {
                esc = true; q = cq + 2; continue;
            }
            q = cq;
            break;
        }
        c->p = S.ab + pos + 1; c->n = q - (pos + 1);
        c->flags = 1 | (esc ? 2 : 0) | 8;
        ++q;
        if (q < S.endb && !tpx_mw_test_d(S, q, delim, chk)) c->flags |= 4;
        int spec = 0;
        int nd = tpx_mw_next_d(S, q, delim, chk, &spec);
        if (nd < S.endb) { S.more = true; S.pos = nd + 1; } else S.pos = S.endb;
        return;
    }
    int spec = 0;
    int nd = tpx_mw_next_d(S, pos, delim, chk, &spec);
    c->p = S.ab + pos; c->n = nd - pos;
    if (spec) c->flags |= 8;
    if (nd < S.endb) { S.more = true; S.pos = nd + 1; } else S.pos = S.endb;
}

// typed cell parse (cells path semantics: python-whitespace trim + fast_atoX,
// CellSourceTaskBuilder + Runtime.cc:319 wrappers)
__device__ __forceinline__ int tpx_cell_i64(const tpx_cell& c, long long* out) {
    const char* a = c.p;
    const char* b = c.p + c.n;
    while (a < b && tpx_is_pyws(*a)) ++a;
    while (b > a && tpx_is_pyws(*(b - 1))) --b;
    return tpx_fast_atoi64(a, b, out);
}

__device__ __forceinline__ int tpx_cell_f64(const tpx_cell& c, double* out) {
    const char* a = c.p;
    const char* b = c.p + c.n;
    while (a < b && tpx_is_pyws(*a)) ++a;
    while (b > a && tpx_is_pyws(*(b - 1))) --b;
    return tpx_fast_atod(a, b, out);
}

// fast_atob semantics (StringUtils.cc:186): case-insensitive
// true/t/yes/y/1 | false/f/no/n/0
#define EC_BOOLPARSE_ 54
__device__ __forceinline__ int tpx_cell_bool(const tpx_cell& c, bool* out) {
    const char* a = c.p;
    const char* b = c.p + c.n;
    while (a < b && tpx_is_pyws(*a)) ++a;
    while (b > a && tpx_is_pyws(*(b - 1))) --b;
    long long n = b - a;
    char buf[8];
    if (n < 1 || n > 5) return EC_BOOLPARSE_;
    for (long long i = 0; i < n; ++i) {
        char ch = a[i];
        buf[i] = (ch >= 'A' && ch <= 'Z') ? ch + 32 : ch;
    }
    auto eq = [&](const char* s, long long l) {
        if (n != l) return false;
        for (long long i = 0; i < l; ++i) if (buf[i] != s[i]) return false;
        return true;
    };
    if (eq("true", 4) || eq("t", 1) || eq("yes", 3) || eq("y", 1) || eq("1", 1)) {
        *out = true; return EC_OK;
    }
    if (eq("false", 5) || eq("f", 1) || eq("no", 2) || eq("n", 1) || eq("0", 1)) {
        *out = false; return EC_OK;
    }
    return EC_BOOLPARSE_;
}

#ifndef TPX_HOST_TEST  // device-only: reduction/hash/scan kernels
// ---- fixed kernels ---------------------------------------------------------------

// ---- aggregate reduction (AggregateFunctions.cc fold -> deterministic device
//      reduce: per-thread grid-stride partials in fixed index order, block tree,
//      then a single-thread final pass — same result every run for a given n) ----

#define TPX_RED_THREADS 256

extern "C" __global__ void tpx_reduce_f64(const double* __restrict__ vals,
                                          const unsigned char* __restrict__ keep,
                                          long long n,
                                          double* __restrict__ partials) {
    __shared__ double sh[TPX_RED_THREADS];
    long long stride = (long long)gridDim.x * blockDim.x;
    double acc = 0.0;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        if (keep[i]) acc += vals[i];
    sh[threadIdx.x] = acc;
    __syncthreads();
    for (int o = TPX_RED_THREADS / 2; o; o >>= 1) {
        if ((int)threadIdx.x < o) sh[threadIdx.x] += sh[threadIdx.x + o];
        __syncthreads();
    }
    if (threadIdx.x == 0) partials[blockIdx.x] = sh[0];
}

extern "C" __global__ void tpx_reduce_f64_final(const double* __restrict__ partials,
                                                long long n,
                                                double* __restrict__ out) {
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        double s = 0.0;
        for (long long i = 0; i < n; ++i) s += partials[i];
        *out = s;
    }
}

extern "C" __global__ void tpx_reduce_i64(const long long* __restrict__ vals,
                                          const unsigned char* __restrict__ keep,
                                          long long n,
                                          long long* __restrict__ partials) {
    __shared__ long long sh[TPX_RED_THREADS];
    long long stride = (long long)gridDim.x * blockDim.x;
    long long acc = 0;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        if (keep[i]) acc += vals[i];
    sh[threadIdx.x] = acc;
    __syncthreads();
    for (int o = TPX_RED_THREADS / 2; o; o >>= 1) {
        if ((int)threadIdx.x < o) sh[threadIdx.x] += sh[threadIdx.x + o];
        __syncthreads();
    }
    if (threadIdx.x == 0) partials[blockIdx.x] = sh[0];
}

extern "C" __global__ void tpx_reduce_i64_final(const long long* __restrict__ partials,
                                                long long n,
                                                long long* __restrict__ out) {
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        long long s = 0;
        for (long long i = 0; i < n; ++i) s += partials[i];
        *out = s;
    }
}

// ---- aggregateByKey hash-reduce (hashmap.cc / int_hashmap.cc analog) ----------
// Open-addressing table, EMPTY key = -1 (table memset 0xFF); real key -1 goes to
// a dedicated special slot. Output order is parity-unpinned.

#define TPX_HK_EMPTY (-1LL)
#define TPX_HK_MAXPROBE 1024

// fnv1a for hash-join string keys (build-side replica in codegen._jhash_bytes)
__device__ __forceinline__ unsigned long long tpx_jhash_bytes(const char* p,
                                                              long long n) {
    unsigned long long h = 1469598103934665603ULL;
    for (long long i = 0; i < n; ++i) {
        h ^= (unsigned char)p[i];
        h *= 1099511628211ULL;
    }
    return h;
}

__device__ __forceinline__ unsigned long long tpx_hash_i64(long long k) {
    unsigned long long x = (unsigned long long)k;
    x ^= x >> 33;
    x *= 0xff51afd7ed558ccdULL;  // murmur3 finalizer (mix quality only)
    x ^= x >> 33;
    x *= 0xc4ceb9fe1a85ec53ULL;
    x ^= x >> 33;
    return x;
}

// string-key variant: the slot key is the (ptr,len) of the FIRST row that
// claimed it (CAS on the pointer; len stored after the win). A reader that
// sees a claimed slot whose len is not yet visible just probes on — the same
// string may then occupy several slots, and the HOST merges the emitted
// (key,val) pairs by string content (sum is commutative), so no cross-lane
// spin is ever needed (a same-wave spin on a diverged writer would deadlock
// under wave-lockstep execution).
__device__ __forceinline__ bool tpx_hk_bytes_eq(const char* a, const char* b,
                                                int n) {
    for (int i = 0; i < n; ++i)
        if (a[i] != b[i]) return false;
    return true;
}

#define TPX_HK_SDEF(NAME, VT, ATOMIC)                                        \
extern "C" __global__ void NAME(const unsigned long long* __restrict__ kptr, \
                                const int* __restrict__ klen,                \
                                const VT* __restrict__ vals,                 \
                                const unsigned char* __restrict__ keep,      \
                                long long n, unsigned long long* tkeys,      \
                                int* tlens, VT* tvals,                       \
                                unsigned long long tmask,                    \
                                unsigned long long* used, int* overflow) {   \
    long long stride = (long long)gridDim.x * blockDim.x;                    \
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;     \
         i < n; i += stride) {                                               \
        if (!keep[i]) continue;                                              \
        unsigned long long p = kptr[i];                                      \
        int len = klen[i];                                                   \
        VT val = vals[i];                                                    \
        unsigned long long h = tpx_jhash_bytes((const char*)p, len) & tmask; \
        int probe = 0;                                                       \
        for (; probe < TPX_HK_MAXPROBE; ++probe) {                           \
            unsigned long long cur = tkeys[h];                               \
            if (cur == 0) {                                                  \
                unsigned long long prev =                                    \
                    atomicCAS(&tkeys[h], 0ULL, p);                           \
                if (prev == 0) {                                             \
                    tlens[h] = len + 1; /* +1: 0 means not-ready */          \
                    __threadfence();                                         \
                    atomicAdd(used, 1ULL);                                   \
                    ATOMIC(&tvals[h], val);                                  \
                    break;                                                   \
                }                                                            \
                cur = prev;                                                  \
            }                                                                \
            int cl = tlens[h];                                               \
            if (cl == len + 1 &&                                             \
                (cur == p || tpx_hk_bytes_eq((const char*)cur,               \
                                             (const char*)p, len))) {        \
                ATOMIC(&tvals[h], val);                                      \
                break;                                                       \
            }                                                                \
            h = (h + 1) & tmask;                                             \
        }                                                                    \
        if (probe == TPX_HK_MAXPROBE) *overflow = 1;                         \
    }                                                                        \
}

#define TPX_HK_ADD_F64(ptr, v) atomicAdd(ptr, v)
#define TPX_HK_ADD_I64(ptr, v) atomicAdd((unsigned long long*)(ptr), \
                                         (unsigned long long)(v))
TPX_HK_SDEF(tpx_hashagg_str_f64, double, TPX_HK_ADD_F64)
TPX_HK_SDEF(tpx_hashagg_str_i64, long long, TPX_HK_ADD_I64)

// emit claimed slots as [ptr,len,valbits] triples (host fetches key bytes and
// merges duplicate slots by string content)
extern "C" __global__ void tpx_hashagg_str_emit(
        const unsigned long long* __restrict__ tkeys,
        const int* __restrict__ tlens,
        const long long* __restrict__ tvals, long long tsize,
        long long* __restrict__ out3, unsigned long long* out_idx) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < tsize; i += stride) {
        if (!tkeys[i] || tlens[i] == 0) continue;
        unsigned long long w = atomicAdd(out_idx, 1ULL);
        out3[w * 3 + 0] = (long long)tkeys[i];
        out3[w * 3 + 1] = (long long)(tlens[i] - 1);
        out3[w * 3 + 2] = tvals[i];
    }
}

extern "C" __global__ void tpx_hashagg_f64(const long long* __restrict__ keys,
                                           const double* __restrict__ vals,
                                           const unsigned char* __restrict__ keep,
                                           long long n, long long* tkeys,
                                           double* tvals,
                                           unsigned long long tmask,
                                           unsigned long long* used,
                                           int* overflow,
                                           unsigned long long* special_cnt,
                                           double* special_val) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        if (!keep[i]) continue;
        long long k = keys[i];
        double v = vals[i];
        if (k == TPX_HK_EMPTY) {
            atomicAdd(special_cnt, 1ULL);
            atomicAdd(special_val, v);
            continue;
        }
        unsigned long long h = tpx_hash_i64(k) & tmask;
        int probe = 0;
        for (; probe < TPX_HK_MAXPROBE; ++probe) {
            long long cur = tkeys[h];
            if (cur == k) { atomicAdd(&tvals[h], v); break; }
            if (cur == TPX_HK_EMPTY) {
                unsigned long long prev = atomicCAS(
                    (unsigned long long*)&tkeys[h],
                    (unsigned long long)TPX_HK_EMPTY, (unsigned long long)k);
                if (prev == (unsigned long long)TPX_HK_EMPTY) {
                    atomicAdd(used, 1ULL);
                    atomicAdd(&tvals[h], v);
                    break;
                }
                if ((long long)prev == k) { atomicAdd(&tvals[h], v); break; }
            }
            h = (h + 1) & tmask;
        }
        if (probe == TPX_HK_MAXPROBE) *overflow = 1;
    }
}

extern "C" __global__ void tpx_hashagg_i64(const long long* __restrict__ keys,
                                           const long long* __restrict__ vals,
                                           const unsigned char* __restrict__ keep,
                                           long long n, long long* tkeys,
                                           long long* tvals,
                                           unsigned long long tmask,
                                           unsigned long long* used,
                                           int* overflow,
                                           unsigned long long* special_cnt,
                                           long long* special_val) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        if (!keep[i]) continue;
        long long k = keys[i];
        long long v = vals[i];
        if (k == TPX_HK_EMPTY) {
            atomicAdd(special_cnt, 1ULL);
            atomicAdd((unsigned long long*)special_val, (unsigned long long)v);
            continue;
        }
        unsigned long long h = tpx_hash_i64(k) & tmask;
        int probe = 0;
        for (; probe < TPX_HK_MAXPROBE; ++probe) {
            long long cur = tkeys[h];
            if (cur == k) {
                atomicAdd((unsigned long long*)&tvals[h], (unsigned long long)v);
                break;
            }
            if (cur == TPX_HK_EMPTY) {
                unsigned long long prev = atomicCAS(
                    (unsigned long long*)&tkeys[h],
                    (unsigned long long)TPX_HK_EMPTY, (unsigned long long)k);
                if (prev == (unsigned long long)TPX_HK_EMPTY) {
                    atomicAdd(used, 1ULL);
                    atomicAdd((unsigned long long*)&tvals[h],
                              (unsigned long long)v);
                    break;
                }
                if ((long long)prev == k) {
                    atomicAdd((unsigned long long*)&tvals[h],
                              (unsigned long long)v);
                    break;
                }
            }
            h = (h + 1) & tmask;
        }
        if (probe == TPX_HK_MAXPROBE) *overflow = 1;
    }
}

// emit table entries as a [numRows][key,val] partition body (16 B rows); the
// numRows header is patched host-side from *out_idx
extern "C" __global__ void tpx_hashagg_emit(const long long* __restrict__ tkeys,
                                            const long long* __restrict__ tvals,
                                            long long tsize,
                                            const unsigned long long* special_cnt,
                                            const long long* special_val,
                                            unsigned char* out,
                                            unsigned long long* out_idx) {
    long long stride = (long long)gridDim.x * blockDim.x;
    long long tid0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (tid0 == 0 && *special_cnt > 0) {
        unsigned long long j = atomicAdd(out_idx, 1ULL);
        long long* w = (long long*)(out + 8 + j * 16);
        w[0] = TPX_HK_EMPTY;
        w[1] = *special_val;
    }
    for (long long i = tid0; i < tsize; i += stride) {
        if (tkeys[i] == TPX_HK_EMPTY) continue;
        unsigned long long j = atomicAdd(out_idx, 1ULL);
        long long* w = (long long*)(out + 8 + j * 16);
        w[0] = tkeys[i];
        w[1] = tvals[i];
    }
}

// exclusive-scan building block: per-block scan of ITEMS_PER_BLOCK i64 items.
// grid-wide composition happens host-side (scan block sums, then add pass).
#define TPX_SCAN_THREADS 256
#define TPX_SCAN_ITEMS 8
#define TPX_SCAN_BLOCK (TPX_SCAN_THREADS * TPX_SCAN_ITEMS)

// one-thread helper: totals of two exclusive scans (in[n-1] + out[n-1]) into a
// 16B slot, so the host fetches both chunk totals with ONE tiny D2H copy
extern "C" __global__ void tpx_pair_total(const long long* __restrict__ a_in,
                                          const long long* __restrict__ a_out,
                                          long long na,
                                          const long long* __restrict__ b_in,
                                          const long long* __restrict__ b_out,
                                          long long nb,
                                          long long* __restrict__ slot) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        slot[0] = na > 0 ? a_in[na - 1] + a_out[na - 1] : 0;
        slot[1] = nb > 0 ? b_in[nb - 1] + b_out[nb - 1] : 0;
    }
}

extern "C" __global__ void tpx_scan_block(const long long* __restrict__ in,
                                          long long* __restrict__ out,
                                          long long* __restrict__ block_sums,
                                          long long n) {
    __shared__ long long sh[TPX_SCAN_THREADS];
    long long base = (long long)blockIdx.x * TPX_SCAN_BLOCK;
    long long vals[TPX_SCAN_ITEMS];
    long long sum = 0;
    #pragma unroll
    for (int k = 0; k < TPX_SCAN_ITEMS; ++k) {
        long long i = base + threadIdx.x * TPX_SCAN_ITEMS + k;
        vals[k] = (i < n) ? in[i] : 0;
        sum += vals[k];
    }
    // wave + block scan of per-thread sums
    sh[threadIdx.x] = sum;
    __syncthreads();
    // simple Hillis-Steele in LDS (256 wide)
    for (int off = 1; off < TPX_SCAN_THREADS; off <<= 1) {
        long long v = (threadIdx.x >= off) ? sh[threadIdx.x - off] : 0;
        __syncthreads();
        sh[threadIdx.x] += v;
        __syncthreads();
    }
    long long excl = sh[threadIdx.x] - sum;  // exclusive prefix of this thread
    if (threadIdx.x == TPX_SCAN_THREADS - 1 && block_sums)
        block_sums[blockIdx.x] = sh[TPX_SCAN_THREADS - 1];
    long long run = excl;
    #pragma unroll
    for (int k = 0; k < TPX_SCAN_ITEMS; ++k) {
        long long i = base + threadIdx.x * TPX_SCAN_ITEMS + k;
        if (i < n) out[i] = run;
        run += vals[k];
    }
}

extern "C" __global__ void tpx_scan_add(long long* __restrict__ data,
                                        const long long* __restrict__ block_offs,
                                        long long n) {
    long long base = (long long)blockIdx.x * TPX_SCAN_BLOCK;
    long long add = block_offs[blockIdx.x];
    for (int k = 0; k < TPX_SCAN_ITEMS; ++k) {
        long long i = base + threadIdx.x + (long long)k * TPX_SCAN_THREADS;
        if (i < n) data[i] += add;
    }
}
#endif  // TPX_HOST_TEST
