// Host-compiled fuzz check of the device runtime's CSV cell walks: the
// mask-based walk (tpx_mwalk) must reproduce tpx_csv_next_cell's cells,
// flags, *more protocol and ASCII gate EXACTLY on arbitrary byte rows.
// Compiled by tests/test_host_rt.py with g++ (TPX_HOST_TEST strips the
// device-only kernels; scalar helpers are plain C).
#define TPX_HOST_TEST 1
#define __HIPRTC__ 1
#define __device__
#define __forceinline__ inline
#define __ffsll(x) __builtin_ffsll(x)
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <stdint.h>
#include <math.h>

static inline unsigned long long atomicAdd(unsigned long long* p,
                                           unsigned long long v) {
    unsigned long long o = *p; *p += v; return o;
}
static inline long long __double_as_longlong(double d) {
    long long v; memcpy(&v, &d, 8); return v;
}
static inline double __longlong_as_double(long long v) {
    double d; memcpy(&d, &v, 8); return d;
}

#include "../tuplex_amd/csrc/tpx_rt.hip.h"

static unsigned long long rng_state = 88172645463325252ULL;
static unsigned long long rnd(void) {
    rng_state ^= rng_state << 13;
    rng_state ^= rng_state >> 7;
    rng_state ^= rng_state << 17;
    return rng_state;
}

int main(void) {
    // incl. delim/quote/CR/LF/high AND the bytes one above each special
    // (',' -> '-', '"' -> '#', '\r' -> 0x0e, '|' -> '}'): the classic SWAR
    // zero-detect falsely flags value+1 bytes after a true match — positional
    // masks must be exact (caught live on flights ",-471.04")
    const char* alpha = "ab,\"\r\n x9\xc3\x00-#\x0e|}";
    int alpha_n = 16;
    char buf[4096 + 128];
    int fails = 0;
    for (int iter = 0; iter < 200000; ++iter) {
        int n = (int)(rnd() % 300);
        int shift = (int)(rnd() % 8);   // arbitrary row alignment
        char* rp = buf + 64 + shift;
        for (int i = 0; i < n; ++i) {
            unsigned r = (unsigned)(rnd() % 100);
            char c = r < 80 ? alpha[rnd() % alpha_n]
                            : (char)('a' + (rnd() % 26));
            if (c == '\n' && (rnd() & 1)) c = 'q';  // fewer hard newlines
            rp[i] = c;
        }
        // strip trailing newline/CR like the generated loader does
        char* rend = rp + n;
        if (rend > rp && rend[-1] == '\n') --rend;
        if (rend > rp && rend[-1] == '\r') --rend;
        char delim = (iter & 7) == 0 ? '|' : ',';
        int chk = delim != ',';
        int ncols = 1 + (int)(rnd() % 12);

        // reference walk
        tpx_cell ref_cells[16];
        bool ref_avail[16];
        unsigned long long hib_ref = 0;
        {
            const char* cur = rp;
            bool m = true, avail = true;
            for (int k = 0; k < ncols; ++k) {
                ref_avail[k] = avail;
                if (!avail) { ref_cells[k] = tpx_cell{rp, 0, 0}; continue; }
                cur = tpx_csv_next_cell(cur, rend, &ref_cells[k], &m, delim,
                                        &hib_ref);
                avail = m;
            }
        }
        // mask walk
        tpx_cell mw_cells[16];
        bool mw_avail[16];
        tpx_mwalk S;
        tpx_mw_init(S, rp, rend);
        {
            bool avail = true;
            for (int k = 0; k < ncols; ++k) {
                mw_avail[k] = avail;
                if (!avail) { mw_cells[k] = tpx_cell{rp, 0, 0}; continue; }
                tpx_mw_cell(S, &mw_cells[k], delim, chk);
                avail = S.more;
            }
        }
        for (int k = 0; k < ncols; ++k) {
            if (ref_avail[k] != mw_avail[k]) {
                printf("iter %d col %d: avail %d vs %d\n", iter, k,
                       (int)ref_avail[k], (int)mw_avail[k]);
                ++fails; break;
            }
            if (!ref_avail[k]) continue;
            if (ref_cells[k].p != mw_cells[k].p ||
                ref_cells[k].n != mw_cells[k].n ||
                ref_cells[k].flags != mw_cells[k].flags) {
                printf("iter %d col %d: cell (%ld,%lld,%d) vs (%ld,%lld,%d) "
                       "row=[%.*s]\n", iter, k,
                       (long)(ref_cells[k].p - rp), ref_cells[k].n,
                       ref_cells[k].flags,
                       (long)(mw_cells[k].p - rp), mw_cells[k].n,
                       mw_cells[k].flags, (int)(rend - rp), rp);
                ++fails; break;
            }
        }
        // ASCII gate: the mask walk must flag at least whenever the byte walk
        // did, and exactly match ground truth when the row was fully consumed
        if (hib_ref && !S.hib) {
            printf("iter %d: hib ref set but mwalk clear\n", iter);
            ++fails;
        }
        if (fails > 5) break;
    }
    if (fails) { printf("FAIL %d\n", fails); return 1; }
    printf("OK\n");
    return 0;
}
