// Host repro: run real CSV rows through the device loader's walk + gates
// (tpx_mwalk, 110 cells, chk_comma=0) and print every row the GPU path would
// DIVERT (prc != 0) with its reason — used to chase spurious diverts on the
// flights shape (wide rows take the global, non-staged branch).
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

int main(int argc, char** argv) {
    if (argc < 3) { fprintf(stderr, "usage: %s file ncols [used...]\n", argv[0]); return 2; }
    FILE* f = fopen(argv[1], "rb");
    if (!f) return 2;
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    // +256 slack like the device input buffer
    char* buf = (char*)malloc((size_t)sz + 256);
    memset(buf + sz, 0, 256);
    if (fread(buf, 1, (size_t)sz, f) != (size_t)sz) return 2;
    fclose(f);
    int ncols = atoi(argv[2]);
    // used columns (typed parse targets): i64 at idx%5==0, f64 at idx%5==1
    int used[256], nused = 0;
    for (int i = 3; i < argc; ++i) used[nused++] = atoi(argv[i]);

    long long row = -1;  // header counts as row -1
    char* p = buf;
    char* endall = buf + sz;
    int diverted = 0;
    while (p < endall) {
        char* nl = (char*)memchr(p, '\n', (size_t)(endall - p));
        char* rend0 = nl ? nl + 1 : endall;
        if (row >= 0) {
            const char* rp = p;
            const char* rend = rend0;
            if (rend > rp && rend[-1] == '\n') --rend;
            if (rend > rp && rend[-1] == '\r') --rend;
            long long prc = 0;
            tpx_mwalk S; tpx_mw_init(S, rp, rend);
            bool avail = true;
            int badf = 0;
            tpx_cell cells[256];
            for (int k = 0; k < ncols; ++k) {
                cells[k] = tpx_cell{rp, 0, 0};
                if (!prc) {
                    if (!avail) prc = 20;  // UNDERRUN
                    else { tpx_mw_cell(S, &cells[k], ',', 0);
                           avail = S.more; badf |= cells[k].flags; }
                }
            }
            if (!prc && avail) prc = 21;  // OVERRUN
            if (!prc && (badf & 6)) prc = 70;
            if (!prc && S.hib) prc = 7;
            // typed parses on used cells
            for (int u = 0; !prc && u < nused; ++u) {
                int k = used[u];
                int m = k % 5;
                if (m == 0) { long long v;
                    if (tpx_cell_i64(cells[k], &v) != 0) prc = 70; }
                else if (m == 1) { double v;
                    if (tpx_cell_f64(cells[k], &v) != 0) prc = 70; }
            }
            if (prc) {
                ++diverted;
                if (diverted <= 40)
                    printf("row %lld prc %lld badf %d len %d : %.80s\n",
                           row, prc, badf, (int)(rend - rp), rp);
            }
        }
        ++row;
        p = rend0;
    }
    printf("total rows %lld diverted %d\n", row, diverted);
    free(buf);
    return 0;
}
