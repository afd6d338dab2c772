/* Microbench: isolate what bounds the LDS-staged candidate-append kernels
 * (k_ord_filter_date shape: 150M int32 rows, ~49% selectivity).
 * Variants: 0 = loads+predicate+popcount only (no append)
 *           1 = + wave prefix sum (no LDS, no output)
 *           2 = full LDS-staged order-preserving append (the product shape)
 *           3 = plain per-lane global atomicAdd-free scatter via ballot
 *               wave_append (order-free)
 */
#include <hip/hip_runtime.h>
#include <stdio.h>
#define WAVE 64

__global__ void k_v0(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     unsigned long long *out)
{
    int64_t nq = n / 4;
    const int4 *d4 = (const int4 *)d;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    unsigned long long my = 0;
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; q < nq;
         q += stride) {
        int4 v = d4[q];
        my += (v.x < c) + (v.y < c) + (v.z < c) + (v.w < c);
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        my += __shfl_down(my, off, WAVE);
    if ((threadIdx.x % WAVE) == 0 && my) atomicAdd(out, my);
}

__global__ void k_v1(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     unsigned long long *out)
{
    int64_t nq = n / 4;
    const int4 *d4 = (const int4 *)d;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int lane = (int)(threadIdx.x % WAVE);
    unsigned long long acc = 0;
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; q < nq;
         q += stride) {
        int4 v = d4[q];
        int mycnt = (v.x < c) + (v.y < c) + (v.z < c) + (v.w < c);
        int incl = mycnt;
        for (int off = 1; off < WAVE; off <<= 1) {
            int up = __shfl_up(incl, off, WAVE);
            if (lane >= off) incl += up;
        }
        acc += __shfl(incl, WAVE - 1, WAVE);
    }
    if (lane == 0) atomicAdd(out, acc);
}

__global__ void k_v2(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     int64_t *cand, int64_t *ncand)
{
    const int BUF = 1024;
    __shared__ int64_t buf[256 / WAVE][BUF];
    int wid = (int)(threadIdx.x / WAVE), lane = (int)(threadIdx.x % WAVE);
    int nbuf = 0;
    int64_t nq = n / 4;
    const int4 *d4 = (const int4 *)d;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;;
         q += stride) {
        bool m[4] = {false, false, false, false};
        int64_t r0 = q * 4;
        if (q < nq) {
            int4 v = d4[q];
            m[0] = v.x < c; m[1] = v.y < c; m[2] = v.z < c; m[3] = v.w < c;
        }
        int mycnt = m[0] + m[1] + m[2] + m[3];
        int incl = mycnt;
        for (int off = 1; off < WAVE; off <<= 1) {
            int up = __shfl_up(incl, off, WAVE);
            if (lane >= off) incl += up;
        }
        int tot = __shfl(incl, WAVE - 1, WAVE);
        if (tot) {
            if (nbuf + tot > BUF) {
                long long bpos = 0;
                if (lane == 0)
                    bpos = (long long)atomicAdd((unsigned long long *)ncand,
                                                (unsigned long long)nbuf);
                bpos = __shfl(bpos, 0, WAVE);
                for (int j = lane; j < nbuf; j += WAVE)
                    cand[bpos + j] = buf[wid][j];
                nbuf = 0;
            }
            int pos = nbuf + incl - mycnt;
#pragma unroll
            for (int j = 0; j < 4; j++)
                if (m[j]) buf[wid][pos++] = r0 + j;
            nbuf += tot;
        }
        if (__all(q >= nq)) break;
    }
    if (nbuf) {
        long long bpos = 0;
        if (lane == 0)
            bpos = (long long)atomicAdd((unsigned long long *)ncand,
                                        (unsigned long long)nbuf);
        bpos = __shfl(bpos, 0, WAVE);
        for (int j = lane; j < nbuf; j += WAVE)
            cand[bpos + j] = buf[wid][j];
    }
}

__global__ void k_v3(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     int64_t *cand, int64_t *ncand)
{
    int64_t nq = n / 4;
    const int4 *d4 = (const int4 *)d;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int lane = (int)(threadIdx.x % WAVE);
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; q < nq;
         q += stride) {
        int4 v = d4[q];
        bool m[4] = {v.x < c, v.y < c, v.z < c, v.w < c};
#pragma unroll
        for (int j = 0; j < 4; j++) {
            unsigned long long mask = __ballot(m[j]);
            if (!mask) continue;
            int leader = __ffsll((long long)mask) - 1;
            int rank = __popcll(mask & ((1ull << lane) - 1ull));
            long long base = 0;
            if (lane == leader)
                base = (long long)atomicAdd((unsigned long long *)ncand,
                                            (unsigned long long)__popcll(mask));
            base = __shfl(base, leader, WAVE);
            if (m[j]) cand[base + rank] = q * 4 + j;
        }
    }
}


/* v4: BUF=2048 (64KB LDS/block, halves flush atomics, 2 blocks/CU) */
__global__ void k_v4(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     int64_t *cand, int64_t *ncand)
{
    const int BUF = 2048;
    __shared__ int64_t buf[256 / WAVE][BUF];
    int wid = (int)(threadIdx.x / WAVE), lane = (int)(threadIdx.x % WAVE);
    int nbuf = 0;
    int64_t nq = n / 4;
    const int4 *d4 = (const int4 *)d;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;;
         q += stride) {
        bool m[4] = {false, false, false, false};
        int64_t r0 = q * 4;
        if (q < nq) {
            int4 v = d4[q];
            m[0] = v.x < c; m[1] = v.y < c; m[2] = v.z < c; m[3] = v.w < c;
        }
        int mycnt = m[0] + m[1] + m[2] + m[3];
        int incl = mycnt;
        for (int off = 1; off < WAVE; off <<= 1) {
            int up = __shfl_up(incl, off, WAVE);
            if (lane >= off) incl += up;
        }
        int tot = __shfl(incl, WAVE - 1, WAVE);
        if (tot) {
            if (nbuf + tot > BUF) {
                long long bpos = 0;
                if (lane == 0)
                    bpos = (long long)atomicAdd((unsigned long long *)ncand,
                                                (unsigned long long)nbuf);
                bpos = __shfl(bpos, 0, WAVE);
                for (int j = lane; j < nbuf; j += WAVE)
                    cand[bpos + j] = buf[wid][j];
                nbuf = 0;
            }
            int pos = nbuf + incl - mycnt;
#pragma unroll
            for (int j = 0; j < 4; j++)
                if (m[j]) buf[wid][pos++] = r0 + j;
            nbuf += tot;
        }
        if (__all(q >= nq)) break;
    }
    if (nbuf) {
        long long bpos = 0;
        if (lane == 0)
            bpos = (long long)atomicAdd((unsigned long long *)ncand,
                                        (unsigned long long)nbuf);
        bpos = __shfl(bpos, 0, WAVE);
        for (int j = lane; j < nbuf; j += WAVE)
            cand[bpos + j] = buf[wid][j];
    }
}

/* v5: register-mask two-phase, one atomic per BLOCK. Fixed 1024 rows/thread
 * window per block (16 u64 mask words); phase 1 computes masks + counts,
 * block reduce + one atomicAdd, phase 2 replays from registers. */
__global__ void k_v5(const int32_t *__restrict__ d, int64_t n, int32_t c,
                     int64_t *cand, int64_t *ncand)
{
    const int MW = 16; /* 1024 rows per thread */
    unsigned long long mask[MW];
    int64_t win = (int64_t)blockDim.x * 1024;
    const int4 *d4 = (const int4 *)d;
    __shared__ int64_t tcnt[256];
    __shared__ int64_t tbase[257];
    for (int64_t w0 = (int64_t)blockIdx.x * win; w0 < n;
         w0 += (int64_t)gridDim.x * win) {
        int my = 0;
#pragma unroll
        for (int mwi = 0; mwi < MW; mwi++) mask[mwi] = 0;
        for (int it = 0; it < 256; it++) {
            int64_t q = (w0 >> 2) + (int64_t)it * blockDim.x + threadIdx.x;
            if (q * 4 >= n) break;
            int4 v = d4[q];
            unsigned long long mm = (unsigned long long)(v.x < c) |
                                    ((unsigned long long)(v.y < c) << 1) |
                                    ((unsigned long long)(v.z < c) << 2) |
                                    ((unsigned long long)(v.w < c) << 3);
            mask[it >> 4] |= mm << ((it & 15) * 4);
            my += __popcll(mm);
        }
        tcnt[threadIdx.x] = my;
        __syncthreads();
        if (threadIdx.x == 0) {
            int64_t tot = 0;
            for (int t = 0; t < (int)blockDim.x; t++) {
                tbase[t] = tot;
                tot += tcnt[t];
            }
            tbase[256] = tot ? (int64_t)atomicAdd((unsigned long long *)ncand,
                                                  (unsigned long long)tot)
                             : 0;
        }
        __syncthreads();
        int64_t pos = tbase[256] + tbase[threadIdx.x];
#pragma unroll
        for (int mwi = 0; mwi < MW; mwi++) {
            unsigned long long mm = mask[mwi];
            while (mm) {
                int b = __ffsll((long long)mm) - 1;
                mm &= mm - 1;
                int it = mwi * 16 + (b >> 2);
                int64_t q = (w0 >> 2) + (int64_t)it * blockDim.x + threadIdx.x;
                cand[pos++] = q * 4 + (b & 3);
            }
        }
        __syncthreads();
    }
}

/* v6: tile-staged compaction — the scatter_ab tile-sort pattern at one
 * bucket. Per 8192-row tile (2 sweeps of 4096, int4 loads, 4 rows/thread):
 * wave shfl prefix -> 16 wave totals scanned by thread 0 -> stable ranks
 * into an LDS stage -> ONE cursor reservation per tile -> linear write-out
 * with consecutive lanes on consecutive addresses (v2/v4/v5 all write
 * per-thread or per-wave runs, strided across the wave — the hypothesis is
 * that THAT, not the atomics, is the 1.3 TB/s plateau). Survivor ids stay
 * ascending within a tile. */
#define V6T 8192
__global__ __launch_bounds__(1024) void k_v6(const int32_t *__restrict__ d,
                                             int64_t n, int32_t c,
                                             int64_t *cand, int64_t *ncand)
{
    __shared__ int64_t stage[V6T];
    __shared__ int wtot[16];
    __shared__ int woff[16];
    __shared__ int sweepbase;
    __shared__ long long gbase;
    int wid = (int)(threadIdx.x / WAVE), lane = (int)(threadIdx.x % WAVE);
    const int4 *d4 = (const int4 *)d;
    int64_t ntiles = (n + V6T - 1) / V6T;
    for (int64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
        int64_t lo = t * (int64_t)V6T;
        int64_t hi = lo + V6T < n ? lo + V6T : n;
        if (threadIdx.x == 0) sweepbase = 0;
        __syncthreads();
        for (int64_t s0 = lo; s0 < hi; s0 += 4096) {
            int64_t r0 = s0 + 4 * (int64_t)threadIdx.x;
            bool m[4] = {false, false, false, false};
            int mycnt = 0;
            if (r0 + 3 < n) {
                int4 v = d4[r0 / 4];
                m[0] = v.x < c; m[1] = v.y < c;
                m[2] = v.z < c; m[3] = v.w < c;
                mycnt = m[0] + m[1] + m[2] + m[3];
            } else {
                for (int j = 0; j < 4 && r0 + j < n; j++) {
                    m[j] = d[r0 + j] < c;
                    mycnt += m[j];
                }
            }
            int incl = mycnt;
            for (int off = 1; off < WAVE; off <<= 1) {
                int up = __shfl_up(incl, off, WAVE);
                if (lane >= off) incl += up;
            }
            if (lane == WAVE - 1) wtot[wid] = incl;
            __syncthreads();
            if (threadIdx.x == 0) {
                int acc = sweepbase;
                for (int w = 0; w < 16; w++) {
                    woff[w] = acc;
                    acc += wtot[w];
                }
                sweepbase = acc;
            }
            __syncthreads();
            int pos = woff[wid] + incl - mycnt;
#pragma unroll
            for (int j = 0; j < 4; j++)
                if (m[j]) stage[pos++] = r0 + j;
            __syncthreads();
        }
        int tot = sweepbase;
        if (threadIdx.x == 0)
            gbase = tot ? (long long)atomicAdd((unsigned long long *)ncand,
                                               (unsigned long long)tot)
                        : 0;
        __syncthreads();
        for (int p = threadIdx.x; p < tot; p += blockDim.x)
            cand[gbase + p] = stage[p];
        __syncthreads();
    }
}

extern "C" int run(int64_t n, int reps)
{
    int32_t *d;
    int64_t *cand, *ncand;
    unsigned long long *out;
    hipMalloc(&d, n * 4);
    hipMalloc(&cand, n * 8);
    hipMalloc(&ncand, 8);
    hipMalloc(&out, 8);
    hipMemset(d, 0x11, n * 4); /* ~49% pass vs cutoff below */
    int32_t cutoff = 0x11111112;
    /* fill with pseudo-random via simple kernel-free trick: memset pattern is
     * constant; instead use alternating via a second memset on half */
    hipMemset((char *)d, 0x22, n * 2); /* first half fails, second passes */
    dim3 grid(2048), blk(256);
    hipDeviceSynchronize();
    for (int v = 0; v < 7; v++) {
        hipEvent_t e0, e1;
        hipEventCreate(&e0); hipEventCreate(&e1);
        float best = 1e9f;
        for (int r = 0; r < reps; r++) {
            hipMemset(ncand, 0, 8);
            hipEventRecord(e0);
            if (v == 0) hipLaunchKernelGGL(k_v0, grid, blk, 0, 0, d, n, cutoff, out);
            if (v == 1) hipLaunchKernelGGL(k_v1, grid, blk, 0, 0, d, n, cutoff, out);
            if (v == 2) hipLaunchKernelGGL(k_v2, grid, blk, 0, 0, d, n, cutoff, cand, ncand);
            if (v == 3) hipLaunchKernelGGL(k_v3, grid, blk, 0, 0, d, n, cutoff, cand, ncand);
            if (v == 4) hipLaunchKernelGGL(k_v4, grid, blk, 0, 0, d, n, cutoff, cand, ncand);
            if (v == 5) hipLaunchKernelGGL(k_v5, dim3((n + 256*1024 - 1) / (256*1024)), blk, 0, 0, d, n, cutoff, cand, ncand);
            if (v == 6) hipLaunchKernelGGL(k_v6, dim3(2048), dim3(1024), 0, 0, d, n, cutoff, cand, ncand);
            hipEventRecord(e1);
            hipEventSynchronize(e1);
            float ms; hipEventElapsedTime(&ms, e0, e1);
            if (ms < best) best = ms;
        }
        printf("variant %d: %.3f ms  (%.0f GB/s read)\n", v, best,
               n * 4.0 / best / 1e6);
    }
    hipDeviceSynchronize();
    return 0;
}

int main() { return run(150000000, 6); }
