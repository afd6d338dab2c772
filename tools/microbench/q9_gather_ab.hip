/*
 * q9_gather_ab.hip — component A/B for the Q9 fused-probe kernel's bound.
 * Measures, over the same 600M-row partkey/orderkey streams as Q9 SF100:
 *   A  stream-only       : read pk+ok (16 B/row), reduce
 *   B  A + bitmap gather : + random 8B lookup in the 2.5 MB part bitmap
 *   C  B + dtab lookup   : + 4B date lookup for the ~6% surviving rows
 *   D  C + payload       : + conditional ep/dc loads + 7-year accumulate
 *                          (= the product kernel's full work)
 * The step where the time jumps is the binding resource. Build:
 *   hipcc -O3 --offload-arch=gfx950 -o q9_gather_ab q9_gather_ab.hip
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include "../../oracle/otbx_gen.h"

#define CHK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

__global__ void k_gen(int64_t *pk, int64_t *ok, double *ep, double *dc,
                      int64_t n, int64_t nparts)
{
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += stride) {
        pk[i] = otbx_li_partkey(OTBX_DEFAULT_SEED, i, nparts);
        ok[i] = otbx_li_orderkey(i);
        ep[i] = otbx_li_extendedprice(OTBX_DEFAULT_SEED, i);
        dc[i] = otbx_li_discount(OTBX_DEFAULT_SEED, i);
    }
}

__global__ void k_bitmap(unsigned long long *bm, int64_t nparts)
{
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < nparts; i += stride)
        if (otbx_part_type(OTBX_DEFAULT_SEED, (uint64_t)i) % 17 == 0)
            atomicOr(&bm[i >> 6], 1ull << (i & 63));
}

__global__ void k_dtab(int32_t *dt, int64_t norders)
{
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < norders; i += stride)
        dt[i] = otbx_ord_orderdate(OTBX_DEFAULT_SEED, (uint64_t)i);
}

template <int MODE>   /* 0=A 1=B 2=C 3=D */
__global__ void k_probe(const int64_t *__restrict__ pk,
                        const int64_t *__restrict__ ok,
                        const double *__restrict__ ep,
                        const double *__restrict__ dc,
                        const unsigned long long *__restrict__ bm,
                        const int32_t *__restrict__ dt, int64_t n,
                        int64_t nparts, int64_t norders,
                        unsigned long long *out, double *outd)
{
    const longlong2 *pk2 = (const longlong2 *)pk;
    const longlong2 *ok2 = (const longlong2 *)ok;
    unsigned long long local = 0;
    double acc[7];
    unsigned cnt7[7];
#pragma unroll
    for (int y = 0; y < 7; y++) { acc[y] = 0; cnt7[y] = 0; }
    int64_t nq = n / 4;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; q < nq;
         q += stride) {
        longlong2 pa = pk2[2 * q], pb = pk2[2 * q + 1];
        longlong2 oa = ok2[2 * q], ob2 = ok2[2 * q + 1];
        int64_t pks[4] = {pa.x, pa.y, pb.x, pb.y};
        int64_t oks[4] = {oa.x, oa.y, ob2.x, ob2.y};
#pragma unroll
        for (int j = 0; j < 4; j++) {
            if (MODE == 0) { local += (unsigned long long)(pks[j] + oks[j]); continue; }
            int64_t pidx = pks[j] - 1;
            bool m = pidx >= 0 && pidx < nparts &&
                     ((bm[pidx >> 6] >> (pidx & 63)) & 1ull);
            if (MODE == 1) { local += m; continue; }
            int64_t oidx = oks[j] - 1;
            bool okb = m && oidx >= 0 && oidx < norders;
            int32_t date = dt[okb ? oidx : 0];
            if (!okb || date == 0) continue;
            if (MODE == 2) { local += (unsigned)date; continue; }
            int32_t y = otbx_year_of_day(date);
            int64_t i = 4 * q + j;
            double rev = ep[i] * (1.0 - dc[i]);
#pragma unroll
            for (int yy = 0; yy < 7; yy++) {
                acc[yy] += (yy == y) ? rev : 0.0;
                cnt7[yy] += (yy == y);
            }
        }
    }
    double tot = 0;
#pragma unroll
    for (int y = 0; y < 7; y++) { tot += acc[y]; local += cnt7[y]; }
    if (local) atomicAdd(out, local);
    if (tot != 0.0) atomicAdd(outd, tot);
}

int main()
{
    const int64_t N = 600000000, NP = N / 30, NO = N / 4;
    int64_t *pk, *ok; double *ep, *dc;
    unsigned long long *bm; int32_t *dt;
    unsigned long long *out; double *outd;
    CHK(hipMalloc(&pk, N * 8)); CHK(hipMalloc(&ok, N * 8));
    CHK(hipMalloc(&ep, N * 8)); CHK(hipMalloc(&dc, N * 8));
    CHK(hipMalloc(&bm, NP / 8 + 64)); CHK(hipMalloc(&dt, NO * 4));
    CHK(hipMalloc(&out, 8)); CHK(hipMalloc(&outd, 8));
    CHK(hipMemset(bm, 0, NP / 8 + 64));
    hipLaunchKernelGGL(k_gen, dim3(2048), dim3(256), 0, 0, pk, ok, ep, dc, N, NP);
    hipLaunchKernelGGL(k_bitmap, dim3(2048), dim3(256), 0, 0, bm, NP);
    hipLaunchKernelGGL(k_dtab, dim3(2048), dim3(256), 0, 0, dt, NO);
    CHK(hipDeviceSynchronize());
    hipEvent_t e0, e1; CHK(hipEventCreate(&e0)); CHK(hipEventCreate(&e1));
    const char *names[4] = {"A stream16B", "B +bitmap", "C +dtab", "D +payload"};
    for (int mode = 0; mode < 4; mode++) {
        for (int rep = 0; rep < 4; rep++) {
            CHK(hipMemset(out, 0, 8)); CHK(hipMemset(outd, 0, 8));
            CHK(hipEventRecord(e0));
            switch (mode) {
            case 0: hipLaunchKernelGGL(k_probe<0>, dim3(2048), dim3(256), 0, 0, pk, ok, ep, dc, bm, dt, N, NP, NO, out, outd); break;
            case 1: hipLaunchKernelGGL(k_probe<1>, dim3(2048), dim3(256), 0, 0, pk, ok, ep, dc, bm, dt, N, NP, NO, out, outd); break;
            case 2: hipLaunchKernelGGL(k_probe<2>, dim3(2048), dim3(256), 0, 0, pk, ok, ep, dc, bm, dt, N, NP, NO, out, outd); break;
            case 3: hipLaunchKernelGGL(k_probe<3>, dim3(2048), dim3(256), 0, 0, pk, ok, ep, dc, bm, dt, N, NP, NO, out, outd); break;
            }
            CHK(hipEventRecord(e1)); CHK(hipEventSynchronize(e1));
            float ms; CHK(hipEventElapsedTime(&ms, e0, e1));
            unsigned long long h; CHK(hipMemcpy(&h, out, 8, hipMemcpyDeviceToHost));
            if (rep == 3)
                printf("%-12s %7.3f ms  %6.1f Grows/s  check=%llu\n",
                       names[mode], ms, N / ms / 1e6, h);
        }
    }
    return 0;
}
