/* tools/kernel_ab.hip — standalone A/B harness for newview INNER_INNER
 * variants on gfx950 (dev tool; not part of the product library).
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -ffp-contract=off \
 *        tools/kernel_ab.hip -o gpurun_out/kernel_ab
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

#define N_SITES 1000000L
#define BLOCK 256

#define TWOTOTHE256 \
  115792089237316195423570985008687907853269984665640564039457584007913129639936.0
#define MINLIK (1.0 / TWOTOTHE256)

__device__ __forceinline__ void nv_core(const double4 xl, const double4 xr,
                                        const double *sL, const double *sR,
                                        const double *sEV, int cat,
                                        double &a0, double &a1, double &a2,
                                        double &a3) {
  double u1[4], u2[4];
#pragma unroll
  for (int l = 0; l < 4; l++) {
    const double *pl = &sL[cat * 16 + l * 4];
    const double *pr = &sR[cat * 16 + l * 4];
    u1[l] = (xl.x * pl[0] + xl.y * pl[1]) + (xl.z * pl[2] + xl.w * pl[3]);
    u2[l] = (xr.x * pr[0] + xr.y * pr[1]) + (xr.z * pr[2] + xr.w * pr[3]);
  }
  a0 = a1 = a2 = a3 = 0;
#pragma unroll
  for (int l = 0; l < 4; l++) {
    const double t = u1[l] * u2[l];
    a0 += t * sEV[l * 4 + 0];
    a1 += t * sEV[l * 4 + 1];
    a2 += t * sEV[l * 4 + 2];
    a3 += t * sEV[l * 4 + 3];
  }
}

/* V0 — production kernel shape: grid-stride capped at 4096 blocks */
template <int MODE>
__global__ __launch_bounds__(BLOCK) void k_ii(
    const double *__restrict__ x1, const double *__restrict__ x2,
    double *__restrict__ x3, const double *__restrict__ P,
    const double *__restrict__ EV, const int *__restrict__ wgt, long n,
    unsigned int *__restrict__ scalerInc) {
  __shared__ double sL[64], sR[64], sEV[16];
  const int tid = threadIdx.x;
  if (tid < 64) {
    sL[tid] = P[tid];
    sR[tid] = P[64 + tid];
  }
  if (tid < 16) sEV[tid] = EV[tid];
  __syncthreads();
  const long units = n * 4;
  const int lane = tid & 63;
  for (long idx = (long)blockIdx.x * BLOCK + tid; idx < units;
       idx += (long)gridDim.x * BLOCK) {
    const double4 xl = *reinterpret_cast<const double4 *>(&x1[idx * 4]);
    const double4 xr = *reinterpret_cast<const double4 *>(&x2[idx * 4]);
    const int cat = (int)(idx & 3);
    double a0, a1, a2, a3;
    nv_core(xl, xr, sL, sR, sEV, cat, a0, a1, a2, a3);
    const bool small = (fabs(a0) < MINLIK) & (fabs(a1) < MINLIK) &
                       (fabs(a2) < MINLIK) & (fabs(a3) < MINLIK);
    const unsigned long long m = __ballot(small);
    if (((m >> (lane & ~3)) & 0xFULL) == 0xFULL) {
      a0 *= TWOTOTHE256; a1 *= TWOTOTHE256; a2 *= TWOTOTHE256;
      a3 *= TWOTOTHE256;
      if ((lane & 3) == 0) atomicAdd(scalerInc, (unsigned)wgt[idx >> 2]);
    }
    if (MODE == 2) { /* nontemporal store */
      __builtin_nontemporal_store(a0, &x3[idx * 4 + 0]);
      __builtin_nontemporal_store(a1, &x3[idx * 4 + 1]);
      __builtin_nontemporal_store(a2, &x3[idx * 4 + 2]);
      __builtin_nontemporal_store(a3, &x3[idx * 4 + 3]);
    } else {
      *reinterpret_cast<double4 *>(&x3[idx * 4]) =
          make_double4(a0, a1, a2, a3);
    }
  }
}

/* V3 — two units per thread, site-pair per lane (64 B/lane) */
__global__ __launch_bounds__(BLOCK) void k_ii_x2(
    const double *__restrict__ x1, const double *__restrict__ x2,
    double *__restrict__ x3, const double *__restrict__ P,
    const double *__restrict__ EV, const int *__restrict__ wgt, long n,
    unsigned int *__restrict__ scalerInc) {
  __shared__ double sL[64], sR[64], sEV[16];
  const int tid = threadIdx.x;
  if (tid < 64) {
    sL[tid] = P[tid];
    sR[tid] = P[64 + tid];
  }
  if (tid < 16) sEV[tid] = EV[tid];
  __syncthreads();
  const long pairs = n * 2; /* each thread: 2 consecutive (site,cat) units */
  const int lane = tid & 63;
  for (long pi = (long)blockIdx.x * BLOCK + tid; pi < pairs;
       pi += (long)gridDim.x * BLOCK) {
    const long idx = pi * 2;
    const double4 xl0 = *reinterpret_cast<const double4 *>(&x1[idx * 4]);
    const double4 xr0 = *reinterpret_cast<const double4 *>(&x2[idx * 4]);
    const double4 xl1 = *reinterpret_cast<const double4 *>(&x1[idx * 4 + 4]);
    const double4 xr1 = *reinterpret_cast<const double4 *>(&x2[idx * 4 + 4]);
    const int cat0 = (int)(idx & 3);
    double b0, b1, b2, b3, c0, c1, c2, c3;
    nv_core(xl0, xr0, sL, sR, sEV, cat0, b0, b1, b2, b3);
    nv_core(xl1, xr1, sL, sR, sEV, cat0 + 1, c0, c1, c2, c3);
    const bool small = (fabs(b0) < MINLIK) & (fabs(b1) < MINLIK) &
                       (fabs(b2) < MINLIK) & (fabs(b3) < MINLIK) &
                       (fabs(c0) < MINLIK) & (fabs(c1) < MINLIK) &
                       (fabs(c2) < MINLIK) & (fabs(c3) < MINLIK);
    const unsigned long long m = __ballot(small);
    if (((m >> (lane & ~1)) & 0x3ULL) == 0x3ULL) {
      b0 *= TWOTOTHE256; b1 *= TWOTOTHE256; b2 *= TWOTOTHE256;
      b3 *= TWOTOTHE256; c0 *= TWOTOTHE256; c1 *= TWOTOTHE256;
      c2 *= TWOTOTHE256; c3 *= TWOTOTHE256;
      if ((lane & 1) == 0) atomicAdd(scalerInc, (unsigned)wgt[pi >> 1]);
    }
    *reinterpret_cast<double4 *>(&x3[idx * 4]) = make_double4(b0, b1, b2, b3);
    *reinterpret_cast<double4 *>(&x3[idx * 4 + 4]) =
        make_double4(c0, c1, c2, c3);
  }
}

/* pure-stream ceiling probe: same bytes, no math */
__global__ __launch_bounds__(BLOCK) void k_copy(
    const double *__restrict__ x1, const double *__restrict__ x2,
    double *__restrict__ x3, long n) {
  const long units = n * 4;
  for (long idx = (long)blockIdx.x * BLOCK + threadIdx.x; idx < units;
       idx += (long)gridDim.x * BLOCK) {
    const double4 a = *reinterpret_cast<const double4 *>(&x1[idx * 4]);
    const double4 b = *reinterpret_cast<const double4 *>(&x2[idx * 4]);
    *reinterpret_cast<double4 *>(&x3[idx * 4]) =
        make_double4(a.x + b.x, a.y + b.y, a.z + b.z, a.w + b.w);
  }
}

#define CHK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("ERR %s %s\n", #x, hipGetErrorString(e)); exit(1); } } while (0)

template <typename F>
double timeit(F f, int reps) {
  hipEvent_t a, b;
  CHK(hipEventCreate(&a));
  CHK(hipEventCreate(&b));
  f(); /* warm */
  CHK(hipDeviceSynchronize());
  CHK(hipEventRecord(a, 0));
  for (int i = 0; i < reps; i++) f();
  CHK(hipEventRecord(b, 0));
  CHK(hipEventSynchronize(b));
  float ms;
  CHK(hipEventElapsedTime(&ms, a, b));
  return ms / reps;
}

extern "C" void run_appended_probes();

int main() {
  const long n = N_SITES;
  double *x1, *x2, *x3, *P, *EV;
  int *wgt;
  unsigned int *inc;
  CHK(hipMalloc(&x1, n * 16 * 8));
  CHK(hipMalloc(&x2, n * 16 * 8));
  CHK(hipMalloc(&x3, n * 16 * 8));
  CHK(hipMalloc(&P, 128 * 8));
  CHK(hipMalloc(&EV, 16 * 8));
  CHK(hipMalloc(&wgt, n * 4));
  CHK(hipMalloc(&inc, 4));
  /* init */
  double *h = (double *)malloc(n * 16 * 8);
  for (long i = 0; i < n * 16; i++) h[i] = 0.1 + (i % 97) * 0.009;
  CHK(hipMemcpy(x1, h, n * 16 * 8, hipMemcpyHostToDevice));
  CHK(hipMemcpy(x2, h, n * 16 * 8, hipMemcpyHostToDevice));
  double hp[128], hev[16];
  for (int i = 0; i < 128; i++) hp[i] = 0.2 + i * 0.003;
  for (int i = 0; i < 16; i++) hev[i] = 0.3 + i * 0.01;
  CHK(hipMemcpy(P, hp, sizeof(hp), hipMemcpyHostToDevice));
  CHK(hipMemcpy(EV, hev, sizeof(hev), hipMemcpyHostToDevice));
  CHK(hipMemset(wgt, 1, n * 4));
  CHK(hipMemset(inc, 0, 4));

  const long units = n * 4;
  struct Cfg { const char *name; int grid; int mode; } cfgs[] = {
      {"V0 gridstride4096", 4096, 0},
      {"V0 grid2048", 2048, 0},
      {"V0 grid8192", 8192, 0},
      {"V0 exactgrid", (int)((units + BLOCK - 1) / BLOCK), 0},
      {"V2 nontemporal4096", 4096, 2},
      {"V2 nontemporal8192", 8192, 2},
  };
  printf("DNA II newview, %ld sites, 388 B/site algorithmic\n", n);
  for (auto &c : cfgs) {
    double ms;
    if (c.mode == 0)
      ms = timeit([&] {
        hipLaunchKernelGGL(k_ii<0>, dim3(c.grid), dim3(BLOCK), 0, 0, x1, x2,
                           x3, P, EV, wgt, n, inc);
      }, 20);
    else
      ms = timeit([&] {
        hipLaunchKernelGGL(k_ii<2>, dim3(c.grid), dim3(BLOCK), 0, 0, x1, x2,
                           x3, P, EV, wgt, n, inc);
      }, 20);
    printf("%-22s %8.2f us  %6.2f TB/s\n", c.name, ms * 1e3,
           388.0 * n / (ms * 1e-3) / 1e12);
  }
  for (int grid : {2048, 4096, 8192, (int)((n * 2 + BLOCK - 1) / BLOCK)}) {
    double ms = timeit([&] {
      hipLaunchKernelGGL(k_ii_x2, dim3(grid), dim3(BLOCK), 0, 0, x1, x2, x3,
                         P, EV, wgt, n, inc);
    }, 20);
    printf("V3 x2/thread g%-7d %8.2f us  %6.2f TB/s\n", grid, ms * 1e3,
           388.0 * n / (ms * 1e-3) / 1e12);
  }
  {
    double ms = timeit([&] {
      hipLaunchKernelGGL(k_copy, dim3(4096), dim3(BLOCK), 0, 0, x1, x2, x3,
                         n);
    }, 20);
    printf("%-22s %8.2f us  %6.2f TB/s (stream ceiling, 384 B/site)\n",
           "copy-add ceiling", ms * 1e3, 384.0 * n / (ms * 1e-3) / 1e12);
  }
  run_appended_probes();
  return 0;
}

/* ===== appended probes: protein II variants + f64 MFMA rate ============= */

/* P1: production protein shape (interleaved cats, LDS P/EV) */
__global__ __launch_bounds__(BLOCK) void kp_base(
    const double *__restrict__ x1, const double *__restrict__ x2,
    double *__restrict__ x3, const double *__restrict__ P,
    const double *__restrict__ EV, long n) {
  __shared__ double sL[1600], sR[1600], sEV[400];
  const int tid = threadIdx.x;
  for (int j = tid; j < 1600; j += BLOCK) { sL[j] = P[j]; sR[j] = P[1600+j]; }
  for (int j = tid; j < 400; j += BLOCK) sEV[j] = EV[j];
  __syncthreads();
  const long units = n * 4;
  for (long idx = (long)blockIdx.x * BLOCK + tid; idx < units;
       idx += (long)gridDim.x * BLOCK) {
    const int cat = (int)(idx & 3);
    double xl[20], xr[20], acc[20];
#pragma unroll
    for (int s2 = 0; s2 < 20; s2 += 4) {
      const double4 a = *reinterpret_cast<const double4 *>(&x1[idx*20+s2]);
      const double4 b = *reinterpret_cast<const double4 *>(&x2[idx*20+s2]);
      xl[s2]=a.x; xl[s2+1]=a.y; xl[s2+2]=a.z; xl[s2+3]=a.w;
      xr[s2]=b.x; xr[s2+1]=b.y; xr[s2+2]=b.z; xr[s2+3]=b.w;
    }
#pragma unroll
    for (int s2 = 0; s2 < 20; s2++) acc[s2] = 0;
    for (int l = 0; l < 20; l++) {
      double t0=0,t1=0,t2=0,t3=0, r0=0,r1=0,r2=0,r3=0;
#pragma unroll
      for (int c = 0; c < 20; c += 4) {
        t0 += xl[c]*sL[cat*400+l*20+c];   t1 += xl[c+1]*sL[cat*400+l*20+c+1];
        t2 += xl[c+2]*sL[cat*400+l*20+c+2]; t3 += xl[c+3]*sL[cat*400+l*20+c+3];
        r0 += xr[c]*sR[cat*400+l*20+c];   r1 += xr[c+1]*sR[cat*400+l*20+c+1];
        r2 += xr[c+2]*sR[cat*400+l*20+c+2]; r3 += xr[c+3]*sR[cat*400+l*20+c+3];
      }
      const double t = ((t0+t1)+(t2+t3)) * ((r0+r1)+(r2+r3));
#pragma unroll
      for (int s2 = 0; s2 < 20; s2++) acc[s2] += t * sEV[l*20+s2];
    }
#pragma unroll
    for (int s2 = 0; s2 < 20; s2 += 4)
      *reinterpret_cast<double4 *>(&x3[idx*20+s2]) =
          make_double4(acc[s2], acc[s2+1], acc[s2+2], acc[s2+3]);
  }
}

/* P2: cat-uniform wave -> P/EV via uniform (scalar) global loads */
__global__ __launch_bounds__(BLOCK) void kp_scalar(
    const double *__restrict__ x1, const double *__restrict__ x2,
    double *__restrict__ x3, const double *__restrict__ P,
    const double *__restrict__ EV, long n) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int cat = wave & 3; /* uniform per wave */
  const long nw = (long)gridDim.x * (BLOCK / 64 / 4); /* site-groups of 64 */
  const long wg = (long)blockIdx.x * (BLOCK / 64 / 4) + (wave >> 2);
  const double *Lc = P + cat * 400;
  const double *Rc = P + 1600 + cat * 400;
  for (long base = wg * 64; base < n; base += nw * 64) {
    const long site = base + lane;
    if (site >= n) break;
    const long idx = site * 4 + cat;
    double xl[20], xr[20], acc[20];
#pragma unroll
    for (int s2 = 0; s2 < 20; s2 += 4) {
      const double4 a = *reinterpret_cast<const double4 *>(&x1[idx*20+s2]);
      const double4 b = *reinterpret_cast<const double4 *>(&x2[idx*20+s2]);
      xl[s2]=a.x; xl[s2+1]=a.y; xl[s2+2]=a.z; xl[s2+3]=a.w;
      xr[s2]=b.x; xr[s2+1]=b.y; xr[s2+2]=b.z; xr[s2+3]=b.w;
    }
#pragma unroll
    for (int s2 = 0; s2 < 20; s2++) acc[s2] = 0;
    for (int l = 0; l < 20; l++) {
      double t0=0,t1=0,t2=0,t3=0, r0=0,r1=0,r2=0,r3=0;
#pragma unroll
      for (int c = 0; c < 20; c += 4) {
        t0 += xl[c]*Lc[l*20+c];     t1 += xl[c+1]*Lc[l*20+c+1];
        t2 += xl[c+2]*Lc[l*20+c+2]; t3 += xl[c+3]*Lc[l*20+c+3];
        r0 += xr[c]*Rc[l*20+c];     r1 += xr[c+1]*Rc[l*20+c+1];
        r2 += xr[c+2]*Rc[l*20+c+2]; r3 += xr[c+3]*Rc[l*20+c+3];
      }
      const double t = ((t0+t1)+(t2+t3)) * ((r0+r1)+(r2+r3));
#pragma unroll
      for (int s2 = 0; s2 < 20; s2++) acc[s2] += t * EV[l*20+s2];
    }
#pragma unroll
    for (int s2 = 0; s2 < 20; s2 += 4)
      *reinterpret_cast<double4 *>(&x3[idx*20+s2]) =
          make_double4(acc[s2], acc[s2+1], acc[s2+2], acc[s2+3]);
  }
}

/* f64 MFMA 16x16x4 throughput probe */
typedef double d4v __attribute__((ext_vector_type(4)));
__global__ __launch_bounds__(256) void k_mfma64(double *out, double a0) {
  d4v c = {0, 0, 0, 0};
  double a = a0 + threadIdx.x, b = a0 * 1.0001 + threadIdx.x;
  d4v c1 = {1, 1, 1, 1}, c2 = {2, 2, 2, 2}, c3 = {3, 3, 3, 3};
  for (int i = 0; i < 2048; i++) {
    c = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
    c1 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c1, 0, 0, 0);
    c2 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c2, 0, 0, 0);
    c3 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c3, 0, 0, 0);
  }
  out[threadIdx.x + blockIdx.x * 256] = c[0] + c1[1] + c2[2] + c3[3];
}

/* f64 VALU FMA probe */
__global__ __launch_bounds__(256) void k_valu64(double *out, double a0) {
  double a = a0 + threadIdx.x, b = a0 * 1.0001;
  double c0=0,c1=1,c2=2,c3=3,c4=4,c5=5,c6=6,c7=7;
  for (int i = 0; i < 4096; i++) {
    c0 = fma(a, b, c0); c1 = fma(a, b, c1); c2 = fma(a, b, c2);
    c3 = fma(a, b, c3); c4 = fma(a, b, c4); c5 = fma(a, b, c5);
    c6 = fma(a, b, c6); c7 = fma(a, b, c7);
  }
  out[threadIdx.x + blockIdx.x * 256] = c0+c1+c2+c3+c4+c5+c6+c7;
}

extern "C" void run_appended_probes() {
  const long n = 200000;
  double *x1, *x2, *x3, *P, *EV;
  CHK(hipMalloc(&x1, n * 80 * 8));
  CHK(hipMalloc(&x2, n * 80 * 8));
  CHK(hipMalloc(&x3, n * 80 * 8));
  CHK(hipMalloc(&P, 3200 * 8));
  CHK(hipMalloc(&EV, 400 * 8));
  double *h = (double *)malloc(n * 80 * 8);
  for (long i = 0; i < n * 80; i++) h[i] = 0.1 + (i % 97) * 0.009;
  CHK(hipMemcpy(x1, h, n * 80 * 8, hipMemcpyHostToDevice));
  CHK(hipMemcpy(x2, h, n * 80 * 8, hipMemcpyHostToDevice));
  double hp[3200], hev[400];
  for (int i = 0; i < 3200; i++) hp[i] = 0.2 + (i % 53) * 0.003;
  for (int i = 0; i < 400; i++) hev[i] = 0.3 + (i % 31) * 0.01;
  CHK(hipMemcpy(P, hp, sizeof(hp), hipMemcpyHostToDevice));
  CHK(hipMemcpy(EV, hev, sizeof(hev), hipMemcpyHostToDevice));

  printf("\nprotein II (no scaling branch), %ld sites, 1924 B/site\n", n);
  for (int grid : {2048, 4096, 8192}) {
    double ms = timeit([&] {
      hipLaunchKernelGGL(kp_base, dim3(grid), dim3(BLOCK), 0, 0, x1, x2, x3,
                         P, EV, n);
    }, 10);
    printf("P1 lds base  g%-6d %8.1f us  %6.2f TB/s  %5.1f Gsites/s\n", grid,
           ms * 1e3, 1920.0 * n / (ms * 1e-3) / 1e12, n / (ms * 1e-3) / 1e9);
  }
  for (int grid : {2048, 4096, 8192, 12500}) {
    double ms = timeit([&] {
      hipLaunchKernelGGL(kp_scalar, dim3(grid), dim3(BLOCK), 0, 0, x1, x2,
                         x3, P, EV, n);
    }, 10);
    printf("P2 scalarP   g%-6d %8.1f us  %6.2f TB/s  %5.1f Gsites/s\n", grid,
           ms * 1e3, 1920.0 * n / (ms * 1e-3) / 1e12, n / (ms * 1e-3) / 1e9);
  }
  /* verify P2 == P1 bitwise */
  {
    double *x4;
    CHK(hipMalloc(&x4, n * 80 * 8));
    hipLaunchKernelGGL(kp_base, dim3(4096), dim3(BLOCK), 0, 0, x1, x2, x3, P, EV, n);
    hipLaunchKernelGGL(kp_scalar, dim3(4096), dim3(BLOCK), 0, 0, x1, x2, x4, P, EV, n);
    CHK(hipDeviceSynchronize());
    double *h3 = (double *)malloc(n * 80 * 8), *h4 = (double *)malloc(n * 80 * 8);
    CHK(hipMemcpy(h3, x3, n * 80 * 8, hipMemcpyDeviceToHost));
    CHK(hipMemcpy(h4, x4, n * 80 * 8, hipMemcpyDeviceToHost));
    long bad = 0;
    for (long i = 0; i < n * 80; i++) if (h3[i] != h4[i]) bad++;
    printf("P1 vs P2 bit-diff count: %ld\n", bad);
  }
  double *out;
  CHK(hipMalloc(&out, 8192 * 256 * 8));
  {
    double ms = timeit([&] {
      hipLaunchKernelGGL(k_mfma64, dim3(2048), dim3(256), 0, 0, out, 1.0001);
    }, 5);
    /* flops: grid*256 threads /64 lanes = waves; per wave iter: 4 mfma * 2048 flop */
    double waves = 2048.0 * 256 / 64;
    double fl = waves * 2048.0 * 4 * 2048;
    printf("f64 MFMA 16x16x4: %8.2f ms -> %6.1f TFLOP/s\n", ms, fl / (ms * 1e-3) / 1e12);
  }
  {
    double ms = timeit([&] {
      hipLaunchKernelGGL(k_valu64, dim3(2048), dim3(256), 0, 0, out, 1.0001);
    }, 5);
    double fl = 2048.0 * 256 * 4096.0 * 8 * 2;
    printf("f64 VALU fma:     %8.2f ms -> %6.1f TFLOP/s\n", ms, fl / (ms * 1e-3) / 1e12);
  }
}
