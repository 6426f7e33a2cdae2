// Microprobe: back-to-back v_mfma_i32_16x16x64_i8 issue rate by
// occupancy and accumulator count on gfx950 (round-2: the rs2 kernel's
// compute-only ablation sits 3.2x above the 20.4 cyc/instr rate derived
// from the 3944-TOPS peak — this pins the real per-wave issue rate at
// 1 wave/SIMD with a 256-register accumulator file, vs 4 waves/SIMD).
//
// Build: hipcc -O3 --offload-arch=gfx950 probe_mfma_rate.hip -o probe_mfma_rate
// Run on the GPU box: ./probe_mfma_rate

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <ctime>

typedef int v4i __attribute__((ext_vector_type(4)));

// NACC accumulators round-robin, ITER*NACC MFMAs per wave.
template <int NACC, int WPE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma_rate_kernel(const int* __restrict__ in, long* __restrict__ cyc,
                      int* __restrict__ sink, int iter) {
    v4i acc[NACC];
    v4i a, b;
    int lane = threadIdx.x & 63;
    a = v4i{in[lane], in[lane + 64], in[lane + 128], in[lane + 192]};
    b = v4i{in[lane + 1], in[lane + 65], in[lane + 129], in[lane + 193]};
#pragma unroll
    for (int i = 0; i < NACC; ++i)
        acc[i] = v4i{in[i], 0, 0, 0};
    __syncthreads();
    long t0 = __builtin_amdgcn_s_memtime();
    for (int it = 0; it < iter; ++it) {
#pragma unroll
        for (int i = 0; i < NACC; ++i)
            acc[i] = __builtin_amdgcn_mfma_i32_16x16x64_i8(a, b, acc[i],
                                                           0, 0, 0);
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int i = 0; i < NACC; ++i) s += acc[i][0] + acc[i][3];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = t1 - t0;
    }
}

// Kernel-pattern variant: 8x8 DIVERSE operand pairs (fa[ta], fb[tb])
// over 64 accumulators — exactly the rs2 burst's register pattern.
// Prices MFMA operand-diversity vs the fixed-operand chain above.
template <int WPE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma_pairs_kernel(const int* __restrict__ in, long* __restrict__ cyc,
                       int* __restrict__ sink, int iter) {
    v4i acc[8][8];
    v4i fa[8], fb[8];
    int lane = threadIdx.x & 63;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        fa[u] = v4i{in[lane + u], in[lane + u + 64], in[lane + u + 128],
                    in[lane + u + 192]};
        fb[u] = v4i{in[lane + u + 1], in[lane + u + 65],
                    in[lane + u + 129], in[lane + u + 193]};
    }
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 8; ++y) acc[x][y] = v4i{in[x], 0, 0, in[y]};
    __syncthreads();
    long t0 = __builtin_amdgcn_s_memtime();
    for (int it = 0; it < iter; ++it) {
#pragma unroll
        for (int ta = 0; ta < 8; ++ta)
#pragma unroll
            for (int tb = 0; tb < 8; ++tb)
                acc[ta][tb] = __builtin_amdgcn_mfma_i32_16x16x64_i8(
                    fa[ta], fb[tb], acc[ta][tb], 0, 0, 0);
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 8; ++y) s += acc[x][y][0] + acc[x][y][3];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = t1 - t0;
    }
}

// Kernel-convergence variants: reproduce the in-situ slowdown by adding
// the rs2 kernel's environment piece by piece.
//   LDSALLOC: declare the kernel's 69632-B LDS (occupancy via LDS cap)
//   OUTER:    20-tile outer loop with per-"slab" __syncthreads
template <int LDSALLOC, int OUTER, int WPE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma_pairs_env_kernel(const int* __restrict__ in,
                           long* __restrict__ cyc, int* __restrict__ sink,
                           int iter) {
    extern __shared__ signed char dyn_lds[];
    v4i acc[8][8];
    v4i fa[8], fb[8];
    int lane = threadIdx.x & 63;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        fa[u] = v4i{in[lane + u], in[lane + u + 64], in[lane + u + 128],
                    in[lane + u + 192]};
        fb[u] = v4i{in[lane + u + 1], in[lane + u + 65],
                    in[lane + u + 129], in[lane + u + 193]};
    }
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 8; ++y) acc[x][y] = v4i{in[x], 0, 0, in[y]};
    if (LDSALLOC && threadIdx.x == 0) dyn_lds[0] = (signed char)in[0];
    __syncthreads();
    long t0 = __builtin_amdgcn_s_memtime();
    for (int tile = 0; tile < (OUTER ? 20 : 1); ++tile) {
        for (int it = 0; it < iter; ++it) {
#pragma unroll
            for (int ta = 0; ta < 8; ++ta)
#pragma unroll
                for (int tb = 0; tb < 8; ++tb)
                    acc[ta][tb] = __builtin_amdgcn_mfma_i32_16x16x64_i8(
                        fa[ta], fb[tb], acc[ta][tb], 0, 0, 0);
            if (OUTER) __syncthreads();
        }
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 8; ++y) s += acc[x][y][0] + acc[x][y][3];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = (t1 - t0) / (OUTER ? 20 : 1);
    }
}

// rsm geometry: fa[8] x fb[4] -> 32 accumulators (128 regs) at
// 2 waves/SIMD — do co-resident waves hide the operand-switch stalls?
template <int WPE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma_pairs2_kernel(const int* __restrict__ in, long* __restrict__ cyc,
                        int* __restrict__ sink, int iter) {
    v4i acc[8][4];
    v4i fa[8], fb[4];
    int lane = threadIdx.x & 63;
#pragma unroll
    for (int u = 0; u < 8; ++u)
        fa[u] = v4i{in[lane + u], in[lane + u + 64], in[lane + u + 128],
                    in[lane + u + 192]};
#pragma unroll
    for (int u = 0; u < 4; ++u)
        fb[u] = v4i{in[lane + u + 1], in[lane + u + 65],
                    in[lane + u + 129], in[lane + u + 193]};
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 4; ++y) acc[x][y] = v4i{in[x], 0, 0, in[y]};
    __syncthreads();
    long t0 = __builtin_amdgcn_s_memtime();
    for (int it = 0; it < iter; ++it) {
#pragma unroll
        for (int ta = 0; ta < 8; ++ta)
#pragma unroll
            for (int tb = 0; tb < 4; ++tb)
                acc[ta][tb] = __builtin_amdgcn_mfma_i32_16x16x64_i8(
                    fa[ta], fb[tb], acc[ta][tb], 0, 0, 0);
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int x = 0; x < 8; ++x)
#pragma unroll
        for (int y = 0; y < 4; ++y) s += acc[x][y][0] + acc[x][y][3];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = t1 - t0;
    }
}

// v_mfma_i32_32x32x32_i8: 4x the ops per instruction, so any per-
// instruction operand-switch stall amortizes 4x.  Fixed vs diverse.
typedef int v16i __attribute__((ext_vector_type(16)));

template <int WPE, int DIVERSE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma32_kernel(const int* __restrict__ in, long* __restrict__ cyc,
                   int* __restrict__ sink, int iter) {
    v16i acc[4][4];
    v4i fa[4], fb[4];
    int lane = threadIdx.x & 63;
#pragma unroll
    for (int u = 0; u < 4; ++u) {
        fa[u] = v4i{in[lane + u], in[lane + u + 64], in[lane + u + 128],
                    in[lane + u + 192]};
        fb[u] = v4i{in[lane + u + 1], in[lane + u + 65],
                    in[lane + u + 129], in[lane + u + 193]};
    }
#pragma unroll
    for (int x = 0; x < 4; ++x)
#pragma unroll
        for (int y = 0; y < 4; ++y)
#pragma unroll
            for (int e = 0; e < 16; ++e) acc[x][y][e] = in[x + y + e];
    __syncthreads();
    long t0 = __builtin_amdgcn_s_memtime();
    for (int it = 0; it < iter; ++it) {
#pragma unroll
        for (int ta = 0; ta < 4; ++ta)
#pragma unroll
            for (int tb = 0; tb < 4; ++tb)
                acc[ta][tb] = __builtin_amdgcn_mfma_i32_32x32x32_i8(
                    DIVERSE ? fa[ta] : fa[0], DIVERSE ? fb[tb] : fb[0],
                    acc[ta][tb], 0, 0, 0);
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int x = 0; x < 4; ++x)
#pragma unroll
        for (int y = 0; y < 4; ++y) s += acc[x][y][0] + acc[x][y][15];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = t1 - t0;
    }
}

// Same, with 2 ds_read_b64_tr8 per MFMA interleaved (the rs2 fragment
// read mix) to price the co-issue.
typedef int v2i __attribute__((ext_vector_type(2)));
typedef __attribute__((address_space(3))) v2i* lds_v2i;

template <int NACC, int WPE>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(WPE)))
void mfma_tr8_kernel(const int* __restrict__ in, long* __restrict__ cyc,
                     int* __restrict__ sink, int iter) {
    __shared__ signed char lds[64][272];
    v4i acc[NACC];
    v4i a, b;
    int tid = threadIdx.x;
    int lane = tid & 63;
    a = v4i{in[lane], in[lane + 64], in[lane + 128], in[lane + 192]};
    b = v4i{in[lane + 1], in[lane + 65], in[lane + 129], in[lane + 193]};
#pragma unroll
    for (int i = 0; i < NACC; ++i)
        acc[i] = v4i{in[i], 0, 0, 0};
    for (int i = tid; i < 64 * 272 / 4; i += 256)
        ((__attribute__((address_space(3))) int*)&lds[0][0])[i] = in[i & 255];
    __syncthreads();
    int tr_row = 8 * (lane >> 4) + ((lane & 15) >> 1);
    int tr_half = lane & 1;
    const signed char* base =
        (const signed char*)&lds[0][0] + tr_row * 272 + 8 * tr_half;
    long t0 = __builtin_amdgcn_s_memtime();
    for (int it = 0; it < iter; ++it) {
#pragma unroll
        for (int i = 0; i < NACC; ++i) {
            v2i lo = __builtin_amdgcn_ds_read_tr8_b64_v2i32(
                (lds_v2i)(base + 16 * (i & 15)));
            v2i hi = __builtin_amdgcn_ds_read_tr8_b64_v2i32(
                (lds_v2i)(base + 32 * 272 + 16 * (i & 15)));
            v4i f = v4i{lo[0], lo[1], hi[0], hi[1]};
            acc[i] = __builtin_amdgcn_mfma_i32_16x16x64_i8(a, f, acc[i],
                                                           0, 0, 0);
        }
    }
    long t1 = __builtin_amdgcn_s_memtime();
    int s = 0;
#pragma unroll
    for (int i = 0; i < NACC; ++i) s += acc[i][0] + acc[i][3];
    if (threadIdx.x == 0) {
        sink[blockIdx.x] = s;
        cyc[blockIdx.x] = t1 - t0;
    }
}

bool g_zero_data = false;

template <typename K>
void run(const char* name, K kern, int blocks, int iter, int nacc,
         int ldsbytes = 0) {
    int* in;
    long* cyc;
    int* sink;
    (void)hipMalloc(&in, 4096);
    (void)hipMalloc(&cyc, blocks * sizeof(long));
    (void)hipMalloc(&sink, blocks * sizeof(int));
    if (g_zero_data) {
        (void)hipMemset(in, 1, 4096);
    } else {
        unsigned char rnd[4096];
        unsigned s = 12345;
        for (int i = 0; i < 4096; ++i) {
            s = s * 1664525u + 1013904223u;
            rnd[i] = (unsigned char)(s >> 24);
        }
        (void)hipMemcpy(in, rnd, 4096, hipMemcpyHostToDevice);
    }
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), ldsbytes, 0, in, cyc,
                       sink, iter);
    (void)hipDeviceSynchronize();
    timespec w0, w1;
    clock_gettime(CLOCK_MONOTONIC, &w0);
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), ldsbytes, 0, in, cyc,
                       sink, iter);
    (void)hipDeviceSynchronize();
    clock_gettime(CLOCK_MONOTONIC, &w1);
    double wall = (w1.tv_sec - w0.tv_sec) + 1e-9 * (w1.tv_nsec - w0.tv_nsec);
    long* h = new long[blocks];
    (void)hipMemcpy(h, cyc, blocks * sizeof(long), hipMemcpyDeviceToHost);
    long mx = 0, mn = (long)1e18;
    for (int i = 0; i < blocks; ++i) {
        if (h[i] > mx) mx = h[i];
        if (h[i] < mn) mn = h[i];
    }
    double per = (double)mx / ((double)iter * nacc);
    // chip-wide sustained rate from wall (valid for the non-OUTER
    // kernels where total equiv-MFMAs = blocks*4*iter*nacc; OUTER
    // variants are 20x this — noted per row)
    double tot = (double)blocks * 4.0 * iter * nacc;
    double tops = tot * 32768.0 / wall / 1e12;
    printf("%-28s blocks=%4d nacc=%3d  cyc/MFMA max %.2f min %.2f  "
           "wall %.2f ms (%.0f equiv-TOPS)\n",
           name, blocks, nacc, per, (double)mn / ((double)iter * nacc),
           wall * 1e3, tops);
    delete[] h;
    (void)hipFree(in); (void)hipFree(cyc); (void)hipFree(sink);
}

int main() {
    int iter = 2000;
    // Random operand data by default: low-toggle (memset-1) operands let
    // the chip clock ~15% higher and overstate sustainable rates
    // (PROBE_ZERO_DATA=1 restores the old behaviour for comparison).
    g_zero_data = getenv("PROBE_ZERO_DATA") != nullptr;
    // pure MFMA chains
    run("occ1 nacc64 (rs2 shape)", mfma_rate_kernel<64, 1>, 256, iter, 64);
    run("occ1 nacc16", mfma_rate_kernel<16, 1>, 256, iter, 16);
    run("occ2 nacc32", mfma_rate_kernel<32, 2>, 512, iter, 32);
    run("occ2 nacc16", mfma_rate_kernel<16, 2>, 512, iter, 16);
    run("occ4 nacc16 (rs shape)", mfma_rate_kernel<16, 4>, 1024, iter, 16);
    run("occ4 nacc8", mfma_rate_kernel<8, 4>, 1024, iter, 8);
    // diverse 8x8 operand pairs (the rs2 burst register pattern)
    run("pairs occ1 (rs2 burst)", mfma_pairs_kernel<1>, 256, iter, 64);
    // SUSTAINED power envelope: long full-chip burst on random data
    run("SUSTAINED pairs 10x", mfma_pairs_kernel<1>, 256, 20000, 64);
    run("SUSTAINED fixed 10x", mfma_rate_kernel<64, 1>, 256, 20000, 64);
    run("env base (pairs occ1)", mfma_pairs_env_kernel<0, 0, 1>, 256,
        iter, 64);
    run("env +lds69k", mfma_pairs_env_kernel<1, 0, 1>, 256, iter, 64,
        69632);
    run("env +outer/sync", mfma_pairs_env_kernel<0, 1, 1>, 256, 500, 64);
    run("env +lds+outer", mfma_pairs_env_kernel<1, 1, 1>, 256, 500, 64,
        69632);
    run("pairs2 occ2 (rsm 8x4)", mfma_pairs2_kernel<2>, 512, iter, 32);
    run("pairs2 occ1 (8x4)", mfma_pairs2_kernel<1>, 256, iter, 32);
    // 32x32x32 i8 (16 i32 acc per instr; cyc shown PER 16x16x64-EQUIV,
    // i.e. raw cyc/instr divided by 2 — compare directly with rows above)
    run("mfma32 occ1 fixed", mfma32_kernel<1, 0>, 256, iter, 32);
    run("mfma32 occ1 diverse 4x4", mfma32_kernel<1, 1>, 256, iter, 32);
    // MFMA + 2x tr8 per MFMA (the fragment-read mix)
    run("tr8 occ1 nacc64", mfma_tr8_kernel<64, 1>, 256, iter, 64);
    run("tr8 occ2 nacc32", mfma_tr8_kernel<32, 2>, 512, iter, 32);
    run("tr8 occ4 nacc16", mfma_tr8_kernel<16, 4>, 1024, iter, 16);
    return 0;
}
