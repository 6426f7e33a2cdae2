// Layout probe for the i8-MFMA cherk kernel (dev tool, not shipped in the
// library).  Establishes on real hardware:
//   1. mfma_i32_16x16x64_i8 A/B fragment layouts (hypothesis H1: lane l
//      holds A[row=l&15][k=16*(l>>4)+b] / B[col=l&15][k=16*(l>>4)+b],
//      b = byte index 0..15; C/D: col=lane&15, row=(lane>>4)*4+reg).
//   2. ds_read_tr8_b64 gather pattern (dump of which LDS byte each lane
//      byte receives, for uniform and per-lane bases).
// Build: make probe   Run: ./probe_mfma

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

typedef int v4i __attribute__((ext_vector_type(4)));
typedef int v2i __attribute__((ext_vector_type(2)));
typedef __attribute__((address_space(3))) v2i* lds_v2i;

#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

// --- 1. verify H1 for 16x16x64 ---------------------------------------------
__global__ void mfma_h1_kernel(const signed char* A,  // [16][64]
                               const signed char* B,  // [64][16]
                               int* D) {              // [16][16]
    int lane = threadIdx.x;
    v4i av, bv, cv{};
    signed char ab[16], bb[16];
    for (int b = 0; b < 16; ++b) {
        int row = lane & 15, kA = 16 * (lane >> 4) + b;
        ab[b] = A[row * 64 + kA];
        int col = lane & 15, kB = 16 * (lane >> 4) + b;
        bb[b] = B[kB * 16 + col];
    }
    __builtin_memcpy(&av, ab, 16);
    __builtin_memcpy(&bv, bb, 16);
    cv = __builtin_amdgcn_mfma_i32_16x16x64_i8(av, bv, cv, 0, 0, 0);
    for (int r = 0; r < 4; ++r) {
        int col = lane & 15, row = (lane >> 4) * 4 + r;
        D[row * 16 + col] = cv[r];
    }
}

// Variant H2: A/B k-group order swapped (k = (lane>>4) + 4*b)
__global__ void mfma_h2_kernel(const signed char* A, const signed char* B,
                               int* D) {
    int lane = threadIdx.x;
    v4i av, bv, cv{};
    signed char ab[16], bb[16];
    for (int b = 0; b < 16; ++b) {
        int kA = (lane >> 4) + 4 * b;
        ab[b] = A[(lane & 15) * 64 + kA];
        bb[b] = B[kA * 16 + (lane & 15)];
    }
    __builtin_memcpy(&av, ab, 16);
    __builtin_memcpy(&bv, bb, 16);
    cv = __builtin_amdgcn_mfma_i32_16x16x64_i8(av, bv, cv, 0, 0, 0);
    for (int r = 0; r < 4; ++r)
        D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = cv[r];
}

// --- 1b. verify H1 for 32x32x32 --------------------------------------------
typedef int v16i __attribute__((ext_vector_type(16)));
__global__ void mfma32_h1_kernel(const signed char* A,  // [32][32]
                                 const signed char* B,  // [32][32]
                                 int* D) {              // [32][32]
    int lane = threadIdx.x;
    v4i av, bv;
    v16i cv{};
    signed char ab[16], bb[16];
    for (int b = 0; b < 16; ++b) {
        int kA = 16 * (lane >> 5) + b;
        ab[b] = A[(lane & 31) * 32 + kA];
        bb[b] = B[kA * 32 + (lane & 31)];
    }
    __builtin_memcpy(&av, ab, 16);
    __builtin_memcpy(&bv, bb, 16);
    cv = __builtin_amdgcn_mfma_i32_32x32x32_i8(av, bv, cv, 0, 0, 0);
    for (int r = 0; r < 16; ++r) {
        int col = lane & 31;
        int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        D[row * 32 + col] = cv[r];
    }
}

// --- 2. tr8 gather dump ------------------------------------------------------
// Fill 1024 LDS bytes with known values; each lane does a tr8 read with a
// configurable base; dump the 8 received bytes (low+high runs decode addr).
__global__ void tr8_dump_kernel(int mode, int which, unsigned char* out) {
    __shared__ unsigned char lds[1024];
    for (int i = threadIdx.x; i < 1024; i += 64) {
        lds[i] = which == 0 ? (unsigned char)(i & 0xFF)
                            : (unsigned char)(i >> 8);
    }
    __syncthreads();
    int lane = threadIdx.x;
    unsigned addr;
    if (mode == 0) addr = 0;                       // uniform base
    else if (mode == 1) addr = lane * 8;           // linear per-lane
    else addr = (lane >> 4) * 128;                 // per-group base
    v2i r = __builtin_amdgcn_ds_read_tr8_b64_v2i32((lds_v2i)&lds[addr]);
    unsigned char bytes[8];
    __builtin_memcpy(bytes, &r, 8);
    for (int b = 0; b < 8; ++b) out[lane * 8 + b] = bytes[b];
}

int main() {
    srand(42);
    std::vector<signed char> A(16 * 64), B(64 * 16), A32(32 * 32), B32(32 * 32);
    for (auto& v : A) v = (signed char)(rand() % 255 - 127);
    for (auto& v : B) v = (signed char)(rand() % 255 - 127);
    for (auto& v : A32) v = (signed char)(rand() % 255 - 127);
    for (auto& v : B32) v = (signed char)(rand() % 255 - 127);

    signed char *dA, *dB, *dA32, *dB32;
    int* dD;
    CHECK(hipMalloc(&dA, A.size()));
    CHECK(hipMalloc(&dB, B.size()));
    CHECK(hipMalloc(&dA32, A32.size()));
    CHECK(hipMalloc(&dB32, B32.size()));
    CHECK(hipMalloc(&dD, 32 * 32 * 4));
    CHECK(hipMemcpy(dA, A.data(), A.size(), hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dB, B.data(), B.size(), hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dA32, A32.data(), A32.size(), hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dB32, B32.data(), B32.size(), hipMemcpyHostToDevice));

    // host reference 16x16
    std::vector<int> ref(16 * 16), got(16 * 16);
    for (int i = 0; i < 16; ++i)
        for (int j = 0; j < 16; ++j) {
            int s = 0;
            for (int k = 0; k < 64; ++k) s += (int)A[i * 64 + k] * B[k * 16 + j];
            ref[i * 16 + j] = s;
        }
    for (int h = 0; h < 2; ++h) {
        CHECK(hipMemset(dD, 0, 16 * 16 * 4));
        if (h == 0) hipLaunchKernelGGL(mfma_h1_kernel, 1, 64, 0, 0, dA, dB, dD);
        else hipLaunchKernelGGL(mfma_h2_kernel, 1, 64, 0, 0, dA, dB, dD);
        CHECK(hipMemcpy(got.data(), dD, 16 * 16 * 4, hipMemcpyDeviceToHost));
        int bad = 0;
        for (int i = 0; i < 256; ++i) bad += got[i] != ref[i];
        printf("mfma_16x16x64 H%d: %s (%d/256 mismatches)\n", h + 1,
               bad ? "FAIL" : "PASS", bad);
    }

    std::vector<int> ref32(32 * 32), got32(32 * 32);
    for (int i = 0; i < 32; ++i)
        for (int j = 0; j < 32; ++j) {
            int s = 0;
            for (int k = 0; k < 32; ++k)
                s += (int)A32[i * 32 + k] * B32[k * 32 + j];
            ref32[i * 32 + j] = s;
        }
    CHECK(hipMemset(dD, 0, 32 * 32 * 4));
    hipLaunchKernelGGL(mfma32_h1_kernel, 1, 64, 0, 0, dA32, dB32, dD);
    CHECK(hipMemcpy(got32.data(), dD, 32 * 32 * 4, hipMemcpyDeviceToHost));
    int bad32 = 0;
    for (int i = 0; i < 1024; ++i) bad32 += got32[i] != ref32[i];
    printf("mfma_32x32x32 H1: %s (%d/1024 mismatches)\n",
           bad32 ? "FAIL" : "PASS", bad32);

    // tr8 dump
    unsigned char* dout;
    CHECK(hipMalloc(&dout, 64 * 8));
    std::vector<unsigned char> lo(64 * 8), hi(64 * 8);
    for (int mode = 0; mode < 3; ++mode) {
        hipLaunchKernelGGL(tr8_dump_kernel, 1, 64, 0, 0, mode, 0, dout);
        CHECK(hipMemcpy(lo.data(), dout, 64 * 8, hipMemcpyDeviceToHost));
        hipLaunchKernelGGL(tr8_dump_kernel, 1, 64, 0, 0, mode, 1, dout);
        CHECK(hipMemcpy(hi.data(), dout, 64 * 8, hipMemcpyDeviceToHost));
        printf("tr8 mode %d: lane -> source byte addresses\n", mode);
        for (int l = 0; l < 64; l += 1) {
            printf("  lane %2d:", l);
            for (int b = 0; b < 8; ++b) {
                int addr = lo[l * 8 + b] | (hi[l * 8 + b] << 8);
                printf(" %4d", addr);
            }
            printf("\n");
            if (l == 3) { l = 14; printf("  ...\n"); }
            else if (l == 19) { l = 31; printf("  ...\n"); }
            else if (l == 35) { l = 62; printf("  ...\n"); }
        }
    }
    printf("probe done\n");
    return 0;
}
