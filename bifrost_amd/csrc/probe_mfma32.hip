// Layout verification for v_mfma_i32_32x32x32_i8 (round 2): assumed
//   A (v4i, 16 B/lane): lane l -> A[row = l&31][k = 16*(l>>5) + e]
//   B (v4i):            lane l -> B[col = l&31][k = 16*(l>>5) + e]
//   D (v16i):           lane l, reg r -> D[row = 8*(r>>2) + 4*(l>>5)
//                                          + (r&3)][col = l&31]
// with D[i][j] = sum_k A[i][k]*B[j][k] (the 16x16x64 semantic scaled).
// The host side builds the matrices under that assumption, computes the
// reference product, and compares element-wise; on mismatch it dumps
// the full D so the real layout can be fitted offline.
//
// Build: hipcc -O3 --offload-arch=gfx950 probe_mfma32.hip -o probe_mfma32

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

typedef int v4i __attribute__((ext_vector_type(4)));
typedef int v16i __attribute__((ext_vector_type(16)));

__global__ void mfma32_layout_kernel(const signed char* __restrict__ abytes,
                                     const signed char* __restrict__ bbytes,
                                     int* __restrict__ dout) {
    int lane = threadIdx.x & 63;
    v4i a, b;
    signed char tmp[16];
    for (int e = 0; e < 16; ++e) tmp[e] = abytes[lane * 16 + e];
    __builtin_memcpy(&a, tmp, 16);
    for (int e = 0; e < 16; ++e) tmp[e] = bbytes[lane * 16 + e];
    __builtin_memcpy(&b, tmp, 16);
    v16i acc{};
    acc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a, b, acc, 0, 0, 0);
    for (int r = 0; r < 16; ++r) dout[lane * 16 + r] = acc[r];
}

int main() {
    signed char ha[64 * 16], hb[64 * 16];
    srand(12345);
    for (int i = 0; i < 64 * 16; ++i) {
        ha[i] = (signed char)(rand() % 255 - 127);
        hb[i] = (signed char)(rand() % 255 - 127);
    }
    signed char *da, *db;
    int* dd;
    (void)hipMalloc(&da, sizeof(ha));
    (void)hipMalloc(&db, sizeof(hb));
    (void)hipMalloc(&dd, 64 * 16 * sizeof(int));
    (void)hipMemcpy(da, ha, sizeof(ha), hipMemcpyHostToDevice);
    (void)hipMemcpy(db, hb, sizeof(hb), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(mfma32_layout_kernel, dim3(1), dim3(64), 0, 0, da,
                       db, dd);
    (void)hipDeviceSynchronize();
    int hd[64 * 16];
    (void)hipMemcpy(hd, dd, sizeof(hd), hipMemcpyDeviceToHost);

    // host reference under the assumed layout
    static int A[32][32], B[32][32], D[32][32];
    for (int l = 0; l < 64; ++l)
        for (int e = 0; e < 16; ++e) {
            A[l & 31][16 * (l >> 5) + e] = ha[l * 16 + e];
            B[l & 31][16 * (l >> 5) + e] = hb[l * 16 + e];
        }
    for (int i = 0; i < 32; ++i)
        for (int j = 0; j < 32; ++j) {
            int s = 0;
            for (int k = 0; k < 32; ++k) s += A[i][k] * B[j][k];
            D[i][j] = s;
        }
    int bad = 0;
    for (int l = 0; l < 64 && bad < 8; ++l)
        for (int r = 0; r < 16; ++r) {
            int row = 8 * (r >> 2) + 4 * (l >> 5) + (r & 3);
            int col = l & 31;
            if (hd[l * 16 + r] != D[row][col]) {
                if (bad < 8)
                    printf("MISMATCH lane %d reg %d: got %d, D[%d][%d]=%d\n",
                           l, r, hd[l * 16 + r], row, col, D[row][col]);
                ++bad;
            }
        }
    if (!bad) {
        printf("LAYOUT OK: A row=l&31 k=16*(l>>5)+e; D row=8*(r>>2)+"
               "4*(l>>5)+(r&3) col=l&31\n");
        return 0;
    }
    // dump for offline fitting
    printf("DUMP D (lane-major, 16 regs per lane):\n");
    for (int l = 0; l < 64; ++l) {
        printf("l%02d:", l);
        for (int r = 0; r < 16; ++r) printf(" %d", hd[l * 16 + r]);
        printf("\n");
    }
    printf("DUMP Abytes:\n");
    for (int l = 0; l < 64; ++l) {
        printf("a%02d:", l);
        for (int e = 0; e < 16; ++e) printf(" %d", (int)ha[l * 16 + e]);
        printf("\n");
    }
    printf("DUMP Bbytes:\n");
    for (int l = 0; l < 64; ++l) {
        printf("b%02d:", l);
        for (int e = 0; e < 16; ++e) printf(" %d", (int)hb[l * 16 + e]);
        printf("\n");
    }
    return 1;
}
