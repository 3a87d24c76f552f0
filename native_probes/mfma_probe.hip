// Fragment-layout probe for v_mfma_f32_16x16x32_bf16 on gfx950.
//
// Verifies the assumed lane->element mappings with ASYMMETRIC integer
// operands (exact in bf16/fp32, so any layout error is a hard mismatch,
// not a tolerance issue):
//   A (16x32): row = lane & 15, k = (lane >> 4) * 8 + i   (i in [0,8))
//   B (32x16): col = lane & 15, k = (lane >> 4) * 8 + i
//   D (16x16): col = lane & 15, row = (lane >> 4) * 4 + reg
//
// Build: hipcc --offload-arch=gfx950 -O2 mfma_probe.hip -o mfma_probe
// Run on an MI355X box: ./mfma_probe   (prints PASS/FAIL)

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

__global__ void probe_kernel(const __bf16* A, const __bf16* B, float* D) {
    const int lane = threadIdx.x;
    bf16x8 a_frag, b_frag;
    const int row = lane & 15;
    const int col = lane & 15;
    const int k0 = (lane >> 4) * 8;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
        a_frag[i] = A[row * 32 + (k0 + i)];
        b_frag[i] = B[(k0 + i) * 16 + col];
    }
    floatx4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int out_row = (lane >> 4) * 4 + r;
        D[out_row * 16 + col] = acc[r];
    }
}

int main() {
    __bf16 hA[16 * 32], hB[32 * 16];
    float ref[16 * 16];
    for (int r = 0; r < 16; ++r)
        for (int k = 0; k < 32; ++k) hA[r * 32 + k] = (__bf16)(float)((r * 7 + k * 3) % 8);
    for (int k = 0; k < 32; ++k)
        for (int c = 0; c < 16; ++c) hB[k * 16 + c] = (__bf16)(float)((k * 5 + c * 11) % 8);
    for (int r = 0; r < 16; ++r)
        for (int c = 0; c < 16; ++c) {
            float s = 0.f;
            for (int k = 0; k < 32; ++k) s += (float)((r * 7 + k * 3) % 8) * (float)((k * 5 + c * 11) % 8);
            ref[r * 16 + c] = s;
        }
    __bf16 *dA, *dB;
    float* dD;
    (void)hipMalloc(&dA, sizeof(hA));
    (void)hipMalloc(&dB, sizeof(hB));
    (void)hipMalloc(&dD, sizeof(float) * 256);
    (void)hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
    (void)hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD);
    float out[256];
    (void)hipMemcpy(out, dD, sizeof(out), hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 256; ++i)
        if (out[i] != ref[i]) {
            if (bad < 8) printf("mismatch [%d,%d]: got %g want %g\n", i / 16, i % 16, out[i], ref[i]);
            ++bad;
        }
    printf(bad ? "MFMA PROBE FAIL (%d mismatches)\n" : "MFMA PROBE PASS\n", bad);
    return bad ? 1 : 0;
}
