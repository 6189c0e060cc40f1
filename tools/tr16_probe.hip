// Empirical semantics probe for gfx950 ds_read_b64_tr_b16.
// Fills LDS with element-index values, has each lane issue one transpose
// read at a configurable per-lane address, and dumps the 4 bf16 element
// indices each lane received.
//
// Build & run on the GPU box:
//   hipcc --offload-arch=gfx950 tools/tr16_probe.hip -o /tmp/tr16_probe
//   /tmp/tr16_probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__global__ void tr16_probe(float* out, int addr_mode) {
    __shared__ __bf16 s[4096];
    for (int i = threadIdx.x; i < 4096; i += blockDim.x)
        s[i] = (__bf16)(float)i;
    __syncthreads();
    int l = threadIdx.x;
    if (l >= 64) return;
    unsigned addr;
    switch (addr_mode) {
        case 0: addr = 0; break;                    // uniform base
        case 1: addr = l * 4; break;                // contiguous b64 per lane
        case 2: addr = (l >> 4) * 64; break;        // per-16-group base
        case 3: addr = (l & 15) * 4 + (l >> 4) * 256; break;
        default: addr = l * 8; break;
    }
    auto* p = (__attribute__((address_space(3))) bf16x4*)&s[addr];
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
#pragma unroll
    for (int j = 0; j < 4; ++j)
        out[l * 4 + j] = (float)v[j];
}

int main() {
    float* out;
    hipMalloc(&out, 64 * 4 * sizeof(float));
    float host[256];
    for (int mode = 0; mode < 4; ++mode) {
        hipMemset(out, 0, sizeof(host));
        hipLaunchKernelGGL(tr16_probe, dim3(1), dim3(64), 0, 0, out, mode);
        hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
        printf("mode %d:\n", mode);
        int lanes[] = {0, 1, 2, 3, 15, 16, 17, 31, 32, 48, 63};
        for (int li = 0; li < 11; ++li) {
            int l = lanes[li];
            printf("  lane %2d: %5.0f %5.0f %5.0f %5.0f\n", l,
                   host[l * 4], host[l * 4 + 1], host[l * 4 + 2],
                   host[l * 4 + 3]);
        }
    }
    return 0;
}
