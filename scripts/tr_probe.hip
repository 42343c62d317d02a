// Standalone probe: empirical element mapping of gfx950 ds_read_b64_tr_b16.
// Fills LDS with lds[i] = i (bf16-sized payload, stored as uint16 raw), each
// lane supplies addr = 8 * lane (4 contiguous uint16 per lane), then dumps
// the 4 elements each lane receives.  Expected (guide T10): out[l][j] =
// lds[(l & 15) + 16 * j + 64 * (l >> 4)].
//   hipcc --offload-arch=gfx950 scripts/tr_probe.hip -o /tmp/tr_probe && /tmp/tr_probe
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void k(unsigned short* out) {
    __shared__ __attribute__((aligned(16))) unsigned short lds[512];
    const int tid = threadIdx.x;
    for (int i = tid; i < 512; i += 64) lds[i] = (unsigned short)i;
    __syncthreads();
    unsigned addr = (unsigned)(unsigned long long)(const void*)&lds[tid * 4];
    unsigned long long v;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(v) : "v"(addr));
    for (int j = 0; j < 4; ++j)
        out[tid * 4 + j] = (unsigned short)(v >> (16 * j));
}

int main() {
    unsigned short* d;
    hipMalloc(&d, 64 * 4 * 2);
    hipLaunchKernelGGL(k, dim3(1), dim3(64), 0, 0, d);
    unsigned short h[256];
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    int bad = 0;
    for (int l = 0; l < 64; ++l) {
        for (int j = 0; j < 4; ++j) {
            int expect = (l & 15) + 16 * j + 64 * (l >> 4);
            if (h[l * 4 + j] != expect) ++bad;
        }
        if (l < 4 || l == 16)
            printf("lane %2d: %3d %3d %3d %3d\n", l, h[l*4], h[l*4+1],
                   h[l*4+2], h[l*4+3]);
    }
    printf(bad ? "MISMATCH vs guide formula: %d cells\n" : "matches guide formula\n", bad);
    return bad != 0;
}
