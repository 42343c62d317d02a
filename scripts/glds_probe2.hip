// Probe for the 8-wave conv design (run: hipcc -o p glds_probe2.hip && ./p):
//  (1) does raw_ptr_buffer_load_lds with an out-of-bounds voffset write ZERO
//      to the LDS destination (the padding trick for implicit-GEMM glds)?
//  (2) does __builtin_amdgcn_global_load_lds respect the EXEC mask (skip
//      inactive lanes' 16B chunks)?
#include <hip/hip_runtime.h>

#include <cstdio>

__global__ void k_oob(const float* g, float* out, unsigned nbytes) {
    __shared__ float lds[256];
    for (int i = threadIdx.x; i < 256; i += blockDim.x) lds[i] = -7.0f;
    __syncthreads();
    auto rsrc = __builtin_amdgcn_make_buffer_rsrc((void*)g, 0, nbytes, 0x00020000);
    // lanes >= 32: voffset far beyond nbytes -> expect 0 written to LDS
    unsigned voff = threadIdx.x * 16 + (threadIdx.x >= 32 ? 1u << 30 : 0);
    __builtin_amdgcn_raw_ptr_buffer_load_lds(
        rsrc, (__attribute__((address_space(3))) void*)&lds[threadIdx.x * 4],
        16, voff, 0, 0, 0);
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    out[threadIdx.x * 4] = lds[threadIdx.x * 4];
}

__global__ void k_exec(const float* g, float* out) {
    __shared__ float lds[256];
    for (int i = threadIdx.x; i < 256; i += blockDim.x) lds[i] = -7.0f;
    __syncthreads();
    if (threadIdx.x < 32) {  // half the wave inactive for the glds
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)
                (g + threadIdx.x * 4),
            (__attribute__((address_space(3))) unsigned int*)
                &lds[threadIdx.x * 4],
            16, 0, 0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    out[threadIdx.x * 4] = lds[threadIdx.x * 4];
}

int main() {
    float* g;
    float* out;
    hipMalloc(&g, 64 * 16);
    hipMalloc(&out, 256 * 4);
    float host[256];
    for (int i = 0; i < 256; ++i) host[i] = 100.f + i;
    hipMemcpy(g, host, sizeof(host), hipMemcpyHostToDevice);

    k_oob<<<1, 64>>>(g, out, 64 * 16);
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("oob: lane0=%g lane31=%g lane32=%g lane63=%g  -> %s\n",
           host[0], host[31 * 4], host[32 * 4], host[63 * 4],
           (host[32 * 4] == 0.f && host[63 * 4] == 0.f) ? "OOB=ZERO ok"
           : host[32 * 4] == -7.f ? "OOB=SKIPPED (stale lds)" : "OOB=??");

    k_exec<<<1, 64>>>(g, out);
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("exec: lane0=%g lane31=%g lane32=%g lane63=%g -> %s\n",
           host[0], host[31 * 4], host[32 * 4], host[63 * 4],
           host[32 * 4] == -7.f ? "EXEC respected (inactive skipped)"
                                : "inactive lanes WROTE");
    return 0;
}
