// Empirical semantics probe for gfx950 ds_read_b64_tr_b16.
// Fills LDS with lds[i] = i (ushort), has each lane supply an address,
// dumps what each lane's 4 output elements contain. The printed mapping
// pins down the (lane, elem) -> LDS element relation we build the conv
// wgrad tr-images on. Build: hipcc --offload-arch=gfx950 -O2 probe_tr16.hip -o probe_tr16
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;

// addr_mode: 0 = base + lane*8B (lane-linear)
//            1 = base + (lane&15)*8B            (group-constant rows)
//            2 = base + lane*8B + 32B           (shifted by one 16-ushort row)
//            3 = base + (lane>>2)*8B            (quad-shared addresses)
__global__ void probe_tr(ushort* out, int addr_mode) {
    __shared__ ushort lds[4096];
    for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (ushort)i;
    __syncthreads();
    const int l = threadIdx.x & 63;
    int off;
    switch (addr_mode) {
        case 1: off = (l & 15) * 4; break;
        case 2: off = l * 4 + 16; break;
        case 3: off = (l >> 2) * 4; break;
        default: off = l * 4; break;
    }
    bf16x4t v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4t*)&lds[off]);
    ushort4 o = *reinterpret_cast<ushort4*>(&v);
    if (threadIdx.x < 64) reinterpret_cast<ushort4*>(out)[l] = o;
}

int main() {
    ushort* d;
    hipMalloc(&d, 64 * 4 * sizeof(ushort));
    ushort h[256];
    for (int mode = 0; mode < 4; ++mode) {
        hipLaunchKernelGGL(probe_tr, dim3(1), dim3(64), 0, 0, d, mode);
        hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
        hipDeviceSynchronize();
        printf("== addr_mode %d (lane: elem0 elem1 elem2 elem3) ==\n", mode);
        for (int l = 0; l < 64; ++l)
            printf("%2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
    }
    return 0;
}
