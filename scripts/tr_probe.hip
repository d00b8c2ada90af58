// Probe: pin ds_read_b64_tr_b16's lane/element mapping on gfx950.
//
// LDS u16[1024] is filled with value == its own index.  Each lane issues
// one tr read at a configurable per-lane address pattern and dumps its
// four u16 results; the host prints lane -> source-index table, from
// which the fragment-builder mapping is derived (same methodology as the
// round-1 fed_mfma_probe for the MFMA fragment maps).
//
// Build+run (GPU box): hipcc --offload-arch=gfx950 -O3 tr_probe.hip -o tr_probe && ./tr_probe
#include <hip/hip_runtime.h>
#include <stdio.h>

__global__ void k_tr_probe(unsigned short* out, int pattern) {
    __shared__ unsigned short lds[1024];
    const int lane = threadIdx.x & 63;
    for (int i = threadIdx.x; i < 1024; i += blockDim.x)
        lds[i] = (unsigned short)i;
    __syncthreads();

    // per-lane element offset into the u16 image
    int eoff;
    switch (pattern) {
        case 0: eoff = lane * 4; break;                       // 8-B pieces, linear
        case 1: eoff = (lane & 15) + (lane >> 4) * 64; break; // canonical formula
        case 2: eoff = (lane >> 4) * 64 + (lane & 15) * 4; break;
        case 3: eoff = lane * 4; break;  // control via plain ds_read_b64
        default: eoff = lane * 4; break;
    }
    // DS vaddr is a 32-bit LDS byte address (static __shared__ starts at 0)
    const unsigned a32 = (unsigned)((const char*)&lds[eoff] - (const char*)&lds[0]);
    unsigned long long v;
    if (pattern == 3) {
        // control: plain b64 through the identical asm path
        asm volatile("ds_read_b64 %0, %1\n\ts_waitcnt lgkmcnt(0) vmcnt(0)"
                     : "=v"(v) : "v"(a32) : "memory");
    } else {
        asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0) vmcnt(0)"
                     : "=v"(v) : "v"(a32) : "memory");
    }
    // a plain C read of the array keeps the fill alive (the asm only sees
    // an integer address, so LLVM otherwise proves the stores dead)
    out[256 + threadIdx.x] = lds[threadIdx.x];
    if (threadIdx.x < 64) {
        out[lane * 4 + 0] = (unsigned short)(v & 0xffff);
        out[lane * 4 + 1] = (unsigned short)((v >> 16) & 0xffff);
        out[lane * 4 + 2] = (unsigned short)((v >> 32) & 0xffff);
        out[lane * 4 + 3] = (unsigned short)((v >> 48) & 0xffff);
    }
}

int main() {
    unsigned short* out_d;
    (void)hipMalloc(&out_d, (256 + 64) * 2 * 2);
    unsigned short out_h[256];
    for (int pat = 0; pat < 4; ++pat) {
        hipLaunchKernelGGL(k_tr_probe, dim3(1), dim3(64), 0, 0, out_d, pat);
        if (hipDeviceSynchronize() != hipSuccess) { printf("pat %d: launch failed\n", pat); continue; }
        (void)hipMemcpy(out_h, out_d, sizeof(out_h), hipMemcpyDeviceToHost);
        printf("pattern %d:\n", pat);
        for (int l = 0; l < 64; ++l) {
            printf("  lane %2d: %4d %4d %4d %4d\n", l,
                   out_h[l * 4], out_h[l * 4 + 1], out_h[l * 4 + 2], out_h[l * 4 + 3]);
        }
    }
    return 0;
}
