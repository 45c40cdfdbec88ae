// Probe for gfx950 ds_read_b64_tr_b16 semantics (guide T10): fill LDS with
// identity shorts, point each lane at a candidate address, dump what lands
// in which lane/element. Standalone tool (not part of the extension).
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void tr16_probe(short* out, const int* addr_scheme, int nscheme) {
  __shared__ short lds[4096];  // 8 KiB
  const int tid = threadIdx.x;
  for (int i = tid; i < 4096; i += 64) lds[i] = (short)i;
  __syncthreads();
  for (int s = 0; s < nscheme; ++s) {
    // candidate address for this lane, in BYTES into lds
    unsigned addr;
    switch (addr_scheme[s]) {
      case 0: addr = tid * 8; break;                       // linear 8B/lane
      case 1: addr = (tid & 15) * 8 + (tid >> 4) * 128; break;
      case 2: addr = (tid & 3) * 32 + ((tid >> 2) & 3) * 8 +
                     (tid >> 4) * 128; break;
      default: addr = tid * 8; break;
    }
    unsigned base = (unsigned)(size_t)&lds[0];
    unsigned a = base + addr;
    unsigned r0, r1;
    asm volatile("ds_read_b64_tr_b16 v[20:21], %2\n\t"
                 "s_waitcnt lgkmcnt(0)\n\t"
                 "v_mov_b32 %0, v20\n\t"
                 "v_mov_b32 %1, v21"
                 : "=v"(r0), "=v"(r1) : "v"(a) : "v20", "v21");
    short* o = out + ((long)s * 64 + tid) * 5;
    o[0] = (short)(addr / 2);  // element index the lane pointed at
    o[1] = (short)(r0 & 0xffff);
    o[2] = (short)(r0 >> 16);
    o[3] = (short)(r1 & 0xffff);
    o[4] = (short)(r1 >> 16);
  }
}

int main() {
  const int nscheme = 3;
  int h_sch[nscheme] = {0, 1, 2};
  int* d_sch;
  short* d_out;
  hipMalloc(&d_sch, sizeof(h_sch));
  hipMalloc(&d_out, nscheme * 64 * 5 * sizeof(short));
  hipMemcpy(d_sch, h_sch, sizeof(h_sch), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(tr16_probe, dim3(1), dim3(64), 0, 0, d_out, d_sch,
                     nscheme);
  short h_out[nscheme * 64 * 5];
  hipMemcpy(h_out, d_out, sizeof(h_out), hipMemcpyDeviceToHost);
  hipError_t err = hipGetLastError();
  if (err != hipSuccess) {
    printf("ERROR: %s\n", hipGetErrorString(err));
    return 1;
  }
  for (int s = 0; s < nscheme; ++s) {
    printf("== scheme %d (lane: addr_elem -> e0 e1 e2 e3)\n", s);
    for (int l = 0; l < 64; ++l) {
      short* o = h_out + ((long)s * 64 + l) * 5;
      printf("l%02d:%4d -> %4d %4d %4d %4d\n", l, o[0], o[1], o[2], o[3],
             o[4]);
    }
  }
  return 0;
}
