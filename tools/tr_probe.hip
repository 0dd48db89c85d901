// Probe for ds_read_tr16_b64 lane semantics on gfx950 (the transpose-read
// used by LDS-image GEMM recipes).  Build:
//   hipcc --offload-arch=gfx950 -O2 tools/tr_probe.hip -o tools/tr_probe
//
// MEASURED RESULT (MI355X, ROCm 7.2): with each lane L passing the address
// of its own 4-element group (addr stride 4 elements), the instruction
// returns, per 16-lane quarter:
//     out[L] = { e(L), e(L+16), e(L+32), e(L+48) }
// over the quarter's 64 collectively-loaded elements — equivalently
//     out[L][j] = slot (L%4) of the word loaded by lane (L/4 + 4j)
// i.e. column L of the quarter's [16 lanes][4 shorts] block: a plain
// k-major LDS slab transposes in the read crossbar.  Addresses are
// per-lane (the shuffle operates on loaded data, not on addresses).
#include <hip/hip_runtime.h>
#include <cstdio>
typedef short shortx4 __attribute__((ext_vector_type(4)));

__global__ void probe(short* out, int stride_el) {
  __shared__ short lds[64 * 64];
  for (int i = threadIdx.x; i < 64 * 64; i += 64) lds[i] = i;  // value == element index
  __syncthreads();
  auto p = (__attribute__((address_space(3))) shortx4*)(lds + threadIdx.x * stride_el);
  shortx4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
  for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = v[j];
}

int main() {
  short* d;
  (void)hipMalloc(&d, 64 * 4 * 2);
  for (int stride : {0, 4}) {
    probe<<<1, 64>>>(d, stride);
    short h[256];
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("=== per-lane addr stride %d elements ===\n", stride);
    for (int l = 0; l < 64; ++l)
      printf("lane %2d: %5d %5d %5d %5d\n", l, h[l * 4], h[l * 4 + 1], h[l * 4 + 2],
             h[l * 4 + 3]);
  }
  return 0;
}
