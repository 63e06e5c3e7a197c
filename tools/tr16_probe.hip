// Probe ds_read_b64_tr_b16 lane semantics on gfx950.
// Fills LDS with shorts = their own element index, issues one tr16 read per
// lane at a chosen per-lane address, prints which LDS elements land in each
// lane's 4 result slots.  Build & run (GPU box):
//   hipcc --offload-arch=gfx950 tools/tr16_probe.hip -o /tmp/tr16_probe && /tmp/tr16_probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef short s16x4 __attribute__((ext_vector_type(4)));

__global__ void probe(short* out, const int* addr) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4*)&lds[addr[threadIdx.x]]);
  *(s16x4*)&out[threadIdx.x * 4] = v;
}

static void run(const char* name, int (*f)(int)) {
  int ha[64];
  for (int l = 0; l < 64; ++l) ha[l] = f(l);
  int* da; short* dout;
  (void)hipMalloc(&da, sizeof(ha));
  (void)hipMalloc(&dout, 64 * 4 * sizeof(short));
  (void)hipMemcpy(da, ha, sizeof(ha), hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dout, da);
  short ho[256];
  (void)hipMemcpy(ho, dout, sizeof(ho), hipMemcpyDeviceToHost);
  printf("== %s ==\n", name);
  for (int l = 0; l < 64; ++l)
    printf("lane %2d addr %4d -> %4d %4d %4d %4d\n", l, ha[l],
           ho[4 * l], ho[4 * l + 1], ho[4 * l + 2], ho[4 * l + 3]);
  (void)hipFree(da);
  (void)hipFree(dout);
}

int main() {
  run("addr = lane*4 (element)", [](int l) { return l * 4; });
  run("addr = 0 (all lanes same)", [](int l) { (void)l; return 0; });
  run("addr = (lane&3)*64 + (lane>>2)*4", [](int l) {
    return (l & 3) * 64 + (l >> 2) * 4;
  });
  return 0;
}
