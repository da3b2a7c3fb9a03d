// hipModuleLoad runner for the hand-edited .s kernel — proves the
// compile->.s->hand-edit->assemble->load toolchain path end to end
// (the round-3 asm-forward build recipe).
// Build: hipcc -O2 tools/asm_probe/load_run.cpp -o tools/asm_probe/load_run
// Run (repo root): ./tools/asm_probe/load_run
#include <hip/hip_runtime.h>
#include <stdio.h>

int main() {
  hipModule_t mod;
  hipFunction_t fn;
  if (hipModuleLoad(&mod, "tools/asm_probe/edited.hsaco") != hipSuccess) {
    printf("FAIL: module load\n");
    return 1;
  }
  if (hipModuleGetFunction(&fn, mod, "trivial_kernel") != hipSuccess) {
    printf("FAIL: get function\n");
    return 1;
  }
  float* out;
  (void)hipMalloc(&out, 256 * 4);
  struct {
    void* out;
    float x;
  } args = {out, 1.5f};
  size_t size = sizeof(args);
  void* cfg[] = {HIP_LAUNCH_PARAM_BUFFER_POINTER, &args,
                 HIP_LAUNCH_PARAM_BUFFER_SIZE, &size,
                 HIP_LAUNCH_PARAM_END};
  if (hipModuleLaunchKernel(fn, 1, 1, 1, 256, 1, 1, 0, 0, nullptr, cfg) !=
      hipSuccess) {
    printf("FAIL: launch\n");
    return 1;
  }
  float h[256];
  (void)hipMemcpy(h, out, sizeof(h), hipMemcpyDeviceToHost);
  // expected: 1.5 + i + 100 (the hand-inserted v_add)
  for (int i = 0; i < 256; ++i) {
    if (h[i] != 1.5f + i + 100.f) {
      printf("FAIL: h[%d] = %f (want %f)\n", i, h[i], 1.5f + i + 100.f);
      return 1;
    }
  }
  printf("PASS: hand-edited .s kernel assembled, loaded and ran "
         "(out[7] = %.1f)\n", h[7]);
  return 0;
}
