#include <hip/hip_runtime.h>
#include <stdio.h>
#include <string.h>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "%s -> %d (%s)\n", #x, e, hipGetErrorString(e)); \
  return 1; } } while (0)

int main(int argc, char** argv) {
  if (argc < 2) return 2;
  hipIpcMemHandle_t h;
  memset(&h, 0, sizeof(h));
  for (size_t i = 0; i < sizeof(h); i++) {
    unsigned v;
    sscanf(argv[1] + 2 * i, "%02x", &v);
    ((unsigned char*)&h)[i] = (unsigned char)v;
  }
  CHECK(hipSetDevice(0));
  void* ptr = NULL;
  CHECK(hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess));
  unsigned char out[16];
  CHECK(hipMemcpy(out, ptr, 16, hipMemcpyDeviceToHost));
  for (int i = 0; i < 16; i++)
    if (out[i] != (unsigned char)i) { fprintf(stderr, "byte %d mismatch\n", i); return 3; }
  CHECK(hipIpcCloseMemHandle(ptr));
  printf("CHILD_OK\n");
  return 0;
}
