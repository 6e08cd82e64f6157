// Probe: can RCCL init a 2-rank communicator with both ranks on ONE device
// (in-process loopback)? If yes, the native RCCL engine's full message loop
// is testable on a single-GPU box.
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <cstdio>

#define CHECK(x) do { auto e = (x); if (e != ncclSuccess) { \
  printf("FAIL %s -> %s\n", #x, ncclGetErrorString(e)); return 1; } } while (0)
#define HCHECK(x) do { auto e = (x); if (e != hipSuccess) { \
  printf("HIPFAIL %s -> %s\n", #x, hipGetErrorString(e)); return 1; } } while (0)

int main() {
  ncclUniqueId id;
  CHECK(ncclGetUniqueId(&id));
  ncclComm_t comms[2];
  HCHECK(hipSetDevice(0));
  CHECK(ncclGroupStart());
  CHECK(ncclCommInitRank(&comms[0], 2, id, 0));
  CHECK(ncclCommInitRank(&comms[1], 2, id, 1));
  CHECK(ncclGroupEnd());
  printf("init ok\n");
  float *a, *b;
  HCHECK(hipMalloc(&a, 1024));
  HCHECK(hipMalloc(&b, 1024));
  HCHECK(hipMemset(a, 1, 1024));
  hipStream_t s0, s1;
  HCHECK(hipStreamCreate(&s0));
  HCHECK(hipStreamCreate(&s1));
  CHECK(ncclGroupStart());
  CHECK(ncclSend(a, 256, ncclFloat, 1, comms[0], s0));
  CHECK(ncclRecv(b, 256, ncclFloat, 0, comms[1], s1));
  CHECK(ncclGroupEnd());
  HCHECK(hipStreamSynchronize(s0));
  HCHECK(hipStreamSynchronize(s1));
  unsigned char h[8];
  HCHECK(hipMemcpy(h, b, 8, hipMemcpyDeviceToHost));
  printf("sendrecv ok, byte=%d\n", (int)h[0]);
  return 0;
}
