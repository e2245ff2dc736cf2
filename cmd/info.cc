// nezha-info: device/topology dump + RCCL comm self-check.
#include <cstdio>

#include <hip/hip_runtime.h>

#include "nezha.h"
#include "src/util/util.h"

int main(int argc, const char** argv) {
  nz::Flags flags;
  flags.parse(argc - 1, argv + 1);
  printf("%s\n", nz_version());
  int n = nz_hip_device_count();
  printf("HIP devices: %d\n", n);
  for (int i = 0; i < n; ++i) {
    hipDeviceProp_t prop;
    if (hipGetDeviceProperties(&prop, i) != hipSuccess) continue;
    printf("  [%d] %s  gcnArch=%s  CUs=%d  LDS/CU=%zuKB  HBM=%.0fGB  clk=%dMHz\n",
           i, prop.name, prop.gcnArchName, prop.multiProcessorCount,
           prop.maxSharedMemoryPerMultiProcessor / 1024,
           prop.totalGlobalMem / 1e9, prop.clockRate / 1000);
  }
  if (n > 1) {
    // xGMI topology: P2P access + link type/hops between device pairs
    printf("P2P topology (link:hops, '-' = no direct access):\n     ");
    for (int j = 0; j < n; ++j) printf("%5d", j);
    printf("\n");
    for (int i = 0; i < n; ++i) {
      printf("  %2d ", i);
      for (int j = 0; j < n; ++j) {
        if (i == j) {
          printf("    .");
          continue;
        }
        int access = 0;
        (void)hipDeviceCanAccessPeer(&access, i, j);
        uint32_t link = 0, hops = 0;
        if (access &&
            hipExtGetLinkTypeAndHopCount(i, j, &link, &hops) == hipSuccess)
          printf("  %u:%u", link, hops);
        else
          printf("    -");
      }
      printf("\n");
    }
  }
  if (flags.get_bool("comm-check", false)) {
    if (n == 0) {
      printf("comm-check: no GPU\n");
      return 1;
    }
    nz_ctx* ctx = nz_ctx_new(NZ_HIP, 0);
    char uid[256];
    if (nz_comm_get_uid(uid) != 0) {
      printf("comm-check FAILED: %s\n", nz_last_error());
      return 1;
    }
    nz_comm* comm = nz_comm_init_rccl(ctx, 1, 0, uid);
    if (!comm) {
      printf("comm-check FAILED: %s\n", nz_last_error());
      return 1;
    }
    float v = 2.5f;
    nz_comm_allreduce_scalar(comm, &v, 0);
    printf("comm-check OK (nranks=1 all-reduce: %.1f)\n", v);
    nz_comm_free(comm);
    nz_ctx_free(ctx);
  }
  return 0;
}
