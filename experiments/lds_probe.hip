/* Probe: dynamic-LDS launch limits on gfx950 + a bounded repro of the
 * skew-join shape. Diagnostic for the staged-scatter wedge. */
#include <hip/hip_runtime.h>

#include <cstdio>

__global__ void lds_touch(int n, int* out)
{
  extern __shared__ int buf[];
  buf[threadIdx.x] = threadIdx.x;
  __syncthreads();
  if (threadIdx.x == 0) *out = buf[n % blockDim.x];
}

int main()
{
  hipDeviceProp_t prop;
  hipGetDeviceProperties(&prop, 0);
  printf("sharedMemPerBlock=%zu maxSharedMemoryPerMultiProcessor=%d\n",
         prop.sharedMemPerBlock, prop.maxSharedMemoryPerMultiProcessor);
  int* out;
  hipMalloc(&out, 4);
  for (size_t lds : {65536UL, 65552UL, 69632UL, 77824UL, 131072UL, 163840UL}) {
    hipLaunchKernelGGL(lds_touch, dim3(4), dim3(1024), lds, 0, 7, out);
    hipError_t e1 = hipGetLastError();
    hipError_t e2 = hipDeviceSynchronize();
    printf("lds=%zu launch=%s sync=%s\n", lds, hipGetErrorString(e1), hipGetErrorString(e2));
  }
  return 0;
}
