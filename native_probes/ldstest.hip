#include <hip/hip_runtime.h>
#include <cstdio>

extern __shared__ unsigned char lds[];

__global__ __launch_bounds__(512, 1) void k(float* out, int n) {
    float* buf = reinterpret_cast<float*>(lds);
    for (int i = threadIdx.x; i < n; i += blockDim.x) buf[i] = (float)i;
    __syncthreads();
    float acc = 0;
    for (int i = threadIdx.x; i < n; i += blockDim.x) acc += buf[i];
    atomicAdd(out, acc);
}

int main() {
    hipDeviceProp_t prop;
    (void)hipGetDeviceProperties(&prop, 0);
    printf("sharedMemPerBlock=%zu maxSharedMemoryPerMultiProcessor=%zu\n",
           prop.sharedMemPerBlock, prop.maxSharedMemoryPerMultiProcessor);
    float* out;
    (void)hipMalloc(&out, 4);
    for (size_t lds_bytes : {64ul*1024, 80ul*1024, 100ul*1024, 128ul*1024, 160ul*1024}) {
        (void)hipMemset(out, 0, 4);
        hipError_t e1 = hipFuncSetAttribute((const void*)k, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);
        hipLaunchKernelGGL(k, dim3(4), dim3(512), lds_bytes, 0, out, (int)(lds_bytes/4));
        hipError_t e2 = hipDeviceSynchronize();
        float host = 0;
        (void)hipMemcpy(&host, out, 4, hipMemcpyDeviceToHost);
        int n = (int)(lds_bytes/4);
        double expect = 4.0 * ((double)n*(n-1)/2.0);
        printf("lds=%zuKB setattr=%s launch=%s sum_ok=%d\n", lds_bytes/1024,
               hipGetErrorString(e1), hipGetErrorString(e2), host == (float)expect);
    }
    return 0;
}
