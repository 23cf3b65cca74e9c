// Microbenchmark: issue/throughput rates of the integer ops that make up
// 4x64 Montgomery mul on gfx950 (v_mad_u64_u32, v_mul_lo/hi_u32, 64-bit
// add, 32-bit add chains, v_mov) — measured, not guessed.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
using u32=uint32_t; using u64=uint64_t;

#define ITER 4096
#define UNROLL 16

// independent mads (ILP): 4 chains
__global__ void k_mad64_ilp(u64* out, u64 seed) {
    u64 a=seed+threadIdx.x, b=seed^threadIdx.x, c=seed|1, d=seed+7;
    u32 y=(u32)(seed>>5)|3;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL/4;u++){
            a = (u64)(u32)a*y + a; b = (u64)(u32)b*y + b; c = (u64)(u32)c*y + c; d = (u64)(u32)d*y + d;
        }
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = a+b+c+d;
}
// fully dependent mad chain
__global__ void k_mad64_dep(u64* out, u64 seed) {
    u64 a=seed+threadIdx.x;
    u32 y=(u32)(seed>>5)|3;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL;u++) a = (u64)(u32)a*y + a;
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = a;
}
// 32-bit mul_lo / mul_hi pairs, independent
__global__ void k_mul32_ilp(u64* out, u64 seed) {
    u32 a=(u32)seed+threadIdx.x, b=(u32)(seed>>7)|1, c=(u32)(seed>>13)|5, d=(u32)seed|9;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL/4;u++){
            a = a*b + 1; b = __umulhi(b,c) + b; c = c*d + 3; d = __umulhi(d,a) + d;
        }
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = (u64)a+b+c+d;
}
// 64-bit adds, independent
__global__ void k_add64_ilp(u64* out, u64 seed) {
    u64 a=seed+threadIdx.x, b=seed^threadIdx.x, c=seed|1, d=seed+7;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL/4;u++){ a+=b; b+=c; c+=d; d+=a; }
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = a+b+c+d;
}
// 32-bit adds
__global__ void k_add32_ilp(u64* out, u64 seed) {
    u32 a=(u32)seed+threadIdx.x, b=(u32)seed^threadIdx.x, c=(u32)seed|1, d=(u32)seed+7;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL/4;u++){ a+=b; b+=c; c+=d; d+=a; }
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = (u64)a+b+c+d;
}
// the V3 mont-mul inner step shape: c += (u64)x*b + t  (mad + add64 chain)
__global__ void k_ciosstep(u64* out, u64 seed) {
    u64 c=seed+threadIdx.x;
    u32 x=(u32)(seed>>3)|1, b=(u32)(seed>>5)|3, t=(u32)seed;
    for (int i=0;i<ITER;i++){
#pragma unroll
        for(int u=0;u<UNROLL;u++){ c = (c>>32) + (u64)x*b + t; t = (u32)c; }
    }
    out[blockIdx.x*blockDim.x+threadIdx.x] = c+t;
}

static double run(void(*k)(u64*,u64), const char* name, int ops_per_inner) {
    u64* d; (void)hipMalloc(&d, (size_t)2048*256*8);
    hipEvent_t e0,e1; (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
    // warm
    hipLaunchKernelGGL(k, dim3(2048), dim3(256), 0, 0, d, 12345ull);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0,0);
    hipLaunchKernelGGL(k, dim3(2048), dim3(256), 0, 0, d, 12345ull);
    (void)hipEventRecord(e1,0);
    (void)hipDeviceSynchronize();
    float ms; (void)hipEventElapsedTime(&ms,e0,e1);
    double lanes = 2048.0*256;
    double ops = lanes * ITER * UNROLL * ops_per_inner / (double)UNROLL; // ITER*UNROLL ops per lane
    ops = lanes * (double)ITER * UNROLL;
    double tops = ops / (ms/1e3) / 1e12;
    printf("%-14s %8.3f ms  %7.2f T lane-ops/s\n", name, ms, tops);
    (void)hipFree(d);
    return tops;
}
int main(){
    run(k_mad64_ilp,  "mad64_ilp", 1);
    run(k_mad64_dep,  "mad64_dep", 1);
    run(k_mul32_ilp,  "mul32_ilp", 1);
    run(k_add64_ilp,  "add64_ilp", 1);
    run(k_add32_ilp,  "add32_ilp", 1);
    run(k_ciosstep,   "cios_step", 1);
    return 0;
}
