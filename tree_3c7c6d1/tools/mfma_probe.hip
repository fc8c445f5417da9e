// Standalone probe: verify the lane->element map of v_mfma_f32_16x16x32_bf16
#include <cstring>
// on gfx950. Computes C = A(16x32) @ B(32x16) with asymmetric random inputs
// and compares against a CPU fp32 reference under two candidate maps:
//   map0: A[i=l&15][k=(l>>4)*8+t], B[k=(l>>4)*8+t][j=l&15]   (contiguous 8)
//   map1: A[i=l&15][k=(l>>4)*4+t  (t<4), 16+(l>>4)*4+t-4]    (two 16-halves)
// C/D map assumed: col=lane&15, row=(lane>>4)*4+reg.
// Build: hipcc --offload-arch=gfx950 -O3 tools/mfma_probe.hip -o /tmp/mfma_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;


__device__ __forceinline__ unsigned short tobf(float f){
    union{float f;unsigned u;}c;c.f=f;unsigned lsb=(c.u>>16)&1u;c.u+=0x7fffu+lsb;return (unsigned short)(c.u>>16);
}

template<int MAP>
__global__ void probe(const float* A, const float* B, float* C){
    int l = threadIdx.x;
    short8 a, b;
    for (int t=0;t<8;++t){
        int i = l & 15, kq = l >> 4, k;
        if (MAP==0) k = kq*8 + t;
        else        k = (t<4) ? kq*4 + t : 16 + kq*4 + (t-4);
        a[t] = (short)tobf(A[i*32 + k]);
        b[t] = (short)tobf(B[k*16 + i]);   // B[k][j=i]
    }
    floatx4 acc = {0,0,0,0};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    int col = l & 15, r0 = (l>>4)*4;
    for (int j=0;j<4;++j) C[(r0+j)*16 + col] = acc[j];
}

// ---- probe 2: ds_read_tr16_b64 gather map ---------------------------------
// LDS filled with lds[e] = e; each lane reads at addr = base + lane*8B and
// we print which element index each (lane, j) slot received.
typedef __attribute__((ext_vector_type(4))) short short4_;
__global__ void probe_tr(unsigned short* out){
    __shared__ unsigned short lds[512];
    int l = threadIdx.x;
    for (int i = l; i < 512; i += 64) lds[i] = (unsigned short)i;
    __syncthreads();
    short4_ v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) short4_*)&lds[l*4]);
    for (int j=0;j<4;++j) out[l*4+j] = (unsigned short)v[j];
}

int main(){
    float *A, *B, *C;
    hipMallocManaged(&A, 16*32*4); hipMallocManaged(&B, 32*16*4); hipMallocManaged(&C, 16*16*4);
    srand(42);
    for(int i=0;i<16*32;++i) A[i] = (rand()%17-8)*0.25f;
    for(int i=0;i<32*16;++i) B[i] = (rand()%23-11)*0.125f + (i%16)*0.01f; // asymmetric
    // CPU ref (bf16-rounded inputs)
    auto bf=[&](float f){unsigned u;memcpy(&u,&f,4);unsigned l=(u>>16)&1u;u+=0x7fff+l;u=(u>>16)<<16;float r;memcpy(&r,&u,4);return r;};
    float ref[256];
    for(int i=0;i<16;++i)for(int j=0;j<16;++j){float s=0;for(int k=0;k<32;++k)s+=bf(A[i*32+k])*bf(B[k*16+j]);ref[i*16+j]=s;}
    unsigned short* T; hipMallocManaged(&T, 64*4*2);
    hipLaunchKernelGGL(probe_tr, dim3(1), dim3(64), 0, 0, T);
    hipDeviceSynchronize();
    printf("tr16 map (lane: e0 e1 e2 e3):\n");
    for(int l=0;l<64;++l){printf("%2d: %3d %3d %3d %3d%s",l,T[l*4],T[l*4+1],T[l*4+2],T[l*4+3], (l%4==3)?"\n":"   ");}
    for (int MAP=0; MAP<2; ++MAP){
        if (MAP==0) hipLaunchKernelGGL(probe<0>, dim3(1), dim3(64), 0, 0, A,B,C);
        else        hipLaunchKernelGGL(probe<1>, dim3(1), dim3(64), 0, 0, A,B,C);
        hipDeviceSynchronize();
        double maxerr=0; for(int i=0;i<256;++i) maxerr = fmax(maxerr, fabs((double)C[i]-ref[i]));
        printf("MAP%d: maxerr=%g -> %s\n", MAP, maxerr, maxerr < 1e-2 ? "PASS" : "FAIL");
    }
    return 0;
}

