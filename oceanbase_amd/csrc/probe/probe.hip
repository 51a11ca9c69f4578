// scratch perf probes (not product code): layer-by-layer bandwidth isolation
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#define WG 256

// A: plain grid-stride dwordx4 streaming read
extern "C" __global__ void p_stream(const uint4* __restrict__ g, size_t n16,
                                    unsigned long long* sink) {
  uint64_t acc = 0;
  for (size_t i = (size_t)blockIdx.x * WG + threadIdx.x; i < n16;
       i += (size_t)gridDim.x * WG) {
    uint4 v = g[i];
    acc += v.x + v.y + v.z + v.w;
  }
  if (acc == 0xdeadbeefdeadbeefull) *sink = acc;
}

// B: block-structured DMA stage + trivial consume (mimics filter pipeline)
extern "C" __global__ __launch_bounds__(WG, 2) void p_stage(
    const uint8_t* __restrict__ buf, const uint64_t* __restrict__ offs,
    uint32_t n_blocks, uint32_t blk16, unsigned long long* sink) {
  __shared__ uint8_t lds[2 * 17408 + 32];
  uint64_t acc = 0;
  uint32_t par = 0;
  auto issue = [&](uint32_t b, uint8_t* dst) {
    const uint8_t* src = buf + offs[b];
    for (uint32_t i = threadIdx.x; i < blk16; i += WG)
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(src + (size_t)i * 16),
          (__attribute__((address_space(3))) uint32_t*)(dst + (size_t)i * 16),
          16, 0, 0);
  };
  if (blockIdx.x < n_blocks) issue(blockIdx.x, lds);
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    uint32_t b2 = b + gridDim.x;
    if (b2 < n_blocks) issue(b2, lds + (par ^ 1) * 17408);
    const uint64_t* w = (const uint64_t*)(lds + par * 17408);
    // trivial consume: each thread reads 8 u64 spread over the block
    for (uint32_t i = threadIdx.x; i < blk16 * 2; i += WG) acc += w[i];
    par ^= 1;
    __syncthreads();
  }
  if (acc == 0xdeadbeefdeadbeefull) *sink = acc;
}

// C: like B but with per-row bit unpack + compare + ballot (filter shape)
extern "C" __global__ __launch_bounds__(WG, 2) void p_filter(
    const uint8_t* __restrict__ buf, const uint64_t* __restrict__ offs,
    uint32_t n_blocks, uint32_t blk16, uint32_t rows_per_block,
    unsigned long long* sink) {
  __shared__ uint8_t lds[2 * 17408 + 32];
  __shared__ unsigned long long wg_cnt;
  if (threadIdx.x == 0) wg_cnt = 0;
  uint32_t par = 0;
  auto issue = [&](uint32_t b, uint8_t* dst) {
    const uint8_t* src = buf + offs[b];
    for (uint32_t i = threadIdx.x; i < blk16; i += WG)
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(src + (size_t)i * 16),
          (__attribute__((address_space(3))) uint32_t*)(dst + (size_t)i * 16),
          16, 0, 0);
  };
  if (blockIdx.x < n_blocks) issue(blockIdx.x, lds);
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    uint32_t b2 = b + gridDim.x;
    if (b2 < n_blocks) issue(b2, lds + (par ^ 1) * 17408);
    const uint8_t* base = lds + par * 17408;
    const uint32_t iters = (rows_per_block + WG - 1) / WG;
    for (uint32_t it = 0; it < iters; it++) {
      uint32_t r = it * WG + threadIdx.x;
      bool pass = r < rows_per_block;
      if (pass) {
        uint64_t bitpos = 80 * 8 + (uint64_t)r * 64;  // 8B/row after header
        const uint64_t* w = (const uint64_t*)base;
        uint64_t widx = bitpos >> 6;
        uint64_t v = w[widx];
        pass = (int64_t)v < 24;
      }
      uint64_t m = __ballot(pass);
      if ((threadIdx.x & 63) == 0 && m)
        atomicAdd(&wg_cnt, (unsigned long long)__popcll(m));
    }
    par ^= 1;
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(sink, wg_cnt);
}

int main(int argc, char** argv) {
  size_t MB = 800;
  size_t bytes = MB << 20;
  uint8_t* d;
  (void)hipMalloc(&d, bytes + 64);
  (void)hipMemset(d, 1, bytes);
  unsigned long long* sink;
  (void)hipMalloc(&sink, 8);
  // block table: 16KB blocks
  uint32_t blk_bytes = 16384;
  uint32_t n_blocks = bytes / blk_bytes;
  std::vector<uint64_t> offs(n_blocks);
  for (uint32_t i = 0; i < n_blocks; i++) offs[i] = (uint64_t)i * blk_bytes;
  uint64_t* d_offs;
  (void)hipMalloc(&d_offs, n_blocks * 8);
  (void)hipMemcpy(d_offs, offs.data(), n_blocks * 8, hipMemcpyHostToDevice);

  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
  auto bench = [&](const char* name, auto launch) {
    launch();  // warmup
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0);
    for (int i = 0; i < 5; i++) launch();
    (void)hipEventRecord(e1);
    (void)hipDeviceSynchronize();
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    ms /= 5;
    printf("%-28s %8.3f ms  %8.1f GB/s\n", name, ms, bytes / (ms * 1e6));
  };
  for (int grid : {2048, 4096, 8192}) {
    char nm[64];
    snprintf(nm, 64, "A stream grid=%d", grid);
    bench(nm, [&] {
      hipLaunchKernelGGL(p_stream, dim3(grid), dim3(WG), 0, 0,
                         (const uint4*)d, bytes / 16, sink);
    });
  }
  for (int grid : {2048, 4096, 8192}) {
    char nm[64];
    snprintf(nm, 64, "B stage grid=%d", grid);
    bench(nm, [&] {
      hipLaunchKernelGGL(p_stage, dim3(grid), dim3(WG), 0, 0, d, d_offs,
                         n_blocks, blk_bytes / 16, sink);
    });
  }
  for (int grid : {4096}) {
    char nm[64];
    snprintf(nm, 64, "C filter grid=%d", grid);
    bench(nm, [&] {
      hipLaunchKernelGGL(p_filter, dim3(grid), dim3(WG), 0, 0, d, d_offs,
                         n_blocks, blk_bytes / 16, 2038, sink);
    });
  }
  return 0;
}
