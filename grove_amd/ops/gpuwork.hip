// _gpuwork — MI355X (gfx950/CDNA4) HIP kernels: the GPU pod payload.
//
// Grove schedules inference pods; this framework's GPU-node agent runs each scheduled
// pod's payload on its assigned GPU. The payload is real CDNA4 work (not a sleep):
//   * mfma_gemm_bf16 — LDS-tiled bf16 GEMM on v_mfma_f32_32x32x16_bf16 matrix cores:
//     128×128 block tile, BK=64 double-buffered LDS, XOR-swizzled (cb ^ (row&7)) images
//     for conflict-free ds_read_b128 fragment reads, XCD-aware block swizzle
//     (blockIdx → tile map groups tiles per XCD for L2 reuse, 8 XCDs).
//   * stream_triad — float4 HBM streaming (bandwidth probe + memory-heavy payload).
//
// Numerics: tests/test_gpu_kernels.py checks mfma_gemm_bf16 against a torch fp32
// reference (A=I + asymmetric-B layout traps included).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64
#define BM 128
#define BN 128
#define BK BKT

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;   // 4 VGPRs (A/B fragment)
using f32x16 = __attribute__((ext_vector_type(16))) float;   // 32x32 accumulator
using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef short short8 __attribute__((ext_vector_type(8)));

// LDS image: [rows][8 pieces of 16B], piece column cb stored at cb ^ (row & 7).
// A row is 64 bf16 = 128 B = 8 pieces; the XOR spreads the b128 fragment reads
// (fixed cb over 32 rows) across all banks.
template <int BKT>
__device__ __forceinline__ int swz(int row, int cb) {
    // Bank swizzle for conflict-free ds_read_b128: fold row bit 3+ in so rows that
    // share (row & 7) but differ at bit 3 (the b128 lane-group aliases, e.g. rows 12
    // and 20 in group {0-3,12-15,20-27}) land on different banks. Verified: 16 lanes
    // x 4 dwords cover all 64 banks in both b128 lane groups.
    int e = (row ^ (row >> 3)) & 7;
    return ((cb ^ e) & (BKT / 8 - 1)) | (cb & ~7);
}

template <int BKT>
__device__ __forceinline__ int lds_off(int row, int cb) {
    return row * BKT + (swz<BKT>(row, cb) << 3);
}

// One workgroup = 256 threads = 4 waves arranged 2x2; each wave owns a 64x64 output
// quadrant = 2x2 MFMA 32x32 tiles.
template <int BKT>
__global__ __launch_bounds__(256, 2)
void mfma_gemm_bf16_kernel(const bf16* __restrict__ A,   // [M][K] row-major
                           const bf16* __restrict__ Bt,  // [N][K] row-major (B^T)
                           float* __restrict__ C,        // [M][N]
                           int M, int N, int K) {
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    // buffer b: A at b*(BM+BN)*BK, B at that + BM*BK (single __shared__ object —
    // guide §5 trap 4(a): a second __shared__ object de-pipelines the k-loop)
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };

    const int tiles_n = N / BN;
    // XCD-aware swizzle: consecutive blocks land on XCDs round-robin (b % 8); map so
    // the 8 tiles resident on one XCD at a time form a 8(M)x1(N) column sharing the
    // same B tile slab in that XCD's L2.
    int bid = blockIdx.x;
    const int GROUP = 8;
    int num_pid_m = M / BM;
    int group_size = min(GROUP, num_pid_m);
    int pids_per_group = group_size * tiles_n;
    int group = bid / pids_per_group;
    int in_group = bid % pids_per_group;
    int tile_m = group * GROUP + (in_group % group_size);
    int tile_n = in_group / group_size;

    const int tid = threadIdx.x;
    const int wave = tid / WAVE;          // 0..3
    const int lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;  // wave quadrant (2x2)

    const long row0 = (long)tile_m * BM;
    const long col0 = (long)tile_n * BN;

    // Staging via 16-B LDS-DMA (global_load_lds, guide §5 ladder step 3): thread t
    // covers LDS piece p = t + i*256 (row = p/8, cb_slot = p%8). The glds LDS
    // destination is wave-uniform-base + lane*16 and our LDS image is lane-linear in
    // p, so the XOR swizzle is applied on the SOURCE address (cb_src = cb_slot ^
    // (row&7)) while LDS stays linear — guide §5 rule 21. Compute on buf overlaps the
    // DMA into buf^1; __syncthreads() drains it (vmcnt(0)) each K-tile.
    const int wave_piece0 = (tid / WAVE) * WAVE;  // wave-uniform piece base
    auto stage = [&](int buf, long kk) {
        const bf16* gA = A + row0 * K + kk;
        const bf16* gB = Bt + col0 * K + kk;
        __bf16* dA = sAp(buf);
        __bf16* dB = sBp(buf);
        constexpr int PPR = BKT / 8;  // 16-B pieces per row
#pragma unroll
        for (int i = 0; i < (BM * BKT / 8) / 256; ++i) {
            int p = tid + i * 256;
            int row = p / PPR;
            int cb_slot = p % PPR;
            int cb_src = swz<BKT>(row, cb_slot);  // XOR is involutive: source swizzle = image swizzle
            int base = wave_piece0 + i * 256;  // uniform across the wave
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)(
                    gA + (long)row * K + (cb_src << 3)),
                (__attribute__((address_space(3))) uint32_t*)(dA + base * 8),
                16, 0, 0);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)(
                    gB + (long)row * K + (cb_src << 3)),
                (__attribute__((address_space(3))) uint32_t*)(dB + base * 8),
                16, 0, 0);
        }
    };

    f32x16 acc[2][2] = {};   // [mt][nt] 32x32 tiles
    bf16x8 afrag[2], bfrag[2];

    stage(0, 0);
    __syncthreads();

    const int a_row = wr * 64 + (lane & 31);        // two m-tiles: +0 / +32
    const int b_row = wc * 64 + (lane & 31);
    const int k_half = lane >> 5;                   // 0/1 → k piece within step

    for (long kk = 0; kk < K; kk += BK) {
        int buf = (kk / BK) & 1;
        if (kk + BK < K) stage(buf ^ 1, kk + BK);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {      // 4 MFMA k-steps of 16
            int cb = (ks << 1) | k_half;            // 16-bf16 step = two 8-elt pieces
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                afrag[mt] = *(const bf16x8*)(sAp(buf) + lds_off<BKT>(a_row + mt * 32, cb));
#pragma unroll
            for (int nt = 0; nt < 2; ++nt)
                bfrag[nt] = *(const bf16x8*)(sBp(buf) + lds_off<BKT>(b_row + nt * 32, cb));
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 2; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[mt], bfrag[nt], acc[mt][nt], 0, 0, 0);
        }
        __syncthreads();
    }

    // Epilogue: C/D map for 32x32: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
    const int c_col_base = wc * 64 + (lane & 31);
    const int c_row_lane = 4 * (lane >> 5);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int r = wr * 64 + mt * 32 + (reg & 3) + 8 * (reg >> 2) + c_row_lane;
                int cl = c_col_base + nt * 32;
                C[(row0 + r) * (long)N + col0 + cl] = acc[mt][nt][reg];
            }
        }
}

__global__ void stream_triad_kernel(const float4* __restrict__ a,
                                    const float4* __restrict__ b,
                                    float4* __restrict__ c, long n4, float s) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n4; i += stride) {
        float4 x = a[i], y = b[i];
        c[i] = make_float4(x.x + s * y.x, x.y + s * y.y, x.z + s * y.z, x.w + s * y.w);
    }
}

// ---------------------------------------------------------------- host wrappers

static void check_dims(int64_t M, int64_t N, int64_t K) {
    TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % 64 == 0,
                "mfma_gemm_bf16 requires M%128==0, N%128==0, K%64==0 (got ",
                M, "x", N, "x", K, ")");
}

torch::Tensor mfma_gemm_bf16(torch::Tensor a, torch::Tensor bt) {
    TORCH_CHECK(a.is_cuda() && bt.is_cuda(), "inputs must be on GPU");
    TORCH_CHECK(a.dtype() == torch::kBFloat16 && bt.dtype() == torch::kBFloat16,
                "inputs must be bf16");
    TORCH_CHECK(a.is_contiguous() && bt.is_contiguous(), "inputs must be contiguous");
    int64_t M = a.size(0), K = a.size(1), N = bt.size(0);
    TORCH_CHECK(bt.size(1) == K, "shape mismatch: A[M,K] Bt[N,K]");
    check_dims(M, N, K);
    auto c = torch::empty({M, N}, a.options().dtype(torch::kFloat32));
    dim3 grid((M / BM) * (N / BN)), block(256);
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(HIP_KERNEL_NAME(mfma_gemm_bf16_kernel<64>), grid, block, 0, stream,
                       (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
                       c.data_ptr<float>(), (int)M, (int)N, (int)K);
    return c;
}

// Pod payload: `iters` GEMM steps on pre-allocated buffers; returns achieved TFLOP/s.
double burn_gemm_v(int64_t m, int64_t n, int64_t k, int64_t iters, int64_t bk) {
    check_dims(m, n, k);
    auto opt = torch::TensorOptions().dtype(torch::kBFloat16).device(torch::kCUDA);
    auto a = torch::randn({m, k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto bt = torch::randn({n, k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto c = torch::empty({m, n}, opt.dtype(torch::kFloat32));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    dim3 grid((m / BM) * (n / BN)), block(256);
    auto launch = [&] {
        if (bk == 128)
            hipLaunchKernelGGL(HIP_KERNEL_NAME(mfma_gemm_bf16_kernel<128>), grid,
                               block, 0, stream,
                               (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
                               c.data_ptr<float>(), (int)m, (int)n, (int)k);
        else
            hipLaunchKernelGGL(HIP_KERNEL_NAME(mfma_gemm_bf16_kernel<64>), grid,
                               block, 0, stream,
                               (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
                               c.data_ptr<float>(), (int)m, (int)n, (int)k);
    };
    launch();  // warmup
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t0 = std::chrono::steady_clock::now();
    for (int64_t i = 0; i < iters; ++i) launch();
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t1 = std::chrono::steady_clock::now();
    double secs = std::chrono::duration<double>(t1 - t0).count();
    return 2.0 * m * n * k * iters / secs / 1e12;
}

// HBM bandwidth probe; returns GB/s (3 streams: 2 read + 1 write).
double stream_triad(int64_t n_floats, int64_t iters) {
    TORCH_CHECK(n_floats % 4 == 0, "n_floats must be a multiple of 4");
    auto opt = torch::TensorOptions().dtype(torch::kFloat32).device(torch::kCUDA);
    auto a = torch::randn({n_floats}, opt);
    auto b = torch::randn({n_floats}, opt);
    auto c = torch::empty({n_floats}, opt);
    long n4 = n_floats / 4;
    hipStream_t stream = at::hip::getCurrentHIPStream();
    // ≫256 workgroups to fill 256 CUs across 8 XCDs
    int blocks = std::min<long>(8192, (n4 + 255) / 256);
    auto launch = [&] {
        hipLaunchKernelGGL(stream_triad_kernel, dim3(blocks), dim3(256), 0, stream,
                           (const float4*)a.data_ptr(), (const float4*)b.data_ptr(),
                           (float4*)c.data_ptr(), n4, 1.5f);
    };
    launch();
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t0 = std::chrono::steady_clock::now();
    for (int64_t i = 0; i < iters; ++i) launch();
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t1 = std::chrono::steady_clock::now();
    double secs = std::chrono::duration<double>(t1 - t0).count();
    return 12.0 * n_floats * iters / secs / 1e9;  // 3 × 4 B per element
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "grove_amd MI355X pod-payload kernels (MFMA bf16 GEMM, HBM stream)";
    m.def("mfma_gemm_bf16", &mfma_gemm_bf16, "C[M,N]=A[M,K]@Bt[N,K]^T (bf16 in, fp32 out)");
    m.def("burn_gemm",
          [](int64_t m_, int64_t n_, int64_t k_, int64_t it) {
              return burn_gemm_v(m_, n_, k_, it, 64);
          }, "run iters GEMM steps; returns TFLOP/s");
    m.def("burn_gemm_v", &burn_gemm_v, "burn_gemm with explicit BK (64 or 128)");
    m.def("stream_triad", &stream_triad, "HBM triad; returns GB/s");
}
