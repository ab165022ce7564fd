// _gpuwork — MI355X (gfx950/CDNA4) HIP kernels: the GPU pod payload.
//
// Grove schedules inference pods; this framework's GPU-node agent runs each scheduled
// pod's payload on its assigned GPU. The payload is real CDNA4 work (not a sleep):
//   * mfma_gemm_bf16 — LDS-tiled bf16 GEMM on v_mfma_f32_32x32x16_bf16 matrix cores.
//     Two tilings, picked by shape:
//       - 256x256 block tile, 8 waves (512 thr), 2x the arithmetic intensity of the
//         128 tile (the kernel is staging-bandwidth-bound, so intensity is the lever)
//       - 128x128 block tile, 4 waves (256 thr) for shapes not divisible by 256
//     Both: BK=64 double-buffered LDS staged by 16-B global_load_lds (LDS-DMA, the
//     DMA of tile t+1 overlaps the MFMAs of tile t; the __syncthreads() vmcnt(0)
//     drain coincides with the data dependency, so nothing is wasted), conflict-free
//     XOR bank swizzle (row^(row>>3) folded in — PMC-verified 0 LDS conflicts), and
//     an XCD-aware blockIdx→tile map (8-tile M-columns share a B slab per XCD L2).
//     Measured (1xMI355X): 3-buffer raw-barrier pipelining was tried and REGRESSED
//     (96-160 KB LDS drops to 1 WG/CU, losing block-level overlap) — see git history.
//     Register-level fragment double-buffering (ks+1 ds_reads issued under ks's
//     MFMAs) measured +15% @4096³ (975 → 1124 TF/s); a 128x256/BK=32 2-WG/CU
//     variant and bare s_setprio both REGRESSED (experiments/gemm_exp.hip).
//   * stream_triad — float4 HBM streaming (bandwidth probe + memory-heavy payload).
//
// Numerics: tests/test_gpu_kernels.py checks both tilings against a torch fp32
// reference (A=I + asymmetric-B layout traps included).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64
#define BK 64

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;   // 4 VGPRs (A/B fragment)
using f32x16 = __attribute__((ext_vector_type(16))) float;   // 32x32 accumulator

// Bank swizzle for conflict-free ds_read_b128: fold row bit 3+ in so rows that share
// (row & 7) but differ at bit 3 (the b128 lane-group aliases, e.g. rows 12 and 20 in
// group {0-3,12-15,20-27}) land on different banks. Verified: 16 lanes x 4 dwords
// cover all 64 banks in both b128 lane groups; SQ_LDS_BANK_CONFLICT measured 0.
__device__ __forceinline__ int swz(int row, int cb) {
    return (cb ^ ((row ^ (row >> 3)) & 7)) & 7;
}

__device__ __forceinline__ int lds_off(int row, int cb) {
    return row * BK + (swz(row, cb) << 3);   // element (bf16) offset, 8 pieces/row
}

// XCD-aware blockIdx→tile map: consecutive blocks land on XCDs round-robin (b % 8);
// group tiles into 8(M)x1(N) columns so one XCD's resident tiles share a B slab in
// its (non-coherent, per-XCD) L2.
__device__ __forceinline__ void tile_map(int bid, int num_pid_m, int tiles_n,
                                         int& tile_m, int& tile_n) {
    const int GROUP = 8;
    int group_size = min(GROUP, num_pid_m);
    int pids_per_group = group_size * tiles_n;
    int group = bid / pids_per_group;
    int in_group = bid % pids_per_group;
    tile_m = group * GROUP + (in_group % group_size);
    tile_n = in_group / group_size;
}

// Rectangular per-XCD map for LARGE grids (mode 1): with the column map above, at
// 8192³ each XCD's resident blocks share one M-row and walk ALL tiles_n, so every
// XCD streams the whole B matrix from HBM each dispatch wave (~33 K-slabs/XCD).
// Here a 256-block window gives the blocks resident on one XCD (bid % 8, the
// hardware round-robin) an m_per_xcd x n_win RECTANGLE of tiles: the XCD's L2
// re-serves m_per_xcd A-slabs + n_win B-slabs (e.g. 4+8=12 slabs instead of 33).
// Host side guarantees divisibility before selecting mode 1.
__device__ __forceinline__ void tile_map_rect(int bid, int tiles_m, int tiles_n,
                                              int m_per_xcd, int n_win,
                                              int& tile_m, int& tile_n) {
    int mwin_tiles = 8 * m_per_xcd;          // M tiles covered per window
    int win = bid >> 8;                      // 256 blocks per window
    int idx = bid & 255;
    int xcd = idx & 7;
    int slot = idx >> 3;                     // 0 .. m_per_xcd*n_win - 1
    int nwins = tiles_n / n_win;             // windows per M band
    tile_m = (win / nwins) * mwin_tiles + xcd * m_per_xcd + slot % m_per_xcd;
    tile_n = (win % nwins) * n_win + slot / m_per_xcd;
}

// Stage one (rows x BK) bf16 tile via 16-B LDS-DMA. The glds LDS destination is
// wave-uniform-base + lane*16 and the LDS image is lane-linear in piece index, so the
// XOR swizzle is applied on the SOURCE address (guide §5 rule 21: swizzled images via
// pre-swizzled global addresses, LDS stays linear).
template <int THREADS, int AUX = 0>
__device__ __forceinline__ void stage_tile(const bf16* __restrict__ g, long ld,
                                           __bf16* __restrict__ dst, int rows,
                                           int tid, int wave_piece0) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {           // rows*8 pieces / THREADS == 8 iterations
        if (i >= (rows * 8) / THREADS)
            break;
        int p = tid + i * THREADS;
        int row = p >> 3;
        int cb_src = swz(row, p & 7);
        int base = wave_piece0 + i * THREADS;  // uniform across the wave
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)(
                g + (long)row * ld + (cb_src << 3)),
            (__attribute__((address_space(3))) uint32_t*)(dst + base * 8),
            16, 0, AUX);
    }
}

// ---------------------------------------------------------------- 256x256 tile
// 8 waves as 4(row)x2(col); each wave computes 64x128 = 2x4 MFMA 32x32 tiles
// (128 accumulator VGPRs). Arithmetic intensity 128 FLOP/B staged — double the 128
// tile — which is what moves a staging-bound GEMM.
__global__ __launch_bounds__(512, 1)
void mfma_gemm_bf16_256_kernel(const bf16* __restrict__ A,   // [M][K] row-major
                               const bf16* __restrict__ Bt,  // [N][K] row-major
                               float* __restrict__ C,        // [M][N]
                               int M, int N, int K,
                               int m_per_xcd, int n_win) {   // 0,0 = column map
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };

    int tile_m, tile_n;
    if (m_per_xcd)
        tile_map_rect(blockIdx.x, M / BM, N / BN, m_per_xcd, n_win,
                      tile_m, tile_n);
    else
        tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;      // 4(row) x 2(col)
    const long row0 = (long)tile_m * BM;
    const long col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;

    auto stage = [&](int buf, long kk) {
        // nt (aux=2) on the A stream was measured neutral@4096 / -2.5%@8192 —
        // default cache policy kept for both operands
        stage_tile<512>(A + row0 * K + kk, K, sAp(buf), BM, tid, wave_piece0);
        stage_tile<512>(Bt + col0 * K + kk, K, sBp(buf), BN, tid, wave_piece0);
    };

    f32x16 acc[2][4] = {};
    // register-level fragment double-buffer: ks+1's LDS reads issue while ks's MFMAs
    // run, so the pre-cluster s_waitcnt never waits on just-issued ds_reads
    // (measured +15% @4096³: 975 → 1124 TF/s; +5% @8192³)
    bf16x8 afrag[2][2], bfrag[2][4];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 128 + (lane & 31);
    const int k_half = lane >> 5;
    auto load_frags = [&](int pb, int buf, int ks) {
        int cb = (ks << 1) | k_half;
#pragma unroll
        for (int mt = 0; mt < 2; ++mt)
            afrag[pb][mt] = *(const bf16x8*)(sAp(buf) + lds_off(a_row + mt * 32, cb));
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
            bfrag[pb][nt] = *(const bf16x8*)(sBp(buf) + lds_off(b_row + nt * 32, cb));
    };

    stage(0, 0);
    __syncthreads();
    load_frags(0, 0, 0);
    for (long kk = 0; kk < K; kk += BK) {
        int buf = (kk / BK) & 1;
        if (kk + BK < K) stage(buf ^ 1, kk + BK);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {
            int pb = ks & 1;
            if (ks + 1 < BK / 16)
                load_frags(pb ^ 1, buf, ks + 1);
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[pb][mt], bfrag[pb][nt], acc[mt][nt], 0, 0, 0);
        }
        __syncthreads();
        if (kk + BK < K) load_frags(0, buf ^ 1, 0);
    }

    // C/D map for 32x32: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
    const int c_row_lane = 4 * (lane >> 5);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int r = wr * 64 + mt * 32 + (reg & 3) + 8 * (reg >> 2) + c_row_lane;
                int cl = wc * 128 + nt * 32 + (lane & 31);
                C[(row0 + r) * (long)N + col0 + cl] = acc[mt][nt][reg];
            }
}

// ---------------------------------------------------------------- 128x128 tile
// 4 waves as 2x2; each wave 64x64 = 2x2 MFMA tiles. General-shape fallback.
__global__ __launch_bounds__(256, 2)
void mfma_gemm_bf16_128_kernel(const bf16* __restrict__ A,
                               const bf16* __restrict__ Bt,
                               float* __restrict__ C,
                               int M, int N, int K) {
    constexpr int BM = 128, BN = 128;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };

    int tile_m, tile_n;
    tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;
    const long row0 = (long)tile_m * BM;
    const long col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;

    auto stage = [&](int buf, long kk) {
        stage_tile<256>(A + row0 * K + kk, K, sAp(buf), BM, tid, wave_piece0);
        stage_tile<256>(Bt + col0 * K + kk, K, sBp(buf), BN, tid, wave_piece0);
    };

    f32x16 acc[2][2] = {};
    // same register-level fragment double-buffer as the 256 tile
    bf16x8 afrag[2][2], bfrag[2][2];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 64 + (lane & 31);
    const int k_half = lane >> 5;
    auto load_frags = [&](int pb, int buf, int ks) {
        int cb = (ks << 1) | k_half;
#pragma unroll
        for (int mt = 0; mt < 2; ++mt)
            afrag[pb][mt] = *(const bf16x8*)(sAp(buf) + lds_off(a_row + mt * 32, cb));
#pragma unroll
        for (int nt = 0; nt < 2; ++nt)
            bfrag[pb][nt] = *(const bf16x8*)(sBp(buf) + lds_off(b_row + nt * 32, cb));
    };

    stage(0, 0);
    __syncthreads();
    load_frags(0, 0, 0);
    for (long kk = 0; kk < K; kk += BK) {
        int buf = (kk / BK) & 1;
        if (kk + BK < K) stage(buf ^ 1, kk + BK);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {
            int pb = ks & 1;
            if (ks + 1 < BK / 16)
                load_frags(pb ^ 1, buf, ks + 1);
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 2; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[pb][mt], bfrag[pb][nt], acc[mt][nt], 0, 0, 0);
        }
        __syncthreads();
        if (kk + BK < K) load_frags(0, buf ^ 1, 0);
    }

    const int c_row_lane = 4 * (lane >> 5);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 2; ++nt)
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int r = wr * 64 + mt * 32 + (reg & 3) + 8 * (reg >> 2) + c_row_lane;
                int cl = wc * 64 + nt * 32 + (lane & 31);
                C[(row0 + r) * (long)N + col0 + cl] = acc[mt][nt][reg];
            }
}

// ---------------------------------------------------------------- decode GEMV
// y[n] = W[n][k] · x[k], bf16 weights / fp32 accumulate — the batch-1 decode-step
// shape (weight-streaming, HBM-bound: the guide's GEMV rule is load straight to
// VGPRs, deep unroll, no LDS round trip). One workgroup per 64 output rows; each
// wave owns one row block and strides K with 8 bf16 (16 B) per lane per step.
__global__ __launch_bounds__(256)
void decode_gemv_kernel(const bf16* __restrict__ W,   // [N][K] row-major
                        const bf16* __restrict__ x,   // [K]
                        float* __restrict__ y, int N, int K) {
    const int lane = threadIdx.x % WAVE;
    const int wave = threadIdx.x / WAVE;
    const int row = blockIdx.x * 4 + wave;            // 4 waves -> 4 rows per block
    if (row >= N) return;
    const bf16* w = W + (long)row * K;
    float acc = 0.f;
    // 8 bf16 per lane per step => 64 lanes cover 512 elements per wave-step
    for (int k0 = lane * 8; k0 < K; k0 += WAVE * 8) {
        bf16x8 wv = *(const bf16x8*)(w + k0);
        bf16x8 xv = *(const bf16x8*)(x + k0);
#pragma unroll
        for (int i = 0; i < 8; ++i)
            acc += (float)wv[i] * (float)xv[i];
    }
    // wave reduction (64 lanes) via LDS-free shuffles
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, 64);
    if (lane == 0) y[row] = acc;
}

torch::Tensor decode_gemv(torch::Tensor w, torch::Tensor x) {
    TORCH_CHECK(w.is_cuda() && x.is_cuda() && w.dtype() == torch::kBFloat16
                && x.dtype() == torch::kBFloat16, "bf16 GPU tensors required");
    TORCH_CHECK(w.is_contiguous() && x.is_contiguous(), "contiguous required");
    int64_t N = w.size(0), K = w.size(1);
    TORCH_CHECK(x.numel() == K && K % 512 == 0, "x must be [K], K%512==0");
    auto y = torch::empty({N}, w.options().dtype(torch::kFloat32));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(decode_gemv_kernel, dim3((N + 3) / 4), dim3(256), 0, stream,
                       (const bf16*)w.data_ptr(), (const bf16*)x.data_ptr(),
                       y.data_ptr<float>(), (int)N, (int)K);
    return y;
}

// Decode payload: `iters` GEMV steps over an [n,k] weight matrix (one transformer
// projection worth of weight streaming per step); returns achieved GB/s.
double burn_decode(int64_t n, int64_t k, int64_t iters) {
    auto opt = torch::TensorOptions().dtype(torch::kBFloat16).device(torch::kCUDA);
    auto w = torch::randn({n, k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto x = torch::randn({k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto y = torch::empty({n}, opt.dtype(torch::kFloat32));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    auto launch = [&] {
        hipLaunchKernelGGL(decode_gemv_kernel, dim3((n + 3) / 4), dim3(256), 0,
                           stream, (const bf16*)w.data_ptr(),
                           (const bf16*)x.data_ptr(), y.data_ptr<float>(),
                           (int)n, (int)k);
    };
    launch();
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t0 = std::chrono::steady_clock::now();
    for (int64_t i = 0; i < iters; ++i) launch();
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t1 = std::chrono::steady_clock::now();
    double secs = std::chrono::duration<double>(t1 - t0).count();
    return 2.0 * n * k * iters / secs / 1e9;  // weight bytes streamed
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

static void launch_gemm(const torch::Tensor& a, const torch::Tensor& bt,
                        torch::Tensor& c, int64_t M, int64_t N, int64_t K,
                        int64_t map_mode = -1) {
    hipStream_t stream = at::hip::getCurrentHIPStream();
    // 256-tile needs >=256 workgroups to fill the 256-CU chip (one WG per CU at
    // 128 KB LDS); smaller grids run the 128-tile at 4x the block count.
    if (M % 256 == 0 && N % 256 == 0 && (M / 256) * (N / 256) >= 256) {
        int tiles_m = (int)(M / 256), tiles_n = (int)(N / 256);
        // rectangular per-XCD map when the grid divides into 256-block windows
        // (large K streams benefit; the column map wins at <= 16x16 grids)
        int m_per_xcd = 0, n_win = 0;
        int mpx = std::min<int>(4, std::max(1, tiles_m / 8));
        int nw = 256 / (8 * mpx);
        bool want_rect = (map_mode == 1) ||
                         (map_mode < 0 && (int64_t)tiles_m * tiles_n > 256);
        if (want_rect && map_mode != 0 &&
            tiles_m % (8 * mpx) == 0 && tiles_n % nw == 0) {
            m_per_xcd = mpx; n_win = nw;
        }
        dim3 grid((M / 256) * (N / 256)), block(512);
        hipLaunchKernelGGL(mfma_gemm_bf16_256_kernel, grid, block, 0, stream,
                           (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
                           c.data_ptr<float>(), (int)M, (int)N, (int)K,
                           m_per_xcd, n_win);
    } else {
        dim3 grid((M / 128) * (N / 128)), block(256);
        hipLaunchKernelGGL(mfma_gemm_bf16_128_kernel, grid, block, 0, stream,
                           (const bf16*)a.data_ptr(), (const bf16*)bt.data_ptr(),
                           c.data_ptr<float>(), (int)M, (int)N, (int)K);
    }
}

static void check_dims(int64_t M, int64_t N, int64_t K) {
    TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
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
    launch_gemm(a, bt, c, M, N, K);
    return c;
}

// Pod payload: `iters` GEMM steps on pre-allocated buffers; returns achieved TFLOP/s.
double burn_gemm(int64_t m, int64_t n, int64_t k, int64_t iters,
                 int64_t map_mode) {
    check_dims(m, n, k);
    auto opt = torch::TensorOptions().dtype(torch::kBFloat16).device(torch::kCUDA);
    auto a = torch::randn({m, k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto bt = torch::randn({n, k}, opt.dtype(torch::kFloat32)).to(torch::kBFloat16);
    auto c = torch::empty({m, n}, opt.dtype(torch::kFloat32));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    launch_gemm(a, bt, c, m, n, k, map_mode);  // warmup
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    auto t0 = std::chrono::steady_clock::now();
    for (int64_t i = 0; i < iters; ++i) launch_gemm(a, bt, c, m, n, k, map_mode);
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
    m.def("burn_gemm", &burn_gemm, "run iters GEMM steps; returns TFLOP/s",
          pybind11::arg("m"), pybind11::arg("n"), pybind11::arg("k"),
          pybind11::arg("iters"), pybind11::arg("map_mode") = -1);
    m.def("burn_gemm_v",
          [](int64_t m_, int64_t n_, int64_t k_, int64_t it, int64_t) {
              return burn_gemm(m_, n_, k_, it, -1);
          }, "compat alias for burn_gemm (bk arg ignored)");
    m.def("stream_triad", &stream_triad, "HBM triad; returns GB/s");
    m.def("decode_gemv", &decode_gemv, "y[N]=W[N,K]@x[K] (bf16 in, fp32 out)");
    m.def("burn_decode", &burn_decode,
          "iters decode GEMV steps; returns weight-stream GB/s");
}
