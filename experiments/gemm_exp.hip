// GEMM experiments (standalone, no torch) — gfx950
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
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
                               int M, int N, int K) {
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };

    int tile_m, tile_n;
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
    bf16x8 afrag[2], bfrag[4];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 128 + (lane & 31);
    const int k_half = lane >> 5;

    stage(0, 0);
    __syncthreads();
    for (long kk = 0; kk < K; kk += BK) {
        int buf = (kk / BK) & 1;
        if (kk + BK < K) stage(buf ^ 1, kk + BK);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {
            int cb = (ks << 1) | k_half;
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                afrag[mt] = *(const bf16x8*)(sAp(buf) + lds_off(a_row + mt * 32, cb));
#pragma unroll
            for (int nt = 0; nt < 4; ++nt)
                bfrag[nt] = *(const bf16x8*)(sBp(buf) + lds_off(b_row + nt * 32, cb));
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[mt], bfrag[nt], acc[mt][nt], 0, 0, 0);
        }
        __syncthreads();
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

__global__ __launch_bounds__(512, 1)
void mfma_gemm_v1_setprio(const bf16* __restrict__ A,   // [M][K] row-major
                               const bf16* __restrict__ Bt,  // [N][K] row-major
                               float* __restrict__ C,        // [M][N]
                               int M, int N, int K) {
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };

    int tile_m, tile_n;
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
    bf16x8 afrag[2], bfrag[4];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 128 + (lane & 31);
    const int k_half = lane >> 5;

    stage(0, 0);
    __syncthreads();
    for (long kk = 0; kk < K; kk += BK) {
        int buf = (kk / BK) & 1;
        if (kk + BK < K) stage(buf ^ 1, kk + BK);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {
            int cb = (ks << 1) | k_half;
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                afrag[mt] = *(const bf16x8*)(sAp(buf) + lds_off(a_row + mt * 32, cb));
#pragma unroll
            for (int nt = 0; nt < 4; ++nt)
                bfrag[nt] = *(const bf16x8*)(sBp(buf) + lds_off(b_row + nt * 32, cb));
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[mt], bfrag[nt], acc[mt][nt], 0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
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


// ---------------------------------------------------------------- V2: register
// fragment double-buffer — load ks+1 fragments while ks computes, so the waitcnt
// before the MFMA cluster never waits on just-issued ds_reads.
__global__ __launch_bounds__(512, 1)
void mfma_gemm_v2_regdb(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                        float* __restrict__ C, int M, int N, int K) {
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };
    int tile_m, tile_n;
    tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;
    const long row0 = (long)tile_m * BM, col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;
    auto stage = [&](int buf, long kk) {
        stage_tile<512>(A + row0 * K + kk, K, sAp(buf), BM, tid, wave_piece0);
        stage_tile<512>(Bt + col0 * K + kk, K, sBp(buf), BN, tid, wave_piece0);
    };
    f32x16 acc[2][4] = {};
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

// ---------------------------------------------------------------- V2: register
// fragment double-buffer — load ks+1 fragments while ks computes, so the waitcnt
// before the MFMA cluster never waits on just-issued ds_reads.
__global__ __launch_bounds__(512, 1)
void mfma_gemm_v4_regdb_prio(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                        float* __restrict__ C, int M, int N, int K) {
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };
    int tile_m, tile_n;
    tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;
    const long row0 = (long)tile_m * BM, col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;
    auto stage = [&](int buf, long kk) {
        stage_tile<512>(A + row0 * K + kk, K, sAp(buf), BM, tid, wave_piece0);
        stage_tile<512>(Bt + col0 * K + kk, K, sBp(buf), BN, tid, wave_piece0);
    };
    f32x16 acc[2][4] = {};
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
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[pb][mt], bfrag[pb][nt], acc[mt][nt], 0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
        if (kk + BK < K) load_frags(0, buf ^ 1, 0);
    }
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

// ---------------------------------------------------------------- V3: 128x256 tile,
// BK=32, 256 threads (4 waves as 2x2, each 64x128 = 2x4 MFMA tiles), LDS 48 KB →
// 2 workgroups/CU (VGPR-bound), so one WG's staging hides under the other's MFMAs.
#define BK32 32
__device__ __forceinline__ int swz32(int row, int cb) {
    return (cb ^ ((row >> 2) & 3)) & 3;   // 4 pieces/row; rows sharing a bank-16
}                                          // group (row%4) get distinct pieces

__device__ __forceinline__ int lds_off32(int row, int cb) {
    return row * BK32 + (swz32(row, cb) << 3);
}

template <int THREADS, int ROWS>
__device__ __forceinline__ void stage_tile32(const bf16* __restrict__ g, long ld,
                                             __bf16* __restrict__ dst, int tid,
                                             int wave_piece0) {
#pragma unroll
    for (int i = 0; i < (ROWS * 4) / THREADS; ++i) {
        int p = tid + i * THREADS;
        int row = p >> 2;
        int cb_src = swz32(row, p & 3);
        int base = wave_piece0 + i * THREADS;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)(
                g + (long)row * ld + (cb_src << 3)),
            (__attribute__((address_space(3))) uint32_t*)(dst + base * 8),
            16, 0, 0);
    }
}

__global__ __launch_bounds__(256, 2)
void mfma_gemm_v3_2wg(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                      float* __restrict__ C, int M, int N, int K) {
    constexpr int BM = 128, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK32 + BN * BK32)];
    auto sAp = [&](int b) { return smem + b * (BM * BK32 + BN * BK32); };
    auto sBp = [&](int b) { return smem + b * (BM * BK32 + BN * BK32) + BM * BK32; };
    int tile_m, tile_n;
    tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;      // 2x2 waves
    const long row0 = (long)tile_m * BM, col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;
    auto stage = [&](int buf, long kk) {
        stage_tile32<256, BM>(A + row0 * K + kk, K, sAp(buf), tid, wave_piece0);
        stage_tile32<256, BN>(Bt + col0 * K + kk, K, sBp(buf), tid, wave_piece0);
    };
    f32x16 acc[2][4] = {};
    bf16x8 afrag[2], bfrag[4];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 128 + (lane & 31);
    const int k_half = lane >> 5;
    stage(0, 0);
    __syncthreads();
    for (long kk = 0; kk < K; kk += BK32) {
        int buf = (kk / BK32) & 1;
        if (kk + BK32 < K) stage(buf ^ 1, kk + BK32);
#pragma unroll
        for (int ks = 0; ks < BK32 / 16; ++ks) {
            int cb = (ks << 1) | k_half;
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                afrag[mt] = *(const bf16x8*)(sAp(buf) + lds_off32(a_row + mt * 32, cb));
#pragma unroll
            for (int nt = 0; nt < 4; ++nt)
                bfrag[nt] = *(const bf16x8*)(sBp(buf) + lds_off32(b_row + nt * 32, cb));
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[mt], bfrag[nt], acc[mt][nt], 0, 0, 0);
        }
        __syncthreads();
    }
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

// ---------------------------------------------------------------- V5: split-K for
// small grids — a 2048³ GEMM has only (2048/256)² = 64 tiles (25% of 256 CUs);
// SPLIT=4 K-partitions give 256 WGs. Each WG accumulates its K-quarter and
// atomicAdds into C (fp32); C must be zeroed first.
__global__ __launch_bounds__(512, 1)
void mfma_gemm_v5_splitk(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                         float* __restrict__ C, int M, int N, int K, int split) {
    constexpr int BM = 256, BN = 256;
    __shared__ __bf16 smem[2 * (BM * BK + BN * BK)];
    auto sAp = [&](int b) { return smem + b * (BM * BK + BN * BK); };
    auto sBp = [&](int b) { return smem + b * (BM * BK + BN * BK) + BM * BK; };
    int tiles = (M / BM) * (N / BN);
    int tile_id = blockIdx.x % tiles;
    int kpart = blockIdx.x / tiles;
    int tile_m, tile_n;
    tile_map(tile_id, M / BM, N / BN, tile_m, tile_n);
    long kspan = (long)K / split;
    long k0 = kpart * kspan, k1 = k0 + kspan;
    const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;
    const long row0 = (long)tile_m * BM, col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;
    auto stage = [&](int buf, long kk) {
        stage_tile<512>(A + row0 * K + kk, K, sAp(buf), BM, tid, wave_piece0);
        stage_tile<512>(Bt + col0 * K + kk, K, sBp(buf), BN, tid, wave_piece0);
    };
    f32x16 acc[2][4] = {};
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
    stage(0, k0);
    __syncthreads();
    load_frags(0, 0, 0);
    for (long kk = k0; kk < k1; kk += BK) {
        int buf = ((kk - k0) / BK) & 1;
        if (kk + BK < k1) stage(buf ^ 1, kk + BK);
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
        if (kk + BK < k1) load_frags(0, buf ^ 1, 0);
    }
    const int c_row_lane = 4 * (lane >> 5);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int r = wr * 64 + mt * 32 + (reg & 3) + 8 * (reg >> 2) + c_row_lane;
                int cl = wc * 128 + nt * 32 + (lane & 31);
                atomicAdd(&C[(row0 + r) * (long)N + col0 + cl], acc[mt][nt][reg]);
            }
}


// ---------------------------------------------------------------- V6: small-shape
// deep-prefetch — 128x128 tile, BK=32, 4 LDS buffers (16 KB each, 64 KB total →
// 2 WGs/CU), raw-barrier pipeline keeping 3 buffers of LDS-DMA in flight. Targets
// latency-bound small GEMMs (2048³ payload shape: only 8 kk iterations at BK=256-tile).
#define V6_NBUF 4
#define V6_LOADS_PER_BUF 4          // per thread: 2 A pieces + 2 B pieces
// s_waitcnt imm: vmcnt[3:0], expcnt[6:4]=7 (no wait), lgkmcnt[11:8]=15 (no wait)
#define V6_WAITCNT(vm) __builtin_amdgcn_s_waitcnt(0xF00 | 0x70 | (vm))

__global__ __launch_bounds__(256, 2)
void mfma_gemm_v6_deep(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                       float* __restrict__ C, int M, int N, int K) {
    constexpr int BM = 128, BN = 128;
    __shared__ __bf16 smem[V6_NBUF * (BM * BK32 + BN * BK32)];
    auto sAp = [&](int b) { return smem + b * (BM * BK32 + BN * BK32); };
    auto sBp = [&](int b) { return smem + b * (BM * BK32 + BN * BK32) + BM * BK32; };
    int tile_m, tile_n;
    tile_map(blockIdx.x, M / BM, N / BN, tile_m, tile_n);
    const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
    const int wr = wave >> 1, wc = wave & 1;
    const long row0 = (long)tile_m * BM, col0 = (long)tile_n * BN;
    const int wave_piece0 = wave * WAVE;
    auto stage = [&](int buf, long kk) {
        stage_tile32<256, BM>(A + row0 * K + kk, K, sAp(buf), tid, wave_piece0);
        stage_tile32<256, BN>(Bt + col0 * K + kk, K, sBp(buf), tid, wave_piece0);
    };
    f32x16 acc[2][2] = {};
    bf16x8 afrag[2], bfrag[2];
    const int a_row = wr * 64 + (lane & 31);
    const int b_row = wc * 64 + (lane & 31);
    const int k_half = lane >> 5;
    const int nkk = K / BK32;
    const int prologue = nkk < V6_NBUF ? nkk : V6_NBUF;
    for (int i = 0; i < prologue; ++i)
        stage(i, (long)i * BK32);
    for (int i = 0; i < nkk; ++i) {
        int buf = i % V6_NBUF;
        // in flight beyond buffer i: min(NBUF-1, nkk-1-i) newer buffers
        int newer = nkk - 1 - i;
        if (newer > V6_NBUF - 1) newer = V6_NBUF - 1;
        switch (newer * V6_LOADS_PER_BUF) {   // imm must be a constant
            case 0: V6_WAITCNT(0); break;
            case 4: V6_WAITCNT(4); break;
            case 8: V6_WAITCNT(8); break;
            default: V6_WAITCNT(12); break;
        }
        __builtin_amdgcn_s_barrier();        // all threads' buffer-i loads landed
#pragma unroll
        for (int ks = 0; ks < BK32 / 16; ++ks) {
            int cb = (ks << 1) | k_half;
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                afrag[mt] = *(const bf16x8*)(sAp(buf) + lds_off32(a_row + mt * 32, cb));
#pragma unroll
            for (int nt = 0; nt < 2; ++nt)
                bfrag[nt] = *(const bf16x8*)(sBp(buf) + lds_off32(b_row + nt * 32, cb));
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 2; ++nt)
                    acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        afrag[mt], bfrag[nt], acc[mt][nt], 0, 0, 0);
        }
        if (i + V6_NBUF < nkk) {
            __builtin_amdgcn_s_barrier();    // everyone done READING slot buf
            stage(buf, (long)(i + V6_NBUF) * BK32);
        }
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

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("hip err %s\n", hipGetErrorString(e)); exit(1);} } while(0)

typedef void (*kfn)(const bf16*, const bf16*, float*, int, int, int);

float run(void(*k)(const bf16*,const bf16*,float*,int,int,int), const char* name,
          const bf16* A, const bf16* Bt, float* C, int M, int N, int K, int iters) {
    dim3 grid((M/256)*(N/256)), block(512);
    hipLaunchKernelGGL(k, grid, block, 0, 0, A, Bt, C, M, N, K);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
    hipEventRecord(t0);
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(k, grid, block, 0, 0, A, Bt, C, M, N, K);
    hipEventRecord(t1);
    HIP_CHECK(hipDeviceSynchronize());
    float ms; hipEventElapsedTime(&ms, t0, t1); ms /= iters;
    double tf = 2.0 * M * N * K / (ms * 1e-3) / 1e12;
    printf("%-24s %8.3f ms  %8.1f TF/s\n", name, ms, tf);
    return ms;
}

int main(int argc, char** argv) {
    int M = argc > 1 ? atoi(argv[1]) : 4096;
    int N = M, K = M, iters = argc > 2 ? atoi(argv[2]) : 20;
    std::vector<__hip_bfloat16> hA((size_t)M*K), hB((size_t)N*K);
    for (size_t i = 0; i < hA.size(); ++i) hA[i] = __hip_bfloat16((float)((i*7+3)%13) * 0.1f - 0.6f);
    for (size_t i = 0; i < hB.size(); ++i) hB[i] = __hip_bfloat16((float)((i*5+1)%11) * 0.1f - 0.5f);
    bf16 *A, *Bt; float *C;
    HIP_CHECK(hipMalloc(&A, hA.size()*2)); HIP_CHECK(hipMalloc(&Bt, hB.size()*2));
    HIP_CHECK(hipMalloc(&C, (size_t)M*N*4));
    HIP_CHECK(hipMemcpy(A, hA.data(), hA.size()*2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(Bt, hB.data(), hB.size()*2, hipMemcpyHostToDevice));
    float base = run(mfma_gemm_bf16_256_kernel, "v0_baseline", A, Bt, C, M, N, K, iters);
    // correctness spot check vs v0: compare a few elements between variants
    std::vector<float> c0(4096);
    HIP_CHECK(hipMemcpy(c0.data(), C, 4096*4, hipMemcpyDeviceToHost));
    auto check = [&](const char* name) {
        std::vector<float> c1(4096);
        HIP_CHECK(hipMemcpy(c1.data(), C, 4096*4, hipMemcpyDeviceToHost));
        for (int i = 0; i < 4096; ++i)
            if (fabsf(c0[i]-c1[i]) > 1e-3f * (1.0f + fabsf(c0[i]))) {
                printf("MISMATCH %s @%d %f vs %f\n", name, i, c0[i], c1[i]); exit(1);
            }
        printf("numerics ok (%s == v0 on probe)\n", name);
    };
    run(mfma_gemm_v1_setprio, "v1_setprio", A, Bt, C, M, N, K, iters);
    check("v1");
    run(mfma_gemm_v2_regdb, "v2_regdb", A, Bt, C, M, N, K, iters);
    check("v2");
    run(mfma_gemm_v4_regdb_prio, "v4_regdb_prio", A, Bt, C, M, N, K, iters);
    check("v4");
    {   // v3: different launch geometry
        dim3 grid((M/128)*(N/256)), block(256);
        hipLaunchKernelGGL(mfma_gemm_v3_2wg, grid, block, 0, 0, A, Bt, C, M, N, K);
        HIP_CHECK(hipDeviceSynchronize());
        hipEvent_t t0, t1; (void)hipEventCreate(&t0); (void)hipEventCreate(&t1);
        (void)hipEventRecord(t0);
        for (int i = 0; i < iters; ++i)
            hipLaunchKernelGGL(mfma_gemm_v3_2wg, grid, block, 0, 0, A, Bt, C, M, N, K);
        (void)hipEventRecord(t1);
        HIP_CHECK(hipDeviceSynchronize());
        float ms; (void)hipEventElapsedTime(&ms, t0, t1); ms /= iters;
        printf("%-24s %8.3f ms  %8.1f TF/s\n", "v3_2wg_128x256_bk32", ms,
               2.0 * M * N * K / (ms * 1e-3) / 1e12);
        check("v3");
    }
    {   // v5 split-K (small-grid shapes): zero C then atomic-accumulate
        int tiles = (M/256)*(N/256);
        int split = tiles >= 256 ? 1 : 256 / tiles;
        if ((long)K % ((long)split * 64) == 0) {
            dim3 grid(tiles * split), block(512);
            auto runv5 = [&]() {
                (void)hipMemsetAsync(C, 0, (size_t)M*N*4, 0);
                hipLaunchKernelGGL(mfma_gemm_v5_splitk, grid, block, 0, 0,
                                   A, Bt, C, M, N, K, split);
            };
            runv5(); HIP_CHECK(hipDeviceSynchronize());
            hipEvent_t t0, t1; (void)hipEventCreate(&t0); (void)hipEventCreate(&t1);
            (void)hipEventRecord(t0);
            for (int i = 0; i < iters; ++i) runv5();
            (void)hipEventRecord(t1);
            HIP_CHECK(hipDeviceSynchronize());
            float ms; (void)hipEventElapsedTime(&ms, t0, t1); ms /= iters;
            printf("%-24s %8.3f ms  %8.1f TF/s (split=%d)\n", "v5_splitk", ms,
                   2.0 * M * N * K / (ms * 1e-3) / 1e12, split);
            check("v5");
        }
    }
    {   // v6 deep-prefetch small-shape kernel: its own geometry (128x128)
        dim3 grid((M/128)*(N/128)), block(256);
        hipLaunchKernelGGL(mfma_gemm_v6_deep, grid, block, 0, 0, A, Bt, C, M, N, K);
        HIP_CHECK(hipDeviceSynchronize());
        hipEvent_t t0, t1; (void)hipEventCreate(&t0); (void)hipEventCreate(&t1);
        (void)hipEventRecord(t0);
        for (int i = 0; i < iters; ++i)
            hipLaunchKernelGGL(mfma_gemm_v6_deep, grid, block, 0, 0, A, Bt, C, M, N, K);
        (void)hipEventRecord(t1);
        HIP_CHECK(hipDeviceSynchronize());
        float ms; (void)hipEventElapsedTime(&ms, t0, t1); ms /= iters;
        printf("%-24s %8.3f ms  %8.1f TF/s\n", "v6_deep_128_bk32", ms,
               2.0 * M * N * K / (ms * 1e-3) / 1e12);
        check("v6");
    }
    printf("baseline %.3f ms\n", base);
    return 0;
}

