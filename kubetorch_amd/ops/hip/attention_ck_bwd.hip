// CK-tile FMHA backward instantiation: bf16, head_dim 128, causal, batch
// mode, GQA-native (nhead_ratio_qk — K/V are read per KV-head directly, no
// host-side repeat_interleave), no bias/dropout, non-deterministic dq.
//
// Three kernels, as in the CK bwd design:
//   1. FmhaBwdOGradDotOKernel   : D = rowsum(dO * O)            [B,Hq,S] f32
//   2. FmhaBwdDQDKDVKernel      : main loop (K/V resident per N0-block of
//      keys, M-loop over queries); dq accumulated atomically into an fp32
//      buffer, dk/dv written per Q-head (host sums groups)
//   3. FmhaBwdConvertQGradKernel: dq_acc f32 -> dq bf16
//
// TWO pipeline instantiations, runtime-selected (pipe arg):
//   0 "std":    kUseTrLoad=false -> BlockFmhaBwdDQDKDVPipelineKRKTRVRIGLP,
//               classic 32x32x16 warp tiles (the widely-validated CK
//               codegen config for hd128)
//   1 "trload": kUseTrLoad=true -> BlockFmhaBwdDQDKDVPipelineTrLoadKRKTRVR
//               (gfx950 ds_read_b64_tr_b16 transposed-fragment loads),
//               16x16x32 warp tiles (kCM0PerLane==1 constraint)
// Replaces torch's AITER asm bwd path (dense over expanded KV) —
// profiles/ROUND2.md lever #1.
#include <ck_tile/core.hpp>
#include <ck_tile/host/kernel_launch.hpp>
#include <ck_tile/ops/epilogue.hpp>
#include <ck_tile/ops/fmha.hpp>

namespace {

using bf16 = ck_tile::bf16_t;

using BwdTraits = ck_tile::TileFmhaBwdTraits<
    0 /* kPadHeadDimQ: exact 128 */, 0 /* kPadHeadDimV */,
    ck_tile::BlockAttentionBiasEnum::NO_BIAS, false /* kHasBiasGrad */>;

using BwdMask = ck_tile::SimplifiedGenericAttentionMask<true>;
using BwdDropout = ck_tile::BlockDropoutBwd<false, false, false>;

// hd128 bwd tile: K/V block 128 keys resident, 32-query M-loop.
// BlockTile = <M0, N0, K0, K1, K2, K3, K4, QKHeaddim, VHeaddim>
// constraints: kM0 == kK1 == kK3 (contraction over the M tile).
// kK0 and kK2 MUST equal the head dim: the KR/VR pipelines issue gemm0
// (Q@K^T) and gemm2 (dO@V^T) as a SINGLE block-gemm over a register tile
// read as (kN0, kK0)/(kN0, kK2) — a smaller kK0 silently contracts only
// the first kK0 of 128 head dims (diagnosed on-box: chunk regression
// showed coeffs [1,1,1,1,0,...] at kK0=32, gpurun_out/bwd_probe4.log).
// Only gemm4 (dS@K over kN0 keys) has an explicit k4 loop.
// Block-warp M dims follow gemm1/3 M=N0 (dV/dK partition keys across
// warps), gemm0/2/4 M=M0.
template <bool UseTrLoad, int WtM, int WtN, int WtK, int M0 = 32,
          int N0 = 128>
struct BwdConfig {
  using Shape = ck_tile::TileFmhaBwdShape<
      ck_tile::sequence<M0, N0, 128, M0, 128, M0, 32, 128, 128>,
      ck_tile::sequence<1, 4, 1>, ck_tile::sequence<WtM, WtN, WtK>,  // gemm0 S
      ck_tile::sequence<4, 1, 1>, ck_tile::sequence<WtM, WtN, WtK>,  // gemm1 dV
      ck_tile::sequence<1, 4, 1>, ck_tile::sequence<WtM, WtN, WtK>,  // gemm2 dP
      ck_tile::sequence<4, 1, 1>, ck_tile::sequence<WtM, WtN, WtK>,  // gemm3 dK
      ck_tile::sequence<1, 4, 1>, ck_tile::sequence<WtM, WtN, WtK>>; // gemm4 dQ

  using Problem = ck_tile::BlockFmhaBwdPipelineProblem<
      bf16 /* Q */, bf16 /* K */, bf16 /* V */, bf16 /* Gemm */,
      float /* LSE */, float /* Acc */, float /* D */, bf16 /* Bias */,
      uint8_t /* RandVal */, bf16 /* O */, bf16 /* OGrad */, bf16 /* QGrad */,
      bf16 /* KGrad */, bf16 /* VGrad */, bf16 /* BiasGrad */, Shape,
      false /* kIsGroupMode */, false /* kIsDeterministic */, BwdMask,
      BwdDropout, UseTrLoad, BwdTraits>;

  using Pipeline = ck_tile::BlockFmhaBwdDQDKDVPipeline<Problem>;

  using KGradEpilogue = ck_tile::Default2DEpilogue<
      ck_tile::Default2DEpilogueProblem<float, bf16, false, false>>;
  using VGradEpilogue = ck_tile::Default2DEpilogue<
      ck_tile::Default2DEpilogueProblem<float, bf16, false, false>>;

  using Kernel = ck_tile::FmhaBwdDQDKDVKernel<Pipeline, KGradEpilogue,
                                              VGradEpilogue>;
};

using StdKernel = BwdConfig<false, 32, 32, 16>::Kernel;
using TrKernel = BwdConfig<true, 16, 16, 32>::Kernel;
using TrKernel64 = BwdConfig<true, 16, 16, 32, 64>::Kernel;  // pipe=2
// pipe=3: N0=64 halves the register-resident K/KT/V tiles (occupancy 2
// candidate; 2x the key-block grid)
using TrKernelN64 = BwdConfig<true, 16, 16, 32, 32, 64>::Kernel;

// --- D = rowsum(dO*O) -------------------------------------------------------
using DotTraits = ck_tile::TileFmhaBwdOGradDotOTraits<
    false /* kPadSeqLenQ: S % 64 == 0 gate */, false /* kPadHeadDimV */>;
using DotProblem = ck_tile::BlockFmhaBwdOGradDotOPipelineProblem<
    bf16 /* O */, bf16 /* OGrad */, float /* D */, 64 /* kBlockSize */,
    128 /* kVHeaddim */, false /* kIsGroupMode */, DotTraits>;
using DotKernel =
    ck_tile::FmhaBwdOGradDotOKernel<ck_tile::BlockFmhaBwdOGradDotO<DotProblem>>;

// --- dq_acc f32 -> dq bf16 --------------------------------------------------
using CvtTraits = ck_tile::TileFmhaBwdConvertQGradTraits<
    false /* kPadSeqLenQ */, false /* kPadHeadDimQ */>;
using CvtProblem = ck_tile::BlockFmhaBwdConvertQGradPipelineProblem<
    float /* Acc */, bf16 /* QGrad */, 256 /* kBlockSize */, 64 /* kM0 */,
    128 /* kN0 */, 128 /* kQKHeaddim */, false /* kIsGroupMode */,
    false /* kIsDeterministic */, CvtTraits>;
using CvtKernel =
    ck_tile::FmhaBwdConvertQGradKernel<ck_tile::BlockFmhaBwdConvertQGrad<CvtProblem>>;

template <typename K, typename... Args>
void launch(hipStream_t stream, dim3 grid, Args&&... args) {
  ck_tile::stream_config cfg{};
  cfg.stream_id_ = stream;
  cfg.cold_niters_ = 0;
  cfg.nrepeat_ = 1;
  ck_tile::launch_kernel(
      cfg, ck_tile::make_kernel<K::kBlockPerCu>(
               K{}, grid, K::BlockSize(), 0, std::forward<Args>(args)...));
}

template <typename MainKernel>
void run_bwd(const void* q, const void* k, const void* v, const void* o,
             const void* do_, const void* lse, void* d, void* dq_acc, void* dq,
             void* dk, void* dv, int B, int Hq, int Hkv, int S, float scale,
             int mask_mode, hipStream_t stream) {
  const ck_tile::index_t D = 128;
  const ck_tile::index_t sq = (ck_tile::index_t)S * D;   // nhead stride q-side
  const ck_tile::index_t bq = (ck_tile::index_t)Hq * sq; // batch stride q-side
  const ck_tile::index_t bk = (ck_tile::index_t)Hkv * sq;

  {  // 1: D = rowsum(dO * O)
    auto kargs = DotKernel::MakeKargs(
        o, do_, d, 1.0f /* p_undrop */, S, D,
        D /* stride_do */, D /* stride_o */, sq /* nhead_stride_do */,
        sq /* nhead_stride_o */, S /* nhead_stride_d */,
        bq /* batch_stride_do */, bq /* batch_stride_o */,
        (ck_tile::index_t)Hq * S /* batch_stride_d */);
    launch<DotKernel>(stream, DotKernel::GridSize(B, Hq, S), kargs);
  }
  {  // 2: dq_acc / dk / dv
    auto kargs = MainKernel::MakeKargsImpl(
        q, k, v, nullptr /* bias */, lse, do_, d, nullptr /* randval */,
        dk, dv, nullptr /* dbias */, dq_acc,
        S /* seqlen_q */, S /* seqlen_k */, D, D, Hq, Hq / Hkv, scale,
        D /* stride_q */, D /* stride_k */, D /* stride_v */,
        0 /* stride_bias */, 0 /* stride_randval */, D /* stride_do */,
        D /* stride_dq_acc */, D /* stride_dk */, D /* stride_dv */,
        0 /* stride_dbias */,
        sq /* nhead_stride_q */, sq /* nhead_stride_k */,
        sq /* nhead_stride_v */, 0, 0, sq /* nhead_stride_do */,
        S /* nhead_stride_lsed */, sq /* nhead_stride_dq_acc */,
        sq /* nhead_stride_dk */, sq /* nhead_stride_dv */, 0,
        bq /* batch_stride_q */, bk /* batch_stride_k */,
        bk /* batch_stride_v */, 0, 0, bq /* batch_stride_do */,
        (ck_tile::index_t)Hq * S /* batch_stride_lsed */,
        bq /* batch_stride_dq_acc */, bq /* batch_stride_dk */,
        bq /* batch_stride_dv */, 0,
        0 /* split_stride_dq_acc */,
        (ck_tile::index_t)(mask_mode == 2 ? 0 : -1) /* window_left */,
        (ck_tile::index_t)(mask_mode == 2 ? -1 : 0) /* window_right */,
        (ck_tile::index_t)(mask_mode == 1
            ? ck_tile::GenericAttentionMaskEnum::MASK_FROM_BOTTOM_RIGHT
            : ck_tile::GenericAttentionMaskEnum::MASK_FROM_TOP_LEFT),
        0.0f /* p_drop */, std::make_pair<uint64_t, uint64_t>(0, 0));
    launch<MainKernel>(stream, MainKernel::GridSize(B, Hq, S), kargs);
  }
  {  // 3: dq = bf16(dq_acc)
    auto kargs = CvtKernel::MakeKargs(
        dq_acc, dq, S, S, D, D /* stride_dq */, D /* stride_dq_acc */,
        sq /* nhead_stride_dq */, sq /* nhead_stride_dq_acc */,
        bq /* batch_stride_dq */, bq /* batch_stride_dq_acc */,
        0 /* split_stride_dq_acc */);
    launch<CvtKernel>(stream, CvtKernel::GridSize(B, Hq, S), kargs);
  }
}

}  // namespace

// All tensors contiguous [B, H, S, 128] bf16 (Hq for q/o/do/dq, Hkv for
// k/v; dk/dv are Hq-EXPANDED — the caller group-sums). lse/d: [B, Hq, S]
// f32. dq_acc: [B, Hq, S, 128] f32 ZEROED by the caller (atomic accum).
// mask_mode selects the causal-mask karg convention (runtime A/B):
//   0: window (-1, 0), MASK_FROM_TOP_LEFT     (matches the fwd kernels)
//   1: window (-1, 0), MASK_FROM_BOTTOM_RIGHT
//   2: window (0, -1), MASK_FROM_TOP_LEFT     (anti-causal)
// pipe: 0 = std (KRKTRVR IGLP, 32x32x16), 1 = trload (gfx950, 16x16x32).
extern "C" void kt_attn_bwd_ck(const void* q, const void* k, const void* v,
                               const void* o, const void* do_, const void* lse,
                               void* d, void* dq_acc, void* dq, void* dk,
                               void* dv, int B, int Hq, int Hkv, int S,
                               float scale, int mask_mode, int pipe,
                               hipStream_t stream) {
  if (pipe == 3) {
    run_bwd<TrKernelN64>(q, k, v, o, do_, lse, d, dq_acc, dq, dk, dv, B, Hq,
                         Hkv, S, scale, mask_mode, stream);
  } else if (pipe == 2) {
    run_bwd<TrKernel64>(q, k, v, o, do_, lse, d, dq_acc, dq, dk, dv, B, Hq,
                        Hkv, S, scale, mask_mode, stream);
  } else if (pipe == 1) {
    run_bwd<TrKernel>(q, k, v, o, do_, lse, d, dq_acc, dq, dk, dv, B, Hq, Hkv,
                      S, scale, mask_mode, stream);
  } else {
    run_bwd<StdKernel>(q, k, v, o, do_, lse, d, dq_acc, dq, dk, dv, B, Hq, Hkv,
                       S, scale, mask_mode, stream);
  }
}
