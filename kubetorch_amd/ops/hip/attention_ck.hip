// CK-tile FMHA forward instantiation for the Llama training shape:
// bf16, head_dim 128, causal, batch mode, GQA, LSE output (training).
//
// composable_kernel's ck_tile FMHA is AMD's optimized attention (the
// header-only form of the AITER kernels); like hipBLASLt for GEMM this is
// the library path — we instantiate the exact variant the model needs and
// pair it with torch's AITER backward in ops.flash_attention.
// (Upstream dispatch `fmha_fwd()` is codegen at CK build time and is not
// shipped in /opt/rocm, so we instantiate the kernel template directly.)
#include <ck_tile/core.hpp>
#include <ck_tile/host/kernel_launch.hpp>
#include <ck_tile/ops/epilogue.hpp>
#include <ck_tile/ops/fmha.hpp>

namespace {

using qkv_t = ck_tile::bf16_t;

// hdim-128 bf16 tile config (CK codegen default for this head dim):
// M0=128 N0=128 K0=32 N1=128 K1=32 K0L=128; 4 warps; 32x32x16 warp gemm.
using FmhaShape = ck_tile::TileFmhaShape<
    ck_tile::sequence<128, 128, 32, 128, 32, 128>,
    ck_tile::sequence<4, 1, 1>, ck_tile::sequence<32, 32, 16>,
    ck_tile::sequence<4, 1, 1>, ck_tile::sequence<32, 32, 16>,
    true /* V row-major */>;

using FmhaTraits = ck_tile::TileFmhaTraits<
    true /* kPadSeqLenQ (async pipeline requires the padded views) */,
    true /* kPadSeqLenK */,
    true /* kPadHeadDimQ */,
    true /* kPadHeadDimV */,
    false /* kHasLogitsSoftCap */,
    ck_tile::BlockAttentionBiasEnum::NO_BIAS,
    false /* kHasBiasGrad */,
    true /* kStoreLSE: training */,
    false /* kHasDropout */,
    false /* kDoFp8StaticQuant */>;

using FmhaMask = ck_tile::SimplifiedGenericAttentionMask<true /* masking */>;

using FmhaPipelineProblem = ck_tile::BlockFmhaPipelineProblem<
    qkv_t /* Q */, qkv_t /* K */, qkv_t /* V */,
    float /* Sacc */, float /* SMPLCompute */,
    qkv_t /* Bias */, uint8_t /* RandValOutput */,
    float /* LSE */, qkv_t /* P */, float /* Oacc */, qkv_t /* O */,
    FmhaShape,
    false /* kIsGroupMode */,
    ck_tile::StandardAttention,
    FmhaMask,
    false /* kUseTrLoad */,
    FmhaTraits>;

using FmhaPipeline = ck_tile::BlockFmhaPipelineQRKSVSAsync<FmhaPipelineProblem>;

using FmhaEpilogue = ck_tile::Default2DEpilogue<
    ck_tile::Default2DEpilogueProblem<float, qkv_t, false /* kPadM */,
                                      false /* kPadN */>>;

using Kernel = ck_tile::FmhaFwdKernel<FmhaPipeline, FmhaEpilogue>;

using FmhaPipelineProblemTr = ck_tile::BlockFmhaPipelineProblem<
    qkv_t, qkv_t, qkv_t, float, float, qkv_t, uint8_t, float, qkv_t, float,
    qkv_t, FmhaShape, false, ck_tile::StandardAttention, FmhaMask,
    true /* kUseTrLoad: gfx950 ds_read_tr path */, FmhaTraits>;
using FmhaPipelineTr = ck_tile::BlockFmhaPipelineQRKSVSAsync<FmhaPipelineProblemTr>;
using KernelTr = ck_tile::FmhaFwdKernel<FmhaPipelineTr, FmhaEpilogue>;

template <typename K>
void run_fmha(const void* q, const void* k, const void* v, void* o,
              void* lse, int B, int Hq, int Hkv, int S, float scale,
              hipStream_t stream) {
  const ck_tile::index_t D = 128;
  auto kargs = K::MakeKargs(
      q, k, v, nullptr, nullptr, lse, o, S, S, D, D, Hq, Hq / Hkv, scale,
      1.0f, 1.0f, 0.0f, D, D, D, 0, 0, D,
      (ck_tile::index_t)S * D, (ck_tile::index_t)S * D,
      (ck_tile::index_t)S * D, 0, 0, S, (ck_tile::index_t)S * D,
      (ck_tile::index_t)Hq * S * D, (ck_tile::index_t)Hkv * S * D,
      (ck_tile::index_t)Hkv * S * D, 0, 0, (ck_tile::index_t)Hq * S,
      (ck_tile::index_t)Hq * S * D, -1, 0,
      (ck_tile::index_t)ck_tile::GenericAttentionMaskEnum::MASK_FROM_TOP_LEFT,
      0.0f, false, std::make_tuple<uint64_t, uint64_t>(0, 0));
  dim3 grid = K::GridSize(B, Hq, S, D, false);
  ck_tile::stream_config cfg{};
  cfg.stream_id_ = stream;
  cfg.cold_niters_ = 0;
  cfg.nrepeat_ = 1;
  ck_tile::launch_kernel(cfg, ck_tile::make_kernel<K::kBlockPerCu>(
                                  K{}, grid, K::kBlockSize, 0, kargs));
}

}  // namespace

extern "C" void kt_attn_fwd_ck_tr(const void* q, const void* k,
                                  const void* v, void* o, void* lse, int B,
                                  int Hq, int Hkv, int S, float scale,
                                  hipStream_t stream) {
  run_fmha<KernelTr>(q, k, v, o, lse, B, Hq, Hkv, S, scale, stream);
}

extern "C" void kt_attn_fwd_ck(const void* q, const void* k, const void* v,
                               void* o, void* lse, int B, int Hq, int Hkv,
                               int S, float scale, hipStream_t stream) {
  // layouts: q/o [B, Hq, S, 128]; k/v [B, Hkv, S, 128]; lse [B, Hq, S] fp32
  const ck_tile::index_t D = 128;
  auto kargs = Kernel::MakeKargs(
      q, k, v,
      nullptr /* bias */, nullptr /* rand_val */, lse, o,
      S /* seqlen_q */, S /* seqlen_k */, D /* hdim_q */, D /* hdim_v */,
      Hq /* num_head_q */, Hq / Hkv /* nhead_ratio_qk */,
      scale /* scale_s */, 1.0f /* scale_p */, 1.0f /* scale_o */,
      0.0f /* logits_soft_cap */,
      D /* stride_q */, D /* stride_k */, D /* stride_v */,
      0 /* stride_bias */, 0 /* stride_randval */, D /* stride_o */,
      (ck_tile::index_t)S * D /* nhead_stride_q */,
      (ck_tile::index_t)S * D /* nhead_stride_k */,
      (ck_tile::index_t)S * D /* nhead_stride_v */,
      0 /* nhead_stride_bias */, 0 /* nhead_stride_randval */,
      S /* nhead_stride_lse */,
      (ck_tile::index_t)S * D /* nhead_stride_o */,
      (ck_tile::index_t)Hq * S * D /* batch_stride_q */,
      (ck_tile::index_t)Hkv * S * D /* batch_stride_k */,
      (ck_tile::index_t)Hkv * S * D /* batch_stride_v */,
      0 /* batch_stride_bias */, 0 /* batch_stride_randval */,
      (ck_tile::index_t)Hq * S /* batch_stride_lse */,
      (ck_tile::index_t)Hq * S * D /* batch_stride_o */,
      -1 /* window_size_left */, 0 /* window_size_right */,
      (ck_tile::index_t)ck_tile::GenericAttentionMaskEnum::MASK_FROM_TOP_LEFT,
      0.0f /* p_drop */, false /* s_randval */,
      std::make_tuple<uint64_t, uint64_t>(0, 0));

  dim3 grid = Kernel::GridSize(B, Hq, S, D, false);
  ck_tile::stream_config cfg{};
  cfg.stream_id_ = stream;
  cfg.cold_niters_ = 0;
  cfg.nrepeat_ = 1;
  ck_tile::launch_kernel(
      cfg, ck_tile::make_kernel<Kernel::kBlockPerCu>(
               Kernel{}, grid, Kernel::kBlockSize, 0, kargs));
}
