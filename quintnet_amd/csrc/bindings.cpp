// Torch bindings for the quintnet_amd CDNA4 kernel library (gfx950).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#include <vector>

// launcher decls (defined in the .hip TUs)
template <typename T>
void layernorm_fwd_launch(const T*, const T*, const T*, const T*, T*, T*, float*, float*, int, int, float, hipStream_t);
template <typename T>
void layernorm_bwd_launch(const T*, const T*, const T*, const float*, const float*, const T*, T*, float*, float*, int, int, hipStream_t);
template <typename T>
void softmax_fwd_launch(const T*, T*, long long, int, int, float, int, hipStream_t);
template <typename T>
void softmax_bwd_launch(const T*, const T*, T*, long long, int, float, hipStream_t);
template <typename T>
void ce_fwd_launch(const T*, const long long*, float*, float*, long long, int, long long, hipStream_t);
template <typename T>
void ce_bwd_launch(const T*, const long long*, const float*, const float*, const long long*, T*, long long, int, long long, float, hipStream_t);
template <typename T>
void act_bwd_launch(const T*, const T*, T*, long long, int, hipStream_t);
template <typename T>
void colsum_launch(const T*, float*, long long, int, hipStream_t);
long long colsum_bf16_chunks(long long rows, int N);
int wgrad_tn_splits(int M, int N, int K);
void wgrad_tn_launch(const unsigned short*, const unsigned short*, float*,
                     unsigned short*, int, int, int, int, hipStream_t);
bool colsum_bf16_launch(const unsigned short*, float*, unsigned short*,
                        long long, int, hipStream_t);
template <typename T>
void dropout_fwd_launch(const T*, T*, unsigned char*, long long, float, unsigned, hipStream_t);
template <typename T>
void dropout_bwd_launch(const T*, T*, const unsigned char*, long long, float, hipStream_t);
template <typename T>
void embedding_pair_fwd_launch(const long long*, const T*, const T*, T*, long long, int, int, hipStream_t);
template <typename T>
void embedding_pair_bwd_launch(const long long*, const T*, float*, float*, long long, int, int, hipStream_t);
template <typename TP, typename TG>
void adamw_launch(TP*, float*, const TG*, float*, float*, const long long*, long long, float, float, float, float, float, int, hipStream_t);
template <typename T>
void sumsq_launch(const T*, long long, float*, hipStream_t);
std::vector<at::Tensor> gemm_bias_gelu_aux(at::Tensor x, at::Tensor w,
                                           at::Tensor bias);  // blaslt.cpp
void gemm_nt_launch(const unsigned short*, const unsigned short*, const unsigned short*,
                    unsigned short*, unsigned short*, int, int, int, int, hipStream_t,
                    int mode = 0);
struct AttnStrides {
  long long qB, qH, qT, kB, kH, kT, vB, vH, vT, oB, oH, oT;
};
void attn_fwd_launch(const unsigned short*, const unsigned short*, const unsigned short*,
                     unsigned short*, float*, int, int, int, int, int, float,
                     int, const AttnStrides&, hipStream_t);
void attn_delta_launch(const unsigned short*, const unsigned short*, float*, int, int, int,
                       long long, long long, long long, long long, long long, long long,
                       hipStream_t);
void attn_bwd_dq_launch(const unsigned short*, const unsigned short*, const unsigned short*,
                        const unsigned short*, const float*, const float*, unsigned short*,
                        int, int, int, int, int, float, int, const AttnStrides&,
                        long long, long long, long long, hipStream_t);
void attn_bwd_dkv_launch(const unsigned short*, const unsigned short*, const unsigned short*,
                         const unsigned short*, const float*, const float*, unsigned short*,
                         unsigned short*, int, int, int, int, int, float, int, const AttnStrides&,
                         long long, long long, long long, long long, long long, long long,
                         long long, long long, long long, hipStream_t);

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

const unsigned short* bf16p(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr<at::BFloat16>());
}
unsigned short* bf16p_mut(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr<at::BFloat16>());
}

#define CHECK_GPU(t) TORCH_CHECK(t.is_cuda(), #t " must be on GPU")

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> gemm_nt(torch::Tensor x, torch::Tensor w,
                                   c10::optional<torch::Tensor> bias, int64_t act,
                                   int64_t mode) {
  CHECK_GPU(x); CHECK_GPU(w);
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16,
              "gemm_nt: bf16 only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "gemm_nt: shape mismatch");
  TORCH_CHECK(x.size(1) % 8 == 0, "gemm_nt: K must be a multiple of 8");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  int64_t M = xc.size(0), K = xc.size(1), N = wc.size(0);
  auto out = torch::empty({M, N}, xc.options());
  torch::Tensor pre;
  unsigned short* prep = nullptr;
  if (act != 0) {
    pre = torch::empty({M, N}, xc.options());
    prep = bf16p_mut(pre);
  }
  const unsigned short* bp = nullptr;
  torch::Tensor bc;
  if (bias.has_value() && bias->defined()) {
    bc = bias->contiguous();
    TORCH_CHECK(bc.dtype() == torch::kBFloat16, "bias must be bf16");
    bp = bf16p(bc);
  }
  gemm_nt_launch(bf16p(xc), bf16p(wc), bp, bf16p_mut(out), prep,
                 (int)M, (int)N, (int)K, (int)act, cur_stream(), (int)mode);
  if (act != 0) return {out, pre};
  return {out};
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps,
                                         c10::optional<torch::Tensor> residual) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  int64_t H = xc.size(-1);
  int64_t rows = xc.numel() / H;
  auto y = torch::empty_like(xc);
  auto mean = torch::empty({rows}, xc.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({rows}, xc.options().dtype(torch::kFloat32));
  auto wc = w.contiguous(); auto bc = b.contiguous();
  torch::Tensor rc, sum_out;
  bool has_res = residual.has_value() && residual->defined();
  if (has_res) {
    rc = residual->contiguous();
    sum_out = torch::empty_like(xc);
  }
  if (xc.dtype() == torch::kBFloat16) {
    layernorm_fwd_launch<unsigned short>(
        bf16p(xc), has_res ? bf16p(rc) : nullptr, bf16p(wc), bf16p(bc),
        bf16p_mut(y), has_res ? bf16p_mut(sum_out) : nullptr,
        mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)rows, (int)H,
        (float)eps, cur_stream());
  } else {
    layernorm_fwd_launch<float>(
        xc.data_ptr<float>(), has_res ? rc.data_ptr<float>() : nullptr,
        wc.data_ptr<float>(), bc.data_ptr<float>(), y.data_ptr<float>(),
        has_res ? sum_out.data_ptr<float>() : nullptr, mean.data_ptr<float>(),
        rstd.data_ptr<float>(), (int)rows, (int)H, (float)eps, cur_stream());
  }
  if (has_res) return {y, mean, rstd, sum_out};
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd,
                                         c10::optional<torch::Tensor> dsum) {
  CHECK_GPU(x);
  auto dyc = dy.contiguous(); auto xc = x.contiguous(); auto wc = w.contiguous();
  int64_t H = xc.size(-1);
  int64_t rows = xc.numel() / H;
  auto dx = torch::empty_like(xc);
  auto dwf = torch::zeros({H}, xc.options().dtype(torch::kFloat32));
  auto dbf = torch::zeros({H}, xc.options().dtype(torch::kFloat32));
  torch::Tensor dsc;
  bool has_ds = dsum.has_value() && dsum->defined();
  if (has_ds) dsc = dsum->contiguous();
  if (xc.dtype() == torch::kBFloat16) {
    layernorm_bwd_launch<unsigned short>(
        bf16p(dyc), bf16p(xc), bf16p(wc), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), has_ds ? bf16p(dsc) : nullptr, bf16p_mut(dx),
        dwf.data_ptr<float>(), dbf.data_ptr<float>(), (int)rows, (int)H,
        cur_stream());
  } else {
    layernorm_bwd_launch<float>(
        dyc.data_ptr<float>(), xc.data_ptr<float>(), wc.data_ptr<float>(),
        mean.data_ptr<float>(), rstd.data_ptr<float>(),
        has_ds ? dsc.data_ptr<float>() : nullptr, dx.data_ptr<float>(),
        dwf.data_ptr<float>(), dbf.data_ptr<float>(), (int)rows, (int)H,
        cur_stream());
  }
  return {dx, dwf.to(w.dtype()), dbf.to(w.dtype())};
}

// ---------------------------------------------------------------------------
torch::Tensor softmax_fwd(torch::Tensor scores, double scale, bool causal) {
  CHECK_GPU(scores);
  auto sc = scores.contiguous();
  int64_t S = sc.size(-1);
  int64_t Tq = sc.size(-2);
  long long rows = sc.numel() / S;
  auto out = torch::empty_like(sc);
  if (sc.dtype() == torch::kBFloat16) {
    softmax_fwd_launch<unsigned short>(bf16p(sc), bf16p_mut(out), rows, (int)Tq,
                                       (int)S, (float)scale, causal ? 1 : 0,
                                       cur_stream());
  } else {
    softmax_fwd_launch<float>(sc.data_ptr<float>(), out.data_ptr<float>(), rows,
                              (int)Tq, (int)S, (float)scale, causal ? 1 : 0,
                              cur_stream());
  }
  return out;
}

torch::Tensor softmax_bwd(torch::Tensor p, torch::Tensor dp, double scale) {
  CHECK_GPU(p);
  auto pc = p.contiguous(); auto dpc = dp.contiguous();
  int64_t S = pc.size(-1);
  long long rows = pc.numel() / S;
  auto ds = torch::empty_like(pc);
  if (pc.dtype() == torch::kBFloat16) {
    softmax_bwd_launch<unsigned short>(bf16p(pc), bf16p(dpc), bf16p_mut(ds), rows,
                                       (int)S, (float)scale, cur_stream());
  } else {
    softmax_bwd_launch<float>(pc.data_ptr<float>(), dpc.data_ptr<float>(),
                              ds.data_ptr<float>(), rows, (int)S, (float)scale,
                              cur_stream());
  }
  return ds;
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor target,
                                             int64_t ignore_index) {
  CHECK_GPU(logits);
  auto lc = logits.contiguous();
  auto tc = target.contiguous();
  TORCH_CHECK(tc.dtype() == torch::kInt64, "target must be int64");
  int64_t V = lc.size(-1);
  long long rows = lc.numel() / V;
  auto lse = torch::empty({(int64_t)rows}, lc.options().dtype(torch::kFloat32));
  auto loss_sum = torch::zeros({}, lc.options().dtype(torch::kFloat32));
  if (lc.dtype() == torch::kBFloat16) {
    ce_fwd_launch<unsigned short>(bf16p(lc), (const long long*)tc.data_ptr<int64_t>(),
                                  lse.data_ptr<float>(), loss_sum.data_ptr<float>(),
                                  rows, (int)V, ignore_index, cur_stream());
  } else {
    ce_fwd_launch<float>(lc.data_ptr<float>(), (const long long*)tc.data_ptr<int64_t>(),
                         lse.data_ptr<float>(), loss_sum.data_ptr<float>(), rows,
                         (int)V, ignore_index, cur_stream());
  }
  auto n_valid = (tc != ignore_index).sum();
  return {loss_sum, n_valid, lse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor logits, torch::Tensor target,
                                torch::Tensor lse, torch::Tensor n_valid,
                                int64_t ignore_index,
                                c10::optional<torch::Tensor> grad_scale) {
  CHECK_GPU(logits);
  auto lc = logits.contiguous();
  auto tc = target.contiguous();
  int64_t V = lc.size(-1);
  long long rows = lc.numel() / V;
  auto dl = torch::empty_like(lc);
  auto nv = n_valid.to(torch::kInt64).contiguous();
  TORCH_CHECK(nv.is_cuda() && nv.numel() == 1, "n_valid must be a GPU scalar");
  const long long* nvp = (const long long*)nv.data_ptr<int64_t>();
  float inv_n = 1.0f;
  const float* gp = nullptr;
  torch::Tensor gs;
  if (grad_scale.has_value() && grad_scale->defined()) {
    gs = grad_scale->to(torch::kFloat32).contiguous();
    TORCH_CHECK(gs.numel() == 1, "grad_scale must be a scalar tensor");
    gp = gs.data_ptr<float>();
  }
  if (lc.dtype() == torch::kBFloat16) {
    ce_bwd_launch<unsigned short>(bf16p(lc), (const long long*)tc.data_ptr<int64_t>(),
                                  lse.data_ptr<float>(), gp, nvp, bf16p_mut(dl),
                                  rows, (int)V, ignore_index, inv_n, cur_stream());
  } else {
    ce_bwd_launch<float>(lc.data_ptr<float>(), (const long long*)tc.data_ptr<int64_t>(),
                         lse.data_ptr<float>(), gp, nvp, dl.data_ptr<float>(),
                         rows, (int)V, ignore_index, inv_n, cur_stream());
  }
  return dl;
}

torch::Tensor embedding_pair_fwd(torch::Tensor ids, torch::Tensor wte,
                                 torch::Tensor wpe) {
  CHECK_GPU(ids);
  auto ic = ids.contiguous();
  TORCH_CHECK(ic.dtype() == torch::kInt64, "ids must be int64");
  int64_t Tlen = ic.size(-1);
  long long rows = ic.numel();
  int64_t H = wte.size(1);
  auto out = torch::empty({ic.size(0), Tlen, H}, wte.options());
  if (wte.dtype() == torch::kBFloat16)
    embedding_pair_fwd_launch<unsigned short>(
        (const long long*)ic.data_ptr<int64_t>(), bf16p(wte), bf16p(wpe),
        bf16p_mut(out), rows, (int)Tlen, (int)H, cur_stream());
  else
    embedding_pair_fwd_launch<float>(
        (const long long*)ic.data_ptr<int64_t>(), wte.data_ptr<float>(),
        wpe.data_ptr<float>(), out.data_ptr<float>(), rows, (int)Tlen, (int)H,
        cur_stream());
  return out;
}

std::vector<torch::Tensor> embedding_pair_bwd(torch::Tensor ids, torch::Tensor dout,
                                              int64_t vocab, int64_t n_pos) {
  CHECK_GPU(ids);
  auto ic = ids.contiguous();
  auto dc = dout.contiguous();
  int64_t Tlen = ic.size(-1);
  long long rows = ic.numel();
  int64_t H = dc.size(-1);
  auto dwte = torch::zeros({vocab, H}, dc.options().dtype(torch::kFloat32));
  auto dwpe = torch::zeros({n_pos, H}, dc.options().dtype(torch::kFloat32));
  if (dc.dtype() == torch::kBFloat16)
    embedding_pair_bwd_launch<unsigned short>(
        (const long long*)ic.data_ptr<int64_t>(), bf16p(dc),
        dwte.data_ptr<float>(), dwpe.data_ptr<float>(), rows, (int)Tlen, (int)H,
        cur_stream());
  else
    embedding_pair_bwd_launch<float>(
        (const long long*)ic.data_ptr<int64_t>(), dc.data_ptr<float>(),
        dwte.data_ptr<float>(), dwpe.data_ptr<float>(), rows, (int)Tlen, (int)H,
        cur_stream());
  return {dwte.to(dout.dtype()), dwpe.to(dout.dtype())};
}

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p, int64_t seed) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  auto y = torch::empty_like(xc);
  auto mask = torch::empty({xc.numel()}, xc.options().dtype(torch::kUInt8));
  if (xc.dtype() == torch::kBFloat16)
    dropout_fwd_launch<unsigned short>(bf16p(xc), bf16p_mut(y), mask.data_ptr<unsigned char>(),
                                       xc.numel(), (float)p, (unsigned)seed, cur_stream());
  else
    dropout_fwd_launch<float>(xc.data_ptr<float>(), y.data_ptr<float>(), mask.data_ptr<unsigned char>(),
                              xc.numel(), (float)p, (unsigned)seed, cur_stream());
  return {y, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p) {
  CHECK_GPU(dy);
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(dyc);
  if (dyc.dtype() == torch::kBFloat16)
    dropout_bwd_launch<unsigned short>(bf16p(dyc), bf16p_mut(dx), mask.data_ptr<unsigned char>(),
                                       dyc.numel(), (float)p, cur_stream());
  else
    dropout_bwd_launch<float>(dyc.data_ptr<float>(), dx.data_ptr<float>(), mask.data_ptr<unsigned char>(),
                              dyc.numel(), (float)p, cur_stream());
  return dx;
}

torch::Tensor wgrad_tn(torch::Tensor dy, torch::Tensor x) {
  CHECK_GPU(dy); CHECK_GPU(x);
  TORCH_CHECK(dy.dtype() == torch::kBFloat16 && x.dtype() == torch::kBFloat16,
              "wgrad_tn: bf16 only");
  auto dyc = dy.contiguous(); auto xc = x.contiguous();
  TORCH_CHECK(dyc.dim() == 2 && xc.dim() == 2 && dyc.size(0) == xc.size(0),
              "wgrad_tn: [M,N],[M,K] expected");
  int64_t M = dyc.size(0), N = dyc.size(1), K = xc.size(1);
  TORCH_CHECK(N % 128 == 0 && K % 128 == 0 && M % 64 == 0,
              "wgrad_tn: shape not eligible");
  int splits = wgrad_tn_splits((int)M, (int)N, (int)K);
  auto slab = torch::empty({splits, N, K}, dyc.options().dtype(torch::kFloat32));
  auto out = torch::empty({N, K}, dyc.options());
  wgrad_tn_launch(bf16p(dyc), bf16p(xc), slab.data_ptr<float>(), bf16p_mut(out),
                  (int)M, (int)N, (int)K, splits, cur_stream());
  return out;
}

torch::Tensor colsum(torch::Tensor x) {
  CHECK_GPU(x);
  auto xc = x.contiguous();
  int64_t N = xc.size(-1);
  long long rows = xc.numel() / N;
  if (xc.dtype() == torch::kBFloat16 && (N & 7) == 0) {
    long long chunks = colsum_bf16_chunks(rows, (int)N);
    auto part = torch::empty({chunks, N}, xc.options().dtype(torch::kFloat32));
    auto out = torch::empty({N}, xc.options());
    if (colsum_bf16_launch(bf16p(xc), part.data_ptr<float>(), bf16p_mut(out),
                           rows, (int)N, cur_stream()))
      return out;
  }
  auto out = torch::zeros({N}, xc.options().dtype(torch::kFloat32));
  if (xc.dtype() == torch::kBFloat16)
    colsum_launch<unsigned short>(bf16p(xc), out.data_ptr<float>(), rows, (int)N, cur_stream());
  else
    colsum_launch<float>(xc.data_ptr<float>(), out.data_ptr<float>(), rows, (int)N, cur_stream());
  return out.to(x.dtype());
}

torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor pre, int64_t act) {
  CHECK_GPU(dy);
  auto dyc = dy.contiguous();
  auto prec = pre.contiguous();
  TORCH_CHECK(dyc.numel() == prec.numel(), "act_bwd size mismatch");
  auto dx = torch::empty_like(dyc);
  if (dyc.dtype() == torch::kBFloat16)
    act_bwd_launch<unsigned short>(bf16p(dyc), bf16p(prec), bf16p_mut(dx),
                                   dyc.numel(), (int)act, cur_stream());
  else
    act_bwd_launch<float>(dyc.data_ptr<float>(), prec.data_ptr<float>(),
                          dx.data_ptr<float>(), dyc.numel(), (int)act, cur_stream());
  return dx;
}

// ---------------------------------------------------------------------------
void adamw_step(torch::Tensor param, torch::Tensor master, torch::Tensor grad,
                torch::Tensor m, torch::Tensor v, int64_t step, double lr,
                double beta1, double beta2, double eps, double wd,
                c10::optional<torch::Tensor> step_dev) {
  CHECK_GPU(param);
  long long n = param.numel();
  TORCH_CHECK(master.numel() == n && grad.numel() == n && m.numel() == n && v.numel() == n,
              "adamw_step: size mismatch");
  auto st = cur_stream();
  float* mp = master.data_ptr<float>();
  float* m1 = m.data_ptr<float>();
  float* m2 = v.data_ptr<float>();
  const long long* sd = nullptr;
  if (step_dev.has_value() && step_dev->defined())
    sd = (const long long*)step_dev->data_ptr<int64_t>();
  bool pbf = param.dtype() == torch::kBFloat16;
  bool gbf = grad.dtype() == torch::kBFloat16;
  if (pbf && gbf)
    adamw_launch<unsigned short, unsigned short>(bf16p_mut(param), mp, bf16p(grad), m1, m2, sd, n,
        (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, (int)step, st);
  else if (pbf && !gbf)
    adamw_launch<unsigned short, float>(bf16p_mut(param), mp, grad.data_ptr<float>(), m1, m2, sd, n,
        (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, (int)step, st);
  else if (!pbf && gbf)
    adamw_launch<float, unsigned short>(param.data_ptr<float>(), mp, bf16p(grad), m1, m2, sd, n,
        (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, (int)step, st);
  else
    adamw_launch<float, float>(param.data_ptr<float>(), mp, grad.data_ptr<float>(), m1, m2, sd, n,
        (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, (int)step, st);
}

torch::Tensor multi_tensor_sumsq(std::vector<torch::Tensor> tensors) {
  TORCH_CHECK(!tensors.empty(), "empty tensor list");
  auto out = torch::zeros({}, tensors[0].options().dtype(torch::kFloat32));
  auto st = cur_stream();
  for (auto& t : tensors) {
    auto tc = t.contiguous();
    if (tc.dtype() == torch::kBFloat16)
      sumsq_launch<unsigned short>(bf16p(tc), tc.numel(), out.data_ptr<float>(), st);
    else
      sumsq_launch<float>(tc.data_ptr<float>(), tc.numel(), out.data_ptr<float>(), st);
  }
  return out;
}


// ---------------------------------------------------------------------------
// fused flash attention (head_dim 64, bf16, strided [B,H,T,D] views)
namespace {
void check_attn_view(const torch::Tensor& t, int64_t B, int64_t H, int64_t T) {
  TORCH_CHECK(t.dim() == 4 && t.size(0) == B && t.size(1) == H && t.size(2) == T &&
              t.size(3) == 64, "attn: bad view shape");
  TORCH_CHECK(t.stride(3) == 1, "attn: head_dim must be contiguous");
  TORCH_CHECK(t.stride(2) % 8 == 0 && t.stride(1) % 8 == 0 && t.stride(0) % 8 == 0,
              "attn: strides must be 16B-aligned");
  TORCH_CHECK(t.dtype() == torch::kBFloat16, "attn: bf16 only");
}
const unsigned short* ap(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr<at::BFloat16>());
}
unsigned short* ap_mut(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr<at::BFloat16>());
}
}  // namespace

torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       torch::Tensor out, double scale, bool causal,
                       int64_t q_offset) {
  int64_t B = q.size(0), H = q.size(1), Tq = q.size(2), Tk = k.size(2);
  TORCH_CHECK(Tq % 128 == 0 && Tk % 128 == 0 && q_offset % 128 == 0,
              "attn: Tq/Tk/q_offset must be multiples of 128");
  TORCH_CHECK(q_offset >= 0 && q_offset + Tq <= Tk,
              "attn: q rows [q_offset, q_offset+Tq) must lie within [0, Tk)");
  check_attn_view(q, B, H, Tq); check_attn_view(k, B, H, Tk);
  check_attn_view(v, B, H, Tk); check_attn_view(out, B, H, Tq);
  auto lse2 = torch::empty({B * H, Tq}, q.options().dtype(torch::kFloat32));
  AttnStrides st{q.stride(0), q.stride(1), q.stride(2), k.stride(0), k.stride(1),
                 k.stride(2), v.stride(0), v.stride(1), v.stride(2), out.stride(0),
                 out.stride(1), out.stride(2)};
  attn_fwd_launch(ap(q), ap(k), ap(v), ap_mut(out), lse2.data_ptr<float>(),
                  (int)B, (int)H, (int)Tq, (int)Tk, (int)q_offset, (float)scale,
                  causal ? 1 : 0, st, cur_stream());
  return lse2;
}

void attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
              torch::Tensor out, torch::Tensor dout, torch::Tensor lse2,
              torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
              double scale, bool causal, int64_t q_offset) {
  int64_t B = q.size(0), H = q.size(1), Tq = q.size(2), Tk = k.size(2);
  check_attn_view(dout, B, H, Tq); check_attn_view(dq, B, H, Tq);
  check_attn_view(dk, B, H, Tk); check_attn_view(dv, B, H, Tk);
  auto delta = torch::empty({B * H, Tq}, q.options().dtype(torch::kFloat32));
  attn_delta_launch(ap(dout), ap(out), delta.data_ptr<float>(), (int)B, (int)H,
                    (int)Tq, dout.stride(0), dout.stride(1), dout.stride(2),
                    out.stride(0), out.stride(1), out.stride(2), cur_stream());
  AttnStrides st{q.stride(0), q.stride(1), q.stride(2), k.stride(0), k.stride(1),
                 k.stride(2), v.stride(0), v.stride(1), v.stride(2), dq.stride(0),
                 dq.stride(1), dq.stride(2)};
  attn_bwd_dq_launch(ap(q), ap(k), ap(v), ap(dout), lse2.data_ptr<float>(),
                     delta.data_ptr<float>(), ap_mut(dq), (int)B, (int)H,
                     (int)Tq, (int)Tk, (int)q_offset,
                     (float)scale, causal ? 1 : 0, st, dout.stride(0),
                     dout.stride(1), dout.stride(2), cur_stream());
  attn_bwd_dkv_launch(ap(q), ap(k), ap(v), ap(dout), lse2.data_ptr<float>(),
                      delta.data_ptr<float>(), ap_mut(dk), ap_mut(dv), (int)B,
                      (int)H, (int)Tq, (int)Tk, (int)q_offset, (float)scale,
                      causal ? 1 : 0, st,
                      dout.stride(0), dout.stride(1), dout.stride(2),
                      dk.stride(0), dk.stride(1), dk.stride(2), dv.stride(0),
                      dv.stride(1), dv.stride(2), cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_nt", &gemm_nt, "bf16 MFMA NT GEMM with fused bias/activation",
        py::arg("x"), py::arg("w"), py::arg("bias") = c10::nullopt,
        py::arg("act") = 0, py::arg("mode") = 0);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("softmax_fwd", &softmax_fwd);
  m.def("softmax_bwd", &softmax_bwd);
  m.def("cross_entropy_fwd", &cross_entropy_fwd);
  m.def("cross_entropy_bwd", &cross_entropy_bwd);
  m.def("act_bwd", &act_bwd, "fused activation backward");
  m.def("colsum", &colsum, "column sum (bias grad)");
  m.def("wgrad_tn", &wgrad_tn, "split-K TN weight gradient (tr-read MFMA)");
  m.def("dropout_fwd", &dropout_fwd);
  m.def("embedding_pair_fwd", &embedding_pair_fwd);
  m.def("embedding_pair_bwd", &embedding_pair_bwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("attn_fwd", &attn_fwd, "fused flash attention forward (D=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("out"),
        py::arg("scale"), py::arg("causal"), py::arg("q_offset") = 0);
  m.def("attn_bwd", &attn_bwd, "fused flash attention backward (D=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("out"),
        py::arg("dout"), py::arg("lse2"), py::arg("dq"), py::arg("dk"),
        py::arg("dv"), py::arg("scale"), py::arg("causal"),
        py::arg("q_offset") = 0);
  m.def("adamw_step", &adamw_step);
  m.def("gemm_bias_gelu_aux", &gemm_bias_gelu_aux,
        "hipBLASLt GEMM + fused bias/GELU epilogue (aux = pre-activation)");
  m.def("multi_tensor_sumsq", &multi_tensor_sumsq);
  m.attr("gfx") = "gfx950";
}
