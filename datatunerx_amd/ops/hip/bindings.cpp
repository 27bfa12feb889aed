// Torch bindings for the gfx950 HIP kernel library.
// All tensors must be contiguous; bf16 compute dtype, fp32 reductions.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

// ---- launchers (defined in the .hip files) ----
void launch_rmsnorm_fwd(const void*, const void*, void*, float*, int, int,
                        float, hipStream_t);
int rmsnorm_bwd_nblocks(int M);
void launch_rmsnorm_bwd(const void*, const void*, const void*, const float*,
                        const void*, void*, float*, float*, int, int,
                        hipStream_t);
void launch_rope(const void*, const float*, const float*, void*, long, int,
                 int, int, int, int, const int*, int, hipStream_t);
void launch_swiglu_fwd(const void*, const void*, void*, long, hipStream_t);
void launch_swiglu_bwd(const void*, const void*, const void*, void*, void*,
                       long, hipStream_t);
void launch_xent_fwd(const void*, const long*, float*, float*, long, int,
                     long, hipStream_t);
void launch_xent_bwd(const void*, const long*, const float*, const float*,
                     void*, long, int, long, hipStream_t);
void launch_dequant_int8(const void*, const float*, void*, long, int,
                         hipStream_t);
void launch_dequant_int4(const void*, const float*, void*, long, int, int,
                         hipStream_t);
void launch_xent_lse_merge(const void*, const long*, float*, float*,
                           float*, long, int, long, long, hipStream_t);
void launch_xent_dlogits(const void*, const long*, const float*, void*,
                         long, int, long, long, hipStream_t);
void launch_adamw(void*, float*, const float*, float*, float*, long, float,
                  float, float, float, float, int, hipStream_t);
void launch_l2_norm(const float*, float*, float*, long, hipStream_t);
int lora_contract_ksplit(int K, long M);
void launch_dropout_mask(void*, long, unsigned long long, float,
                         hipStream_t);
void launch_lora_contract(const void*, const void*, const void*, float*,
                          float*, long, int, int, unsigned long long,
                          float, hipStream_t);
void launch_lora_expand_add(void*, const float*, const void*, const void*,
                            long, int, int, float, unsigned long long,
                            float, hipStream_t);
int lora_wgrad_splitm(int K, int r, long M);
void launch_lora_wgrad(const float*, const void*, const void*, float*,
                       float*, long, int, int, float, unsigned long long,
                       float, hipStream_t);
void launch_attn_fwd(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, int, int, float, int, hipStream_t);
void launch_transpose_sd(const void*, void*, int, int, int, int,
                         hipStream_t);
int attn_decode_nsplit(int Skv);
void launch_gemv_bf16(const void*, const void*, void*, int, int,
                      hipStream_t);
void launch_gemm_nt(const void*, const void*, const void*, void*, long,
                    int, int, hipStream_t);
void launch_attn_decode(const void*, const void*, const void*,
                        const int*, float*, void*, float*, int, int, int,
                        int, int, float, int, hipStream_t);
void launch_attn_delta(const void*, const void*, float*, long, int, int,
                       int, hipStream_t);
void launch_attn_bwd_dkdv(const void*, const void*, const void*,
                          const void*, const float*, const float*,
                          void*, void*, int, int, int, int, int, int,
                          float, int, hipStream_t);
void launch_attn_bwd_dq(const void*, const void*, const void*,
                        const void*, const float*, const float*, void*,
                        int, int, int, int, int, int, float, int,
                        hipStream_t);

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

void check_bf16_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ------------------------------------------------------------- RMSNorm
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int H = (int)x.size(-1);
  const long M = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "H % 8");
  auto y = torch::empty_like(x);
  auto inv = torch::empty({M}, x.options().dtype(torch::kFloat));
  launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     inv.data_ptr<float>(), (int)M, H, (float)eps,
                     cur_stream());
  return {y, inv};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor inv,
                                       c10::optional<torch::Tensor> dres) {
  check_bf16_contig(dy, "dy");
  check_bf16_contig(x, "x");
  const int H = (int)x.size(-1);
  const long M = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw = torch::empty({H}, x.options().dtype(torch::kFloat));
  const int nb = rmsnorm_bwd_nblocks((int)M);
  auto part = torch::empty({nb, H}, x.options().dtype(torch::kFloat));
  const void* drp = nullptr;
  if (dres.has_value()) {
    check_bf16_contig(*dres, "dres");
    TORCH_CHECK(dres->numel() == x.numel(), "dres shape");
    drp = dres->data_ptr();
  }
  launch_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     inv.data_ptr<float>(), drp, dx.data_ptr(),
                     part.data_ptr<float>(), dw.data_ptr<float>(), (int)M, H,
                     cur_stream());
  return {dx, dw};
}

// ---------------------------------------------------------------- RoPE
torch::Tensor rope(torch::Tensor x, torch::Tensor cosb, torch::Tensor sinb,
                   long pos0, bool backward,
                   c10::optional<torch::Tensor> pos_dev) {
  check_bf16_contig(x, "x");
  TORCH_CHECK(x.dim() == 4, "x must be [B,S,H,D]");
  TORCH_CHECK(cosb.scalar_type() == torch::kFloat && cosb.is_contiguous());
  const int B = (int)x.size(0), S = (int)x.size(1), H = (int)x.size(2),
            D = (int)x.size(3);
  TORCH_CHECK(D % 8 == 0, "D % 8");
  const int* pd = nullptr;
  int pos_per_b = 0;
  if (pos_dev.has_value()) {
    TORCH_CHECK(pos_dev->scalar_type() == torch::kInt &&
                pos_dev->is_cuda(), "pos_dev must be int32 on GPU");
    pd = pos_dev->data_ptr<int>();
    pos_per_b = pos_dev->numel() > 1;
    if (pos_per_b)
      TORCH_CHECK(pos_dev->numel() == B, "pos_dev must be [1] or [B]");
    TORCH_CHECK(cosb.size(0) >= S, "rope table too short");
  } else {
    TORCH_CHECK(cosb.size(0) >= pos0 + S, "rope table too short");
  }
  auto y = torch::empty_like(x);
  launch_rope(x.data_ptr(), cosb.data_ptr<float>(), sinb.data_ptr<float>(),
              y.data_ptr(), B, S, H, D, (int)pos0, backward ? 1 : 0, pd,
              pos_per_b, cur_stream());
  return y;
}

// --------------------------------------------------------------- SwiGLU
torch::Tensor swiglu_fwd(torch::Tensor gate, torch::Tensor up) {
  check_bf16_contig(gate, "gate");
  check_bf16_contig(up, "up");
  TORCH_CHECK(gate.numel() % 8 == 0);
  auto out = torch::empty_like(gate);
  launch_swiglu_fwd(gate.data_ptr(), up.data_ptr(), out.data_ptr(),
                    gate.numel(), cur_stream());
  return out;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dout, torch::Tensor gate,
                                      torch::Tensor up) {
  check_bf16_contig(dout, "dout");
  auto dg = torch::empty_like(gate);
  auto du = torch::empty_like(up);
  launch_swiglu_bwd(dout.data_ptr(), gate.data_ptr(), up.data_ptr(),
                    dg.data_ptr(), du.data_ptr(), gate.numel(),
                    cur_stream());
  return {dg, du};
}

// -------------------------------------------------------- cross entropy
std::vector<torch::Tensor> xent_fwd(torch::Tensor logits,
                                    torch::Tensor targets, long ignore) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == torch::kLong);
  const int V = (int)logits.size(-1);
  const long N = logits.numel() / V;
  TORCH_CHECK(V % 8 == 0, "V % 8");
  auto loss = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat));
  launch_xent_fwd(logits.data_ptr(), targets.data_ptr<long>(),
                  loss.data_ptr<float>(), lse.data_ptr<float>(), N, V,
                  ignore, cur_stream());
  return {loss, lse};
}

torch::Tensor xent_bwd(torch::Tensor logits, torch::Tensor targets,
                       torch::Tensor lse, torch::Tensor dloss, long ignore) {
  check_bf16_contig(logits, "logits");
  const int V = (int)logits.size(-1);
  const long N = logits.numel() / V;
  auto dlogits = torch::empty_like(logits);
  launch_xent_bwd(logits.data_ptr(), targets.data_ptr<long>(),
                  lse.data_ptr<float>(), dloss.data_ptr<float>(),
                  dlogits.data_ptr(), N, V, ignore, cur_stream());
  return dlogits;
}

// ----------------------------------------------------------------- LoRA
// Optional `mask` (bf16, same shape as x / y) fuses the PEFT-style
// input dropout into the kernels (no mask-multiply materialization).
static const void* opt_mask(const c10::optional<torch::Tensor>& m,
                            long numel, const char* name) {
  if (!m.has_value()) return nullptr;
  check_bf16_contig(*m, name);
  TORCH_CHECK(m->numel() == numel, name, " shape mismatch");
  return m->data_ptr();
}

torch::Tensor lora_contract(torch::Tensor x, torch::Tensor w,
                            c10::optional<torch::Tensor> mask,
                            int64_t seed, double keep) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int K = (int)x.size(-1);
  const long M = x.numel() / K;
  const int r = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K, "w [r,K] mismatch");
  TORCH_CHECK(r <= 64, "r <= 64");
  TORCH_CHECK(K % 8 == 0);
  auto t = torch::empty({M, r}, x.options().dtype(torch::kFloat));
  const int nsplit = lora_contract_ksplit(K, M);
  torch::Tensor part;
  float* part_ptr = nullptr;
  if (nsplit > 1) {
    part = torch::empty({nsplit, M, r}, x.options().dtype(torch::kFloat));
    part_ptr = part.data_ptr<float>();
  }
  launch_lora_contract(x.data_ptr(), w.data_ptr(),
                       opt_mask(mask, x.numel(), "mask"), part_ptr,
                       t.data_ptr<float>(), M, K, r,
                       (unsigned long long)seed, (float)keep,
                       cur_stream());
  return t;
}

void lora_expand_add(torch::Tensor y, torch::Tensor t, torch::Tensor w,
                     double scale, c10::optional<torch::Tensor> mask,
                     int64_t seed, double keep) {
  check_bf16_contig(y, "y");
  check_bf16_contig(w, "w");
  const int N = (int)y.size(-1);
  const long M = y.numel() / N;
  const int r = (int)w.size(1);
  TORCH_CHECK(w.size(0) == N, "w [N,r] mismatch");
  TORCH_CHECK(t.scalar_type() == torch::kFloat && t.is_contiguous());
  auto wt = w.t().contiguous();          // [r,N] for coalesced rows
  TORCH_CHECK(keep >= 1.0 || (r <= 16 && N % 8 == 0 && M >= 64),
              "RNG dropout needs the r<=16 fast path; materialize the "
              "mask for this shape");
  launch_lora_expand_add(y.data_ptr(), t.data_ptr<float>(), wt.data_ptr(),
                         opt_mask(mask, y.numel(), "mask"), M, N, r,
                         (float)scale, (unsigned long long)seed,
                         (float)keep, cur_stream());
}

torch::Tensor lora_wgrad(torch::Tensor t, torch::Tensor x, double scale,
                         c10::optional<torch::Tensor> mask,
                         int64_t seed, double keep) {
  check_bf16_contig(x, "x");
  TORCH_CHECK(t.scalar_type() == torch::kFloat && t.is_contiguous());
  const int K = (int)x.size(-1);
  const long M = x.numel() / K;
  const int r = (int)t.size(-1);
  auto out = torch::empty({r, K}, x.options().dtype(torch::kFloat));
  const int sm = lora_wgrad_splitm(K, r, M);
  auto part = torch::empty({sm, r, K}, x.options().dtype(torch::kFloat));
  launch_lora_wgrad(t.data_ptr<float>(), x.data_ptr(),
                    opt_mask(mask, x.numel(), "mask"),
                    part.data_ptr<float>(), out.data_ptr<float>(), M, K, r,
                    (float)scale, (unsigned long long)seed, (float)keep,
                    cur_stream());
  return out;
}

// Materialize the RNG dropout mask (bit-identical to what the fused
// MODE==2 kernels consume) — used by tests and the r>16 fallback.
torch::Tensor dropout_mask(int64_t M, int64_t K, int64_t seed, double keep,
                           torch::Tensor like) {
  TORCH_CHECK((M * K) % 8 == 0, "M*K % 8 == 0");
  auto m = torch::empty({M, K}, like.options().dtype(torch::kBFloat16));
  launch_dropout_mask(m.data_ptr(), M * K, (unsigned long long)seed,
                      (float)keep, cur_stream());
  return m;
}

// ---------------------------------------------------------------- AdamW
void adamw(torch::Tensor p, torch::Tensor master, torch::Tensor grad,
           torch::Tensor m, torch::Tensor v, double lr, double b1, double b2,
           double eps, double wd, long step) {
  check_bf16_contig(p, "p");
  TORCH_CHECK(master.scalar_type() == torch::kFloat);
  TORCH_CHECK(grad.scalar_type() == torch::kFloat);
  TORCH_CHECK(p.numel() % 4 == 0, "flat param buffer must be padded to 4");
  launch_adamw(p.data_ptr(), master.data_ptr<float>(),
               grad.data_ptr<float>(), m.data_ptr<float>(),
               v.data_ptr<float>(), p.numel(), (float)lr, (float)b1,
               (float)b2, (float)eps, (float)wd, (int)step, cur_stream());
}

torch::Tensor l2_norm(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat &&
              x.is_contiguous());
  auto ws = torch::empty({1024}, x.options());
  auto out = torch::empty({1}, x.options());
  launch_l2_norm(x.data_ptr<float>(), ws.data_ptr<float>(),
                 out.data_ptr<float>(), x.numel(), cur_stream());
  return out;
}

// ------------------------------------------------------------ attention
// [B,S,H,D] -> [B,H,D,S]
torch::Tensor transpose_sd(torch::Tensor x) {
  check_bf16_contig(x, "x");
  TORCH_CHECK(x.dim() == 4, "x must be [B,S,H,D]");
  const int B = (int)x.size(0), S = (int)x.size(1), H = (int)x.size(2),
            D = (int)x.size(3);
  auto xt = torch::empty({B, H, D, S}, x.options());
  launch_transpose_sd(x.data_ptr(), xt.data_ptr(), B, S, H, D,
                      cur_stream());
  return xt;
}

// q,k: [B,S,H,D] BSHD; vt: [B,Hkv,D,Skv] (pre-transposed V).
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    double scale) {
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            D = (int)q.size(3);
  const int Skv = (int)k.size(1), Hkv = (int)k.size(2);
  TORCH_CHECK(v.size(1) == Skv && v.size(2) == Hkv && v.size(3) == D,
              "v must be BSHD like k");
  TORCH_CHECK(D == 64 || D == 128, "D must be 64 or 128");
  TORCH_CHECK(Hq % Hkv == 0, "GQA group");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat));
  launch_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr<float>(), B, Hq, Hkv, S, Skv, D,
                  (float)scale, causal ? 1 : 0, cur_stream());
  return {o, lse};
}

// C[M,N] = A[M,K] @ B[N,K]^T (+ optional bf16 src) — the hand-written
// 256x256-tile MFMA GEMM (gemm.hip). Requires N % 256 == 0, K % 64 == 0,
// K >= 128; any M (SRSRC-clamped tail).
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> src) {
  check_bf16_contig(a, "a");
  check_bf16_contig(b, "b");
  const int K = (int)b.size(1), N = (int)b.size(0);
  const long M = a.numel() / K;
  TORCH_CHECK(a.size(-1) == K, "gemm_nt: inner dims");
  TORCH_CHECK(N % 256 == 0 && K % 64 == 0 && K >= 128,
              "gemm_nt: unsupported shape N=", N, " K=", K);
  auto sizes = a.sizes().vec();
  sizes[sizes.size() - 1] = N;
  auto c = torch::empty(sizes, a.options());
  const void* sp = nullptr;
  if (src.has_value()) {
    check_bf16_contig(*src, "src");
    TORCH_CHECK(src->numel() == M * N, "gemm_nt: src shape");
    sp = src->data_ptr();
  }
  launch_gemm_nt(a.data_ptr(), b.data_ptr(), sp, c.data_ptr(), M, N, K,
                 cur_stream());
  return c;
}

// weight-only dequant (int8 per-row / int4 group-64) -> bf16
torch::Tensor dequant_int8(torch::Tensor q, torch::Tensor scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kChar &&
              q.is_contiguous());
  const int K = (int)q.size(1);
  const long N = q.size(0);
  TORCH_CHECK(K % 8 == 0, "K % 8");
  auto out = torch::empty({N, K},
                          q.options().dtype(torch::kBFloat16));
  launch_dequant_int8(q.data_ptr(), scale.data_ptr<float>(),
                      out.data_ptr(), N, K, cur_stream());
  return out;
}

torch::Tensor dequant_int4(torch::Tensor packed, torch::Tensor scale,
                           long group) {
  TORCH_CHECK(packed.is_cuda() && packed.scalar_type() == torch::kByte &&
              packed.is_contiguous());
  const int K = (int)packed.size(1) * 2;
  const long N = packed.size(0);
  TORCH_CHECK(K % 8 == 0 && group % 8 == 0, "K/group alignment");
  auto out = torch::empty({N, K},
                          packed.options().dtype(torch::kBFloat16));
  launch_dequant_int4(packed.data_ptr(), scale.data_ptr<float>(),
                      out.data_ptr(), N, K, (int)group, cur_stream());
  return out;
}

// chunked-vocab CE: online LSE merge over one logits chunk (in-place
// m/l/tgt update) and the per-chunk dlogits for the dX sweep.
void xent_lse_merge(torch::Tensor logits, torch::Tensor targets,
                    torch::Tensor m_run, torch::Tensor l_run,
                    torch::Tensor tgt, long v0, long ignore_index) {
  check_bf16_contig(logits, "logits");
  const int Vc = (int)logits.size(-1);
  const long N = logits.numel() / Vc;
  TORCH_CHECK(Vc % 8 == 0, "Vc % 8");
  launch_xent_lse_merge(logits.data_ptr(), targets.data_ptr<long>(),
                        m_run.data_ptr<float>(), l_run.data_ptr<float>(),
                        tgt.data_ptr<float>(), N, Vc, v0, ignore_index,
                        cur_stream());
}

torch::Tensor xent_dlogits(torch::Tensor logits, torch::Tensor targets,
                           torch::Tensor lse, long v0, long ignore_index) {
  check_bf16_contig(logits, "logits");
  const int Vc = (int)logits.size(-1);
  const long N = logits.numel() / Vc;
  TORCH_CHECK(Vc % 8 == 0, "Vc % 8");
  auto dl = torch::empty_like(logits);
  launch_xent_dlogits(logits.data_ptr(), targets.data_ptr<long>(),
                      lse.data_ptr<float>(), dl.data_ptr(), N, Vc, v0,
                      ignore_index, cur_stream());
  return dl;
}

// y[1,N] = x[1,K] @ W[N,K]^T (decode GEMV; fp32 accumulate)
torch::Tensor gemv(torch::Tensor x, torch::Tensor w) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int K = (int)w.size(1), N = (int)w.size(0);
  TORCH_CHECK(x.numel() == K, "gemv wants a single row");
  TORCH_CHECK(K % 8 == 0, "K % 8");
  auto sizes = x.sizes().vec();
  sizes[sizes.size() - 1] = N;
  auto y = torch::empty(sizes, x.options());
  launch_gemv_bf16(x.data_ptr(), w.data_ptr(), y.data_ptr(), N, K,
                   cur_stream());
  return y;
}

// Decode fast path: q [B,1,Hq,D] against the KV cache (flash-decoding
// split-KV partials; no V transpose needed).
std::vector<torch::Tensor> attn_decode(torch::Tensor q, torch::Tensor k,
                                       torch::Tensor v, double scale,
                                       c10::optional<torch::Tensor>
                                           len_dev) {
  check_bf16_contig(q, "q");
  const int B = (int)q.size(0), Hq = (int)q.size(2), D = (int)q.size(3);
  const int Skv = (int)k.size(1), Hkv = (int)k.size(2);
  // accept KV-cache VIEWS (prefix of a preallocated cache): inner
  // strides must be dense; the batch stride only matters for B > 1
  auto dense_kv = [&](const torch::Tensor& t, const char* n) {
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16,
                n, " must be bf16 GPU");
    TORCH_CHECK(t.stride(3) == 1 && t.stride(2) == D &&
                t.stride(1) == (long)Hkv * D,
                n, " must be s/h/d-dense");
    if (B > 1) TORCH_CHECK(t.is_contiguous(), n, " batch>1 needs contig");
  };
  dense_kv(k, "k");
  dense_kv(v, "v");
  TORCH_CHECK(q.size(1) == 1, "decode path wants S == 1");
  TORCH_CHECK(D == 64 || D == 128, "D must be 64 or 128");
  const int ns = attn_decode_nsplit(Skv);
  auto part = torch::empty({(long)B * Hq, ns, D + 2},
                           q.options().dtype(torch::kFloat));
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, 1}, q.options().dtype(torch::kFloat));
  const int* ld = nullptr;
  int len_stride = 0;
  if (len_dev.has_value()) {
    TORCH_CHECK(len_dev->scalar_type() == torch::kInt &&
                len_dev->is_cuda(), "len_dev must be int32 on GPU");
    ld = len_dev->data_ptr<int>();
    len_stride = len_dev->numel() > 1;
    if (len_stride)
      TORCH_CHECK(len_dev->numel() == B, "len_dev must be [1] or [B]");
  }
  launch_attn_decode(q.data_ptr(), k.data_ptr(), v.data_ptr(), ld,
                     part.data_ptr<float>(), o.data_ptr(),
                     lse.data_ptr<float>(), B, Hq, Hkv, Skv, D,
                     (float)scale, len_stride, cur_stream());
  return {o, lse};
}

// All BSHD; v untransposed. Pre-transposes Q/K/dO internally.
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dO, torch::Tensor lse,
                                    bool causal, double scale) {
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  auto dO_c = dO.contiguous();
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            D = (int)q.size(3);
  const int Skv = (int)k.size(1), Hkv = (int)k.size(2);
  auto delta = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat));
  launch_attn_delta(dO_c.data_ptr(), o.data_ptr(), delta.data_ptr<float>(),
                    (long)B * S * Hq, Hq, S, D, cur_stream());
  // dkdv and dq gather their B-fragments from the ROW-major LDS tiles
  // with ds_read_b64_tr_b16 (tr_bfrag) — no pre-transposed Q/K/dO
  // copies, no transpose_sd pre-passes in the backward at all
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  launch_attn_bwd_dkdv(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       dO_c.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dk.data_ptr(),
                       dv.data_ptr(), B, Hq, Hkv, S, Skv, D,
                       (float)scale, causal ? 1 : 0, cur_stream());
  launch_attn_bwd_dq(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                     dO_c.data_ptr(), lse.data_ptr<float>(),
                     delta.data_ptr<float>(), dq.data_ptr(), B, Hq, Hkv, S,
                     Skv, D, (float)scale, causal ? 1 : 0, cur_stream());
  return {dq, dk, dv};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd, py::arg("dy"), py::arg("x"),
        py::arg("w"), py::arg("inv"), py::arg("dres") = py::none());
  m.def("rope", &rope, py::arg("x"), py::arg("cosb"), py::arg("sinb"),
        py::arg("pos0"), py::arg("backward"),
        py::arg("pos_dev") = py::none());
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("xent_fwd", &xent_fwd);
  m.def("xent_bwd", &xent_bwd);
  m.def("xent_lse_merge", &xent_lse_merge);
  m.def("dequant_int8", &dequant_int8);
  m.def("dequant_int4", &dequant_int4);
  m.def("xent_dlogits", &xent_dlogits);
  m.def("lora_contract", &lora_contract, py::arg("x"), py::arg("w"),
        py::arg("mask") = py::none(), py::arg("seed") = 0,
        py::arg("keep") = 1.0);
  m.def("lora_expand_add", &lora_expand_add, py::arg("y"), py::arg("t"),
        py::arg("w"), py::arg("scale"), py::arg("mask") = py::none(),
        py::arg("seed") = 0, py::arg("keep") = 1.0);
  m.def("lora_wgrad", &lora_wgrad, py::arg("t"), py::arg("x"),
        py::arg("scale"), py::arg("mask") = py::none(),
        py::arg("seed") = 0, py::arg("keep") = 1.0);
  m.def("dropout_mask", &dropout_mask, py::arg("m"), py::arg("k"),
        py::arg("seed"), py::arg("keep"), py::arg("like"));
  m.def("adamw", &adamw);
  m.def("l2_norm", &l2_norm);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("transpose_sd", &transpose_sd);
  m.def("gemv", &gemv);
  m.def("gemm_nt", &gemm_nt, py::arg("a"), py::arg("b"),
        py::arg("src") = py::none());
  m.def("attn_decode", &attn_decode, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("scale"), py::arg("len_dev") = py::none());
}
