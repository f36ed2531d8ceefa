// Python bindings for the deepspeed_amd CDNA4 kernels.
//
// Uses torch-ROCm's native HIP stream API (c10::hip) — no CUDA shims.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <tuple>

extern "C" void ds_fused_adam_flat(float* p, const void* g, int grad_dtype,
                                   float* m, float* v, void* p16, long long n,
                                   float lr, float beta1, float beta2, float eps,
                                   float weight_decay, int step, float inv_scale,
                                   int adamw, void* stream);
extern "C" void ds_fused_lion(float* p, const void* g, int grad_dtype,
                              float* m, void* p16, long long n, float lr,
                              float beta1, float beta2, float weight_decay,
                              float inv_scale, void* stream);
extern "C" void ds_fused_lamb(float* p, const void* g, int grad_dtype,
                              float* m, float* v, float* u, float* norms2,
                              void* p16, long long n, float lr, float beta1,
                              float beta2, float eps, float weight_decay,
                              int step, float max_coeff, float min_coeff,
                              float inv_scale, void* stream);
extern "C" void ds_norm_fwd(const void* x, const void* w, const void* b,
                            void* y, float* invrms, float* mean, int rows,
                            int H, float eps, int ln, int dtype, void* stream);
extern "C" void ds_norm_bwd(const void* dy, const void* x, const void* w,
                            const float* invrms, const float* mean, void* dx,
                            float* dw, float* db, int rows, int H, int ln,
                            int dtype, void* stream);
extern "C" void ds_rope(void* x, const float* cos_table, const float* sin_table,
                        const int* positions, long long batch, long long seq,
                        long long heads, long long dim, int bwd, int dtype,
                        void* stream);
extern "C" void ds_gated_act_fwd(const void* gate, const void* up, void* out,
                                 long long n, int act, int dtype, void* stream);
extern "C" void ds_gated_act_bwd(const void* dout, const void* gate,
                                 const void* up, void* dgate, void* dup,
                                 long long n, int act, int dtype, void* stream);
extern "C" void ds_nhwc_bias_add(const void* act, const void* bias,
                                 const void* other, const void* other_bias,
                                 void* out, long long n, int channels,
                                 int dtype, void* stream);
extern "C" void ds_token_move(const void* x, void* y, const int* idx, int B,
                              int S, int K, int D, int gather, int dtype,
                              void* stream);
extern "C" void ds_cpu_adam_flat(float* p, const void* g, int grad_dtype,
                                 float* m, float* v, void* p16, long long n,
                                 float lr, float beta1, float beta2, float eps,
                                 float weight_decay, int step, float inv_scale,
                                 int adamw);
extern "C" void ds_groupwise_quant(const void* x, int dtype, void* q,
                                   float* scales, long long n, int group_size,
                                   int bits, void* stream);
extern "C" void ds_groupwise_dequant(const void* q, const float* scales,
                                     void* out, int dtype, long long n,
                                     int group_size, int bits, void* stream);
extern "C" void ds_fp_quantize(const void* x, int dtype, void* out,
                               float* scales, long long n, int group_size,
                               int bits, int dequant, void* stream);
extern "C" void ds_flash_bwd_dkdv_dbg(const void* q, const void* k,
                                      const void* v, const void* dout,
                                      const void* qt, const void* dot,
                                      const float* lse, const float* delta,
                                      void* dk, void* dv, int B, int S,
                                      int H, int Hkv, float scale,
                                      int variant, void* stream);
extern "C" void ds_paged_decode(const void* q, const void* kpool,
                                const void* vpool, const int* block_table,
                                const int* lens, float* part, float* part_ml,
                                void* o, int n, int H, int Hkv, int BS,
                                int max_blocks, int splits, float scale,
                                void* stream);
extern "C" void ds_transpose_bf16(const void* src, void* dst, int n_batch,
                                  int R, int C, long long row_stride,
                                  int inner, long long inner_stride,
                                  long long outer_stride, void* stream);
extern "C" void ds_fused_softmax(const void* x, const void* mask, void* y,
                                 const float* alibi_slopes, long long rows,
                                 int n, int heads, int sq, int mask_stride,
                                 float scale, int causal, int dtype,
                                 void* stream);
extern "C" void ds_fused_dropout(const void* x, const void* bias,
                                 const void* residual, void* y,
                                 unsigned char* mask_out, long long n,
                                 int cols, float ratio,
                                 unsigned long long seed, int dtype,
                                 void* stream);
extern "C" void ds_dropout_bwd(const void* dy, const unsigned char* mask,
                               void* dx, long long n, float ratio, int dtype,
                               void* stream);
extern "C" void* ds_shm_open(const char* name, int rank, int world,
                             long long max_elems);
extern "C" int ds_shm_allreduce(void* handle, float* data, long long n);
extern "C" void ds_shm_close(void* handle);
extern "C" void* ds_aio_create(long long block_size, int n_threads);
extern "C" void ds_aio_destroy(void* h);
extern "C" int ds_aio_pwrite(void* h, const void* data, long long nbytes,
                             const char* path);
extern "C" int ds_aio_pread(void* h, void* data, long long nbytes,
                            const char* path);
extern "C" int ds_aio_wait(void* h);
extern "C" void ds_flash_fwd(const void* q, const void* k, const void* vt,
                             void* o, void* lse, int B, int S, int H,
                             int Hkv, float scale, int causal, void* stream);
extern "C" void ds_cpu_lion_flat(float* p, const void* g, int grad_dtype,
                                 float* m, void* p16, long long n, float lr,
                                 float beta1, float beta2, float weight_decay,
                                 float inv_scale);
extern "C" void ds_flash_bwd(const void* q, const void* k, const void* v,
                             const void* dout, const void* qt,
                             const void* kt, const void* dot,
                             const float* lse, const float* delta, void* dq,
                             void* dk, void* dv, int B, int S, int H,
                             int Hkv, float scale, int causal, void* stream);
extern "C" void ds_ce_fwd(const void* logits, const long long* labels,
                          float* loss, float* lse, int N, long long V,
                          long long ignore_index, void* stream);
extern "C" void ds_ce_bwd(const void* logits, const long long* labels,
                          const float* lse, const float* gscale,
                          void* dlogits, int N, long long V,
                          long long ignore_index, void* stream);
extern "C" void ds_flash_fwd_dbg(const void* q, const void* k, const void* vt,
                                 void* o, int B, int S, int H, int Hkv,
                                 float scale, int variant, void* stream);

namespace {

int dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default:
      TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

void* cur_stream() {
  return reinterpret_cast<void*>(c10::hip::getCurrentHIPStream().stream());
}

void fused_adam_flat(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                     c10::optional<at::Tensor> p16, double lr, double beta1,
                     double beta2, double eps, double weight_decay,
                     int64_t step, double inv_scale, bool adamw) {
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous() &&
              v.is_contiguous(), "fused_adam_flat: tensors must be contiguous");
  TORCH_CHECK(p.scalar_type() == at::kFloat && m.scalar_type() == at::kFloat &&
              v.scalar_type() == at::kFloat, "p/m/v must be fp32");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel() &&
              p.numel() == v.numel(), "numel mismatch");
  void* p16_ptr = nullptr;
  if (p16.has_value() && p16->defined()) {
    TORCH_CHECK(p16->scalar_type() == at::kBFloat16 && p16->is_contiguous() &&
                p16->numel() == p.numel(), "p16 must be contiguous bf16");
    p16_ptr = p16->data_ptr();
  }
  ds_fused_adam_flat(p.data_ptr<float>(), g.data_ptr(), dtype_code(g),
                     m.data_ptr<float>(), v.data_ptr<float>(), p16_ptr,
                     p.numel(), (float)lr, (float)beta1, (float)beta2,
                     (float)eps, (float)weight_decay, (int)step,
                     (float)inv_scale, adamw ? 1 : 0, cur_stream());
}

void fused_lion(at::Tensor p, at::Tensor g, at::Tensor m,
                c10::optional<at::Tensor> p16, double lr, double beta1,
                double beta2, double weight_decay, double inv_scale) {
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous(),
              "fused_lion: tensors must be contiguous");
  TORCH_CHECK(p.scalar_type() == at::kFloat && m.scalar_type() == at::kFloat,
              "p/m must be fp32");
  void* p16_ptr = nullptr;
  if (p16.has_value() && p16->defined()) p16_ptr = p16->data_ptr();
  ds_fused_lion(p.data_ptr<float>(), g.data_ptr(), dtype_code(g),
                m.data_ptr<float>(), p16_ptr, p.numel(), (float)lr,
                (float)beta1, (float)beta2, (float)weight_decay,
                (float)inv_scale, cur_stream());
}

void fused_lamb(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                at::Tensor u, at::Tensor norms2,
                c10::optional<at::Tensor> p16, double lr, double beta1,
                double beta2, double eps, double weight_decay, int64_t step,
                double max_coeff, double min_coeff, double inv_scale) {
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous() &&
              v.is_contiguous() && u.is_contiguous(),
              "fused_lamb: tensors must be contiguous");
  TORCH_CHECK(norms2.numel() == 2 && norms2.scalar_type() == at::kFloat,
              "norms2 must be fp32[2]");
  void* p16_ptr = nullptr;
  if (p16.has_value() && p16->defined()) p16_ptr = p16->data_ptr();
  ds_fused_lamb(p.data_ptr<float>(), g.data_ptr(), dtype_code(g),
                m.data_ptr<float>(), v.data_ptr<float>(), u.data_ptr<float>(),
                norms2.data_ptr<float>(), p16_ptr, p.numel(), (float)lr,
                (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                (int)step, (float)max_coeff, (float)min_coeff,
                (float)inv_scale, cur_stream());
}

std::vector<at::Tensor> fp_quantize(at::Tensor x, int64_t group_size,
                                    int64_t bits) {
  TORCH_CHECK(x.is_contiguous() && x.is_cuda(), "fp_quantize: cuda contiguous");
  TORCH_CHECK(bits == 4 || bits == 6 || bits == 8 || bits == 12,
              "bits must be 4/6/8/12");
  TORCH_CHECK(group_size % 8 == 0, "group_size % 8 == 0");
  const long long n = x.numel();
  const long long groups = (n + group_size - 1) / group_size;
  const int vper3 = 24 / (int)bits;
  const long long gu = (group_size + vper3 - 1) / vper3;  // units/group
  auto out = at::empty({groups * gu * 3}, x.options().dtype(at::kByte));
  auto scales = at::empty({groups}, x.options().dtype(at::kFloat));
  ds_fp_quantize(x.data_ptr(), dtype_code(x), out.data_ptr(),
                 scales.data_ptr<float>(), n, (int)group_size, (int)bits, 0,
                 cur_stream());
  return {out, scales};
}

at::Tensor fp_dequantize(at::Tensor q, at::Tensor scales, int64_t numel,
                         int64_t group_size, int64_t bits,
                         at::ScalarType out_dtype) {
  TORCH_CHECK(q.is_contiguous() && q.is_cuda(), "fp_dequantize: cuda");
  auto out = at::empty({numel}, q.options().dtype(out_dtype));
  ds_fp_quantize(q.data_ptr(), dtype_code(out), out.data_ptr(),
                 scales.data_ptr<float>(), numel, (int)group_size, (int)bits,
                 1, cur_stream());
  return out;
}

at::Tensor transpose_bf16(at::Tensor src, int64_t n_batch, int64_t R,
                          int64_t C, int64_t row_stride, int64_t inner,
                          int64_t inner_stride, int64_t outer_stride) {
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == at::kBFloat16,
              "transpose_bf16: cuda bf16");
  auto dst = at::empty({n_batch, C, R}, src.options());
  ds_transpose_bf16(src.data_ptr(), dst.data_ptr(), (int)n_batch, (int)R,
                    (int)C, row_stride, (int)inner, inner_stride,
                    outer_stride, cur_stream());
  return dst;
}

at::Tensor fused_softmax(at::Tensor x, c10::optional<at::Tensor> mask,
                         c10::optional<at::Tensor> alibi, int64_t heads,
                         int64_t sq, double scale, bool causal) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "fused_softmax: cuda");
  const int n = x.size(-1);
  const long long rows = x.numel() / n;
  auto y = at::empty_like(x);
  const void* mp = nullptr;
  int mstride = 0;
  if (mask.has_value() && mask->defined()) {
    TORCH_CHECK(mask->scalar_type() == x.scalar_type(), "mask dtype");
    mp = mask->data_ptr();
    mstride = mask->numel() / mask->size(0);
  }
  const float* sl = nullptr;
  if (alibi.has_value() && alibi->defined())
    sl = alibi->data_ptr<float>();
  ds_fused_softmax(x.data_ptr(), mp, y.data_ptr(), sl, rows, n, (int)heads,
                   (int)sq, mstride, (float)scale, causal ? 1 : 0,
                   dtype_code(x), cur_stream());
  return y;
}

std::vector<at::Tensor> fused_dropout(at::Tensor x,
                                      c10::optional<at::Tensor> bias,
                                      c10::optional<at::Tensor> residual,
                                      double ratio, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "fused_dropout: cuda");
  auto y = at::empty_like(x);
  auto m = at::empty({x.numel()}, x.options().dtype(at::kByte));
  const void* bp = bias.has_value() && bias->defined() ? bias->data_ptr()
                                                       : nullptr;
  const void* rp = residual.has_value() && residual->defined()
                       ? residual->data_ptr() : nullptr;
  ds_fused_dropout(x.data_ptr(), bp, rp, y.data_ptr(),
                   m.data_ptr<unsigned char>(), x.numel(),
                   (int)x.size(-1), (float)ratio,
                   (unsigned long long)seed, dtype_code(x), cur_stream());
  return {y, m};
}

at::Tensor dropout_bwd(at::Tensor dy, at::Tensor mask, double ratio) {
  auto dx = at::empty_like(dy);
  ds_dropout_bwd(dy.contiguous().data_ptr(),
                 mask.data_ptr<unsigned char>(), dx.data_ptr(), dy.numel(),
                 (float)ratio, dtype_code(dy), cur_stream());
  return dx;
}

at::Tensor paged_decode(at::Tensor q, at::Tensor kpool, at::Tensor vpool,
                        at::Tensor block_table, at::Tensor lens,
                        int64_t splits, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 &&
              q.is_contiguous() && kpool.is_contiguous() &&
              vpool.is_contiguous(), "paged_decode: bf16 cuda contiguous");
  TORCH_CHECK(block_table.scalar_type() == at::kInt &&
              lens.scalar_type() == at::kInt, "table/lens must be int32");
  const int n = q.size(0), H = q.size(1);
  const int Hkv = kpool.size(1), BS = kpool.size(2);
  TORCH_CHECK(q.size(2) == 128 && kpool.size(3) == 128, "D must be 128");
  const int G = H / Hkv;
  TORCH_CHECK(G == 1 || G == 2 || G == 4 || G == 8, "H/Hkv must be 1/2/4/8");
  const int mb = block_table.size(1);
  auto part = at::empty({(long)n, (long)Hkv, splits, (long)G, 128L},
                        q.options().dtype(at::kFloat));
  auto part_ml = at::empty({(long)n, (long)Hkv, splits, (long)G, 2L},
                           q.options().dtype(at::kFloat));
  auto o = at::empty_like(q);
  ds_paged_decode(q.data_ptr(), kpool.data_ptr(), vpool.data_ptr(),
                  block_table.data_ptr<int>(), lens.data_ptr<int>(),
                  part.data_ptr<float>(), part_ml.data_ptr<float>(),
                  o.data_ptr(), n, H, Hkv, BS, mb, (int)splits, (float)scale,
                  cur_stream());
  return o;
}

class ShmComm {
 public:
  ShmComm(const std::string& name, int64_t rank, int64_t world,
          int64_t max_elems) {
    h_ = ds_shm_open(name.c_str(), (int)rank, (int)world, max_elems);
    TORCH_CHECK(h_ != nullptr, "shm_open failed for group ", name);
  }
  ~ShmComm() {
    if (h_) ds_shm_close(h_);
  }
  void all_reduce(at::Tensor t) {
    TORCH_CHECK(t.device().is_cpu() && t.scalar_type() == at::kFloat &&
                t.is_contiguous(), "shm all_reduce: contiguous fp32 CPU");
    TORCH_CHECK(ds_shm_allreduce(h_, t.data_ptr<float>(), t.numel()) == 0,
                "shm all_reduce failed (tensor larger than max_elems?)");
  }

 private:
  void* h_ = nullptr;
};

void cpu_adam_flat(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                   c10::optional<at::Tensor> p16, double lr, double beta1,
                   double beta2, double eps, double weight_decay, int64_t step,
                   double inv_scale, bool adamw) {
  TORCH_CHECK(!p.is_cuda() && !g.is_cuda(), "cpu_adam_flat: host tensors only");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous() &&
              v.is_contiguous(), "cpu_adam_flat: tensors must be contiguous");
  TORCH_CHECK(p.scalar_type() == at::kFloat && m.scalar_type() == at::kFloat &&
              v.scalar_type() == at::kFloat, "p/m/v must be fp32");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel() &&
              p.numel() == v.numel(), "numel mismatch");
  void* p16_ptr = nullptr;
  if (p16.has_value() && p16->defined()) {
    TORCH_CHECK(p16->scalar_type() == at::kBFloat16 && p16->is_contiguous() &&
                !p16->is_cuda() && p16->numel() == p.numel(),
                "p16 must be contiguous host bf16");
    p16_ptr = p16->data_ptr();
  }
  ds_cpu_adam_flat(p.data_ptr<float>(), g.data_ptr(), dtype_code(g),
                   m.data_ptr<float>(), v.data_ptr<float>(), p16_ptr,
                   p.numel(), (float)lr, (float)beta1, (float)beta2, (float)eps,
                   (float)weight_decay, (int)step, (float)inv_scale,
                   adamw ? 1 : 0);
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> norm_fwd(
    at::Tensor x, at::Tensor w, c10::optional<at::Tensor> b, double eps,
    bool layernorm) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous(), "must be contiguous");
  const int64_t H = x.size(-1);
  const int64_t rows = x.numel() / H;
  TORCH_CHECK(w.numel() == H, "weight shape mismatch");
  auto y = at::empty_like(x);
  auto f32opts = x.options().dtype(at::kFloat);
  auto invrms = at::empty({rows}, f32opts);
  auto mean = layernorm ? at::empty({rows}, f32opts) : at::empty({0}, f32opts);
  const void* bptr = nullptr;
  if (b.has_value() && b->defined()) {
    TORCH_CHECK(b->is_contiguous() && b->numel() == H, "bias shape mismatch");
    bptr = b->data_ptr();
  }
  ds_norm_fwd(x.data_ptr(), w.data_ptr(), bptr, y.data_ptr(),
              invrms.data_ptr<float>(),
              layernorm ? mean.data_ptr<float>() : nullptr, (int)rows, (int)H,
              (float)eps, layernorm ? 1 : 0, dtype_code(x), cur_stream());
  return {y, invrms, mean};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> norm_bwd(
    at::Tensor dy, at::Tensor x, at::Tensor w, at::Tensor invrms,
    c10::optional<at::Tensor> mean, bool layernorm) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous() && w.is_contiguous(),
              "must be contiguous");
  const int64_t H = x.size(-1);
  const int64_t rows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto f32opts = x.options().dtype(at::kFloat);
  auto dw = at::zeros({H}, f32opts);
  auto db = layernorm ? at::zeros({H}, f32opts) : at::empty({0}, f32opts);
  ds_norm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
              invrms.data_ptr<float>(),
              layernorm ? mean->data_ptr<float>() : nullptr, dx.data_ptr(),
              dw.data_ptr<float>(), layernorm ? db.data_ptr<float>() : nullptr,
              (int)rows, (int)H, layernorm ? 1 : 0, dtype_code(x),
              cur_stream());
  return {dx, dw, db};
}

at::Tensor token_gather(at::Tensor x, at::Tensor idx) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous(),
              "token_gather: contiguous [B,S,D] GPU tensor");
  TORCH_CHECK(idx.dim() == 2 && idx.scalar_type() == at::kInt &&
              idx.is_contiguous() && idx.size(0) == x.size(0),
              "token_gather: int32 [B,K] indices");
  const int64_t D = x.size(2);
  const int64_t vec = 16 / x.element_size();
  TORCH_CHECK(D % vec == 0, "token_gather: D % ", vec, " == 0");
  auto y = at::empty({x.size(0), idx.size(1), D}, x.options());
  ds_token_move(x.data_ptr(), y.data_ptr(), idx.data_ptr<int>(),
                (int)x.size(0), (int)x.size(1), (int)idx.size(1), (int)D,
                1, dtype_code(x), cur_stream());
  return y;
}

at::Tensor token_scatter(at::Tensor base, at::Tensor sub, at::Tensor idx) {
  TORCH_CHECK(base.dim() == 3 && sub.dim() == 3 && base.is_cuda() &&
              base.is_contiguous() && sub.is_contiguous() &&
              base.scalar_type() == sub.scalar_type(),
              "token_scatter: contiguous [B,S,D]/[B,K,D] GPU tensors");
  TORCH_CHECK(idx.dim() == 2 && idx.scalar_type() == at::kInt &&
              idx.is_contiguous() && idx.size(1) == sub.size(1),
              "token_scatter: int32 [B,K] indices");
  const int64_t D = base.size(2);
  const int64_t vec = 16 / base.element_size();
  TORCH_CHECK(D % vec == 0, "token_scatter: D % ", vec, " == 0");
  auto y = base.clone();
  ds_token_move(sub.data_ptr(), y.data_ptr(), idx.data_ptr<int>(),
                (int)base.size(0), (int)base.size(1), (int)sub.size(1),
                (int)D, 0, dtype_code(base), cur_stream());
  return y;
}

at::Tensor nhwc_bias_add(at::Tensor act, at::Tensor bias,
                         c10::optional<at::Tensor> other,
                         c10::optional<at::Tensor> other_bias) {
  TORCH_CHECK(act.is_cuda() && act.is_contiguous() && bias.is_contiguous(),
              "nhwc_bias_add: contiguous GPU tensors");
  TORCH_CHECK(act.scalar_type() == at::kBFloat16 ||
              act.scalar_type() == at::kHalf, "nhwc_bias_add: bf16/fp16");
  const int64_t C = bias.numel();
  TORCH_CHECK(C % 8 == 0 && act.numel() % C == 0,
              "nhwc_bias_add: channels % 8 == 0, act a multiple of channels");
  const void* other_p = nullptr;
  const void* ob_p = nullptr;
  if (other.has_value() && other->defined()) {
    TORCH_CHECK(other->is_contiguous() && other->numel() == act.numel() &&
                other->scalar_type() == act.scalar_type(),
                "nhwc_bias_add: other mismatched");
    other_p = other->data_ptr();
  }
  if (other_bias.has_value() && other_bias->defined()) {
    TORCH_CHECK(other_bias->is_contiguous() && other_bias->numel() == C &&
                other_bias->scalar_type() == act.scalar_type(),
                "nhwc_bias_add: other_bias mismatched");
    ob_p = other_bias->data_ptr();
  }
  auto out = at::empty_like(act);
  ds_nhwc_bias_add(act.data_ptr(), bias.data_ptr(), other_p, ob_p,
                   out.data_ptr(), act.numel(), (int)C, dtype_code(act),
                   cur_stream());
  return out;
}

void rope(at::Tensor x, at::Tensor cos_table, at::Tensor sin_table,
          c10::optional<at::Tensor> positions, bool backward) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous(), "rope expects [B,S,H,D]");
  TORCH_CHECK(cos_table.scalar_type() == at::kFloat &&
              sin_table.scalar_type() == at::kFloat, "tables must be fp32");
  const int64_t B = x.size(0), S = x.size(1), Hh = x.size(2), D = x.size(3);
  TORCH_CHECK(cos_table.size(-1) == D / 2, "table dim mismatch");
  TORCH_CHECK(cos_table.size(0) >= S || positions.has_value(),
              "cos table shorter than sequence");
  const int* pos_ptr = nullptr;
  if (positions.has_value() && positions->defined()) {
    TORCH_CHECK(positions->scalar_type() == at::kInt &&
                positions->is_contiguous() && positions->numel() == B * S,
                "positions must be int32 [B,S]");
    pos_ptr = positions->data_ptr<int>();
  }
  ds_rope(x.data_ptr(), cos_table.data_ptr<float>(),
          sin_table.data_ptr<float>(), pos_ptr, B, S, Hh, D, backward ? 1 : 0,
          dtype_code(x), cur_stream());
}

at::Tensor gated_act_fwd(at::Tensor gate, at::Tensor up, int64_t act) {
  TORCH_CHECK(gate.is_contiguous() && up.is_contiguous() &&
              gate.sizes() == up.sizes(), "shape mismatch");
  auto out = at::empty_like(gate);
  ds_gated_act_fwd(gate.data_ptr(), up.data_ptr(), out.data_ptr(),
                   gate.numel(), (int)act, dtype_code(gate), cur_stream());
  return out;
}

std::tuple<at::Tensor, at::Tensor> gated_act_bwd(at::Tensor dout,
                                                 at::Tensor gate,
                                                 at::Tensor up, int64_t act) {
  TORCH_CHECK(dout.is_contiguous() && gate.is_contiguous() &&
              up.is_contiguous(), "must be contiguous");
  auto dgate = at::empty_like(gate);
  auto dup = at::empty_like(up);
  ds_gated_act_bwd(dout.data_ptr(), gate.data_ptr(), up.data_ptr(),
                   dgate.data_ptr(), dup.data_ptr(), dout.numel(), (int)act,
                   dtype_code(gate), cur_stream());
  return {dgate, dup};
}

std::tuple<at::Tensor, at::Tensor> groupwise_quant(at::Tensor x,
                                                   int64_t group_size,
                                                   int64_t bits) {
  TORCH_CHECK(x.is_contiguous() && x.is_cuda(), "x must be contiguous GPU");
  TORCH_CHECK(bits == 8 || bits == 4, "bits must be 4 or 8");
  TORCH_CHECK(group_size % 2 == 0, "group_size must be even");
  const long long n = x.numel();
  const long long groups = (n + group_size - 1) / group_size;
  const long long qbytes = bits == 8 ? n : (n + 1) / 2;
  auto q = at::empty({qbytes}, x.options().dtype(at::kChar));
  auto scales = at::empty({groups}, x.options().dtype(at::kFloat));
  ds_groupwise_quant(x.data_ptr(), dtype_code(x), q.data_ptr(),
                     scales.data_ptr<float>(), n, (int)group_size, (int)bits,
                     cur_stream());
  return {q, scales};
}

at::Tensor groupwise_dequant(at::Tensor q, at::Tensor scales, int64_t numel,
                             int64_t group_size, int64_t bits,
                             at::ScalarType dtype) {
  TORCH_CHECK(q.is_contiguous() && q.is_cuda() && scales.is_contiguous(),
              "q/scales must be contiguous GPU");
  auto out = at::empty({numel}, q.options().dtype(dtype));
  ds_groupwise_dequant(q.data_ptr(), scales.data_ptr<float>(), out.data_ptr(),
                       dtype_code(out), numel, (int)group_size, (int)bits,
                       cur_stream());
  return out;
}

at::Tensor flash_attn_fwd(at::Tensor q, at::Tensor k, at::Tensor vt,
                          double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "flash_fwd: bf16 GPU tensors only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && vt.is_contiguous(),
              "flash_fwd: contiguous tensors required");
  const int B = q.size(0), S = q.size(1), H = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "flash_fwd: head_dim must be 128");
  TORCH_CHECK(S % 32 == 0, "flash_fwd: seq must be a multiple of 32");
  TORCH_CHECK(k.size(1) == S && vt.size(3) == S && vt.size(2) == D &&
              vt.size(1) == Hkv && H % Hkv == 0, "flash_fwd: shape mismatch");
  auto o = at::empty_like(q);
  ds_flash_fwd(q.data_ptr(), k.data_ptr(), vt.data_ptr(), o.data_ptr(),
               nullptr, B, S, H, Hkv, (float)scale, causal ? 1 : 0,
               cur_stream());
  return o;
}

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd_lse(
    at::Tensor q, at::Tensor k, at::Tensor vt, double scale, bool causal) {
  const int B = q.size(0), S = q.size(1), H = q.size(2);
  const int Hkv = k.size(2);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  ds_flash_fwd(q.data_ptr(), k.data_ptr(), vt.data_ptr(), o.data_ptr(),
               lse.data_ptr<float>(), B, S, H, Hkv, (float)scale,
               causal ? 1 : 0, cur_stream());
  return {o, lse};
}

class AioHandle {
 public:
  AioHandle(int64_t block_size, int n_threads)
      : h_(ds_aio_create(block_size, n_threads)) {}
  ~AioHandle() { ds_aio_destroy(h_); }
  void async_pwrite(at::Tensor t, const std::string& path) {
    TORCH_CHECK(!t.is_cuda() && t.is_contiguous(),
                "aio: host contiguous tensors only");
    keep_.push_back(t);  // hold storage until wait()
    TORCH_CHECK(ds_aio_pwrite(h_, t.data_ptr(),
                              t.numel() * t.element_size(),
                              path.c_str()) == 0, "aio pwrite setup failed");
  }
  void async_pread(at::Tensor t, const std::string& path) {
    TORCH_CHECK(!t.is_cuda() && t.is_contiguous(),
                "aio: host contiguous tensors only");
    keep_.push_back(t);
    TORCH_CHECK(ds_aio_pread(h_, t.data_ptr(),
                             t.numel() * t.element_size(),
                             path.c_str()) == 0, "aio pread setup failed");
  }
  int wait() {
    int e;
    {
      pybind11::gil_scoped_release nogil;
      e = ds_aio_wait(h_);
    }
    keep_.clear();
    return e;
  }

 private:
  void* h_;
  std::vector<at::Tensor> keep_;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("flash_bwd_dkdv_dbg",
        [](at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor dout,
           at::Tensor qt, at::Tensor dot, at::Tensor lse, at::Tensor delta,
           at::Tensor dk, at::Tensor dv, double scale, int64_t variant) {
          const int B = q.size(0), H = q.size(1), S = q.size(2);
          const int Hkv = k.size(1);
          ds_flash_bwd_dkdv_dbg(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                                dout.data_ptr(), qt.data_ptr(),
                                dot.data_ptr(), lse.data_ptr<float>(),
                                delta.data_ptr<float>(), dk.data_ptr(),
                                dv.data_ptr(), B, S, H, Hkv, (float)scale,
                                (int)variant, cur_stream());
        },
        "dkdv-only launcher (perf diagnosis)");
  pybind11::class_<ShmComm>(m, "ShmComm")
      .def(pybind11::init<const std::string&, int64_t, int64_t, int64_t>())
      .def("all_reduce", &ShmComm::all_reduce);
  m.def("paged_decode", &paged_decode,
        "split-S flash-decode over a paged KV block table");
  m.def("fused_softmax", &fused_softmax,
        "fused masked/alibi/causal softmax");
  m.def("fused_dropout", &fused_dropout, "fused bias+dropout(+residual)");
  m.def("dropout_bwd", &dropout_bwd, "dropout backward from saved mask");
  m.def("transpose_bf16", &transpose_bf16, "tiled bf16 batched transpose");
  m.def("fp_quantize", &fp_quantize, "groupwise FP4/6/8/12 quantize");
  m.def("fp_dequantize", &fp_dequantize, "groupwise FP dequantize");
  m.def("fused_lion", &fused_lion, "fused Lion step (GPU)");
  m.def("fused_lamb", &fused_lamb, "fused LAMB step (GPU, 2-phase)");
  m.def("fused_adam_flat", &fused_adam_flat,
        "Fused Adam/AdamW on flat fp32 master + 16-bit grad shard");
  m.def("cpu_adam_flat", &cpu_adam_flat,
        "Host Adam/AdamW on flat pinned fp32 master (ZeRO-Offload)");
  m.def("cpu_lion_flat",
        [](at::Tensor p, at::Tensor g, at::Tensor m,
           c10::optional<at::Tensor> p16, double lr, double beta1,
           double beta2, double weight_decay, double inv_scale) {
          TORCH_CHECK(!p.is_cuda() && p.is_contiguous() && g.is_contiguous()
                      && m.is_contiguous(), "cpu_lion: host contiguous");
          TORCH_CHECK(p.scalar_type() == at::kFloat &&
                      m.scalar_type() == at::kFloat, "p/m must be fp32");
          void* p16_ptr = nullptr;
          if (p16.has_value() && p16->defined()) p16_ptr = p16->data_ptr();
          ds_cpu_lion_flat(p.data_ptr<float>(), g.data_ptr(), dtype_code(g),
                           m.data_ptr<float>(), p16_ptr, p.numel(), (float)lr,
                           (float)beta1, (float)beta2, (float)weight_decay,
                           (float)inv_scale);
        },
        "Host Lion on flat fp32 master");
  m.def("norm_fwd", &norm_fwd, "RMSNorm/LayerNorm forward");
  m.def("norm_bwd", &norm_bwd, "RMSNorm/LayerNorm backward");
  m.def("rope", &rope, "Rotary position embedding (in-place)");
  m.def("token_gather", &token_gather,
        "row-coalesced token gather [B,S,D] x [B,K] -> [B,K,D]");
  m.def("token_scatter", &token_scatter,
        "row-coalesced token scatter-back into a cloned base");
  m.def("nhwc_bias_add", &nhwc_bias_add,
        "fused channels-last bias (+residual +residual-bias) add",
        pybind11::arg("act"), pybind11::arg("bias"),
        pybind11::arg("other") = c10::nullopt,
        pybind11::arg("other_bias") = c10::nullopt);
  m.def("gated_act_fwd", &gated_act_fwd, "SwiGLU/GeGLU forward");
  m.def("gated_act_bwd", &gated_act_bwd, "SwiGLU/GeGLU backward");
  m.def("groupwise_quant", &groupwise_quant,
        "Groupwise symmetric int8/int4 quantization");
  m.def("groupwise_dequant", &groupwise_dequant,
        "Groupwise symmetric int8/int4 dequantization");
  m.def("ce_fwd",
        [](at::Tensor logits, at::Tensor labels, int64_t ignore_index) {
          TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() &&
                      logits.scalar_type() == at::kBFloat16 &&
                      labels.scalar_type() == at::kLong, "ce_fwd args");
          const int N = logits.size(0);
          const long long V = logits.size(1);
          auto loss = at::empty({N}, logits.options().dtype(at::kFloat));
          auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
          ds_ce_fwd(logits.data_ptr(), (const long long*)labels.data_ptr<int64_t>(),
                    loss.data_ptr<float>(), lse.data_ptr<float>(), N, V,
                    ignore_index, cur_stream());
          return std::make_tuple(loss, lse);
        },
        "fused CE forward: bf16 logits -> per-token loss + lse");
  m.def("ce_bwd",
        [](at::Tensor logits, at::Tensor labels, at::Tensor lse,
           at::Tensor gscale, int64_t ignore_index) {
          const int N = logits.size(0);
          const long long V = logits.size(1);
          auto d = at::empty_like(logits);
          ds_ce_bwd(logits.data_ptr(), (const long long*)labels.data_ptr<int64_t>(),
                    lse.data_ptr<float>(), gscale.data_ptr<float>(),
                    d.data_ptr(), N, V, ignore_index, cur_stream());
          return d;
        },
        "fused CE backward: dlogits in bf16");
  m.def("flash_attn_fwd_lse", &flash_attn_fwd_lse,
        "flash fwd returning (o, logsumexp) for training");
  m.def("flash_attn_bwd",
        [](at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor dout,
           at::Tensor qt, at::Tensor kt, at::Tensor dot, at::Tensor lse,
           at::Tensor delta, double scale, bool causal) {
          TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 &&
                      q.is_contiguous() && k.is_contiguous() &&
                      v.is_contiguous() && dout.is_contiguous() &&
                      qt.is_contiguous() && kt.is_contiguous() &&
                      dot.is_contiguous(), "flash_bwd: bf16 contiguous GPU");
          TORCH_CHECK(lse.scalar_type() == at::kFloat &&
                      delta.scalar_type() == at::kFloat, "lse/delta fp32");
          const int B = q.size(0), H = q.size(1), S = q.size(2);
          const int Hkv = k.size(1);
          TORCH_CHECK(q.size(3) == 128 && S % 32 == 0,
                      "flash_bwd: D=128, S%32==0");
          auto dq = at::empty_like(q);
          auto dk = at::empty_like(k);
          auto dv = at::empty_like(v);
          ds_flash_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       dout.data_ptr(), qt.data_ptr(), kt.data_ptr(),
                       dot.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                       dv.data_ptr(), B, S, H, Hkv, (float)scale,
                       causal ? 1 : 0, cur_stream());
          return std::make_tuple(dq, dk, dv);
        },
        "MFMA flash-attention backward (BHSD; GPU-validation pending)");
  m.def("flash_attn_fwd_dbg",
        [](at::Tensor q, at::Tensor k, at::Tensor vt, double scale,
           int64_t variant) {
          auto o = at::empty_like(q);
          ds_flash_fwd_dbg(q.data_ptr(), k.data_ptr(), vt.data_ptr(),
                           o.data_ptr(), q.size(0), q.size(1), q.size(2),
                           k.size(2), (float)scale, (int)variant,
                           cur_stream());
          return o;
        },
        "flash fwd ablation (bit0 K-LDS, bit1 V-LDS)");
  m.def("flash_attn_fwd", &flash_attn_fwd,
        "MFMA flash-attention forward (bf16, D=128, GQA, causal)");
  pybind11::class_<AioHandle>(m, "AioHandle")
      .def(pybind11::init<int64_t, int>(),
           pybind11::arg("block_size") = 1 << 20,
           pybind11::arg("n_threads") = 8)
      .def("async_pwrite", &AioHandle::async_pwrite)
      .def("async_pread", &AioHandle::async_pread)
      .def("wait", &AioHandle::wait);
}
