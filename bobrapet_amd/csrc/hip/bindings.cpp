// Torch bindings for the bobrapet_amd CDNA4 kernel library.
//
// This file is the only translation unit that includes torch headers (they
// dominate compile time); the kernels live in *.hip files exposing
// extern "C" launchers over raw pointers + hipStream_t.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

// native-lane hooks (native_engrams.cpp, same .so)
const void* bobra_native_lane_ptr();
bool bobra_native_tensor_get(const std::string& key, at::Tensor* out);
size_t bobra_native_registry_size();
void bobra_native_lane_stats(long*, long*, long*, std::string*);

extern "C" {
void launch_rmsnorm(void*, void*, void*, const void*, int, int, float, bool,
                    hipStream_t);
void launch_silu_mul(void*, const void*, const void*, long, hipStream_t);
void launch_rope(void*, void*, const void*, const void*, int, int, int, int,
                 hipStream_t);
void launch_embed_pool(void*, void*, const void*, const void*, int, int, int,
                       int, hipStream_t);
void launch_add_bf16(void*, const void*, const void*, long, hipStream_t);
void launch_attn_prefill(void*, const void*, const void*, const void*, int,
                         int, int, int, long, float, int, void*, hipStream_t);
void launch_silu_mul_strided(void*, const void*, const void*, long, int, long,
                             hipStream_t);
void launch_rope_qkv(void*, void*, const void*, const void*, const void*, int,
                     int, int, int, long, hipStream_t);
void launch_rope_qkv_decode(void*, void*, void*, const void*, const void*,
                            const void*, int, int, int, int, int, hipStream_t);
void launch_argmax_rows(void*, const void*, int, int, hipStream_t);
void launch_attn_decode(void*, void*, const void*, const void*, const void*,
                        int, int, int, int, int, const void*, float,
                        hipStream_t);
void launch_dbg_mfma(void*, const void*, const void*, hipStream_t);
void launch_attn_prefill_pipe(void*, const void*, const void*, const void*,
                              int, int, int, int, long, float, int, void*,
                              hipStream_t);
void launch_attn_prefill_qrope(void*, const void*, const void*, const void*,
                               int, int, int, int, long, long, const void*,
                               int, float, int, hipStream_t);
void launch_attn_prefill_variant(int, void*, const void*, const void*,
                                 const void*, int, int, int, int, float, int,
                                 hipStream_t);
void launch_gemm_bf16_nt(void*, const void*, const void*, int, int, int,
                         hipStream_t);
void launch_gemv_bf16_nt(void*, const void*, const void*, int, int, int,
                         hipStream_t);
void launch_gemm256(int, void*, const void*, const void*, const void*,
                    const void*, void*, int, int, int, float, float,
                    hipStream_t);
void launch_gemv2(int, void*, const void*, const void*, const void*, int, int,
                  int, float, float, hipStream_t);
void launch_gemm256b_disc(int, void*, const void*, const void*, int, int,
                          int, hipStream_t);
void launch_gemm256w(int, void*, const void*, const void*, const void*,
                     const void*, void*, int, int, int, float, float,
                     hipStream_t);
void launch_gemmsk(int, void*, void*, const void*, const void*, const void*,
                   const void*, void*, int, int, int, float, float,
                   hipStream_t);
long gemmsk_ws_elems(int, int);
void launch_gemm256b(int, void*, const void*, const void*, const void*,
                     const void*, void*, int, int, int, float, float,
                     hipStream_t);
void launch_rowsumsq(void*, const void*, int, int, hipStream_t);
void launch_dbg_attn_core(void*, void*, const void*, const void*, const void*,
                          hipStream_t);
}

namespace {

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream(); }

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0 && H <= 8192, "rmsnorm: H must be %8==0 and <=8192");
  TORCH_CHECK(w.numel() == H, "rmsnorm: weight shape mismatch");
  auto out = torch::empty_like(x);
  const long rows = x.numel() / H;
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), nullptr, w.data_ptr(),
                 (int)rows, H, (float)eps, false, cur_stream());
  return out;
}

torch::Tensor fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                                torch::Tensor w, double eps) {
  // residual += x (written back); returns rmsnorm(residual) * w
  check_bf16(x, "x");
  check_bf16(residual, "residual");
  check_bf16(w, "w");
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0 && H <= 8192, "fused_add_rmsnorm: bad H");
  TORCH_CHECK(residual.sizes() == x.sizes(), "shape mismatch");
  auto out = torch::empty_like(x);
  const long rows = x.numel() / H;
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), residual.data_ptr(),
                 w.data_ptr(), (int)rows, H, (float)eps, true, cur_stream());
  return out;
}

torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up) {
  check_bf16(gate, "gate");
  check_bf16(up, "up");
  TORCH_CHECK(gate.sizes() == up.sizes(), "silu_mul: shape mismatch");
  TORCH_CHECK(gate.numel() % 8 == 0, "silu_mul: numel must be %8==0");
  auto out = torch::empty_like(gate);
  launch_silu_mul(out.data_ptr(), gate.data_ptr(), up.data_ptr(), gate.numel(),
                  cur_stream());
  return out;
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
                  torch::Tensor sin_t) {
  // q: [T, Hq, D], k: [T, Hk, D] bf16; cos/sin: [T, D/2] f32
  check_bf16(q, "q");
  check_bf16(k, "k");
  TORCH_CHECK(cos_t.scalar_type() == torch::kFloat32 && cos_t.is_contiguous(),
              "cos must be f32 contiguous");
  TORCH_CHECK(sin_t.scalar_type() == torch::kFloat32 && sin_t.is_contiguous(),
              "sin must be f32 contiguous");
  const int T = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hk = k.size(1);
  TORCH_CHECK(k.size(0) == T && k.size(2) == D, "rope: q/k shape mismatch");
  TORCH_CHECK(cos_t.size(0) == T && cos_t.size(1) == D / 2, "rope: table shape");
  launch_rope(q.data_ptr(), k.data_ptr(), cos_t.data_ptr(), sin_t.data_ptr(),
              T, Hq, Hk, D, cur_stream());
}

torch::Tensor embed_pool(torch::Tensor table, torch::Tensor ids) {
  check_bf16(table, "table");
  TORCH_CHECK(ids.scalar_type() == torch::kInt32 && ids.is_cuda() &&
                  ids.is_contiguous(),
              "ids must be int32 on GPU");
  const int V = table.size(0), H = table.size(1);
  const int B = ids.size(0), S = ids.size(1);
  TORCH_CHECK(H % 8 == 0 && H <= 8192, "embed_pool: bad H");
  auto out = torch::empty({B, H}, table.options());
  const int nchunk = (S + 7) / 8;  // EMB_SCHUNK partials
  // empty + conditional memset: the sum kernel plain-stores every slot
  // when H is a multiple of the 2048-wide slice (the common case — the
  // per-packet at::native fill was 18% of the stream config's kernel
  // time); only a slice tail needs pre-zeroing, via hipMemsetAsync
  auto pooled =
      torch::empty({(long)B * nchunk, H}, table.options().dtype(torch::kFloat32));
  if (H % 2048 != 0)
    (void)hipMemsetAsync(pooled.data_ptr(), 0,
                         (size_t)B * nchunk * H * sizeof(float), cur_stream());
  launch_embed_pool(out.data_ptr(), pooled.data_ptr(), table.data_ptr(),
                    ids.data_ptr(), B, S, H, V, cur_stream());
  return out;
}

torch::Tensor add_bf16(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  TORCH_CHECK(a.sizes() == b.sizes() && a.numel() % 8 == 0, "add: bad shapes");
  auto out = torch::empty_like(a);
  launch_add_bf16(out.data_ptr(), a.data_ptr(), b.data_ptr(), a.numel(),
                  cur_stream());
  return out;
}

torch::Tensor attn_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                           double scale, bool causal) {
  // q: [B,S,Hq,D=128], k/v: [B,S,Hkv,D] (BSHD: the model's native layout,
  // no transposes on the hot path); v may be a strided view
  check_bf16(q, "q");
  check_bf16(k, "k");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "attn_prefill: D must be 128");
  TORCH_CHECK(Hq % Hkv == 0, "attn_prefill: Hq must be a multiple of Hkv");
  TORCH_CHECK(k.size(1) == S && v.size(1) == S, "attn_prefill: S mismatch");
  // v may be a strided [B,S,Hkv,D] view (head+lane dims contiguous)
  TORCH_CHECK(v.is_cuda() && v.scalar_type() == torch::kBFloat16, "v dtype");
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D, "attn_prefill: v head must be contiguous");
  TORCH_CHECK(v.stride(0) == v.stride(1) * S, "attn_prefill: v batch stride");
  auto out = torch::empty_like(q);
  launch_attn_prefill(out.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                      B, Hq, Hkv, S, (long)v.stride(1), (float)scale,
                      causal ? 1 : 0, nullptr, cur_stream());
  return out;
}

torch::Tensor attn_prefill_qrope(torch::Tensor q, torch::Tensor k,
                                 torch::Tensor v, torch::Tensor inv_freq,
                                 int64_t pos0, double scale, bool causal) {
  // q is a STRIDED view into the fused qkv projection; rope is applied to
  // the Q rows on load (inv_freq [D/2] f32) — no q copy, no q rope kernel
  check_bf16(k, "k");
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16, "q dtype");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "attn_prefill_qrope: D must be 128");
  TORCH_CHECK(Hq % Hkv == 0, "attn_prefill_qrope: Hq % Hkv");
  TORCH_CHECK(k.size(1) == S && v.size(1) == S, "attn_prefill_qrope: S mismatch");
  TORCH_CHECK(q.stride(3) == 1 && q.stride(2) == D, "q head must be contiguous");
  TORCH_CHECK(q.stride(0) == q.stride(1) * S, "q batch stride");
  TORCH_CHECK(v.is_cuda() && v.scalar_type() == torch::kBFloat16, "v dtype");
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D, "v head must be contiguous");
  TORCH_CHECK(v.stride(0) == v.stride(1) * S, "v batch stride");
  TORCH_CHECK(inv_freq.is_cuda() && inv_freq.scalar_type() == torch::kFloat32 &&
                  inv_freq.numel() == D / 2 && inv_freq.is_contiguous(),
              "inv_freq must be f32 [D/2] contiguous");
  auto out = torch::empty({B, S, Hq, D}, k.options());
  launch_attn_prefill_qrope(out.data_ptr(), q.data_ptr(), k.data_ptr(),
                            v.data_ptr(), B, Hq, Hkv, S, (long)v.stride(1),
                            (long)q.stride(1), inv_freq.data_ptr(), (int)pos0,
                            (float)scale, causal ? 1 : 0, cur_stream());
  return out;
}

std::vector<torch::Tensor> rope_k_only(torch::Tensor qkv, int64_t Hq,
                                       int64_t Hk, int64_t D,
                                       torch::Tensor cos_t,
                                       torch::Tensor sin_t) {
  // rope ONLY the K heads out of the fused qkv rows (Q is roped on load
  // inside attn_prefill_qrope); returns rotated contiguous k [T,Hk,D]
  check_bf16(qkv, "qkv");
  const long T = qkv.size(0);
  TORCH_CHECK(qkv.is_contiguous(), "rope_k_only: qkv must be contiguous");
  TORCH_CHECK(cos_t.size(0) == T && cos_t.size(1) == D / 2, "rope_k_only tables");
  auto kout = torch::empty({T, Hk, D}, qkv.options());
  const unsigned short* base =
      (const unsigned short*)qkv.data_ptr() + (long)Hq * D;
  launch_rope_qkv(nullptr, kout.data_ptr(), base, cos_t.data_ptr(),
                  sin_t.data_ptr(), (int)T, 0, (int)Hk, (int)D,
                  (long)qkv.size(1), cur_stream());
  return {kout};
}

torch::Tensor attn_prefill_pipe(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, double scale, bool causal) {
  // pipelined variant (QK(t) in flight over softmax+PV(t-1)) — A/B entry
  check_bf16(q, "q");
  check_bf16(k, "k");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "attn_prefill_pipe: D must be 128");
  TORCH_CHECK(Hq % Hkv == 0, "attn_prefill_pipe: Hq % Hkv");
  TORCH_CHECK(k.size(1) == S && v.size(1) == S, "attn_prefill_pipe: S mismatch");
  TORCH_CHECK(v.is_cuda() && v.scalar_type() == torch::kBFloat16, "v dtype");
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D, "v head must be contiguous");
  TORCH_CHECK(v.stride(0) == v.stride(1) * S, "v batch stride");
  auto out = torch::empty_like(q);
  launch_attn_prefill_pipe(out.data_ptr(), q.data_ptr(), k.data_ptr(),
                           v.data_ptr(), B, Hq, Hkv, S, (long)v.stride(1),
                           (float)scale, causal ? 1 : 0, nullptr,
                           cur_stream());
  return out;
}

std::vector<torch::Tensor> attn_prefill_stats(torch::Tensor q, torch::Tensor k,
                                              torch::Tensor v, double scale,
                                              bool causal) {
  // attn_prefill + per-row softmax stats [B,Hq,S,2] f32 (m, l) in the
  // kernel's exp2 domain — the cross-block merge surface for ring attention
  check_bf16(q, "q");
  check_bf16(k, "k");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "attn_prefill_stats: D must be 128");
  TORCH_CHECK(Hq % Hkv == 0, "attn_prefill_stats: Hq % Hkv");
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D, "v head must be contiguous");
  auto out = torch::empty_like(q);
  auto stats = torch::empty({B, Hq, S, 2}, q.options().dtype(torch::kFloat32));
  launch_attn_prefill(out.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                      B, Hq, Hkv, S, (long)v.stride(1), (float)scale,
                      causal ? 1 : 0, stats.data_ptr(), cur_stream());
  return {out, stats};
}

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          long L, double scale) {
  // q: [B,Hq,D], kc/vc: [B,Hkv,Smax,D]
  check_bf16(q, "q");
  check_bf16(kc, "kc");
  check_bf16(vc, "vc");
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = kc.size(1), Smax = kc.size(2);
  TORCH_CHECK(D == 128, "attn_decode: D must be 128");
  TORCH_CHECK(L <= Smax, "attn_decode: L > Smax");
  auto out = torch::empty_like(q);
  void* ws_ptr = nullptr;
  torch::Tensor ws;
  // two-pass chunked decode (lane-per-row scores + broadcast PV) is the
  // default; BOBRA_DEC_V1=1 forces the single-pass per-(b,h) kernel.
  if (!(getenv("BOBRA_DEC_V1") && getenv("BOBRA_DEC_V1")[0] == '1') &&
      Hq / Hkv <= 8) {
    const int nchunk = (Smax + 255) / 256;
    ws = torch::empty({(long)B * Hkv * nchunk * 8 * 130},
                      q.options().dtype(torch::kFloat32));
    ws_ptr = ws.data_ptr();
  }
  launch_attn_decode(out.data_ptr(), ws_ptr, q.data_ptr(), kc.data_ptr(),
                     vc.data_ptr(), B, Hq, Hkv, Smax, (int)L, nullptr,
                     (float)scale, cur_stream());
  return out;
}

torch::Tensor attn_decode_t(torch::Tensor q, torch::Tensor kc,
                            torch::Tensor vc, torch::Tensor L_dev,
                            double scale) {
  // graph-capturable decode: L read from a device int32 scalar
  check_bf16(q, "q");
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = kc.size(1), Smax = kc.size(2);
  TORCH_CHECK(D == 128, "attn_decode_t: D must be 128");
  TORCH_CHECK(L_dev.scalar_type() == torch::kInt32 && L_dev.is_cuda(),
              "L must be an int32 device scalar");
  auto out = torch::empty_like(q);
  void* ws_ptr = nullptr;
  torch::Tensor ws;
  // two-pass chunked decode (lane-per-row scores + broadcast PV) is the
  // default; BOBRA_DEC_V1=1 forces the single-pass per-(b,h) kernel.
  if (!(getenv("BOBRA_DEC_V1") && getenv("BOBRA_DEC_V1")[0] == '1') &&
      Hq / Hkv <= 8) {
    const int nchunk = (Smax + 255) / 256;
    ws = torch::empty({(long)B * Hkv * nchunk * 8 * 130},
                      q.options().dtype(torch::kFloat32));
    ws_ptr = ws.data_ptr();
  }
  launch_attn_decode(out.data_ptr(), ws_ptr, q.data_ptr(), kc.data_ptr(),
                     vc.data_ptr(), B, Hq, Hkv, Smax, 0, L_dev.data_ptr(),
                     (float)scale, cur_stream());
  return out;
}

torch::Tensor rope_qkv_decode(torch::Tensor qkv, torch::Tensor kc,
                              torch::Tensor vc, torch::Tensor inv_freq,
                              torch::Tensor L_dev, long Hq, long Hkv, long D) {
  // fused decode head prep: rope q (returned contiguous [B,Hq,D]) and
  // rope k + copy v straight into the KV cache at device position L;
  // rope angles computed in-kernel from inv_freq (no host cos/sin tables)
  check_bf16(qkv, "qkv");
  check_bf16(kc, "kc");
  check_bf16(vc, "vc");
  const int B = qkv.size(0);
  const int Smax = kc.size(2);
  TORCH_CHECK(qkv.size(1) == (Hq + 2 * Hkv) * D, "rope_qkv_decode: row size");
  TORCH_CHECK(inv_freq.scalar_type() == torch::kFloat32 &&
                  inv_freq.is_contiguous() && inv_freq.numel() == D / 2,
              "inv_freq must be f32 [D/2] contiguous");
  TORCH_CHECK(L_dev.scalar_type() == torch::kInt32 && L_dev.is_cuda(),
              "L must be an int32 device scalar");
  auto q_out = torch::empty({(long)B, Hq, D}, qkv.options());
  launch_rope_qkv_decode(q_out.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                         qkv.data_ptr(), inv_freq.data_ptr(), L_dev.data_ptr(),
                         B, (int)Hq, (int)Hkv, (int)D, Smax, cur_stream());
  return q_out;
}

torch::Tensor argmax_rows(torch::Tensor x) {
  check_bf16(x, "x");
  const int N = x.size(-1);
  const int M = x.numel() / N;
  auto ids = torch::empty({(long)M}, x.options().dtype(torch::kLong));
  launch_argmax_rows(ids.data_ptr(), x.data_ptr(), M, N, cur_stream());
  return ids;
}

torch::Tensor silu_mul_strided(torch::Tensor gate_up) {
  // gate_up: [..., 2*I] bf16 (contiguous); returns silu(g)*u of shape [..., I]
  check_bf16(gate_up, "gate_up");
  const int two_i = gate_up.size(-1);
  const int inner = two_i / 2;
  TORCH_CHECK(inner % 8 == 0, "silu_mul_strided: I must be %8==0");
  const long rows = gate_up.numel() / two_i;
  auto sizes = gate_up.sizes().vec();
  sizes.back() = inner;
  auto out = torch::empty(sizes, gate_up.options());
  const unsigned short* base = (const unsigned short*)gate_up.data_ptr();
  launch_silu_mul_strided(out.data_ptr(), base, base + inner, rows, inner,
                          (long)two_i, cur_stream());
  return out;
}

std::vector<torch::Tensor> rope_qkv(torch::Tensor qkv, int Hq, int Hk, int D,
                                    torch::Tensor cos_t, torch::Tensor sin_t) {
  // qkv: [T, rowlen] bf16 contiguous; returns rotated contiguous q [T,Hq,D],
  // k [T,Hk,D] (v stays a view into qkv)
  check_bf16(qkv, "qkv");
  const long T = qkv.size(0);
  TORCH_CHECK(cos_t.size(0) == T && cos_t.size(1) == D / 2, "rope_qkv tables");
  auto q = torch::empty({T, (long)Hq, (long)D}, qkv.options());
  auto k = torch::empty({T, (long)Hk, (long)D}, qkv.options());
  launch_rope_qkv(q.data_ptr(), k.data_ptr(), qkv.data_ptr(), cos_t.data_ptr(),
                  sin_t.data_ptr(), (int)T, Hq, Hk, D, (long)qkv.size(1),
                  cur_stream());
  return {q, k};
}

torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b) {
  // a: [M,K] bf16, b: [N,K] bf16 (transposed-weight layout) -> [M,N]
  check_bf16(a, "a");
  check_bf16(b, "b");
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "gemm_nt: K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
              "gemm_nt: M,N must be multiples of 128 and K of 64");
  auto c = torch::empty({M, N}, a.options());
  launch_gemm_bf16_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(), M, N, K,
                      cur_stream());
  return c;
}

void check_gemm256(const torch::Tensor& a, const torch::Tensor& b, int M,
                   int N, int K) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  TORCH_CHECK(b.size(1) == K, "gemm256: K mismatch");
  TORCH_CHECK(N % 256 == 0 && K % 32 == 0,
              "gemm256: N must be a multiple of 256 and K of 32");
}

void launch_gemm256_auto(int epi, void* C, const void* A, const void* B,
                         const void* resid, const void* stat_in,
                         void* stat_out, int M, int N, int K, float stat_mul,
                         float stat_eps, hipStream_t st) {
  if (K % 64 == 0)
    launch_gemm256b(epi, C, A, B, resid, stat_in, stat_out, M, N, K, stat_mul,
                    stat_eps, st);
  else
    launch_gemm256(epi, C, A, B, resid, stat_in, stat_out, M, N, K, stat_mul,
                   stat_eps, st);
}

const float* stat_ptr(const c10::optional<torch::Tensor>& stat, int M) {
  if (!stat.has_value()) return nullptr;
  TORCH_CHECK(stat->scalar_type() == torch::kFloat32 && stat->is_contiguous() &&
                  stat->numel() == M,
              "gemm256: row stat must be f32 [M]");
  return (const float*)stat->data_ptr();
}

torch::Tensor gemm256_nt(torch::Tensor a, torch::Tensor b,
                         c10::optional<torch::Tensor> stat, double stat_mul,
                         double stat_eps) {
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  check_gemm256(a, b, M, N, K);
  auto c = torch::empty({M, N}, a.options());
  launch_gemm256_auto(0, c.data_ptr(), a.data_ptr(), b.data_ptr(), nullptr,
                 stat_ptr(stat, M), nullptr, M, N, K, (float)stat_mul,
                 (float)stat_eps, cur_stream());
  return c;
}

torch::Tensor gemm256_nt_disc(torch::Tensor a, torch::Tensor b,
                              int64_t disc) {
  // probe-only: alternate sync disciplines (1=template pairs, 2=4/tile)
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  check_gemm256(a, b, M, N, K);
  TORCH_CHECK(K % 64 == 0, "gemm256_nt_disc: K % 64 required");
  auto c = torch::empty({M, N}, a.options());
  launch_gemm256b_disc((int)disc, c.data_ptr(), a.data_ptr(), b.data_ptr(), M,
                       N, K, cur_stream());
  return c;
}

// skinny-M GEMM (decode batch 9..32): epi 0 rowscale, 1 swiglu, 2 resid
std::vector<torch::Tensor> gemmsk(torch::Tensor a, torch::Tensor b,
                                  int64_t epi,
                                  c10::optional<torch::Tensor> resid,
                                  c10::optional<torch::Tensor> stat,
                                  double stat_mul, double stat_eps) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(M <= 32, "gemmsk: M must be <= 32");
  TORCH_CHECK(N % 64 == 0 && K % 64 == 0, "gemmsk: N%64, K%64 required");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous(), "gemmsk: contiguous");
  TORCH_CHECK(epi != 2 || (resid && resid->numel() == (long)M * N &&
                           resid->is_contiguous()),
              "gemmsk: resid must be contiguous [M, N]");
  auto c = epi == 1 ? torch::empty({M, N / 2}, a.options())
                    : torch::empty({M, N}, a.options());
  torch::Tensor ws = torch::empty(
      {gemmsk_ws_elems(N, K)}, a.options().dtype(torch::kFloat32));
  torch::Tensor statout;
  void* statout_p = nullptr;
  if (epi == 2) {
    statout = torch::empty({M}, a.options().dtype(torch::kFloat32));
    statout_p = statout.data_ptr();
  }
  launch_gemmsk((int)epi, c.data_ptr(), ws.data_ptr(), a.data_ptr(),
                b.data_ptr(), resid ? resid->data_ptr() : nullptr,
                stat_ptr(stat, M), statout_p, M, N, K, (float)stat_mul,
                (float)stat_eps, cur_stream());
  if (epi == 2) return {c, statout};
  return {c};
}

torch::Tensor gemm256_w(torch::Tensor a, torch::Tensor b, int64_t epi,
                        c10::optional<torch::Tensor> resid,
                        c10::optional<torch::Tensor> stat, double stat_mul,
                        double stat_eps) {
  // 32x32x16-MFMA variant probe (all three epilogues)
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  check_gemm256(a, b, M, N, K);
  TORCH_CHECK(K % 64 == 0, "gemm256_w: K % 64 required");
  auto c = epi == 1 ? torch::empty({M, N / 2}, a.options())
                    : torch::empty({M, N}, a.options());
  torch::Tensor statout;
  void* statout_p = nullptr;
  if (epi == 2) {
    statout = torch::empty({M}, a.options().dtype(torch::kFloat32));
    statout_p = statout.data_ptr();
  }
  launch_gemm256w((int)epi, c.data_ptr(), a.data_ptr(), b.data_ptr(),
                  resid ? resid->data_ptr() : nullptr, stat_ptr(stat, M),
                  statout_p, M, N, K, (float)stat_mul, (float)stat_eps,
                  cur_stream());
  return c;
}

torch::Tensor gemm256_swiglu(torch::Tensor a, torch::Tensor b,
                             c10::optional<torch::Tensor> stat,
                             double stat_mul, double stat_eps) {
  // b rows interleaved (gate_i, up_i); out[M][N/2] = silu(g)*u
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  check_gemm256(a, b, M, N, K);
  auto c = torch::empty({M, N / 2}, a.options());
  launch_gemm256_auto(1, c.data_ptr(), a.data_ptr(), b.data_ptr(), nullptr,
                 stat_ptr(stat, M), nullptr, M, N, K, (float)stat_mul,
                 (float)stat_eps, cur_stream());
  return c;
}

std::vector<torch::Tensor> gemm256_resid(torch::Tensor a, torch::Tensor b,
                                         torch::Tensor resid, bool want_stat) {
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  check_gemm256(a, b, M, N, K);
  check_bf16(resid, "resid");
  TORCH_CHECK(resid.numel() == (long)M * N, "gemm256_resid: resid shape");
  auto c = torch::empty({M, N}, a.options());
  torch::Tensor stat;
  void* stat_p = nullptr;
  if (want_stat) {
    stat = torch::empty({M}, a.options().dtype(torch::kFloat32));
    stat_p = stat.data_ptr();
  }
  launch_gemm256_auto(2, c.data_ptr(), a.data_ptr(), b.data_ptr(), resid.data_ptr(),
                 nullptr, stat_p, M, N, K, 0.0f, 0.0f, cur_stream());
  if (want_stat) return {c, stat};
  return {c};
}

torch::Tensor gemv_norm(torch::Tensor a, torch::Tensor w, double mul,
                        double eps) {
  // out = rsqrt(sumsq(a_m)*mul + eps) * (a @ w.T)  (self-normalizing entry)
  check_bf16(a, "a");
  check_bf16(w, "w");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(M <= 8 && w.size(1) == K && K % 8 == 0, "gemv_norm: bad shape");
  auto c = torch::empty({M, N}, a.options());
  launch_gemv2(1, c.data_ptr(), a.data_ptr(), w.data_ptr(), nullptr, M, N, K,
               (float)mul, (float)eps, cur_stream());
  return c;
}

torch::Tensor gemv_resid(torch::Tensor a, torch::Tensor w,
                         torch::Tensor resid) {
  check_bf16(a, "a");
  check_bf16(w, "w");
  check_bf16(resid, "resid");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(M <= 8 && w.size(1) == K && K % 8 == 0, "gemv_resid: bad shape");
  TORCH_CHECK(resid.numel() == (long)M * N, "gemv_resid: resid shape");
  auto c = torch::empty({M, N}, a.options());
  launch_gemv2(2, c.data_ptr(), a.data_ptr(), w.data_ptr(), resid.data_ptr(),
               M, N, K, 0.0f, 0.0f, cur_stream());
  return c;
}

torch::Tensor gemv_swiglu_norm(torch::Tensor a, torch::Tensor w_i, double mul,
                               double eps) {
  // w_i rows interleaved (gate_i, up_i): out[M, rows/2] = silu(s*g)*(s*u)
  check_bf16(a, "a");
  check_bf16(w_i, "w");
  const int M = a.size(0), K = a.size(1), N = w_i.size(0) / 2;
  TORCH_CHECK(M <= 8 && w_i.size(1) == K && K % 8 == 0 && w_i.size(0) % 2 == 0,
              "gemv_swiglu_norm: bad shape");
  auto c = torch::empty({M, N}, a.options());
  launch_gemv2(3, c.data_ptr(), a.data_ptr(), w_i.data_ptr(), nullptr, M, N, K,
               (float)mul, (float)eps, cur_stream());
  return c;
}

torch::Tensor rowsumsq(torch::Tensor x) {
  check_bf16(x, "x");
  const int K = x.size(-1);
  const int M = x.numel() / K;
  TORCH_CHECK(K % 8 == 0, "rowsumsq: K must be %8==0");
  auto stat = torch::empty({M}, x.options().dtype(torch::kFloat32));
  launch_rowsumsq(stat.data_ptr(), x.data_ptr(), M, K, cur_stream());
  return stat;
}

torch::Tensor gemv_nt(torch::Tensor a, torch::Tensor b) {
  // a: [M<=8, K] bf16, b: [N, K] bf16 -> [M, N]
  check_bf16(a, "a");
  check_bf16(b, "b");
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(M <= 8, "gemv_nt: M must be <= 8");
  TORCH_CHECK(b.size(1) == K && K % 8 == 0, "gemv_nt: K must be %8==0");
  auto c = torch::empty({M, N}, a.options());
  launch_gemv_bf16_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(), M, N, K,
                      cur_stream());
  return c;
}

torch::Tensor attn_prefill_variant(int variant, torch::Tensor q,
                                   torch::Tensor k, torch::Tensor v,
                                   double scale, bool causal) {
  const int B = q.size(0), S = q.size(1), Hq = q.size(2);
  const int Hkv = k.size(2);
  auto out = torch::zeros_like(q);
  launch_attn_prefill_variant(variant, out.data_ptr(), q.data_ptr(),
                              k.data_ptr(), v.data_ptr(), B, Hq, Hkv, S,
                              (float)scale, causal ? 1 : 0, cur_stream());
  return out;
}

torch::Tensor dbg_mfma(torch::Tensor A, torch::Tensor B) {
  auto C = torch::zeros({32, 32}, A.options());
  launch_dbg_mfma(C.data_ptr(), A.data_ptr(), B.data_ptr(), cur_stream());
  return C;
}

std::vector<torch::Tensor> dbg_attn_core(torch::Tensor Q, torch::Tensor K,
                                         torch::Tensor V) {
  auto S = torch::zeros({64, 32}, Q.options().dtype(torch::kFloat32));
  auto O = torch::zeros({32, 128}, Q.options().dtype(torch::kFloat32));
  launch_dbg_attn_core(S.data_ptr(), O.data_ptr(), Q.data_ptr(), K.data_ptr(),
                       V.data_ptr(), cur_stream());
  return {S, O};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "bobrapet_amd hand-written CDNA4 (gfx950) kernels";
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, fused scale)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "residual += x; rmsnorm(residual)");
  m.def("silu_mul", &silu_mul, "silu(gate) * up");
  m.def("rope_inplace", &rope_inplace, "rotary embedding in-place on q,k");
  m.def("embed_pool", &embed_pool, "gather + mean-pool + l2norm");
  m.def("add_bf16", &add_bf16, "a + b");
  m.def("attn_prefill_stats", &attn_prefill_stats,
        "prefill attention + per-row (m,l) softmax stats");
  m.def("attn_prefill", &attn_prefill, "flash attention prefill (MFMA)");
  m.def("attn_prefill_pipe", &attn_prefill_pipe);
  m.def("attn_prefill_qrope", &attn_prefill_qrope);
  m.def("rope_k_only", &rope_k_only);
  m.def("attn_decode", &attn_decode, "decode attention w/ KV cache");
  m.def("rope_qkv_decode", &rope_qkv_decode,
        "fused decode rope + KV-cache append (graph-replayable)");
  m.def("attn_decode_t", &attn_decode_t,
        "decode attention, length from a device scalar (hipGraph-capturable)");
  m.def("dbg_mfma", &dbg_mfma, "layout probe: C=A@B one mfma");
  m.def("attn_prefill_variant", &attn_prefill_variant,
        "ablation: 1=stage 3=+qk/softmax 7=full");
  m.def("gemm_nt", &gemm_nt, "bf16 MFMA GEMM: [M,K] @ [N,K]^T");
  m.def("gemm256_nt_disc", &gemm256_nt_disc);
  m.def("gemm256_w", &gemm256_w);
  m.def("gemmsk", &gemmsk);
  m.def("gemm256_nt", &gemm256_nt,
        "256-tile bf16 MFMA GEMM, optional fused row-scale epilogue");
  m.def("gemm256_swiglu", &gemm256_swiglu,
        "256-tile GEMM with fused SwiGLU epilogue (interleaved gate/up)");
  m.def("gemm256_resid", &gemm256_resid,
        "256-tile GEMM with fused residual-add (+row sumsq) epilogue");
  m.def("rowsumsq", &rowsumsq, "per-row sum of squares (f32)");
  m.def("gemv_norm", &gemv_norm, "self-normalizing decode GEMV (rmsnorm entry folded)");
  m.def("gemv_resid", &gemv_resid, "decode GEMV with fused residual add");
  m.def("gemv_swiglu_norm", &gemv_swiglu_norm,
        "decode GEMV, interleaved gate/up + fused SwiGLU + norm entry");
  m.def("gemv_nt", &gemv_nt, "bf16 weight-streaming GEMV (M<=8)");
  m.def("silu_mul_strided", &silu_mul_strided,
        "silu(gate)*up from a fused [.., 2I] gate_up matrix (no copies)");
  m.def("rope_qkv", &rope_qkv,
        "fused rope: strided q/k heads from the qkv projection -> contiguous");
  m.def("dbg_attn_core", &dbg_attn_core, "layout probe: QK^T + pack + PV");
  m.def("native_lane_capsule", []() {
    return py::capsule(const_cast<void*>(bobra_native_lane_ptr()),
                       "bobra_native_lane");
  });
  m.def("native_tensor_get", [](const std::string& key) -> py::object {
    at::Tensor t;
    if (!bobra_native_tensor_get(key, &t)) return py::none();
    return py::cast(t);
  });
  m.def("native_registry_size", []() { return bobra_native_registry_size(); });
  m.def("native_lane_stats", []() {
    long l, fb, fl;
    std::string err;
    bobra_native_lane_stats(&l, &fb, &fl, &err);
    py::dict d;
    d["launches"] = l;
    d["fallbacks"] = fb;
    d["failures"] = fl;
    d["lastError"] = err;
    return d;
  });
  m.def("argmax_rows", &argmax_rows, "row argmax over bf16 logits");
}
