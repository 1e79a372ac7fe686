// Torch extension bindings for the gfx950 kernel suite + the pinned-host
// snapshot engine (K13, SURVEY.md §2.4: the GPU memory-snapshot engine behind
// the reference's cold-start lifecycle, gpu_snapshot.py:41-53 /
// sglang_snapshot.py:303-312).
//
// The kernels themselves live in the sibling .hip files (plain HIP, no torch
// dependency); this is the only file that includes <torch/extension.h>.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <unordered_map>
#include <vector>

// ---- extern "C" launchers from the .hip translation units ----
extern "C" {
void fa_fwd_bf16(const void*, const void*, const void*, void*, int, int, int,
                 int, int, int, float, int, hipStream_t);
void fa_fwd_strided_bf16(const void*, const void*, const void*, void*, int,
                         int, int, int, int, int, float, int,
                         const long long*, hipStream_t);
void paged_decode_bf16(const void*, const void*, const void*, const int*,
                       const int*, void*, float*, int, int, int, int, int,
                       int, int, float, int, hipStream_t);
void groupnorm_silu_bf16(const void*, void*, float*, const float*,
                         const float*, int, int, long long, int, float, int,
                         hipStream_t);
void layernorm_bf16(const void*, void*, const float*, const float*, long long,
                    int, float, hipStream_t);
void rmsnorm_bf16(const void*, void*, const float*, long long, int, float,
                  hipStream_t);
void cfg_euler_bf16(const void*, const void*, const void*, void*, float, float,
                    long long, hipStream_t);
void silu_mul_bf16(const void*, const void*, void*, long long, hipStream_t);
void geglu_bf16(const void*, const void*, void*, long long, hipStream_t);
void glu_fused_bf16(const void*, void*, long long, long long, int,
                    hipStream_t);
void add_bf16(const void*, const void*, void*, long long, hipStream_t);
void rope_bf16(void*, const float*, const float*, long long, int, int, int,
               long long, long long, long long, const int*, hipStream_t);
void adamw_step(void*, const void*, float*, float*, long long, float, float,
                float, float, float, int, int, hipStream_t);
void gumbel_sample(const float*, unsigned long long*, int*, int, int,
                   float, unsigned long long, hipStream_t);
void softmax_rows(const float*, float*, int, int, hipStream_t);
void tr16_probe(short*, hipStream_t);
void scale_in_dev_bf16(const void*, void*, const float*, const long long*,
                       long long, hipStream_t);
void cfg_euler_dev_bf16(const void*, const void*, const void*, void*,
                        const float*, const long long*, float, long long,
                        hipStream_t);
void advance_step(long long*, hipStream_t);
void conv3x3_bf16(const void*, const void*, const void*, const void*, void*,
                  const float*, const float*, int, int, int, int, int, int,
                  int, int, hipStream_t);
void gn_conv_coeffs_bf16(const void*, float*, const float*, const float*,
                         float*, float*, int, int, int, long long, int, float,
                         hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ---------------------------------------------------------------- attention

torch::Tensor attention(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                        bool causal, double scale) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(v, "v");
  TORCH_CHECK(q.dim() == 4, "q must be [B,H,S,D]");
  int B = q.size(0), Hq = q.size(1), Sq = q.size(2), D = q.size(3);
  int Hkv = k.size(1), Sk = k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  auto o = torch::empty_like(q);
  fa_fwd_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(), B, Hq,
              Hkv, Sq, Sk, D, (float)scale, causal ? 1 : 0, cur_stream());
  return o;
}

torch::Tensor paged_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                           c10::optional<torch::Tensor> block_table,
                           torch::Tensor seq_lens, int64_t block_size,
                           double scale) {
  check_bf16(q, "q");
  // KV cache: bf16, or OCP e4m3 fp8 (the vllm_low_latency FP8-serving role
  // — halves KV bytes per decoded token)
  const bool kv_fp8 = kc.scalar_type() == torch::kFloat8_e4m3fn;
  TORCH_CHECK(kv_fp8 || kc.scalar_type() == torch::kBFloat16,
              "k_cache must be bf16 or float8_e4m3fn");
  TORCH_CHECK(vc.scalar_type() == kc.scalar_type(), "kv cache dtype mismatch");
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous());
  int B = q.size(0), Hq = q.size(1), D = q.size(2);
  int Hkv, max_blocks = 0;
  const int* bt_ptr = nullptr;
  if (block_table.has_value()) {
    Hkv = kc.size(1);  // [blocks, Hkv, block_size, D]
    max_blocks = block_table->size(1);
    bt_ptr = block_table->data_ptr<int>();
  } else {
    Hkv = kc.size(1);  // [B, Hkv, S, D]; block_size = S
    block_size = kc.size(2);
  }
  auto o = torch::empty_like(q);
  // flash-decoding split: fill the chip when B*Hkv is small; cap splits by
  // the longest representable sequence so chunks stay >=256 kv rows
  long long s_bound = block_table.has_value()
      ? (long long)max_blocks * block_size : (long long)kc.size(2);
  int splits = 1;
  // fill the chip for latency-path decodes: allow chunks down to 128 kv
  // rows (B1/Hkv8/S4096 was 128 workgroups = half the CUs idle)
  while (B * Hkv * splits < 512 && (long long)splits * 128 < s_bound &&
         splits < 64)
    splits *= 2;
  torch::Tensor ws;
  float* ws_ptr = nullptr;
  if (splits > 1) {
    ws = torch::empty({(long long)B * Hq * splits * (D + 2)},
                      q.options().dtype(torch::kFloat32));
    ws_ptr = ws.data_ptr<float>();
  }
  paged_decode_bf16(q.data_ptr(), kc.data_ptr(), vc.data_ptr(), bt_ptr,
                    seq_lens.data_ptr<int>(), o.data_ptr(), ws_ptr, B, Hq,
                    Hkv, D, (int)block_size, max_blocks, splits, (float)scale,
                    kv_fp8 ? 1 : 0, cur_stream());
  return o;
}

// Stride-aware attention: q/k/v logically [B,H,S,D] with ANY strides
// (d contiguous); output is written [B,S,H,D]-contiguous so callers in
// sequence-major models (UNet/Llama/Whisper blocks) skip every transpose
// copy around the kernel.
torch::Tensor attention_bshd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                             bool causal, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "head_dim must be contiguous");
  int B = q.size(0), Hq = q.size(1), Sq = q.size(2), D = q.size(3);
  int Hkv = k.size(1), Sk = k.size(2);
  auto o = torch::empty({B, Sq, Hq, D}, q.options());
  long long st[12] = {
      q.stride(0), q.stride(1), q.stride(2),
      k.stride(0), k.stride(1), k.stride(2),
      v.stride(0), v.stride(1), v.stride(2),
      (long long)Sq * Hq * D, D, (long long)Hq * D,
  };
  fa_fwd_strided_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                      B, Hq, Hkv, Sq, Sk, D, (float)scale, causal ? 1 : 0, st,
                      cur_stream());
  return o;
}

// ---------------------------------------------------------------- norms

torch::Tensor groupnorm_silu(torch::Tensor x, torch::Tensor gamma,
                             torch::Tensor beta, int64_t groups, double eps,
                             bool do_silu) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  int N = x.size(0), C = x.size(1);
  long long HW = (long long)x.size(2) * x.size(3);
  auto y = torch::empty_like(x);
  auto ws = torch::zeros({N * groups * 2},
                         x.options().dtype(torch::kFloat32));
  groupnorm_silu_bf16(x.data_ptr(), y.data_ptr(), ws.data_ptr<float>(),
                      gamma.data_ptr<float>(), beta.data_ptr<float>(), N, C,
                      HW, (int)groups, (float)eps, do_silu ? 1 : 0,
                      cur_stream());
  return y;
}

torch::Tensor layernorm(torch::Tensor x, torch::Tensor gamma,
                        torch::Tensor beta, double eps) {
  check_bf16(x, "x");
  int D = x.size(-1);
  long long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  layernorm_bf16(x.data_ptr(), y.data_ptr(), gamma.data_ptr<float>(),
                 beta.data_ptr<float>(), rows, D, (float)eps, cur_stream());
  return y;
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor gamma, double eps) {
  check_bf16(x, "x");
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "rmsnorm needs D % 8 == 0");
  long long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  rmsnorm_bf16(x.data_ptr(), y.data_ptr(), gamma.data_ptr<float>(), rows, D,
               (float)eps, cur_stream());
  return y;
}

// ---------------------------------------------------------------- elementwise

torch::Tensor cfg_euler(torch::Tensor xt, torch::Tensor eps_c,
                        c10::optional<torch::Tensor> eps_u, double guidance,
                        double dsigma) {
  check_bf16(xt, "x_t");
  auto xn = torch::empty_like(xt);
  cfg_euler_bf16(xt.data_ptr(), eps_c.data_ptr(),
                 eps_u.has_value() ? eps_u->data_ptr() : nullptr,
                 xn.data_ptr(), (float)guidance, (float)dsigma, xt.numel(),
                 cur_stream());
  return xn;
}

torch::Tensor silu_mul(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a");
  auto y = torch::empty_like(a);
  silu_mul_bf16(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(),
                cur_stream());
  return y;
}

torch::Tensor geglu(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a");
  auto y = torch::empty_like(a);
  geglu_bf16(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(), cur_stream());
  return y;
}

// act(src[..., :I]) * src[..., I:] without slicing copies — src is the
// fused gate_up / GEGLU projection output [rows, 2I]
torch::Tensor glu_fused(torch::Tensor src, bool gelu) {
  check_bf16(src, "src");
  long long inner = src.size(-1) / 2;
  TORCH_CHECK(src.size(-1) % 2 == 0 && inner % 8 == 0,
              "inner dim must be even and 8-aligned");
  long long rows = src.numel() / (2 * inner);
  auto sizes = src.sizes().vec();
  sizes.back() = inner;
  auto y = torch::empty(sizes, src.options());
  glu_fused_bf16(src.data_ptr(), y.data_ptr(), rows, inner, gelu ? 1 : 0,
                 cur_stream());
  return y;
}

torch::Tensor add_residual(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a");
  auto y = torch::empty_like(a);
  add_bf16(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(), cur_stream());
  return y;
}

void rope_(torch::Tensor qk, torch::Tensor cosv, torch::Tensor sinv,
           c10::optional<torch::Tensor> positions) {
  TORCH_CHECK(qk.is_cuda() && qk.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(qk.dim() == 4 && qk.stride(3) == 1,
              "qk must be [B,H,S,D] with contiguous head_dim");
  long long B = qk.size(0);
  int H = qk.size(1), S = qk.size(2), D = qk.size(3);
  if (positions.has_value())
    TORCH_CHECK(positions->numel() == B * S || positions->numel() == S,
                "positions must be [B*S] (or [S] with B==1)");
  rope_bf16(qk.data_ptr(), cosv.data_ptr<float>(), sinv.data_ptr<float>(), B,
            H, S, D, qk.stride(0), qk.stride(1), qk.stride(2),
            positions.has_value() ? positions->data_ptr<int>() : nullptr,
            cur_stream());
}

void adamw_(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
            double lr, double beta1, double beta2, double eps, double wd,
            int64_t step) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous(), "p must be contiguous GPU");
  bool bf16 = p.scalar_type() == torch::kBFloat16;
  adamw_step(p.data_ptr(), g.data_ptr(), m.data_ptr<float>(),
             v.data_ptr<float>(), p.numel(), (float)lr, (float)beta1,
             (float)beta2, (float)eps, (float)wd, (int)step, bf16 ? 1 : 0,
             cur_stream());
}

// ---------------------------------------------------------------- conv (K3)

torch::Tensor conv3x3(torch::Tensor x, torch::Tensor wr, torch::Tensor bias,
                      c10::optional<torch::Tensor> residual, int64_t K,
                      bool upsample) {
  check_bf16(x, "x");
  check_bf16(wr, "wr");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  TORCH_CHECK(wr.dim() == 3 && wr.size(0) == 9,
              "wr must be [9, Kpad, C16] (repacked 3x3 weights)");
  TORCH_CHECK(bias.scalar_type() == torch::kFloat && bias.is_contiguous(),
              "bias must be fp32 contiguous");
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  if (upsample) { H *= 2; W *= 2; }
  int Kpad = wr.size(1), C16 = wr.size(2);
  TORCH_CHECK(K <= Kpad && C <= C16, "weight pack smaller than conv");
  auto out = torch::empty({N, K, H, W}, x.options());
  const void* resp = nullptr;
  if (residual.has_value()) {
    check_bf16(*residual, "residual");
    TORCH_CHECK(residual->sizes() == out.sizes(), "residual shape mismatch");
    resp = residual->data_ptr();
  }
  conv3x3_bf16(x.data_ptr(), wr.data_ptr(), bias.data_ptr(), resp,
               out.data_ptr(), nullptr, nullptr, N, C, H, W, (int)K, C16,
               Kpad, upsample ? 1 : 0, cur_stream());
  return out;
}

// GroupNorm+SiLU fused INTO the conv's staging read: one stats pass over x,
// per-(n,c) coefficients, then the conv applies silu(x*sc+sh) while staging.
torch::Tensor conv3x3_gn(torch::Tensor x, torch::Tensor wr, torch::Tensor bias,
                         c10::optional<torch::Tensor> residual, int64_t K,
                         bool upsample, torch::Tensor gamma,
                         torch::Tensor beta, int64_t groups, double eps) {
  check_bf16(x, "x");
  check_bf16(wr, "wr");
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long long HW = (long long)H * W;
  int Kpad = wr.size(1), C16 = wr.size(2);
  auto fopt = x.options().dtype(torch::kFloat32);
  auto ws = torch::zeros({(long long)N * groups * 2}, fopt);
  auto scale = torch::empty({(long long)N * C16}, fopt);
  auto shift = torch::empty({(long long)N * C16}, fopt);
  gn_conv_coeffs_bf16(x.data_ptr(), ws.data_ptr<float>(),
                      gamma.data_ptr<float>(), beta.data_ptr<float>(),
                      scale.data_ptr<float>(), shift.data_ptr<float>(), N, C,
                      C16, HW, (int)groups, (float)eps, cur_stream());
  int Ho = upsample ? H * 2 : H, Wo = upsample ? W * 2 : W;
  auto out = torch::empty({N, K, Ho, Wo}, x.options());
  const void* resp = nullptr;
  if (residual.has_value()) {
    check_bf16(*residual, "residual");
    resp = residual->data_ptr();
  }
  conv3x3_bf16(x.data_ptr(), wr.data_ptr(), bias.data_ptr(), resp,
               out.data_ptr(), scale.data_ptr<float>(),
               shift.data_ptr<float>(), N, C, Ho, Wo, (int)K, C16, Kpad,
               upsample ? 1 : 0, cur_stream());
  return out;
}

torch::Tensor sample_gumbel(torch::Tensor logits, double temperature,
                            int64_t seed) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kFloat32,
              "logits must be f32 GPU");
  int rows = logits.size(0), V = logits.size(1);
  auto out = torch::empty({rows}, logits.options().dtype(torch::kInt32));
  auto keys = torch::zeros({rows}, logits.options().dtype(torch::kInt64));
  gumbel_sample(logits.data_ptr<float>(),
                (unsigned long long*)keys.data_ptr<int64_t>(),
                out.data_ptr<int>(), rows, V, (float)temperature,
                (unsigned long long)seed, cur_stream());
  return out;
}

torch::Tensor softmax_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat32);
  int V = x.size(-1);
  long long rows = x.numel() / V;
  auto y = torch::empty_like(x);
  softmax_rows(x.data_ptr<float>(), y.data_ptr<float>(), (int)rows, V,
               cur_stream());
  return y;
}

// graph-capturable denoise-step pieces: sigma schedule + step counter live on
// device so one hipGraph capture serves every step.
void scale_in_dev(torch::Tensor x, torch::Tensor y, torch::Tensor sigmas,
                  torch::Tensor step) {
  scale_in_dev_bf16(x.data_ptr(), y.data_ptr(), sigmas.data_ptr<float>(),
                    (const long long*)step.data_ptr<int64_t>(), x.numel(), cur_stream());
}

void cfg_euler_dev(torch::Tensor xt, torch::Tensor eps_c,
                   c10::optional<torch::Tensor> eps_u, torch::Tensor xn,
                   torch::Tensor sigmas, torch::Tensor step, double guidance) {
  cfg_euler_dev_bf16(xt.data_ptr(), eps_c.data_ptr(),
                     eps_u.has_value() ? eps_u->data_ptr() : nullptr,
                     xn.data_ptr(), sigmas.data_ptr<float>(),
                     (const long long*)step.data_ptr<int64_t>(), (float)guidance, xt.numel(),
                     cur_stream());
}

void step_advance(torch::Tensor step) {
  advance_step((long long*)step.data_ptr<int64_t>(), cur_stream());
}

// ---------------------------------------------------------------- snapshot engine
// Pinned-host weight snapshots: one hipHostMalloc region per snapshot, copies
// overlapped across dedicated streams so restore saturates the PCIe Gen5 link
// (the p50-cold-start headline path, BASELINE.json).

struct SnapRegion {
  void* host = nullptr;
  size_t bytes = 0;
  std::vector<hipStream_t> streams;
  size_t rr = 0;
};

std::unordered_map<int64_t, SnapRegion> g_snaps;
int64_t g_next_snap = 1;

int64_t snap_create(int64_t bytes) {
  SnapRegion r;
  r.bytes = (size_t)bytes;
  TORCH_CHECK(hipHostMalloc(&r.host, r.bytes, hipHostMallocDefault) == hipSuccess,
              "hipHostMalloc failed for ", bytes, " bytes");
  r.streams.resize(4);
  for (auto& s : r.streams)
    TORCH_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking) == hipSuccess);
  int64_t id = g_next_snap++;
  g_snaps[id] = r;
  return id;
}

void snap_save(int64_t id, int64_t offset, torch::Tensor t) {
  auto& r = g_snaps.at(id);
  size_t n = t.numel() * t.element_size();
  TORCH_CHECK((size_t)offset + n <= r.bytes, "snapshot overflow");
  hipStream_t s = r.streams[r.rr++ % r.streams.size()];
  TORCH_CHECK(hipMemcpyAsync((char*)r.host + offset, t.data_ptr(), n,
                             hipMemcpyDeviceToHost, s) == hipSuccess);
}

void snap_restore(int64_t id, int64_t offset, torch::Tensor t) {
  auto& r = g_snaps.at(id);
  size_t n = t.numel() * t.element_size();
  TORCH_CHECK((size_t)offset + n <= r.bytes, "snapshot overflow");
  hipStream_t s = r.streams[r.rr++ % r.streams.size()];
  TORCH_CHECK(hipMemcpyAsync(t.data_ptr(), (char*)r.host + offset, n,
                             hipMemcpyHostToDevice, s) == hipSuccess);
}

void snap_sync(int64_t id) {
  auto& r = g_snaps.at(id);
  for (auto& s : r.streams) TORCH_CHECK(hipStreamSynchronize(s) == hipSuccess);
}

void snap_free(int64_t id) {
  auto it = g_snaps.find(id);
  if (it == g_snaps.end()) return;
  for (auto& s : it->second.streams) hipStreamDestroy(s);
  hipHostFree(it->second.host);
  g_snaps.erase(it);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("attention", &attention, "flash attention fwd bf16 (K1/K5/K7)");
  m.def("attention_bshd", &attention_bshd, "stride-aware attention, BSHD out");
  m.def("paged_decode", &paged_decode, "paged KV decode attention (K6)");
  m.def("groupnorm_silu", &groupnorm_silu);
  m.def("layernorm", &layernorm);
  m.def("rmsnorm", &rmsnorm);
  m.def("cfg_euler", &cfg_euler, "fused CFG + Euler step (K4)");
  m.def("silu_mul", &silu_mul);
  m.def("geglu", &geglu);
  m.def("glu_fused", &glu_fused, "in-place-sliced silu/gelu-mul over a fused [.,2I] projection");
  m.def("add_residual", &add_residual);
  m.def("rope_", &rope_, "in-place RoPE with host cos/sin tables");
  m.def("adamw_", &adamw_, "fused AdamW step (K9)");
  m.def("sample_gumbel", &sample_gumbel, "fused sampling (K8)");
  m.def("conv3x3", &conv3x3, "NCHW implicit-GEMM 3x3 conv, fused bias+res (K3)");
  m.def("conv3x3_gn", &conv3x3_gn, "GroupNorm+SiLU fused into the K3 conv");
  m.def("softmax_fwd", &softmax_fwd);
  m.def("tr16_probe", [](torch::Tensor out) {
    tr16_probe((short*)out.data_ptr(), cur_stream());
  }, "debug: ds_read_b64_tr_b16 lane/element semantics probe");
  m.def("scale_in_dev", &scale_in_dev, "x * 1/sqrt(sigma^2+1), device sigma");
  m.def("cfg_euler_dev", &cfg_euler_dev, "graph-capturable CFG+Euler step");
  m.def("step_advance", &step_advance, "increment device step counter");
  m.def("snap_create", &snap_create, "pinned-host snapshot region (K13)");
  m.def("snap_save", &snap_save);
  m.def("snap_restore", &snap_restore);
  m.def("snap_sync", &snap_sync);
  m.def("snap_free", &snap_free);
}
