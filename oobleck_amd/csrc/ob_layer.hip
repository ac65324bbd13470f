// ob_layer.hip — C-ABI layer objects for the Oobleck hot path on MI355X.
//
// One ob_layer = one fx-shard of the reference's model
// (/root/reference/oobleck/module/sharding.py:12-47): embedding, one
// transformer block, or ln_f+lm_head+loss.  Forward/backward orchestrate the
// hand-written kernels of ob_kernels.hip on the caller's hipStream_t,
// mirroring the semantics of /root/reference/oobleck/execution/layer.py
// (forward :144-145, backward :250-260 — grads ACCUMULATE into the bound
// flat grad buffer) and pipeline.py:169-244 (loss on the last layer).
//
// The activation stash is slot-indexed so several microbatches can be in
// flight, matching deepspeed-style pipe buffers (pipeline.py:556-562).
#include "ob_internal.h"

#include <cmath>
#include <cstring>

static inline hipStream_t S(void* s) { return reinterpret_cast<hipStream_t>(s); }
static inline int64_t i64min(int64_t a, int64_t b) { return a < b ? a : b; }
static inline int64_t i64max(int64_t a, int64_t b) { return a > b ? a : b; }

// ---------------------------------------------------------------------------
// shared backward workspace (single driving thread per GPU by ABI contract)
// ---------------------------------------------------------------------------

struct Workspace {
  float* dp = nullptr;     // [B*nh, S, S]
  float* bsh1 = nullptr;   // [B, S, H]
  float* bsh2 = nullptr;   // [B, S, H]
  float* dqkv = nullptr;   // [B, S, 3H]
  float* b4h = nullptr;    // [B, S, 4H]
  float* t1 = nullptr;     // bf16 [dim, BS] transposed operand (as floats)
  float* t2 = nullptr;
  float* s1 = nullptr;     // side-stream dW transpose buffers (bf16 as
  float* s2 = nullptr;     // floats): the dW family runs on g_side
  int64_t sz_dp = 0, sz_bsh = 0, sz_bsh2 = 0, sz_dqkv = 0, sz_b4h = 0,
          sz_t1 = 0, sz_t2 = 0, sz_s1 = 0, sz_s2 = 0;
};
static Workspace g_ws;

// side stream + events for overlapping the weight-grad family (dW
// transposes + atomic GEMMs + bias colsums: HBM-heavy, independent of
// the dX chain) under the main stream's dX/attention work.  All
// enqueued host-side in order, so event reuse across calls is safe.
struct SideSync {
  hipStream_t stream = nullptr;
  // FRESH event per record (ring): re-recording an event that still has
  // queued waiters serializes those waiters against the future record,
  // which compounds with queue depth (the pp1-overlap stall); a 128-deep
  // ring keeps every in-flight record a distinct object.
  hipEvent_t ring[128] = {};
  int ri = 0;
  bool ready = false;
};
static SideSync g_side;

static hipEvent_t side_evt() {
  hipEvent_t e = g_side.ring[g_side.ri & 127];
  g_side.ri++;
  return e;
}

static int side_init() {
  if (g_side.ready) return 0;
  OB_HIP(hipStreamCreateWithFlags(&g_side.stream, hipStreamNonBlocking));
  for (int i = 0; i < 128; i++)
    OB_HIP(hipEventCreateWithFlags(&g_side.ring[i], hipEventDisableTiming));
  g_side.ready = true;
  return 0;
}

// record a fresh event on `from` and make `to` wait it
#define OB_SIDE_FENCE(from, to)                      \
  do {                                               \
    hipEvent_t ev_ = side_evt();                     \
    OB_HIP(hipEventRecord(ev_, (from)));             \
    OB_HIP(hipStreamWaitEvent((to), ev_, 0));        \
  } while (0)

// ---------------------------------------------------------------------------
// in-step profiler (ob_internal.h declares the family ids + OB_PROF macro).
// Ring of reusable event pairs per family: a slot being reused is first
// drained (sync + accumulate), so memory stays bounded while totals cover
// every region.  Enabled only outside bench.py's timed region.
// ---------------------------------------------------------------------------

struct ProfFam {
  static constexpr int RING = 64;
  hipEvent_t beg[RING] = {}, end[RING] = {};
  bool pending[RING] = {};
  int head = 0;
  double total_ms = 0.0;
  long long count = 0;
};
static struct {
  bool on = false;
  ProfFam fam[OB_PF_NFAM];
} g_prof;

static void prof_drain_slot(ProfFam& f, int i) {
  if (!f.pending[i]) return;
  hipEventSynchronize(f.end[i]);
  float ms = 0.f;
  if (hipEventElapsedTime(&ms, f.beg[i], f.end[i]) == hipSuccess)
    f.total_ms += ms;
  f.pending[i] = false;
}

bool ob_prof_on() { return g_prof.on; }

int ob_prof_beg(int fam, hipStream_t s) {
  ProfFam& f = g_prof.fam[fam];
  const int i = f.head;
  if (!f.beg[i]) {
    hipEventCreate(&f.beg[i]);
    hipEventCreate(&f.end[i]);
  }
  prof_drain_slot(f, i);
  hipEventRecord(f.beg[i], s);
  return i;
}

void ob_prof_end(int fam, int slot, hipStream_t s) {
  ProfFam& f = g_prof.fam[fam];
  hipEventRecord(f.end[slot], s);
  f.pending[slot] = true;
  f.count++;
  f.head = (slot + 1) % ProfFam::RING;
}

extern "C" void ob_profile_enable(int on) { g_prof.on = on != 0; }

extern "C" void ob_profile_reset(void) {
  for (auto& f : g_prof.fam) {
    for (int i = 0; i < ProfFam::RING; i++)
      if (f.pending[i]) {
        hipEventSynchronize(f.end[i]);
        f.pending[i] = false;
      }
    f.total_ms = 0.0;
    f.count = 0;
    f.head = 0;
  }
}

extern "C" int ob_profile_read(int fam, double* total_ms, long long* count) {
  if (fam < 0 || fam >= OB_PF_NFAM) return ob_fail("profile_read: bad fam");
  ProfFam& f = g_prof.fam[fam];
  for (int i = 0; i < ProfFam::RING; i++) prof_drain_slot(f, i);
  *total_ms = f.total_ms;
  *count = f.count;
  return 0;
}

static int ws_ensure(float** buf, int64_t* cur, int64_t need) {
  if (need <= *cur) return 0;
  if (*buf) OB_HIP(hipFree(*buf));
  *buf = nullptr;
  *cur = 0;
  OB_HIP(hipMalloc(buf, need * sizeof(float)));
  *cur = need;
  return 0;
}

// ---------------------------------------------------------------------------
// layer object
// ---------------------------------------------------------------------------

struct ob_layer {
  ob_layer_desc d;
  int B;                    // current microbatch size (<= max_batch)
  float* params = nullptr;  // caller-owned flat fp32
  float* grads = nullptr;
  // stash (extension-owned), one contiguous allocation, slot-strided
  float* stash = nullptr;
  int64_t slot_stride = 0;  // floats per slot
  int64_t* ids = nullptr;   // EMBED: [n_slots, B*S]; FINAL: labels
  // per-kind offsets into stash (floats, within a slot)
  int64_t o_x = 0, o_mean1 = 0, o_rstd1 = 0, o_ln1 = 0, o_qkv = 0, o_p = 0,
          o_attnm = 0, o_hmid = 0, o_mean2 = 0, o_rstd2 = 0, o_ln2 = 0,
          o_u = 0, o_g = 0, o_logits = 0, o_lse = 0;
  // bf16 mode (d.dtype == 1): weight shadows in plain + transposed layouts
  // so every GEMM operand has k contiguous in memory (see
  // ob_kernels_bf16.hip header).  The stash keeps its fp32-sized offsets
  // and bf16 data occupies the first half of each region (2x overcommit,
  // traded for zero layout churn; memory is plentiful at 288 GB).
  __bf16* shadows = nullptr;
  int64_t v_pad = 0;  // vocab rounded up to 8 (16-byte bf16 alignment)
  int64_t sh_qkv = 0, sh_qkv_t = 0, sh_ap = 0, sh_ap_t = 0, sh_fc = 0,
          sh_fc_t = 0, sh_mp = 0, sh_mp_t = 0, sh_lm = 0, sh_lm_t = 0;
  int64_t shadow_count = 0;
  // fused flash-attention path (bf16, head_dim 64, S % 128 == 0): the
  // o_p slot holds only the per-row LSE ([Bm*nh, S] floats) instead of
  // the materialized [Bm*nh, S, S] P.
  bool flash = false;
};

// flash applies when head_dim == 64 and S % 128 == 0 (GPT-2 small and
// XL both qualify); OB_BF16_FLASH=0 falls back to the materialized-P
// path.  Decided at create time: it sizes the stash.
static bool flash_eligible(const ob_layer_desc* d) {
  static const bool off = [] {
    const char* e = getenv("OB_BF16_FLASH");
    return e && e[0] == '0';
  }();
  return !off && d->dtype == 1 && d->n_embd / d->n_head == 64 &&
         d->seq_len % 128 == 0;
}

// parameter offsets (canonical layout, oracle/gpt2_oracle.py::layer_param_spec)
struct BlockParams {
  int64_t ln1_w, ln1_b, w_qkv, b_qkv, w_attnproj, b_attnproj, ln2_w, ln2_b,
      w_fc, b_fc, w_mlpproj, b_mlpproj, total;
};
static BlockParams block_params(int64_t H) {
  BlockParams p;
  int64_t o = 0;
  p.ln1_w = o; o += H;
  p.ln1_b = o; o += H;
  p.w_qkv = o; o += H * 3 * H;
  p.b_qkv = o; o += 3 * H;
  p.w_attnproj = o; o += H * H;
  p.b_attnproj = o; o += H;
  p.ln2_w = o; o += H;
  p.ln2_b = o; o += H;
  p.w_fc = o; o += H * 4 * H;
  p.b_fc = o; o += 4 * H;
  p.w_mlpproj = o; o += 4 * H * H;
  p.b_mlpproj = o; o += H;
  p.total = o;
  return p;
}

extern "C" int64_t ob_layer_param_count(const ob_layer_desc* d) {
  const int64_t H = d->n_embd, V = d->vocab_size, P = d->n_positions;
  switch (d->kind) {
    case OB_KIND_EMBED: return V * H + P * H;
    case OB_KIND_BLOCK: return block_params(H).total;
    case OB_KIND_FINAL: return 2 * H + V * H;
  }
  return -1;
}

extern "C" int ob_layer_create(const ob_layer_desc* d, ob_layer_t* out) {
  if (!d || !out) return ob_fail("create: null arg");
  if (d->kind < 0 || d->kind > 2) return ob_fail("create: bad kind");
  if (d->seq_len > 2048)
    return ob_fail("create: seq_len > 2048 unsupported in round 1");
  ob_layer* l = new ob_layer();
  l->d = *d;
  l->B = d->max_batch;
  const int64_t Bm = d->max_batch, Sq = d->seq_len, H = d->n_embd,
                nh = d->n_head, V = d->vocab_size;
  const int64_t BS = Bm * Sq, BSH = BS * H;
  int64_t o = 0;
  switch (d->kind) {
    case OB_KIND_EMBED:
      l->slot_stride = 0;
      break;
    case OB_KIND_BLOCK:
      l->o_x = o; o += BSH;
      l->o_mean1 = o; o += BS;
      l->o_rstd1 = o; o += BS;
      l->o_ln1 = o; o += BSH;
      l->o_qkv = o; o += BS * 3 * H;
      l->flash = flash_eligible(d);
      l->o_p = o; o += l->flash ? Bm * nh * Sq : Bm * nh * Sq * Sq;
      l->o_attnm = o; o += BSH;
      l->o_hmid = o; o += BSH;
      l->o_mean2 = o; o += BS;
      l->o_rstd2 = o; o += BS;
      l->o_ln2 = o; o += BSH;
      l->o_u = o; o += BS * 4 * H;
      l->o_g = o; o += BS * 4 * H;
      l->slot_stride = o;
      break;
    case OB_KIND_FINAL:
      l->o_x = o; o += BSH;
      l->o_mean1 = o; o += BS;
      l->o_rstd1 = o; o += BS;
      l->o_ln1 = o; o += BSH;
      l->o_logits = o; o += BS * V;
      l->o_lse = o; o += BS;
      l->slot_stride = o;
      break;
  }
  if (l->slot_stride > 0) {
    if (hipMalloc(&l->stash, (size_t)l->slot_stride * d->n_slots *
                                 sizeof(float)) != hipSuccess) {
      delete l;
      return ob_fail("create: stash hipMalloc of %lld floats failed",
                     (long long)(l->slot_stride * d->n_slots));
    }
  }
  if (d->kind == OB_KIND_EMBED || d->kind == OB_KIND_FINAL) {
    if (hipMalloc(&l->ids, (size_t)BS * d->n_slots * sizeof(int64_t)) !=
        hipSuccess) {
      if (l->stash) hipFree(l->stash);
      delete l;
      return ob_fail("create: ids hipMalloc failed");
    }
  }
  // pad vocab to a 256 multiple: makes the lm_head GEMM family
  // (logits fwd, d_lnout, dW_lm via transposes) interior shapes for
  // the glds path AND the 256^2 8-phase kernel (gpt2's 50257 at 128
  // padding lands on 50304 % 256 != 0); shadow pad rows stay zero so
  // padded logits/grads are exactly zero
  l->v_pad = (V + 255) / 256 * 256;
  if (d->dtype == 1) {
    int64_t o2 = 0;
    if (d->kind == OB_KIND_BLOCK) {
      l->sh_qkv = o2; o2 += 3 * H * H;
      l->sh_qkv_t = o2; o2 += 3 * H * H;
      l->sh_ap = o2; o2 += H * H;
      l->sh_ap_t = o2; o2 += H * H;
      l->sh_fc = o2; o2 += 4 * H * H;
      l->sh_fc_t = o2; o2 += 4 * H * H;
      l->sh_mp = o2; o2 += 4 * H * H;
      l->sh_mp_t = o2; o2 += 4 * H * H;
    } else if (d->kind == OB_KIND_FINAL) {
      l->sh_lm = o2; o2 += l->v_pad * H;   // pad rows zeroed at alloc
      l->sh_lm_t = o2; o2 += H * l->v_pad;
    }
    l->shadow_count = o2;
    if (o2 > 0) {
      if (hipMalloc(&l->shadows, (size_t)o2 * sizeof(__bf16)) != hipSuccess)
        return ob_fail("create: shadow hipMalloc failed");
      OB_HIP(hipMemset(l->shadows, 0, (size_t)o2 * sizeof(__bf16)));
    }
  }
  // grow the shared backward workspace
  if (d->kind == OB_KIND_BLOCK) {
    if (!l->flash &&
        ws_ensure(&g_ws.dp, &g_ws.sz_dp, Bm * nh * Sq * Sq))
      return 1;
    if (ws_ensure(&g_ws.dqkv, &g_ws.sz_dqkv, BS * 3 * H)) return 1;
    if (ws_ensure(&g_ws.b4h, &g_ws.sz_b4h, BS * 4 * H)) return 1;
    if (d->dtype == 1) {
      if (ws_ensure(&g_ws.t1, &g_ws.sz_t1, (4 * H * BS + 1) / 2)) return 1;
      if (ws_ensure(&g_ws.t2, &g_ws.sz_t2, (4 * H * BS + 1) / 2)) return 1;
      if (ws_ensure(&g_ws.s1, &g_ws.sz_s1, (4 * H * BS + 1) / 2)) return 1;
      if (ws_ensure(&g_ws.s2, &g_ws.sz_s2, (4 * H * BS + 1) / 2)) return 1;
      if (side_init()) return 1;
    }
  }
  if (d->kind == OB_KIND_FINAL && d->dtype == 1) {
    if (ws_ensure(&g_ws.t1, &g_ws.sz_t1, (l->v_pad * BS + 1) / 2)) return 1;
    if (ws_ensure(&g_ws.t2, &g_ws.sz_t2, (4 * H * BS + 1) / 2)) return 1;
    if (ws_ensure(&g_ws.s1, &g_ws.sz_s1, (l->v_pad * BS + 1) / 2)) return 1;
    if (ws_ensure(&g_ws.s2, &g_ws.sz_s2, (4 * H * BS + 1) / 2)) return 1;
    if (side_init()) return 1;
  }
  if (d->kind != OB_KIND_EMBED) {
    const int64_t need = BSH;
    if (ws_ensure(&g_ws.bsh1, &g_ws.sz_bsh, need)) return 1;
    if (ws_ensure(&g_ws.bsh2, &g_ws.sz_bsh2, need)) return 1;
  }
  *out = l;
  return 0;
}

extern "C" int ob_layer_bind(ob_layer_t l, void* params, void* grads) {
  if (!l) return ob_fail("bind: null layer");
  l->params = (float*)params;
  l->grads = (float*)grads;
  return 0;
}

extern "C" int ob_layer_set_batch(ob_layer_t l, int32_t batch) {
  if (!l) return ob_fail("set_batch: null layer");
  if (batch < 1 || batch > l->d.max_batch)
    return ob_fail("set_batch: %d out of range (max %d)", batch,
                   l->d.max_batch);
  l->B = batch;
  return 0;
}

extern "C" int ob_layer_destroy(ob_layer_t l) {
  if (!l) return 0;
  if (l->stash) hipFree(l->stash);
  if (l->ids) hipFree(l->ids);
  if (l->shadows) hipFree(l->shadows);
  delete l;
  return 0;
}

extern "C" int ob_layer_refresh_weights(ob_layer_t l, void* stream) {
  if (!l) return ob_fail("refresh: null layer");
  if (l->d.dtype != 1) return 0;  // fp32 mode: nothing to do
  if (!l->params) return ob_fail("refresh: params not bound");
  const int64_t H = l->d.n_embd, V = l->d.vocab_size;
  const float* p = l->params;
  __bf16* sh = l->shadows;
  switch (l->d.kind) {
    case OB_KIND_EMBED:
      return 0;  // embedding reads the fp32 master directly
    case OB_KIND_BLOCK: {
      const BlockParams bp = block_params(H);
      if (ob_f32_to_bf16(p + bp.w_qkv, sh + l->sh_qkv, 3 * H * H, stream) ||
          ob_f32_to_bf16_t_ld(p + bp.w_qkv, sh + l->sh_qkv_t, H, 3 * H, H,
                              stream) ||
          ob_f32_to_bf16(p + bp.w_attnproj, sh + l->sh_ap, H * H, stream) ||
          ob_f32_to_bf16_t_ld(p + bp.w_attnproj, sh + l->sh_ap_t, H, H, H,
                              stream) ||
          ob_f32_to_bf16(p + bp.w_fc, sh + l->sh_fc, 4 * H * H, stream) ||
          ob_f32_to_bf16_t_ld(p + bp.w_fc, sh + l->sh_fc_t, H, 4 * H, H,
                              stream) ||
          ob_f32_to_bf16(p + bp.w_mlpproj, sh + l->sh_mp, 4 * H * H, stream) ||
          ob_f32_to_bf16_t_ld(p + bp.w_mlpproj, sh + l->sh_mp_t, 4 * H, H,
                              4 * H, stream))
        return 1;
      return 0;
    }
    case OB_KIND_FINAL:
      if (ob_f32_to_bf16(p + 2 * H, sh + l->sh_lm, V * H, stream) ||
          ob_f32_to_bf16_t_ld(p + 2 * H, sh + l->sh_lm_t, V, H, l->v_pad,
                              stream))
        return 1;
      return 0;
  }
  return ob_fail("refresh: bad kind");
}

// helper: plain strided-batch GEMM call (n1=batch, n2=1) or 2-level.
static int gemm(int tA, int tB, int64_t M, int64_t N, int64_t K, float alpha,
                const float* A, int64_t lda, int64_t sA1, int64_t sA2,
                const float* B, int64_t ldb, int64_t sB1, int64_t sB2,
                float beta, float* C, int64_t ldc, int64_t sC1, int64_t sC2,
                int64_t n1, int64_t n2, const float* bias, const float* R,
                int atomic, int splitk, void* stream) {
  // plain (unbatched, no fused epilogue) fp32 GEMMs to hipBLASLt —
  // covers the dW family too (it accumulates through beta=1)
  static const bool no_lt = [] {
    const char* e = getenv("OB_NO_BLASLT");
    return e && e[0] == '1';
  }();
  if (!no_lt && !bias && !R && !atomic && splitk <= 1 && n1 == 1 &&
      n2 == 1 && M * N >= 512 * 512) {
    const int r = ob_gemm_lt_f32(tA, tB, M, N, K, alpha, A, lda, B, ldb,
                                 beta, C, ldc, stream);
    if (r >= 0) return r;
  }
  return ob_gemm_f32(tA, tB, M, N, K, alpha, A, lda, sA1, sA2, B, ldb, sB1,
                     sB2, beta, C, ldc, sC1, sC2, n1, n2, bias, R, atomic,
                     splitk, stream);
}

// pick a split-K factor for weight-grad GEMMs so the grid fills the chip
// (256 CUs want >= ~512 blocks of 128x128).
static int pick_splitk(int64_t M, int64_t N, int64_t K) {
  const int64_t tiles = ((M + 127) / 128) * ((N + 127) / 128);
  int sk = 1;
  while (sk < 16 && tiles * sk < 512 && (K / (sk * 2)) >= 256) sk *= 2;
  return sk;
}

// narrow-N activation-grad GEMMs (dX = dY @ W^T with N = H): only
// M/128 x H/128 tiles -> the chip is underfilled and the long-K loop is
// latency-bound (measured 57 TF at K=50257 vs 105+ on wide shapes).
// Zero the output and split K across atomicAdd slices instead.
static int gemm_dx_splitk(int tB, int64_t M, int64_t N, int64_t K,
                          float alpha, const float* A, int64_t lda,
                          const float* B, int64_t ldb, float* C, int64_t ldc,
                          void* stream) {
  static const bool no_lt = [] {
    const char* e = getenv("OB_NO_BLASLT");
    return e && e[0] == '1';
  }();
  if (!no_lt) {
    const int r = ob_gemm_lt_f32(0, tB, M, N, K, alpha, A, lda, B, ldb, 0.f,
                                 C, ldc, stream);
    if (r >= 0) return r;
  }
  const int64_t tiles = ((M + 127) / 128) * ((N + 127) / 128);
  int sk = 1;
  while (sk < 16 && tiles * sk < 1024 && (K / (sk * 2)) >= 512) sk *= 2;
  if (sk == 1)
    return ob_gemm_f32(0, tB, M, N, K, alpha, A, lda, 0, 0, B, ldb, 0, 0,
                       0.f, C, ldc, 0, 0, 1, 1, nullptr, nullptr, 0, 1,
                       stream);
  OB_HIP(hipMemsetAsync(C, 0, (size_t)M * ldc * sizeof(float), S(stream)));
  return ob_gemm_f32(0, tB, M, N, K, alpha, A, lda, 0, 0, B, ldb, 0, 0, 0.f,
                     C, ldc, 0, 0, 1, 1, nullptr, nullptr, 1, sk, stream);
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

static int block_forward(ob_layer* l, int slot, const float* in, float* out,
                         void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, nh = l->d.n_head;
  const int64_t B = l->B, BS = B * Sq;
  const int64_t hd = H / nh;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  const BlockParams bp = block_params(H);

  float* x = st + l->o_x;
  OB_HIP(hipMemcpyAsync(x, in, BS * H * sizeof(float), hipMemcpyDeviceToDevice,
                        S(stream)));
  float* ln1 = st + l->o_ln1;
  if (ob_layernorm_fwd_f32(x, p + bp.ln1_w, p + bp.ln1_b, ln1,
                           st + l->o_mean1, st + l->o_rstd1, BS, H, 1e-5f,
                           stream))
    return 1;
  float* qkv = st + l->o_qkv;
  if (gemm(0, 0, BS, 3 * H, H, 1.f, ln1, H, 0, 0, p + bp.w_qkv, 3 * H, 0, 0,
           0.f, qkv, 3 * H, 0, 0, 1, 1, p + bp.b_qkv, nullptr, 0, 1, stream))
    return 1;
  // scores: P[b,h] = Q @ K^T ; Q/K are strided slices of qkv
  float* P = st + l->o_p;
  if (gemm(0, 1, Sq, Sq, hd, 1.f,
           qkv, 3 * H, Sq * 3 * H, hd,            // Q slice
           qkv + H, 3 * H, Sq * 3 * H, hd,        // K slice (stored [S,hd])
           0.f, P, Sq, nh * Sq * Sq, Sq * Sq, B, nh, nullptr, nullptr, 0, 1,
           stream))
    return 1;
  if (ob_softmax_causal_fwd_f32(P, B * nh, Sq, 1.f / sqrtf((float)hd), stream))
    return 1;
  // attn_merged[b,:,h*hd:(h+1)*hd] = P[b,h] @ V[b,h]
  float* am = st + l->o_attnm;
  if (gemm(0, 0, Sq, hd, Sq, 1.f, P, Sq, nh * Sq * Sq, Sq * Sq, qkv + 2 * H,
           3 * H, Sq * 3 * H, hd, 0.f, am, H, Sq * H, hd, B, nh, nullptr,
           nullptr, 0, 1, stream))
    return 1;
  // h_mid = attn_merged @ w_attnproj + b + x (residual)
  float* hmid = st + l->o_hmid;
  if (gemm(0, 0, BS, H, H, 1.f, am, H, 0, 0, p + bp.w_attnproj, H, 0, 0, 0.f,
           hmid, H, 0, 0, 1, 1, p + bp.b_attnproj, x, 0, 1, stream))
    return 1;
  float* ln2 = st + l->o_ln2;
  if (ob_layernorm_fwd_f32(hmid, p + bp.ln2_w, p + bp.ln2_b, ln2,
                           st + l->o_mean2, st + l->o_rstd2, BS, H, 1e-5f,
                           stream))
    return 1;
  float* u = st + l->o_u;
  if (gemm(0, 0, BS, 4 * H, H, 1.f, ln2, H, 0, 0, p + bp.w_fc, 4 * H, 0, 0,
           0.f, u, 4 * H, 0, 0, 1, 1, p + bp.b_fc, nullptr, 0, 1, stream))
    return 1;
  float* gact = st + l->o_g;
  if (ob_gelu_fwd_f32(u, gact, BS * 4 * H, stream)) return 1;
  if (gemm(0, 0, BS, H, 4 * H, 1.f, gact, 4 * H, 0, 0, p + bp.w_mlpproj, H, 0,
           0, 0.f, out, H, 0, 0, 1, 1, p + bp.b_mlpproj, hmid, 0, 1, stream))
    return 1;
  return 0;
}

static int final_forward(ob_layer* l, int slot, const float* in, float* out,
                         const int64_t* labels, void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
  const int64_t B = l->B, BS = B * Sq;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  if (!labels) return ob_fail("final_forward: labels required");

  float* x = st + l->o_x;
  OB_HIP(hipMemcpyAsync(x, in, BS * H * sizeof(float), hipMemcpyDeviceToDevice,
                        S(stream)));
  int64_t* labs = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;
  OB_HIP(hipMemcpyAsync(labs, labels, BS * sizeof(int64_t),
                        hipMemcpyDeviceToDevice, S(stream)));
  float* lnf = st + l->o_ln1;
  if (ob_layernorm_fwd_f32(x, p + 0, p + H, lnf, st + l->o_mean1,
                           st + l->o_rstd1, BS, H, 1e-5f, stream))
    return 1;
  float* logits = st + l->o_logits;
  // logits = ln_out @ w_lm^T   (w_lm stored [V,H])
  if (gemm(0, 1, BS, V, H, 1.f, lnf, H, 0, 0, p + 2 * H, H, 0, 0, 0.f, logits,
           V, 0, 0, 1, 1, nullptr, nullptr, 0, 1, stream))
    return 1;
  OB_HIP(hipMemsetAsync(out, 0, sizeof(float), S(stream)));
  if (ob_ce_fwd_f32(logits, labs, st + l->o_lse, out, B, Sq, V, stream))
    return 1;
  return 0;
}

// ---------------------------------------------------------------------------
// bf16 mixed-precision path (d.dtype == 1): bf16 activations + bf16 MFMA
// GEMMs with fp32 accumulate, fp32 master weights/biases/LN params, fp32
// grads.  Weight operands come from the plain/transposed shadows so every
// GEMM stages its operands with k contiguous (see ob_kernels_bf16.hip).
// ---------------------------------------------------------------------------

static int gemm_bf(int tA, int tB, int64_t M, int64_t N, int64_t K,
                   float alpha, const void* A, int64_t lda, int64_t sA1,
                   int64_t sA2, const void* B, int64_t ldb, int64_t sB1,
                   int64_t sB2, void* C, int64_t ldc, int64_t sC1,
                   int64_t sC2, int64_t n1, int64_t n2, const float* bias,
                   const void* R, int out_kind, int splitk, void* stream) {
  return ob_gemm_bf16(tA, tB, M, N, K, alpha, A, lda, sA1, sA2, B, ldb, sB1,
                      sB2, 0.f, C, ldc, sC1, sC2, n1, n2, bias, R, out_kind,
                      splitk, stream);
}

// weight-grad GEMM for bf16: materialize X^T / dY^T (cheap HBM transpose)
// so the GEMM runs on the fast NT glds path instead of the
// transpose-staged TN case (~150 TF measured there).
static int dw_bf16_ws(const __bf16* act, int64_t actw, const __bf16* dY,
                      int64_t dyw, int64_t BS, float* gout, int64_t ldc,
                      float* w1, float* w2, void* stream) {
  // direct TN via hipBLASLt with beta=1 accumulation into the fp32 flat
  // grads: no transpose materialization, no atomics (measured 694 TF on
  // the fc dW shape vs 436 for transpose + atomic NT)
  static const bool no_lt = [] {
    const char* e = getenv("OB_NO_BLASLT");
    return e && e[0] == '1';
  }();
  if (!no_lt && ldc == dyw) {
    const int r = ob_gemm_lt(1, 0, actw, dyw, BS, 1.f, act, actw, dY, dyw,
                             1.f, gout, ldc, 1, stream);
    if (r >= 0) return r;
  }
  if ((actw % 128) || (dyw % 128) || (BS % 128))
    return gemm_bf(1, 0, actw, dyw, BS, 1.f, act, actw, 0, 0, dY, dyw, 0, 0,
                   gout, ldc, 0, 0, 1, 1, nullptr, nullptr, 2,
                   pick_splitk(actw, dyw, BS), stream);
  __bf16* XT = (__bf16*)w1;
  __bf16* DYT = (__bf16*)w2;
  if (ob_transpose_bf16(act, XT, BS, actw, stream)) return 1;
  if (ob_transpose_bf16(dY, DYT, BS, dyw, stream)) return 1;
  return gemm_bf(0, 1, actw, dyw, BS, 1.f, XT, BS, 0, 0, DYT, BS, 0, 0, gout,
                 ldc, 0, 0, 1, 1, nullptr, nullptr, 2,
                 pick_splitk(actw, dyw, BS), stream);
}

static bool use_flash(const ob_layer* l) { return l->flash; }

static int block_forward_bf16(ob_layer* l, int slot, const __bf16* in,
                              __bf16* out, void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, nh = l->d.n_head;
  const int64_t B = l->B, BS = B * Sq;
  const int64_t hd = H / nh;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  const __bf16* sh = l->shadows;
  const BlockParams bp = block_params(H);

  __bf16* x = (__bf16*)(st + l->o_x);
  OB_HIP(hipMemcpyAsync(x, in, BS * H * sizeof(__bf16),
                        hipMemcpyDeviceToDevice, S(stream)));
  __bf16* ln1 = (__bf16*)(st + l->o_ln1);
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_fwd_bf16(x, p + bp.ln1_w, p + bp.ln1_b, ln1,
                                    st + l->o_mean1, st + l->o_rstd1, BS, H,
                                    1e-5f, stream)))
    return 1;
  __bf16* qkv = (__bf16*)(st + l->o_qkv);
  if (OB_PROF(OB_PF_GEMM_FWD, stream,
              gemm_bf(0, 1, BS, 3 * H, H, 1.f, ln1, H, 0, 0,
                      sh + l->sh_qkv_t, H, 0, 0, qkv, 3 * H, 0, 0, 1, 1,
                      p + bp.b_qkv, nullptr, 0, 1, stream)))
    return 1;
  __bf16* am = (__bf16*)(st + l->o_attnm);
  if (use_flash(l)) {
    // V^T materialization (workspace, transient within this call), then
    // the fused kernel writes O and the per-row LSE (stored where the
    // materialized-P path keeps P).
    __bf16* VTw = (__bf16*)g_ws.t2;
    if (OB_PROF(OB_PF_FLASH_FWD, stream,
                ob_transpose_bf16_b(qkv + 2 * H, VTw, Sq, 64, Sq * 3 * H, 64,
                                    3 * H, B, nh, stream)))
      return 1;
    float* lseP = st + l->o_p;
    if (OB_PROF(OB_PF_FLASH_FWD, stream,
                ob_flash_fwd_bf16(qkv, VTw, am, lseP, B, Sq, H, nh,
                                  1.f / sqrtf((float)hd), stream)))
      return 1;
  } else {
    __bf16* P = (__bf16*)(st + l->o_p);
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(0, 1, Sq, Sq, hd, 1.f, qkv, 3 * H, Sq * 3 * H, hd,
                        qkv + H, 3 * H, Sq * 3 * H, hd, P, Sq, nh * Sq * Sq,
                        Sq * Sq, B, nh, nullptr, nullptr, 0, 1, stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                ob_softmax_causal_fwd_bf16(P, B * nh, Sq,
                                           1.f / sqrtf((float)hd), stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(0, 0, Sq, hd, Sq, 1.f, P, Sq, nh * Sq * Sq, Sq * Sq,
                        qkv + 2 * H, 3 * H, Sq * 3 * H, hd, am, H, Sq * H,
                        hd, B, nh, nullptr, nullptr, 0, 1, stream)))
      return 1;
  }
  __bf16* hmid = (__bf16*)(st + l->o_hmid);
  if (OB_PROF(OB_PF_GEMM_FWD, stream,
              gemm_bf(0, 1, BS, H, H, 1.f, am, H, 0, 0, sh + l->sh_ap_t, H,
                      0, 0, hmid, H, 0, 0, 1, 1, p + bp.b_attnproj, x, 0, 1,
                      stream)))
    return 1;
  __bf16* ln2 = (__bf16*)(st + l->o_ln2);
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_fwd_bf16(hmid, p + bp.ln2_w, p + bp.ln2_b, ln2,
                                    st + l->o_mean2, st + l->o_rstd2, BS, H,
                                    1e-5f, stream)))
    return 1;
  __bf16* u = (__bf16*)(st + l->o_u);
  if (OB_PROF(OB_PF_FC_FWD, stream,
              gemm_bf(0, 1, BS, 4 * H, H, 1.f, ln2, H, 0, 0, sh + l->sh_fc_t,
                      H, 0, 0, u, 4 * H, 0, 0, 1, 1, p + bp.b_fc, nullptr, 0,
                      1, stream)))
    return 1;
  __bf16* gact = (__bf16*)(st + l->o_g);
  if (OB_PROF(OB_PF_ELEM, stream,
              ob_gelu_fwd_bf16(u, gact, BS * 4 * H, stream)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_FWD, stream,
              gemm_bf(0, 1, BS, H, 4 * H, 1.f, gact, 4 * H, 0, 0,
                      sh + l->sh_mp_t, 4 * H, 0, 0, out, H, 0, 0, 1, 1,
                      p + bp.b_mlpproj, hmid, 0, 1, stream)))
    return 1;
  return 0;
}

static int final_forward_bf16(ob_layer* l, int slot, const __bf16* in,
                              float* out, const int64_t* labels,
                              void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
  const int64_t B = l->B, BS = B * Sq;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  if (!labels) return ob_fail("final_forward_bf16: labels required");

  __bf16* x = (__bf16*)(st + l->o_x);
  OB_HIP(hipMemcpyAsync(x, in, BS * H * sizeof(__bf16),
                        hipMemcpyDeviceToDevice, S(stream)));
  int64_t* labs = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;
  OB_HIP(hipMemcpyAsync(labs, labels, BS * sizeof(int64_t),
                        hipMemcpyDeviceToDevice, S(stream)));
  __bf16* lnf = (__bf16*)(st + l->o_ln1);
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_fwd_bf16(x, p + 0, p + H, lnf, st + l->o_mean1,
                                    st + l->o_rstd1, BS, H, 1e-5f, stream)))
    return 1;
  __bf16* logits = (__bf16*)(st + l->o_logits);
  // N = v_pad (zero-padded shadow rows -> padded logits are exactly 0)
  if (OB_PROF(OB_PF_GEMM_FWD, stream,
              gemm_bf(0, 1, BS, l->v_pad, H, 1.f, lnf, H, 0, 0,
                      l->shadows + l->sh_lm, H, 0, 0, logits, l->v_pad, 0, 0,
                      1, 1, nullptr, nullptr, 0, 1, stream)))
    return 1;
  OB_HIP(hipMemsetAsync(out, 0, sizeof(float), S(stream)));
  if (OB_PROF(OB_PF_CE, stream,
              ob_ce_fwd_bf16(logits, labs, st + l->o_lse, out, B, Sq, V,
                             l->v_pad, stream)))
    return 1;
  return 0;
}

static int block_backward_bf16(ob_layer* l, int slot, const __bf16* dout,
                               __bf16* din, void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, nh = l->d.n_head;
  const int64_t B = l->B, BS = B * Sq;
  const int64_t hd = H / nh;
  const float scale = 1.f / sqrtf((float)hd);
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  float* g = l->grads;
  const __bf16* sh = l->shadows;
  const BlockParams bp = block_params(H);

  __bf16* x = (__bf16*)(st + l->o_x);
  __bf16* ln1 = (__bf16*)(st + l->o_ln1);
  __bf16* qkv = (__bf16*)(st + l->o_qkv);
  __bf16* P = (__bf16*)(st + l->o_p);
  __bf16* am = (__bf16*)(st + l->o_attnm);
  __bf16* hmid = (__bf16*)(st + l->o_hmid);
  __bf16* ln2 = (__bf16*)(st + l->o_ln2);
  __bf16* u = (__bf16*)(st + l->o_u);
  __bf16* gact = (__bf16*)(st + l->o_g);

  __bf16* DY4 = (__bf16*)g_ws.b4h;
  __bf16* DLN = (__bf16*)g_ws.bsh1;
  __bf16* DATT = (__bf16*)g_ws.bsh2;
  __bf16* DQKV = (__bf16*)g_ws.dqkv;
  __bf16* DP = (__bf16*)g_ws.dp;

  // The dW family (transposes + atomic GEMMs + bias colsums) has no
  // consumer inside this call and no effect on the dX chain: it runs on
  // g_side, overlapped under the dX/attention work, fenced by events at
  // the producer edges and JOINED at the end of the call (the g_ws
  // buffers it reads are reused by the next microbatch's backward).
  void* const side = (void*)g_side.stream;
  OB_SIDE_FENCE(S(stream), g_side.stream);  // entry state (dout ready)

  OB_HIP(hipMemcpyAsync(din, dout, BS * H * sizeof(__bf16),
                        hipMemcpyDeviceToDevice, S(stream)));
  // ---- MLP ----
  if (OB_PROF(OB_PF_GEMM_DX, stream,
              gemm_bf(0, 1, BS, 4 * H, H, 1.f, dout, H, 0, 0, sh + l->sh_mp,
                      H, 0, 0, DY4, 4 * H, 0, 0, 1, 1, nullptr, nullptr, 0,
                      1, stream)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_DW, side,
              dw_bf16_ws(gact, 4 * H, dout, H, BS, g + bp.w_mlpproj, H,
                         g_ws.s1, g_ws.s2, side)))
    return 1;
  if (OB_PROF(OB_PF_ELEM, side,
              ob_colsum_bf16(dout, g + bp.b_mlpproj, BS, H, side)))
    return 1;
  if (OB_PROF(OB_PF_ELEM, stream,
              ob_gelu_bwd_bf16(u, DY4, DY4, BS * 4 * H, stream)))
    return 1;
  OB_SIDE_FENCE(S(stream), g_side.stream);  // DY4 post-gelu
  if (OB_PROF(OB_PF_GEMM_DW, side,
              dw_bf16_ws(ln2, H, DY4, 4 * H, BS, g + bp.w_fc, 4 * H, g_ws.s1,
                         g_ws.s2, side)))
    return 1;
  if (OB_PROF(OB_PF_ELEM, side,
              ob_colsum_bf16(DY4, g + bp.b_fc, BS, 4 * H, side)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_DX, stream,
              gemm_bf(0, 1, BS, H, 4 * H, 1.f, DY4, 4 * H, 0, 0,
                      sh + l->sh_fc, 4 * H, 0, 0, DLN, H, 0, 0, 1, 1,
                      nullptr, nullptr, 0, 1, stream)))
    return 1;
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_bwd_bf16(hmid, p + bp.ln2_w, st + l->o_mean2,
                                    st + l->o_rstd2, DLN, din, g + bp.ln2_w,
                                    g + bp.ln2_b, BS, H, 1, stream)))
    return 1;
  // ---- attention projection ----
  OB_SIDE_FENCE(S(stream), g_side.stream);  // din post-ln2-bwd
  if (OB_PROF(OB_PF_GEMM_DX, stream,
              gemm_bf(0, 1, BS, H, H, 1.f, din, H, 0, 0, sh + l->sh_ap, H, 0,
                      0, DATT, H, 0, 0, 1, 1, nullptr, nullptr, 0, 1,
                      stream)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_DW, side,
              dw_bf16_ws(am, H, din, H, BS, g + bp.w_attnproj, H, g_ws.s1,
                         g_ws.s2, side)))
    return 1;
  if (OB_PROF(OB_PF_ELEM, side,
              ob_colsum_bf16(din, g + bp.b_attnproj, BS, H, side)))
    return 1;
  // din is re-updated by the final ln1 backward: the main stream must
  // not reach it before the side finished reading din
  hipEvent_t ev_din_done = side_evt();
  OB_HIP(hipEventRecord(ev_din_done, g_side.stream));
  // ---- attention core ----
  if (use_flash(l)) {
    const int64_t BSH = BS * H;
    __bf16* QTw = (__bf16*)g_ws.t1;
    __bf16* KTw = QTw + BSH;
    __bf16* dOTw = KTw + BSH;
    float* Dbuf = g_ws.t1 + (3 * BSH + 1) / 2;
    const float* lseP = st + l->o_p;
    if (OB_PROF(OB_PF_FLASH_BWD, stream,
                ob_transpose_bf16_b(qkv, QTw, Sq, 64, Sq * 3 * H, 64, 3 * H,
                                    B, nh, stream)))
      return 1;
    if (OB_PROF(OB_PF_FLASH_BWD, stream,
                ob_transpose_bf16_b(qkv + H, KTw, Sq, 64, Sq * 3 * H, 64,
                                    3 * H, B, nh, stream)))
      return 1;
    if (OB_PROF(OB_PF_FLASH_BWD, stream,
                ob_transpose_bf16_b(DATT, dOTw, Sq, 64, Sq * H, 64, H, B, nh,
                                    stream)))
      return 1;
    if (OB_PROF(OB_PF_FLASH_BWD, stream,
                ob_flash_dsum_bf16(am, DATT, Dbuf, B, Sq, H, nh, stream)))
      return 1;
    if (OB_PROF(OB_PF_FLASH_BWD, stream,
                ob_flash_bwd_bf16(qkv, QTw, KTw, dOTw, DATT, lseP, Dbuf,
                                  DQKV, B, Sq, H, nh, scale, stream)))
      return 1;
  } else {
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(0, 1, Sq, Sq, hd, 1.f, DATT, H, Sq * H, hd,
                        qkv + 2 * H, 3 * H, Sq * 3 * H, hd, DP, Sq,
                        nh * Sq * Sq, Sq * Sq, B, nh, nullptr, nullptr, 0, 1,
                        stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                ob_softmax_causal_bwd_bf16(P, DP, B * nh, Sq, stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(0, 0, Sq, hd, Sq, scale, DP, Sq, nh * Sq * Sq,
                        Sq * Sq, qkv + H, 3 * H, Sq * 3 * H, hd, DQKV, 3 * H,
                        Sq * 3 * H, hd, B, nh, nullptr, nullptr, 0, 1,
                        stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(1, 0, Sq, hd, Sq, scale, DP, Sq, nh * Sq * Sq,
                        Sq * Sq, qkv, 3 * H, Sq * 3 * H, hd, DQKV + H, 3 * H,
                        Sq * 3 * H, hd, B, nh, nullptr, nullptr, 0, 1,
                        stream)))
      return 1;
    if (OB_PROF(OB_PF_ATTN_MAT, stream,
                gemm_bf(1, 0, Sq, hd, Sq, 1.f, P, Sq, nh * Sq * Sq, Sq * Sq,
                        DATT, H, Sq * H, hd, DQKV + 2 * H, 3 * H, Sq * 3 * H,
                        hd, B, nh, nullptr, nullptr, 0, 1, stream)))
      return 1;
  }
  // ---- QKV projection ----
  OB_SIDE_FENCE(S(stream), g_side.stream);  // DQKV ready
  if (OB_PROF(OB_PF_ELEM, side,
              ob_colsum_bf16(DQKV, g + bp.b_qkv, BS, 3 * H, side)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_DW, side,
              dw_bf16_ws(ln1, H, DQKV, 3 * H, BS, g + bp.w_qkv, 3 * H,
                         g_ws.s1, g_ws.s2, side)))
    return 1;
  if (OB_PROF(OB_PF_GEMM_DX, stream,
              gemm_bf(0, 1, BS, H, 3 * H, 1.f, DQKV, 3 * H, 0, 0,
                      sh + l->sh_qkv, 3 * H, 0, 0, DLN, H, 0, 0, 1, 1,
                      nullptr, nullptr, 0, 1, stream)))
    return 1;
  // ln1 backward accumulates into din: wait for the side's din readers
  OB_HIP(hipStreamWaitEvent(S(stream), ev_din_done, 0));
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_bwd_bf16(x, p + bp.ln1_w, st + l->o_mean1,
                                    st + l->o_rstd1, DLN, din, g + bp.ln1_w,
                                    g + bp.ln1_b, BS, H, 1, stream)))
    return 1;
  // join: the next call reuses DY4/DQKV/din workspaces on the main stream
  OB_SIDE_FENCE(g_side.stream, S(stream));  // join
  return 0;
}

static int final_backward_bf16(ob_layer* l, int slot, const float* dout,
                               __bf16* din, void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
  const int64_t B = l->B, BS = B * Sq;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  float* g = l->grads;
  __bf16* x = (__bf16*)(st + l->o_x);
  __bf16* lnf = (__bf16*)(st + l->o_ln1);
  __bf16* logits = (__bf16*)(st + l->o_logits);
  int64_t* labs = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;
  __bf16* DLN = (__bf16*)g_ws.bsh1;

  if (OB_PROF(OB_PF_CE, stream,
              ob_ce_bwd_bf16(logits, labs, st + l->o_lse, dout, B, Sq, V,
                             l->v_pad, stream)))
    return 1;
  // dW_lm on the side stream (overlaps d_lnout + ln backward below):
  // transpose dlogits and lnf so the GEMM runs on the glds path; tile
  // over v_pad rows, store-guard at V (pad rows of dlogits are 0 anyway,
  // but they have no grad slot).
  {
    OB_SIDE_FENCE(S(stream), g_side.stream);  // dlogits ready
    void* const side = (void*)g_side.stream;
    // direct TN via hipBLASLt: M = V with lda = v_pad skips the pad
    // rows entirely (no store guard, no 826 MB logits transpose)
    static const bool no_lt = [] {
      const char* e = getenv("OB_NO_BLASLT");
      return e && e[0] == '1';
    }();
    int r = -1;
    if (!no_lt)
      r = OB_PROF(OB_PF_GEMM_DW, side,
                  ob_gemm_lt(1, 0, V, H, BS, 1.f, logits, l->v_pad, lnf, H,
                             1.f, g + 2 * H, H, 1, side));
    if (r > 0) return 1;
    if (r < 0) {
      __bf16* DLT = (__bf16*)g_ws.s1;
      __bf16* LNT = (__bf16*)g_ws.s2;
      if (ob_transpose_bf16(logits, DLT, BS, l->v_pad, side)) return 1;
      if (ob_transpose_bf16(lnf, LNT, BS, H, side)) return 1;
      if (OB_PROF(OB_PF_GEMM_DW, side,
                  ob_gemm_bf16_nt_dispatch(DLT, LNT, g + 2 * H, nullptr,
                                           nullptr, l->v_pad, H, BS, BS, BS,
                                           H, 0, 0, 0, 0, 0, 0, 1, 1, 1.f,
                                           0.f, 2, 2, side, V)))
        return 1;
    }
  }
  // d_lnout: K = v_pad (padded dlogits cols and shadow^T cols are zero)
  if (OB_PROF(OB_PF_GEMM_DX, stream,
              gemm_bf(0, 1, BS, H, l->v_pad, 1.f, logits, l->v_pad, 0, 0,
                      l->shadows + l->sh_lm_t, l->v_pad, 0, 0, DLN, H, 0, 0,
                      1, 1, nullptr, nullptr, 0, 1, stream)))
    return 1;
  if (OB_PROF(OB_PF_LN, stream,
              ob_layernorm_bwd_bf16(x, p + 0, st + l->o_mean1,
                                    st + l->o_rstd1, DLN, din, g + 0, g + H,
                                    BS, H, 0, stream)))
    return 1;
  // join: the next backward reuses s1/s2 and reads g on the main stream
  OB_SIDE_FENCE(g_side.stream, S(stream));  // join
  return 0;
}

extern "C" int ob_layer_forward(ob_layer_t l, int32_t slot, const void* in,
                                void* out, const int64_t* labels,
                                void* stream) {
  if (!l) return ob_fail("forward: null layer");
  if (!l->params) return ob_fail("forward: params not bound");
  if (slot < 0 || slot >= l->d.n_slots) return ob_fail("forward: bad slot");
  switch (l->d.kind) {
    case OB_KIND_EMBED: {
      const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
      const int64_t BS = (int64_t)l->B * Sq;
      int64_t* ids = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;
      OB_HIP(hipMemcpyAsync(ids, in, BS * sizeof(int64_t),
                            hipMemcpyDeviceToDevice, S(stream)));
      if (l->d.dtype == 1)
        return OB_PROF(OB_PF_ELEM, stream,
                       ob_embed_fwd_bf16(ids, l->params, l->params + V * H,
                                         out, l->B, Sq, H, stream));
      return OB_PROF(OB_PF_ELEM, stream,
                     ob_embed_fwd_f32(ids, l->params, l->params + V * H,
                                      (float*)out, l->B, Sq, H, stream));
    }
    case OB_KIND_BLOCK:
      if (l->d.dtype == 1)
        return block_forward_bf16(l, slot, (const __bf16*)in, (__bf16*)out,
                                  stream);
      return block_forward(l, slot, (const float*)in, (float*)out, stream);
    case OB_KIND_FINAL:
      if (l->d.dtype == 1)
        return final_forward_bf16(l, slot, (const __bf16*)in, (float*)out,
                                  labels, stream);
      return final_forward(l, slot, (const float*)in, (float*)out, labels,
                           stream);
  }
  return ob_fail("forward: bad kind");
}

// ---------------------------------------------------------------------------
// backward (accumulates into the bound grad buffer)
// ---------------------------------------------------------------------------

static int block_backward(ob_layer* l, int slot, const float* dout, float* din,
                          void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, nh = l->d.n_head;
  const int64_t B = l->B, BS = B * Sq;
  const int64_t hd = H / nh;
  const float scale = 1.f / sqrtf((float)hd);
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  float* g = l->grads;
  const BlockParams bp = block_params(H);

  float* x = st + l->o_x;
  float* ln1 = st + l->o_ln1;
  float* qkv = st + l->o_qkv;
  float* P = st + l->o_p;
  float* am = st + l->o_attnm;
  float* hmid = st + l->o_hmid;
  float* ln2 = st + l->o_ln2;
  float* u = st + l->o_u;
  float* gact = st + l->o_g;

  float* DY4 = g_ws.b4h;    // [BS,4H]
  float* DLN = g_ws.bsh1;   // [BS,H]
  float* DATT = g_ws.bsh2;  // [BS,H]
  float* DQKV = g_ws.dqkv;  // [BS,3H]
  float* DP = g_ws.dp;      // [B*nh,S,S]

  // din starts as d(h_mid) accumulator = dout (residual skip)
  OB_HIP(hipMemcpyAsync(din, dout, BS * H * sizeof(float),
                        hipMemcpyDeviceToDevice, S(stream)));
  // ---- MLP backward ----
  // dg = dout @ w_mlpproj^T
  if (gemm(0, 1, BS, 4 * H, H, 1.f, dout, H, 0, 0, p + bp.w_mlpproj, H, 0, 0,
           0.f, DY4, 4 * H, 0, 0, 1, 1, nullptr, nullptr, 0, 1, stream))
    return 1;
  // dW_mlpproj += g^T @ dout ; db += colsum(dout)
  if (gemm(1, 0, 4 * H, H, BS, 1.f, gact, 4 * H, 0, 0, dout, H, 0, 0, 1.f,
           g + bp.w_mlpproj, H, 0, 0, 1, 1, nullptr, nullptr, 1,
           pick_splitk(4 * H, H, BS), stream))
    return 1;
  if (ob_colsum_f32(dout, g + bp.b_mlpproj, BS, H, stream)) return 1;
  // du = dg * gelu'(u)   (in place on DY4)
  if (ob_gelu_bwd_f32(u, DY4, DY4, BS * 4 * H, stream)) return 1;
  // dW_fc += ln2^T @ du ; db_fc += colsum(du)
  if (gemm(1, 0, H, 4 * H, BS, 1.f, ln2, H, 0, 0, DY4, 4 * H, 0, 0, 1.f,
           g + bp.w_fc, 4 * H, 0, 0, 1, 1, nullptr, nullptr, 1,
           pick_splitk(H, 4 * H, BS), stream))
    return 1;
  if (ob_colsum_f32(DY4, g + bp.b_fc, BS, 4 * H, stream)) return 1;
  // d_ln2out = du @ w_fc^T  (narrow-N: auto split-K)
  if (gemm_dx_splitk(1, BS, H, 4 * H, 1.f, DY4, 4 * H, p + bp.w_fc, 4 * H,
                     DLN, H, stream))
    return 1;
  // ln2 backward: din += dx ; dw/db accumulate
  if (ob_layernorm_bwd_f32(hmid, p + bp.ln2_w, st + l->o_mean2,
                           st + l->o_rstd2, DLN, din, g + bp.ln2_w,
                           g + bp.ln2_b, BS, H, 1, stream))
    return 1;
  // ---- attention projection backward (din now = d_hmid) ----
  if (gemm_dx_splitk(1, BS, H, H, 1.f, din, H, p + bp.w_attnproj, H, DATT,
                     H, stream))
    return 1;
  if (gemm(1, 0, H, H, BS, 1.f, am, H, 0, 0, din, H, 0, 0, 1.f,
           g + bp.w_attnproj, H, 0, 0, 1, 1, nullptr, nullptr, 1,
           pick_splitk(H, H, BS), stream))
    return 1;
  if (ob_colsum_f32(din, g + bp.b_attnproj, BS, H, stream)) return 1;
  // ---- attention core backward ----
  // dP = dO @ V^T
  if (gemm(0, 1, Sq, Sq, hd, 1.f, DATT, H, Sq * H, hd, qkv + 2 * H, 3 * H,
           Sq * 3 * H, hd, 0.f, DP, Sq, nh * Sq * Sq, Sq * Sq, B, nh, nullptr,
           nullptr, 0, 1, stream))
    return 1;
  // dS = softmax_bwd(P, dP)  in place on DP
  if (ob_softmax_causal_bwd_f32(P, DP, B * nh, Sq, stream)) return 1;
  // dQ = scale * dS @ K
  if (gemm(0, 0, Sq, hd, Sq, scale, DP, Sq, nh * Sq * Sq, Sq * Sq, qkv + H,
           3 * H, Sq * 3 * H, hd, 0.f, DQKV, 3 * H, Sq * 3 * H, hd, B, nh,
           nullptr, nullptr, 0, 1, stream))
    return 1;
  // dK = scale * dS^T @ Q
  if (gemm(1, 0, Sq, hd, Sq, scale, DP, Sq, nh * Sq * Sq, Sq * Sq, qkv, 3 * H,
           Sq * 3 * H, hd, 0.f, DQKV + H, 3 * H, Sq * 3 * H, hd, B, nh,
           nullptr, nullptr, 0, 1, stream))
    return 1;
  // dV = P^T @ dO
  if (gemm(1, 0, Sq, hd, Sq, 1.f, P, Sq, nh * Sq * Sq, Sq * Sq, DATT, H,
           Sq * H, hd, 0.f, DQKV + 2 * H, 3 * H, Sq * 3 * H, hd, B, nh,
           nullptr, nullptr, 0, 1, stream))
    return 1;
  // ---- QKV projection backward ----
  if (ob_colsum_f32(DQKV, g + bp.b_qkv, BS, 3 * H, stream)) return 1;
  if (gemm(1, 0, H, 3 * H, BS, 1.f, ln1, H, 0, 0, DQKV, 3 * H, 0, 0, 1.f,
           g + bp.w_qkv, 3 * H, 0, 0, 1, 1, nullptr, nullptr, 1,
           pick_splitk(H, 3 * H, BS), stream))
    return 1;
  if (gemm_dx_splitk(1, BS, H, 3 * H, 1.f, DQKV, 3 * H, p + bp.w_qkv, 3 * H,
                     DLN, H, stream))
    return 1;
  // ln1 backward: din += dx
  if (ob_layernorm_bwd_f32(x, p + bp.ln1_w, st + l->o_mean1, st + l->o_rstd1,
                           DLN, din, g + bp.ln1_w, g + bp.ln1_b, BS, H, 1,
                           stream))
    return 1;
  return 0;
}

static int final_backward(ob_layer* l, int slot, const float* dout, float* din,
                          void* stream) {
  const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
  const int64_t B = l->B, BS = B * Sq;
  float* st = l->stash + (int64_t)slot * l->slot_stride;
  const float* p = l->params;
  float* g = l->grads;
  float* x = st + l->o_x;
  float* lnf = st + l->o_ln1;
  float* logits = st + l->o_logits;  // becomes dlogits in place
  int64_t* labs = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;

  if (ob_ce_bwd_f32(logits, labs, st + l->o_lse, (const float*)dout, B, Sq, V,
                    stream))
    return 1;
  // dW_lm += dlogits^T @ ln_out
  if (gemm(1, 0, V, H, BS, 1.f, logits, V, 0, 0, lnf, H, 0, 0, 1.f, g + 2 * H,
           H, 0, 0, 1, 1, nullptr, nullptr, 1, 1, stream))
    return 1;
  // d_lnout = dlogits @ w_lm  (N=H, K=V: auto split-K)
  if (gemm_dx_splitk(0, BS, H, V, 1.f, logits, V, p + 2 * H, H, g_ws.bsh1, H,
                     stream))
    return 1;
  if (ob_layernorm_bwd_f32(x, p + 0, st + l->o_mean1, st + l->o_rstd1,
                           g_ws.bsh1, din, g + 0, g + H, BS, H, 0, stream))
    return 1;
  return 0;
}

extern "C" int ob_layer_backward(ob_layer_t l, int32_t slot, const void* dout,
                                 void* din, void* stream) {
  if (!l) return ob_fail("backward: null layer");
  if (!l->params || !l->grads) return ob_fail("backward: buffers not bound");
  if (slot < 0 || slot >= l->d.n_slots) return ob_fail("backward: bad slot");
  switch (l->d.kind) {
    case OB_KIND_EMBED: {
      const int64_t Sq = l->d.seq_len, H = l->d.n_embd, V = l->d.vocab_size;
      int64_t* ids = l->ids + (int64_t)slot * (int64_t)l->d.max_batch * Sq;
      if (l->d.dtype == 1)
        return OB_PROF(OB_PF_ELEM, stream,
                       ob_embed_bwd_bf16(ids, dout, l->grads,
                                         l->grads + V * H, l->B, Sq, H,
                                         stream));
      return OB_PROF(OB_PF_ELEM, stream,
                     ob_embed_bwd_f32(ids, (const float*)dout, l->grads,
                                      l->grads + V * H, l->B, Sq, H,
                                      stream));
    }
    case OB_KIND_BLOCK:
      if (!dout || !din) return ob_fail("block backward: dout/din required");
      if (l->d.dtype == 1)
        return block_backward_bf16(l, slot, (const __bf16*)dout, (__bf16*)din,
                                   stream);
      return block_backward(l, slot, (const float*)dout, (float*)din, stream);
    case OB_KIND_FINAL:
      if (!din) return ob_fail("final backward: din required");
      if (l->d.dtype == 1)
        return final_backward_bf16(l, slot, (const float*)dout, (__bf16*)din,
                                   stream);
      return final_backward(l, slot, (const float*)dout, (float*)din, stream);
  }
  return ob_fail("backward: bad kind");
}
