/* include/oobleck_stage.h — the C-ABI drop-in boundary of the MI355X-native
 * rebuild of Oobleck's pipeline-parallel execution hot path.
 *
 * The reference implements this path in Python-on-torch; its internal seams
 * (the interfaces this ABI replaces) are:
 *
 *   - Layer forward/backward over one fx-sharded GPT-2 layer:
 *       /root/reference/oobleck/execution/layer.py:144-145 (forward),
 *       :250-260 (backward: loss.backward() on the last stage, else
 *       autograd.backward(outputs, received grads)).
 *   - The flat contiguous parameter buffer per layer consumed by the
 *     optimizer and by reconfiguration layer-copy:
 *       /root/reference/oobleck/execution/pipeline.py:117-119
 *       (_param_handle.flat_param), engine.py:283-299 (broadcast source).
 *   - The fused AdamW step over those flat buffers:
 *       /root/reference/oobleck/execution/pipeline.py:117-127, 241-244.
 *   - Per-instruction execution handlers the pipeline dispatches into:
 *       /root/reference/oobleck/execution/pipeline.py:169-244.
 *
 * Ownership: the CALLER (the Python host, via torch CUDA tensors) owns the
 * flat parameter / gradient / Adam-state buffers and all activation I/O
 * buffers — so RCCL collectives (grad all-reduce, reconfig broadcast) can
 * run on them directly through torch.distributed without copies.  The
 * extension owns only its internal activation stash and scratch workspace
 * (HIP memory, sized at bind time).
 *
 * Threading: one driving thread per GPU (matches 1 worker = 1 GPU,
 * /root/reference/oobleck/elastic/agent.py:141-174).  Calls are not
 * thread-safe.  All compute is asynchronous on the hipStream_t passed per
 * call; no call synchronizes the device.
 *
 * Errors: every function returns 0 on success, non-zero on failure; the
 * last failure message is available via ob_last_error().  No exceptions
 * cross the ABI.
 *
 * Python-side binding: ctypes over this header.
 */
#ifndef OOBLECK_STAGE_H
#define OOBLECK_STAGE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- layer kinds: the fx-shard grain of the reference -------------------
 * (/root/reference/oobleck/module/sharding.py:12-47: GPT model splits into
 *  [embedding] [transformer.h.i]*L [ln_f + lm_head + loss])              */
enum {
  OB_KIND_EMBED = 0,  /* wte gather + wpe add            ids[B,S] -> f32[B,S,H] */
  OB_KIND_BLOCK = 1,  /* LN1/QKV/attn/proj/LN2/MLP block f32[B,S,H] -> f32[B,S,H] */
  OB_KIND_FINAL = 2   /* ln_f + lm_head + shifted CE     f32[B,S,H] -> f32 loss  */
};

typedef struct ob_layer* ob_layer_t;

typedef struct {
  int32_t kind;         /* OB_KIND_* */
  int32_t n_embd;       /* H */
  int32_t n_head;
  int32_t n_positions;
  int32_t vocab_size;
  int32_t max_batch;    /* microbatch size the stash is sized for */
  int32_t seq_len;      /* S */
  int32_t n_slots;      /* in-flight microbatches (pipeline buffer slots,
                           deepspeed-compatible num_pipe_buffers) */
  int32_t dtype;        /* 0 = fp32; 1 = bf16 activations/GEMMs with fp32
                           master weights + fp32 grads (§8 f4) */
} ob_layer_desc;

/* Number of fp32 elements in the layer's flat parameter buffer.  Layout is
 * the canonical order shared with oracle/gpt2_oracle.py::layer_param_spec. */
int64_t ob_layer_param_count(const ob_layer_desc* desc);

/* Create a layer; allocates the activation stash + workspace on the current
 * HIP device.  Parameters/grads are NOT allocated here — see ob_layer_bind. */
int ob_layer_create(const ob_layer_desc* desc, ob_layer_t* out);

/* Bind caller-owned device fp32 buffers of ob_layer_param_count() elements:
 * params (read/write: forward reads, adam writes) and grads (backward
 * ACCUMULATES, caller zeroes between steps).  Mirrors flat_param /
 * flat_param.grad of the reference's FlatParamHandle (layer.py:96-111). */
int ob_layer_bind(ob_layer_t l, void* params, void* grads);

/* Set the actual microbatch size (<= max_batch) for subsequent calls. */
int ob_layer_set_batch(ob_layer_t l, int32_t batch);

/* Forward one microbatch in stash slot `slot` (0 <= slot < n_slots).
 *   EMBED: in = int64 ids[B,S],  out = f32[B,S,H],   labels ignored
 *   BLOCK: in = f32[B,S,H],      out = f32[B,S,H],   labels ignored
 *   FINAL: in = f32[B,S,H],      labels = int64[B,S], out = f32[1] (mean
 *          shifted CE loss, modeling_gpt2 semantics)
 * Asynchronous on `stream` (a hipStream_t). */
int ob_layer_forward(ob_layer_t l, int32_t slot, const void* in, void* out,
                     const int64_t* labels, void* stream);

/* Backward for the microbatch stashed in `slot`.
 *   FINAL: dout = NULL (seed dloss = 1.0, the last-stage semantics of
 *          layer.py:250-253) or f32[1] scale; din = f32[B,S,H]
 *   BLOCK: dout = f32[B,S,H], din = f32[B,S,H]
 *   EMBED: dout = f32[B,S,H], din ignored (int input)
 * Accumulates into the bound grad buffer (autograd += semantics). */
int ob_layer_backward(ob_layer_t l, int32_t slot, const void* dout, void* din,
                      void* stream);

int ob_layer_destroy(ob_layer_t l);

/* bf16 mode only: refresh the extension's bf16 weight shadows (plain +
 * transposed layouts) from the fp32 master params — call after ob_layer_bind
 * and after every optimizer step. */
int ob_layer_refresh_weights(ob_layer_t l, void* stream);

/* Fused AdamW over a flat buffer (replaces torch AdamW(fused=True) of
 * pipeline.py:117-127; decoupled weight decay + bias correction, eps added
 * after sqrt(v)/sqrt(bc2), matching torch.optim.AdamW).  step is 1-based. */
int ob_adamw_step(void* p, const void* g, void* m, void* v, int64_t n,
                  int32_t step, float lr, float beta1, float beta2, float eps,
                  float weight_decay, void* stream);

/* ---- standalone kernel entry points (parity tests + roofline probes) ---- */

/* C[M,N] (+)= alpha * op(A)[M,K] @ op(B)[K,N] + bias[n] + R[m,n]
 * op(X) = X stored row-major [M,K]/[K,N]; transX=1 means X is stored
 * transposed ([K,M]/[N,K]).  Two-level strided batch: z = i1*n2 + i2,
 * operand offset = i1*stride?1 + i2*stride?2.  beta 0 or 1.  bias/R NULL to
 * skip.  atomic!=0 stores via atomicAdd (used for split-K weight grads). */
int ob_gemm_f32(int transA, int transB, int64_t M, int64_t N, int64_t K,
                float alpha, const void* A, int64_t lda,
                int64_t strideA1, int64_t strideA2,
                const void* B, int64_t ldb,
                int64_t strideB1, int64_t strideB2,
                float beta, void* C, int64_t ldc,
                int64_t strideC1, int64_t strideC2,
                int64_t n1, int64_t n2,
                const void* bias, const void* residual, int atomic, int splitk,
                void* stream);

/* LayerNorm over the last dim: y = (x-mu)*rstd*w + b, rows x H. */
int ob_layernorm_fwd_f32(const void* x, const void* w, const void* b, void* y,
                         void* mean, void* rstd, int64_t rows, int64_t H,
                         float eps, void* stream);
/* dx (+)= LN backward (accumulate if dx_accum); dw,db accumulated atomically. */
int ob_layernorm_bwd_f32(const void* x, const void* w, const void* mean,
                         const void* rstd, const void* dy, void* dx,
                         void* dw, void* db, int64_t rows, int64_t H,
                         int dx_accum, void* stream);

/* Causal row softmax in place on scores[batch, S, S] (row r keeps cols 0..r),
 * scale applied before the max. */
int ob_softmax_causal_fwd_f32(void* scores, int64_t batch, int64_t S,
                              float scale, void* stream);
/* dS = P * (dP - rowsum(dP*P)), in place on dP. */
int ob_softmax_causal_bwd_f32(const void* P, void* dP, int64_t batch,
                              int64_t S, void* stream);

/* gelu_new (tanh approximation, transformers NewGELUActivation). */
int ob_gelu_fwd_f32(const void* u, void* g, int64_t n, void* stream);
int ob_gelu_bwd_f32(const void* u, const void* dg, void* du, int64_t n,
                    void* stream);

/* db[n] += sum_m X[m,n] (bias gradient). */
int ob_colsum_f32(const void* X, void* db, int64_t M, int64_t N, void* stream);

/* ---- bf16 mixed-precision path ------------------------
 * bf16 MFMA GEMM (v_mfma_f32_32x32x16_bf16, fp32 accumulate).  Same
 * call shape as ob_gemm_f32; A/B are bf16; out_kind: 0 = bf16 C,
 * 1 = fp32 C, 2 = fp32 atomicAdd (weight grads / split-K).  lda/ldb must
 * be multiples of 8 (16-byte staging). */
int ob_gemm_bf16(int transA, int transB, int64_t M, int64_t N, int64_t K,
                 float alpha, const void* A, int64_t lda,
                 int64_t strideA1, int64_t strideA2,
                 const void* B, int64_t ldb,
                 int64_t strideB1, int64_t strideB2,
                 float beta, void* C, int64_t ldc,
                 int64_t strideC1, int64_t strideC2,
                 int64_t n1, int64_t n2,
                 const void* bias, const void* residual, int out_kind,
                 int splitk, void* stream);

/* dtype casts (weight shadows / activation conversion). */
int ob_f32_to_bf16(const void* x, void* y, int64_t n, void* stream);
int ob_f32_to_bf16_t(const void* x, void* y, int64_t rows, int64_t cols,
                     void* stream);  /* y[c][r] = x[r][c] */
int ob_bf16_to_f32(const void* x, void* y, int64_t n, void* stream);

/* ---- in-step profiling (measurement only; off by default) ---------------
 * When enabled, each wrapped launch region inside ob_layer_forward/backward
 * and ob_adamw_step is bracketed by HIP events on its launch stream and
 * accumulated per family, so the bench can report the production dispatch's
 * in-step per-launch time (roofline) and a per-family step-time split.
 * Families (ob_profile_read's `fam`): 0 fc_fwd_gemm (the roofline kernel),
 * 1 gemm_fwd_other, 2 gemm_dx, 3 gemm_dw(side stream), 4 flash_fwd,
 * 5 flash_bwd, 6 attn_matmuls(non-flash), 7 layernorm, 8 cross_entropy,
 * 9 elementwise(gelu/colsum/embed), 10 adamw. */
void ob_profile_enable(int on);
void ob_profile_reset(void);
int ob_profile_read(int fam, double* total_ms, long long* count);

const char* ob_last_error(void);

/* Build stamp: returns the gfx arch this library was compiled for. */
const char* ob_build_arch(void);

#ifdef __cplusplus
}
#endif
#endif /* OOBLECK_STAGE_H */
