// Internal shared declarations between ob_kernels.hip and ob_layer.hip.
#pragma once
#include <hip/hip_runtime.h>

#include <cstdarg>
#include <cstdint>
#include <cstdio>

#include "../../include/oobleck_stage.h"

// thread-local last-error buffer (single driving thread per GPU by contract,
// but thread-local keeps the reconfig listener thread safe too).
int ob_fail(const char* fmt, ...);

#define OB_HIP(x)                                                     \
  do {                                                                \
    hipError_t e_ = (x);                                              \
    if (e_ != hipSuccess)                                             \
      return ob_fail("%s:%d: %s", __FILE__, __LINE__,                 \
                     hipGetErrorString(e_));                          \
  } while (0)

#define OB_LAUNCH_CHECK()                                             \
  do {                                                                \
    hipError_t e_ = hipGetLastError();                                \
    if (e_ != hipSuccess)                                             \
      return ob_fail("%s:%d: launch: %s", __FILE__, __LINE__,         \
                     hipGetErrorString(e_));                          \
  } while (0)

// internal launchers used by the layer orchestration (same semantics as the
// public ob_* wrappers but C++ linkage, no error wrapping duplication).
int ob_embed_fwd_f32(const int64_t* ids, const float* wte, const float* wpe,
                     float* out, int64_t B, int64_t S, int64_t H, void* stream);
int ob_embed_bwd_f32(const int64_t* ids, const float* dout, float* dwte,
                     float* dwpe, int64_t B, int64_t S, int64_t H, void* stream);
int ob_ce_fwd_f32(const float* logits, const int64_t* labels, float* lse,
                  float* loss, int64_t B, int64_t S, int64_t V, void* stream);
int ob_ce_bwd_f32(float* logits_to_dlogits, const int64_t* labels,
                  const float* lse, const float* dloss_or_null, int64_t B,
                  int64_t S, int64_t V, void* stream);
