# oracle/gpt2_oracle.py — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
#
# CPU restatement (plain torch fp32 eager ops — this is a floating-point
# path, so the torch fp32 reference is the oracle) of the arithmetic the
# reference executes per fx-sharded layer:
#
#   * layer boundaries: /root/reference/oobleck/module/sharding.py:12-47
#     (GPT split at each transformer.h.{i} + transformer.ln_f), so a GPT-2
#     with L blocks becomes L+2 "layers": [embedding] [block]*L
#     [ln_f + lm_head + loss]  (tests/module/test_model.py:22 in the ref).
#   * the math inside each layer is HF transformers GPT2 (the reference
#     only re-partitions the HF graph; /root/reference/oobleck/module/
#     model.py:71-77 builds AutoModelForPreTraining "gpt2").  Restated here
#     from transformers/models/gpt2/modeling_gpt2.py (v5.15 in this
#     container): Conv1D  y = x@W + b with W [in,out]; eager attention with
#     1/sqrt(head_dim) scaling and causal mask filled with
#     torch.finfo(dtype).min; gelu_new tanh approximation; LayerNorm
#     eps=1e-5; loss = CE(logits[:, :-1], labels[:, 1:]) mean.
#   * forward/backward semantics: /root/reference/oobleck/execution/
#     pipeline.py:169-244 (loss on last stage; non-last stages seed
#     autograd.backward(outputs, received grads), layer.py:250-260).
#
# Deviations from the reference, stated:
#   * dropout is disabled (the reference leaves HF's default 0.1 dropout
#     active in training mode, which makes bitwise parity across
#     implementations impossible; no reference test pins it — SURVEY.md §8c).
#   * weight tying: the reference deep-copies each fx-sharded layer
#     (layer.py:93), so wte (layer 0) and lm_head (last layer) become
#     SEPARATE parameters trained independently.  We match that: the final
#     layer owns its own w_lm [V,H].
#
# Parity pinning: oracle outputs are checked against golden vectors
# generated from transformers.GPT2LMHeadModel (tests/golden/, generation
# script oracle/gen_golden.py) and against torch.optim.AdamW for the
# optimizer math.
from __future__ import annotations

import math
from dataclasses import dataclass

import torch

KIND_EMBED = 0
KIND_BLOCK = 1
KIND_FINAL = 2


@dataclass
class OracleConfig:
    n_embd: int = 768
    n_head: int = 12
    n_layer: int = 12          # transformer blocks
    n_positions: int = 1024
    vocab_size: int = 50257
    layer_norm_eps: float = 1e-5

    @property
    def n_layers_total(self) -> int:
        # sharding.py:12-47: one layer per block + embedding front + ln_f tail
        return self.n_layer + 2

    def layer_kind(self, layer_id: int) -> int:
        if layer_id == 0:
            return KIND_EMBED
        if layer_id == self.n_layers_total - 1:
            return KIND_FINAL
        return KIND_BLOCK


# ---------------------------------------------------------------------------
# Canonical flat-parameter layout (shared contract with the HIP extension,
# include/oobleck_stage.h).  Order within each layer's flat fp32 buffer:
# ---------------------------------------------------------------------------

def layer_param_spec(cfg: OracleConfig, kind: int) -> list[tuple[str, tuple[int, ...]]]:
    H, V, P = cfg.n_embd, cfg.vocab_size, cfg.n_positions
    if kind == KIND_EMBED:
        return [("wte", (V, H)), ("wpe", (P, H))]
    if kind == KIND_BLOCK:
        return [
            ("ln1_w", (H,)), ("ln1_b", (H,)),
            ("w_qkv", (H, 3 * H)), ("b_qkv", (3 * H,)),
            ("w_attnproj", (H, H)), ("b_attnproj", (H,)),
            ("ln2_w", (H,)), ("ln2_b", (H,)),
            ("w_fc", (H, 4 * H)), ("b_fc", (4 * H,)),
            ("w_mlpproj", (4 * H, H)), ("b_mlpproj", (H,)),
        ]
    if kind == KIND_FINAL:
        return [("lnf_w", (H,)), ("lnf_b", (H,)), ("w_lm", (V, H))]
    raise ValueError(f"bad kind {kind}")


def layer_param_numel(cfg: OracleConfig, kind: int) -> int:
    return sum(int(torch.tensor(s).prod()) for _, s in layer_param_spec(cfg, kind))


def unpack_params(cfg: OracleConfig, kind: int, flat: torch.Tensor) -> dict[str, torch.Tensor]:
    """Views into a flat fp32 buffer, canonical order."""
    out, off = {}, 0
    for name, shape in layer_param_spec(cfg, kind):
        n = int(math.prod(shape))
        out[name] = flat[off:off + n].view(shape)
        off += n
    assert off == flat.numel(), f"flat param size mismatch: {off} != {flat.numel()}"
    return out


def init_layer_params(cfg: OracleConfig, kind: int, seed: int,
                      style: str = "gpt2") -> torch.Tensor:
    """Random-init a layer's flat fp32 parameter buffer.

    style="gpt2": HF-like init (normal 0.02 for weights, zeros for biases,
      ones for LN weights) — sane magnitudes for parity tolerances.
    style="reference": uniform [0,1) for everything, as the reference's
      init_tensors does (layer.py:26-37, torch.rand(param.shape)).
    """
    g = torch.Generator().manual_seed(seed)
    flats = []
    for name, shape in layer_param_spec(cfg, kind):
        if style == "reference":
            t = torch.rand(shape, generator=g)
        elif name.startswith(("ln1_w", "ln2_w", "lnf_w")):
            t = torch.ones(shape)
        elif name.endswith("_b") or name.startswith("b_"):
            t = torch.zeros(shape)
        else:
            t = torch.randn(shape, generator=g) * 0.02
        flats.append(t.reshape(-1))
    return torch.cat(flats).float()


# ---------------------------------------------------------------------------
# Per-layer forward math (fp32, no dropout).
# ---------------------------------------------------------------------------

def _layer_norm(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor, eps: float) -> torch.Tensor:
    # nn.LayerNorm semantics: biased variance over the last dim.
    mu = x.mean(dim=-1, keepdim=True)
    var = x.var(dim=-1, keepdim=True, unbiased=False)
    return (x - mu) * torch.rsqrt(var + eps) * w + b


def _gelu_new(x: torch.Tensor) -> torch.Tensor:
    # transformers activations.py NewGELUActivation
    return 0.5 * x * (1.0 + torch.tanh(math.sqrt(2.0 / math.pi) * (x + 0.044715 * x ** 3)))


def _attention(cfg: OracleConfig, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    # eager GPT2Attention: scores scaled by 1/sqrt(head_dim), causal mask
    # filled with finfo.min, fp32 softmax.  q,k,v: [B, nh, S, hd]
    S = q.shape[2]
    w = torch.matmul(q, k.transpose(-1, -2)) / math.sqrt(q.shape[-1])
    mask = torch.tril(torch.ones(S, S, dtype=torch.bool, device=q.device))
    w = torch.where(mask, w, torch.full((), torch.finfo(w.dtype).min, dtype=w.dtype))
    w = torch.softmax(w, dim=-1)
    return torch.matmul(w, v)


def _split_heads(x: torch.Tensor, n_head: int) -> torch.Tensor:
    B, S, H = x.shape
    return x.view(B, S, n_head, H // n_head).permute(0, 2, 1, 3)


def _merge_heads(x: torch.Tensor) -> torch.Tensor:
    B, nh, S, hd = x.shape
    return x.permute(0, 2, 1, 3).reshape(B, S, nh * hd)


def layer_forward(cfg: OracleConfig, kind: int, flat: torch.Tensor,
                  x: torch.Tensor, labels: torch.Tensor | None = None) -> torch.Tensor:
    """One fx-sharded layer's forward.

    kind=EMBED:  x = input_ids int64 [B,S]        -> hidden [B,S,H]
    kind=BLOCK:  x = hidden [B,S,H]               -> hidden [B,S,H]
    kind=FINAL:  x = hidden [B,S,H], labels [B,S] -> scalar loss
                 (labels=None -> logits [B,S,V])
    """
    p = unpack_params(cfg, kind, flat)
    eps = cfg.layer_norm_eps

    if kind == KIND_EMBED:
        S = x.shape[1]
        return p["wte"][x] + p["wpe"][:S]

    if kind == KIND_BLOCK:
        h = x
        a = _layer_norm(h, p["ln1_w"], p["ln1_b"], eps)
        qkv = a @ p["w_qkv"] + p["b_qkv"]
        q, k, v = qkv.split(cfg.n_embd, dim=-1)
        o = _attention(cfg, _split_heads(q, cfg.n_head), _split_heads(k, cfg.n_head),
                       _split_heads(v, cfg.n_head))
        o = _merge_heads(o) @ p["w_attnproj"] + p["b_attnproj"]
        h = h + o
        m = _layer_norm(h, p["ln2_w"], p["ln2_b"], eps)
        m = _gelu_new(m @ p["w_fc"] + p["b_fc"]) @ p["w_mlpproj"] + p["b_mlpproj"]
        return h + m

    if kind == KIND_FINAL:
        hn = _layer_norm(x, p["lnf_w"], p["lnf_b"], eps)
        logits = hn @ p["w_lm"].t()
        if labels is None:
            return logits
        # modeling_gpt2 loss: shift by one, mean CE over B*(S-1)
        return torch.nn.functional.cross_entropy(
            logits[:, :-1].reshape(-1, cfg.vocab_size),
            labels[:, 1:].reshape(-1))

    raise ValueError(f"bad kind {kind}")


# ---------------------------------------------------------------------------
# Whole-model / stage-slice forward+backward (autograd over the restated ops).
# ---------------------------------------------------------------------------

def model_forward(cfg: OracleConfig, flats: list[torch.Tensor], input_ids: torch.Tensor,
                  labels: torch.Tensor | None = None):
    """Run layers [0, n_layers_total). Returns (loss_or_logits, activations)
    where activations[i] is the INPUT to layer i (activations[0]=input_ids)."""
    acts = [input_ids]
    x: torch.Tensor = input_ids
    L = cfg.n_layers_total
    for lid in range(L):
        kind = cfg.layer_kind(lid)
        x = layer_forward(cfg, kind, flats[lid], x,
                          labels if kind == KIND_FINAL else None)
        if lid < L - 1:
            acts.append(x)
    return x, acts


def stage_forward_backward(cfg: OracleConfig, flats: list[torch.Tensor],
                           layer_ids: list[int], x_in: torch.Tensor,
                           labels: torch.Tensor | None = None,
                           dout: torch.Tensor | None = None):
    """Forward+backward over a contiguous slice of layers, mirroring one
    pipeline stage (pipeline.py:169-239): the last stage computes the loss
    and backprops it; other stages seed autograd with the received dout.

    Returns (out, dx_in, grad_flats): out is loss (last stage) or the
    stage's output activation; dx_in is d(loss)/d(x_in) (None for int
    inputs, i.e. the first stage); grad_flats are per-layer flat grads.
    """
    flats = [f.detach().clone().requires_grad_(True) for f in flats]
    x0 = x_in.detach().clone()
    if x0.is_floating_point():
        x0.requires_grad_(True)
    x = x0
    for i, lid in enumerate(layer_ids):
        kind = cfg.layer_kind(lid)
        x = layer_forward(cfg, kind, flats[i], x,
                          labels if kind == KIND_FINAL else None)
    out = x
    if dout is None:
        out.backward()
    else:
        torch.autograd.backward(out, dout)
    dx_in = x0.grad if x0.is_floating_point() else None
    return out.detach(), dx_in, [f.grad for f in flats]


def model_forward_backward(cfg: OracleConfig, flats: list[torch.Tensor],
                           input_ids: torch.Tensor, labels: torch.Tensor):
    loss, dx, grads = stage_forward_backward(
        cfg, flats, list(range(cfg.n_layers_total)), input_ids, labels=labels)
    return loss, grads


# ---------------------------------------------------------------------------
# AdamW (restating torch.optim.AdamW's decoupled weight decay + bias
# correction exactly — the reference uses AdamW(fused=True),
# pipeline.py:117-127; eps is added AFTER sqrt(v)/sqrt(bc2)).
# ---------------------------------------------------------------------------

def adamw_step(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               step: int, lr: float, beta1: float = 0.9, beta2: float = 0.999,
               eps: float = 1e-8, weight_decay: float = 0.0) -> None:
    """In-place AdamW on flat fp32 buffers; step is 1-based."""
    p.mul_(1.0 - lr * weight_decay)
    m.mul_(beta1).add_(g, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
    p.addcdiv_(m, denom, value=-lr / bc1)
