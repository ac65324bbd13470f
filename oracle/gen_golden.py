# oracle/gen_golden.py — generates the committed golden fixtures under
# tests/golden/ by running transformers.GPT2LMHeadModel (the library that
# supplies ALL of the reference's device math — SURVEY.md §5: the reference
# only re-partitions the HF fx graph) side by side with the oracle
# restatement, on this container's transformers 5.15.
#
# Run:  python -m oracle.gen_golden      (from the repo root; needs
# transformers, so it runs HERE — the fixtures, not this dependency, travel
# to the GPU box).
from __future__ import annotations

import pathlib

import numpy as np
import torch

from oracle.gpt2_oracle import (
    KIND_FINAL,
    OracleConfig,
    adamw_step,
    layer_param_spec,
    model_forward,
    stage_forward_backward,
    unpack_params,
)

GOLDEN_DIR = pathlib.Path(__file__).resolve().parent.parent / "tests" / "golden"

# Two small shape classes: one with head_dim=24, one with head_dim=64 (the
# real GPT-2 head_dim).  Arithmetic is shape-independent; small dims keep
# fixtures tiny and the HF run fast.
CASES = [
    dict(name="tiny_hd24", n_embd=96, n_head=4, n_layer=3, n_positions=64,
         vocab_size=211, batch=2, seq=48, seed=42),
    dict(name="tiny_hd64", n_embd=128, n_head=2, n_layer=2, n_positions=96,
         vocab_size=307, batch=2, seq=96, seed=7),
]


def hf_model(cfg: OracleConfig):
    from transformers import GPT2Config, GPT2LMHeadModel
    hcfg = GPT2Config(
        n_embd=cfg.n_embd, n_head=cfg.n_head, n_layer=cfg.n_layer,
        n_positions=cfg.n_positions, vocab_size=cfg.vocab_size,
        resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
        use_cache=False, attn_implementation="eager",
        bos_token_id=0, eos_token_id=0,
    )
    return GPT2LMHeadModel(hcfg).float()


def hf_to_flats(cfg: OracleConfig, model) -> list[torch.Tensor]:
    """Copy HF parameters into the canonical per-layer flat layout.
    NOTE: the final layer gets its OWN copy of lm_head.weight — matching the
    reference's per-layer deepcopy that unties wte/lm_head (layer.py:93)."""
    sd = {k: v.detach().float() for k, v in model.state_dict().items()}
    flats = []
    for lid in range(cfg.n_layers_total):
        kind = cfg.layer_kind(lid)
        parts = []
        for name, shape in layer_param_spec(cfg, kind):
            key = {
                "wte": "transformer.wte.weight",
                "wpe": "transformer.wpe.weight",
                "lnf_w": "transformer.ln_f.weight",
                "lnf_b": "transformer.ln_f.bias",
                "w_lm": "lm_head.weight",
            }.get(name)
            if key is None:
                b = lid - 1
                key = {
                    "ln1_w": f"transformer.h.{b}.ln_1.weight",
                    "ln1_b": f"transformer.h.{b}.ln_1.bias",
                    "w_qkv": f"transformer.h.{b}.attn.c_attn.weight",
                    "b_qkv": f"transformer.h.{b}.attn.c_attn.bias",
                    "w_attnproj": f"transformer.h.{b}.attn.c_proj.weight",
                    "b_attnproj": f"transformer.h.{b}.attn.c_proj.bias",
                    "ln2_w": f"transformer.h.{b}.ln_2.weight",
                    "ln2_b": f"transformer.h.{b}.ln_2.bias",
                    "w_fc": f"transformer.h.{b}.mlp.c_fc.weight",
                    "b_fc": f"transformer.h.{b}.mlp.c_fc.bias",
                    "w_mlpproj": f"transformer.h.{b}.mlp.c_proj.weight",
                    "b_mlpproj": f"transformer.h.{b}.mlp.c_proj.bias",
                }[name]
            t = sd[key]
            assert tuple(t.shape) == tuple(shape), (name, t.shape, shape)
            parts.append(t.reshape(-1))
        flats.append(torch.cat(parts))
    return flats


def gen_case(case: dict) -> None:
    cfg = OracleConfig(n_embd=case["n_embd"], n_head=case["n_head"],
                       n_layer=case["n_layer"], n_positions=case["n_positions"],
                       vocab_size=case["vocab_size"])
    torch.manual_seed(case["seed"])
    model = hf_model(cfg)
    B, S = case["batch"], case["seq"]
    g = torch.Generator().manual_seed(case["seed"])
    ids = torch.randint(0, cfg.vocab_size, (B, S), generator=g)
    labels = ids.clone()          # dataset.py:201 copies input_ids to labels

    # --- HF side: loss + logits + named grads -----------------------------
    model.zero_grad()
    out = model(input_ids=ids, labels=labels, return_dict=True)
    out.loss.backward()
    hf_loss = out.loss.detach()
    hf_logits = out.logits.detach()

    # --- oracle side ------------------------------------------------------
    flats = hf_to_flats(cfg, model)
    logits_o, _ = model_forward(cfg, flats, ids, labels=None)
    loss_o, _, grads_o = stage_forward_backward(
        cfg, flats, list(range(cfg.n_layers_total)), ids, labels=labels)

    # sanity before committing fixtures
    assert torch.allclose(logits_o, hf_logits, rtol=1e-4, atol=1e-4), \
        (logits_o - hf_logits).abs().max()
    assert torch.allclose(loss_o, hf_loss, rtol=1e-5, atol=1e-6)

    # HF grads, mapped into the same flat layout for comparison.  wte and
    # lm_head are TIED in HF, so HF's wte.grad = (embedding grad + lm_head
    # grad); the oracle's untied layout must reproduce it as the SUM of its
    # layer-0 wte grad and its final-layer w_lm grad.
    hf_g = {k: v.grad.detach().float() for k, v in model.named_parameters()
            if v.grad is not None}
    wte_grad_hf = hf_g["transformer.wte.weight"]
    o_emb = unpack_params(cfg, 0, grads_o[0])
    o_fin = unpack_params(cfg, KIND_FINAL, grads_o[-1])
    tied_sum = o_emb["wte"] + o_fin["w_lm"]
    assert torch.allclose(tied_sum, wte_grad_hf, rtol=1e-4, atol=1e-5), \
        (tied_sum - wte_grad_hf).abs().max()

    npz = {
        "cfg": np.array([cfg.n_embd, cfg.n_head, cfg.n_layer, cfg.n_positions,
                         cfg.vocab_size], dtype=np.int64),
        "input_ids": ids.numpy(),
        "hf_loss": hf_loss.numpy(),
        "hf_logits": hf_logits.numpy(),
        "hf_wte_grad": wte_grad_hf.numpy(),
    }
    for lid, (f, gr) in enumerate(zip(flats, grads_o)):
        npz[f"flat_{lid}"] = f.numpy()
        npz[f"grad_{lid}"] = gr.numpy()
    GOLDEN_DIR.mkdir(parents=True, exist_ok=True)
    np.savez_compressed(GOLDEN_DIR / f"{case['name']}.npz", **npz)
    print(f"{case['name']}: loss={hf_loss.item():.6f} "
          f"max|logit diff|={(logits_o - hf_logits).abs().max():.3e}  OK")


def gen_adamw() -> None:
    """Pin the oracle's AdamW restatement against torch.optim.AdamW."""
    g = torch.Generator().manual_seed(3)
    n = 4097
    p0 = torch.randn(n, generator=g)
    hp = dict(lr=3e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01)
    p_t = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p_t], **hp)
    p_o, m_o, v_o = p0.clone(), torch.zeros(n), torch.zeros(n)
    grads = [torch.randn(n, generator=g) for _ in range(4)]
    for step, gr in enumerate(grads, start=1):
        p_t.grad = gr.clone()
        opt.step()
        adamw_step(p_o, gr, m_o, v_o, step, hp["lr"], hp["betas"][0],
                   hp["betas"][1], hp["eps"], hp["weight_decay"])
    assert torch.allclose(p_o, p_t.detach(), rtol=1e-6, atol=1e-7), \
        (p_o - p_t.detach()).abs().max()
    np.savez_compressed(
        GOLDEN_DIR / "adamw.npz",
        p0=p0.numpy(), grads=torch.stack(grads).numpy(),
        p_final=p_t.detach().numpy(), m_final=m_o.numpy(), v_final=v_o.numpy(),
        hyper=np.array([hp["lr"], 0.9, 0.999, hp["eps"], hp["weight_decay"]]))
    print(f"adamw: max|p diff| vs torch = {(p_o - p_t.detach()).abs().max():.3e}  OK")


if __name__ == "__main__":
    torch.set_num_threads(8)
    for case in CASES:
        gen_case(case)
    gen_adamw()
