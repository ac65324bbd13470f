# @gpu layer- and model-level parity: the C-ABI layer objects (the product
# hot path) against the oracle restatement on identical params/inputs.
# These are the parity tests proper (tier contract ③): they call through
# the C-ABI via oobleck_amd.layer.Layer.
import ctypes

import pytest
import torch

from oracle.gpt2_oracle import OracleConfig, adamw_step, layer_forward
from oracle.gpt2_oracle import stage_forward_backward

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")
DEV = torch.device("cuda:0")

TINY24 = dict(n_embd=96, n_head=4, n_layer=3, n_positions=64, vocab_size=211)
TINY64 = dict(n_embd=128, n_head=2, n_layer=2, n_positions=96, vocab_size=307)
SMALLBLK = dict(n_embd=768, n_head=12, n_layer=2, n_positions=1024,
                vocab_size=50257)


def make_layer(mc, lid, flat, max_batch, seq, n_slots=2):
    from oobleck_amd.layer import Layer
    layer = Layer(lid, mc, max_batch, seq, n_slots, DEV)
    layer.flat_param.copy_(flat.to(DEV))
    return layer


def cfgs(d):
    from oobleck_amd.config import ModelConfig
    return ModelConfig(**d), OracleConfig(**d)


def flats_for(oc, seed=42):
    from oracle.gpt2_oracle import init_layer_params
    return [init_layer_params(oc, oc.layer_kind(i), seed * 100 + i)
            for i in range(oc.n_layers_total)]


@requires_gpu
@pytest.mark.parametrize("dims,B,S", [(TINY24, 2, 48), (TINY64, 2, 96),
                                      (SMALLBLK, 2, 256)])
def test_block_layer_parity(dims, B, S):
    mc, oc = cfgs(dims)
    flat = flats_for(oc)[1]
    layer = make_layer(mc, 1, flat, B, S)
    g = torch.Generator().manual_seed(5)
    x = (torch.randn(B, S, oc.n_embd, generator=g) * 0.5)
    dout = torch.randn(B, S, oc.n_embd, generator=g) * 0.1

    xg = x.to(DEV)
    out = torch.empty_like(xg)
    layer.set_batch(B)
    layer.forward_slot(0, xg, out)
    din = torch.empty_like(xg)
    layer.backward_slot(0, dout.to(DEV), din)
    torch.cuda.synchronize()

    ref_out, ref_dx, (ref_grad,) = stage_forward_backward(
        oc, [flat], [1], x, dout=dout)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(din.cpu(), ref_dx, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(layer.flat_grad.cpu(), ref_grad,
                               rtol=1e-3, atol=1e-3)


@requires_gpu
def test_embed_and_final_layer_parity():
    mc, oc = cfgs(TINY24)
    flats = flats_for(oc)
    B, S = 2, 48
    g = torch.Generator().manual_seed(6)
    ids = torch.randint(0, oc.vocab_size, (B, S), generator=g)
    labels = ids.clone()

    # embedding
    emb = make_layer(mc, 0, flats[0], B, S)
    out0 = torch.empty(B, S, oc.n_embd, device=DEV)
    emb.forward_slot(0, ids.to(DEV), out0)
    ref0 = layer_forward(oc, 0, flats[0], ids)
    torch.cuda.synchronize()
    torch.testing.assert_close(out0.cpu(), ref0, rtol=1e-5, atol=1e-6)

    dout = torch.randn(B, S, oc.n_embd, generator=g) * 0.1
    emb.backward_slot(0, dout.to(DEV), None)
    _, _, (ref_grad0,) = stage_forward_backward(oc, [flats[0]], [0], ids,
                                                dout=dout)
    torch.cuda.synchronize()
    torch.testing.assert_close(emb.flat_grad.cpu(), ref_grad0,
                               rtol=1e-4, atol=1e-4)

    # final (ln_f + lm_head + shifted CE)
    fin_lid = oc.n_layers_total - 1
    fin = make_layer(mc, fin_lid, flats[fin_lid], B, S)
    x = (torch.randn(B, S, oc.n_embd, generator=g) * 0.5)
    loss = torch.zeros(1, device=DEV)
    fin.forward_slot(0, x.to(DEV), loss, labels.to(DEV))
    din = torch.empty(B, S, oc.n_embd, device=DEV)
    fin.backward_slot(0, None, din)
    torch.cuda.synchronize()

    ref_loss, ref_dx, (ref_gradf,) = stage_forward_backward(
        oc, [flats[fin_lid]], [fin_lid], x, labels=labels)
    torch.testing.assert_close(loss.cpu()[0], ref_loss, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(din.cpu(), ref_dx, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(fin.flat_grad.cpu(), ref_gradf,
                               rtol=1e-3, atol=1e-4)


@requires_gpu
@pytest.mark.parametrize("dims,B,S", [(TINY24, 2, 48), (TINY64, 2, 96)])
def test_full_model_single_gpu_parity(dims, B, S):
    """Whole model on one GPU as a chain of C-ABI layers: loss + every
    layer's flat grad vs the oracle, then one fused AdamW step vs the
    oracle's AdamW."""
    mc, oc = cfgs(dims)
    flats = flats_for(oc)
    L = oc.n_layers_total
    layers = [make_layer(mc, lid, flats[lid], B, S) for lid in range(L)]
    g = torch.Generator().manual_seed(7)
    ids = torch.randint(0, oc.vocab_size, (B, S), generator=g)
    labels = ids.clone()

    # forward chain
    x = ids.to(DEV)
    acts = []
    for lid, layer in enumerate(layers):
        layer.set_batch(B)
        if lid == L - 1:
            out = torch.zeros(1, device=DEV)
            layer.forward_slot(0, x, out, labels.to(DEV))
        else:
            out = torch.empty(B, S, oc.n_embd, device=DEV)
            layer.forward_slot(0, x, out)
        acts.append(out)
        x = out
    # backward chain
    dout = None
    for lid in range(L - 1, -1, -1):
        din = None if lid == 0 else torch.empty(B, S, oc.n_embd, device=DEV)
        layers[lid].backward_slot(0, dout, din)
        dout = din
    torch.cuda.synchronize()

    ref_loss, _, ref_grads = stage_forward_backward(
        oc, flats, list(range(L)), ids, labels=labels)
    torch.testing.assert_close(acts[-1].cpu()[0], ref_loss,
                               rtol=1e-4, atol=1e-5)
    for lid, layer in enumerate(layers):
        torch.testing.assert_close(layer.flat_grad.cpu(), ref_grads[lid],
                                   rtol=1e-3, atol=1e-3)

    # fused AdamW step parity: oracle Adam on the GPU's OWN grads (grad
    # parity is asserted above; at step 1 Adam ≈ lr*sign(g), so feeding the
    # oracle a slightly different grad would amplify fp noise to 2*lr)
    gpu_grads = [layer.flat_grad.cpu().clone() for layer in layers]
    from oobleck_amd.optimizer import FusedAdamW
    opt = FusedAdamW(layers, lr=1e-3, weight_decay=0.01)
    opt.step()
    torch.cuda.synchronize()
    for lid, layer in enumerate(layers):
        p = flats[lid].clone()
        m = torch.zeros_like(p)
        v = torch.zeros_like(p)
        adamw_step(p, gpu_grads[lid], m, v, 1, 1e-3, weight_decay=0.01)
        torch.testing.assert_close(layer.flat_param.cpu(), p,
                                   rtol=1e-4, atol=1e-5)


@requires_gpu
def test_gpt2_small_block_shape_runs():
    """One real GPT-2-small-shaped block (H=768, S=1024, B=8 — the gpt2.yaml
    microbatch) forwards+backwards without error and with finite outputs."""
    mc, oc = cfgs(SMALLBLK)
    flats = flats_for(oc)
    B, S = 8, 1024
    layer = make_layer(mc, 1, flats[1], B, S)
    x = torch.randn(B, S, 768, device=DEV) * 0.5
    out = torch.empty_like(x)
    layer.forward_slot(0, x, out)
    din = torch.empty_like(x)
    layer.backward_slot(0, torch.randn_like(x) * 0.01, din)
    torch.cuda.synchronize()
    assert torch.isfinite(out).all()
    assert torch.isfinite(din).all()
    assert torch.isfinite(layer.flat_grad).all()


@requires_gpu
@pytest.mark.parametrize("dtype", ["f32", "bf16"])
def test_xl_dims_block_parity(dtype):
    """One GPT-2-XL-dimension block (H=1600, 25 heads — examples/gpt3.yaml,
    BASELINE config[3]) forward+backward parity at B=2, S=256."""
    from oobleck_amd.config import ModelConfig
    from oobleck_amd.layer import Layer
    xl = dict(n_embd=1600, n_head=25, n_layer=2, n_positions=1024,
              vocab_size=50257)
    mc = ModelConfig(**xl)
    oc = OracleConfig(**xl)
    B, S = 2, 256
    from oracle.gpt2_oracle import init_layer_params
    flat = init_layer_params(oc, 1, 99)
    layer = Layer(1, mc, B, S, 1, DEV, dtype=dtype)
    layer.flat_param.copy_(flat.to(DEV))
    layer.refresh_weights()
    g = torch.Generator().manual_seed(12)
    x = torch.randn(B, S, 1600, generator=g) * 0.5
    dout = torch.randn(B, S, 1600, generator=g) * 0.1
    adt = torch.float32 if dtype == "f32" else torch.bfloat16
    xg = x.to(DEV).to(adt)
    out = torch.empty_like(xg)
    layer.forward_slot(0, xg, out)
    din = torch.empty_like(xg)
    layer.backward_slot(0, dout.to(DEV).to(adt), din)
    torch.cuda.synchronize()
    ref_out, ref_dx, (ref_grad,) = stage_forward_backward(
        oc, [flat], [1], x, dout=dout)
    if dtype == "f32":
        torch.testing.assert_close(out.cpu(), ref_out, rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(layer.flat_grad.cpu(), ref_grad,
                                   rtol=1e-3, atol=1e-3)
    else:
        def rel_l2(a, b):
            return ((a - b).norm() / b.norm().clamp_min(1e-12)).item()
        assert rel_l2(out.float().cpu(), ref_out) < 5e-2
        assert rel_l2(layer.flat_grad.cpu(), ref_grad) < 8e-2


@requires_gpu
def test_grad_accumulation_two_microbatches():
    """Backward twice (two microbatches in two slots) must accumulate grads
    exactly like the oracle's sum — the 1F1B per-step semantics."""
    mc, oc = cfgs(TINY24)
    flat = flats_for(oc)[1]
    layer = make_layer(mc, 1, flat, 2, 48, n_slots=2)
    g = torch.Generator().manual_seed(21)
    ref_sum = None
    for slot in range(2):
        x = torch.randn(2, 48, oc.n_embd, generator=g) * 0.5
        dout = torch.randn(2, 48, oc.n_embd, generator=g) * 0.1
        out = torch.empty(2, 48, oc.n_embd, device=DEV)
        layer.forward_slot(slot, x.to(DEV), out)
        din = torch.empty_like(out)
        layer.backward_slot(slot, dout.to(DEV), din)
        _, _, (gr,) = stage_forward_backward(oc, [flat], [1], x, dout=dout)
        ref_sum = gr if ref_sum is None else ref_sum + gr
    torch.cuda.synchronize()
    torch.testing.assert_close(layer.flat_grad.cpu(), ref_sum,
                               rtol=1e-3, atol=1e-3)


@requires_gpu
def test_variable_batch_set_batch():
    """set_batch(B < max_batch) must compute the smaller microbatch
    correctly (heterogeneous-pipeline microbatch sizes)."""
    mc, oc = cfgs(TINY24)
    flat = flats_for(oc)[1]
    layer = make_layer(mc, 1, flat, 4, 48)  # max_batch 4
    g = torch.Generator().manual_seed(22)
    x = torch.randn(2, 48, oc.n_embd, generator=g) * 0.5
    layer.set_batch(2)
    out = torch.empty(2, 48, oc.n_embd, device=DEV)
    layer.forward_slot(0, x.to(DEV), out)
    ref = layer_forward(oc, 1, flat, x)
    torch.cuda.synchronize()
    torch.testing.assert_close(out.cpu(), ref, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_training_loss_decreases():
    """Three full train steps (fwd+bwd+AdamW) on a fixed batch must reduce
    the loss — the end-to-end sanity the reference never asserts."""
    mc, oc = cfgs(TINY24)
    flats = flats_for(oc)
    L = oc.n_layers_total
    layers = [make_layer(mc, lid, flats[lid], 2, 48, n_slots=1)
              for lid in range(L)]
    from oobleck_amd.optimizer import FusedAdamW
    opt = FusedAdamW(layers, lr=1e-3)
    g = torch.Generator().manual_seed(23)
    ids = torch.randint(0, oc.vocab_size, (2, 48), generator=g).to(DEV)
    losses = []
    for _ in range(3):
        x = ids
        for lid, layer in enumerate(layers):
            if lid == L - 1:
                out = torch.zeros(1, device=DEV)
                layer.forward_slot(0, x, out, ids)
            else:
                out = torch.empty(2, 48, oc.n_embd, device=DEV)
                layer.forward_slot(0, x, out)
            x = out
        dout = None
        for lid in range(L - 1, -1, -1):
            din = None if lid == 0 else torch.empty(2, 48, oc.n_embd,
                                                    device=DEV)
            layers[lid].backward_slot(0, dout, din)
            dout = din
        opt.step()
        for layer in layers:
            layer.zero_grads()
        losses.append(x.item())
    assert losses[2] < losses[0], losses
