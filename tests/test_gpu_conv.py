"""Custom MFMA conv stack vs torch/MIOpen references. GPU-only."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from distributed_reinforcement_learning_amd import ops
    assert ops.available()
    return ops.require_ext()


def test_mfma_probe_layout(ext):
    """Pin the 16x16x32 bf16 fragment maps with asymmetric operands
    (guide §3: symmetric inputs hide operand/output transposes)."""
    torch.manual_seed(0)
    A = torch.randn(16, 32, device="cuda").bfloat16()
    B = (torch.arange(32 * 16, device="cuda").reshape(32, 16).float()
         * 0.01 + torch.randn(32, 16, device="cuda")).bfloat16()
    D = ext.mfma_probe(A.view(torch.uint16).contiguous(),
                       B.view(torch.uint16).contiguous())
    ref = A.float() @ B.float()
    assert torch.allclose(D, ref, atol=0.15, rtol=1e-2), \
        f"max err {(D - ref).abs().max().item()}"


def _torch_conv_nhwc(x_f32_nhwc, conv, relu=True):
    w = conv.weight.float()
    b = conv.bias.float()
    y = F.conv2d(x_f32_nhwc.permute(0, 3, 1, 2), w, b,
                 stride=conv.stride)
    if relu:
        y = F.relu(y)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize("layer,ci,co,k,s,hi,ho,u8", [
    (0, 4, 32, 8, 4, 84, 20, True),
    (1, 1, 32, 8, 4, 84, 20, True),
    (2, 32, 64, 4, 2, 20, 9, False),
    (3, 64, 64, 3, 1, 9, 7, False),
])
def test_conv_fwd_layer_parity(ext, layer, ci, co, k, s, hi, ho, u8):
    torch.manual_seed(layer)
    N = 5
    conv = torch.nn.Conv2d(ci, co, k, stride=s).cuda()
    conv_bf = torch.nn.Conv2d(ci, co, k, stride=s).cuda()
    conv_bf.load_state_dict(conv.state_dict())
    conv_bf = conv_bf.to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    if u8:
        x_u8 = torch.randint(0, 256, (N, hi, hi, ci), dtype=torch.uint8,
                             device="cuda")
        x_in = x_u8
        x_ref = x_u8.float() / 255.0
    else:
        x_in = torch.randn(N, hi, hi, ci, device="cuda").bfloat16() * 0.5
        x_ref = x_in.float()
    w_flat = conv_bf.weight.permute(0, 2, 3, 1).reshape(co, -1).contiguous()
    y, _ = ext.conv_fwd(layer, x_in.contiguous(), w_flat,
                        conv_bf.bias.contiguous(), False)
    ref = _torch_conv_nhwc(x_ref, conv)
    err = (y.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=1e-3)
    assert err.mean() / scale < 0.03, \
        f"layer {layer}: rel mean err {(err.mean()/scale).item():.4f}"
    assert (y.float() >= 0).all()  # fused ReLU


def test_full_stack_forward_and_backward_parity():
    """Custom stack (fwd+bwd) vs torch fp32 reference with identical
    weights; gradients compared with bf16 tolerances."""
    from distributed_reinforcement_learning_amd.models.blocks import AtariConvStack
    torch.manual_seed(1)
    N = 6
    stack_ref = AtariConvStack(4).cuda()  # f32, torch path
    stack_bf = AtariConvStack(4).cuda()
    stack_bf.load_state_dict(stack_ref.state_dict())
    stack_bf = stack_bf.to(torch.bfloat16)
    for c in (stack_bf.conv1, stack_bf.conv2, stack_bf.conv3):
        c.to(memory_format=torch.channels_last)

    x_u8 = torch.randint(0, 256, (N, 84, 84, 4), dtype=torch.uint8,
                         device="cuda")
    y_bf = stack_bf(x_u8)  # custom MFMA path (uint8 trigger)
    y_ref = stack_ref(x_u8.float() / 255.0)
    rel = (y_bf.float() - y_ref).abs().mean() / \
        y_ref.abs().mean().clamp(min=1e-3)
    assert rel < 0.05, f"fwd rel err {rel.item():.4f}"

    # backward parity on conv3 weight grads (largest layer)
    g = torch.randn_like(y_ref)
    y_bf.backward(g.to(y_bf.dtype))
    y_ref.backward(g)
    for name, c_bf, c_ref in [("c3", stack_bf.conv3, stack_ref.conv3),
                              ("c2", stack_bf.conv2, stack_ref.conv2),
                              ("c1", stack_bf.conv1, stack_ref.conv1)]:
        gw_bf = c_bf.weight.grad.float()
        gw_ref = c_ref.weight.grad
        rel = (gw_bf - gw_ref).abs().mean() / \
            gw_ref.abs().mean().clamp(min=1e-4)
        assert rel < 0.08, f"{name} wgrad rel err {rel.item():.4f}"
        gb_rel = (c_bf.bias.grad.float() - c_ref.bias.grad).abs().mean() / \
            c_ref.bias.grad.abs().mean().clamp(min=1e-4)
        assert gb_rel < 0.12, f"{name} bias grad rel err {gb_rel.item():.4f}"


def test_impala_agent_uses_custom_conv_and_trains():
    from distributed_reinforcement_learning_amd.agents import impala
    agent = impala.Agent(
        trajectory=8, input_shape=[84, 84, 4], num_action=6,
        lstm_hidden_size=16, discount_factor=0.99, start_learning_rate=1e-3,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.05, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cuda:0", seed=0)
    rng = np.random.default_rng(0)
    B, T, A, H = 4, 8, 6, 16
    out = agent.train(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=np.zeros((B, T), dtype=bool),
        behavior_policy=np.full((B, T, A), 1 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=np.zeros((B, T, H), dtype=np.float32),
        initial_c=np.zeros((B, T, H), dtype=np.float32))
    assert all(np.isfinite(v) for v in out)
