"""HIP kernel parity tests vs plain PyTorch fp32 references. GPU-only."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from distributed_reinforcement_learning_amd import ops
    assert ops.available(), "HIP extension must be built on the GPU box"
    return ops.require_ext()


def test_normalize_f32(ext):
    x = torch.randint(0, 256, (3, 84, 84, 4), dtype=torch.uint8,
                      device="cuda")
    out = ext.normalize_frames_f32(x)
    ref = x.float() / 255.0
    assert torch.equal(out, ref)


def test_normalize_f32_odd_tail(ext):
    x = torch.randint(0, 256, (7, 11, 3), dtype=torch.uint8, device="cuda")
    out = ext.normalize_frames_f32(x)
    assert torch.equal(out, x.float() / 255.0)


def test_normalize_bf16(ext):
    x = torch.randint(0, 256, (5, 84, 84, 1), dtype=torch.uint8,
                      device="cuda")
    out = ext.normalize_frames_bf16(x)
    ref = (x.float() / 255.0).to(torch.bfloat16)
    assert out.dtype == torch.bfloat16
    assert torch.equal(out.view(torch.uint16), ref.view(torch.uint16))


def test_vtrace_scan_matches_cpu(ext):
    torch.manual_seed(0)
    B, T = 32, 18
    deltas = torch.randn(B, T, device="cuda")
    discounts = torch.rand(B, T, device="cuda") * 0.99
    cs = torch.rand(B, T, device="cuda")
    out = ext.vtrace_scan(deltas, discounts, cs)
    # CPU fp32 reference
    from distributed_reinforcement_learning_amd.ops.vtrace_op import vtrace_scan
    ref = vtrace_scan(deltas.cpu(), discounts.cpu(), cs.cpu())
    assert torch.allclose(out.cpu(), ref, atol=1e-5, rtol=1e-5)


def test_lstm_tail_forward_backward_parity(ext):
    torch.manual_seed(1)
    N, H = 64, 32
    gates = torch.randn(N, 4 * H, device="cuda", requires_grad=True)
    c_prev = torch.randn(N, H, device="cuda", requires_grad=True)

    from distributed_reinforcement_learning_amd.ops.lstm_op import lstm_fused_step
    h1, c1 = lstm_fused_step(gates, c_prev, 1.0)

    # eager fp32 reference with fresh leaves
    g2 = gates.detach().clone().requires_grad_(True)
    cp2 = c_prev.detach().clone().requires_grad_(True)
    i, g, f, o = g2.chunk(4, dim=1)
    c2 = torch.sigmoid(f + 1.0) * cp2 + torch.sigmoid(i) * torch.tanh(g)
    h2 = torch.sigmoid(o) * torch.tanh(c2)
    assert torch.allclose(h1, h2, atol=1e-5)
    assert torch.allclose(c1, c2, atol=1e-5)

    gh = torch.randn_like(h1)
    gc = torch.randn_like(c1)
    torch.autograd.backward([h1, c1], [gh, gc])
    torch.autograd.backward([h2, c2], [gh, gc])
    assert torch.allclose(gates.grad, g2.grad, atol=1e-4)
    assert torch.allclose(c_prev.grad, cp2.grad, atol=1e-4)


def test_sq_norm(ext):
    x = torch.randn(1_000_003, device="cuda")
    out = ext.sq_norm(x)
    assert torch.allclose(out.sum(), (x * x).sum(), rtol=1e-4)


def _cpu_rmsprop_golden(p, g, ms, clip, lr, rho, eps):
    norm = g.norm()
    scale = 1.0 if clip <= 0 else float(clip / torch.clamp(norm, min=clip))
    gc = g * scale
    ms2 = rho * ms + (1 - rho) * gc * gc
    p2 = p - lr * gc / torch.sqrt(ms2 + eps)
    return p2, ms2


def test_rmsprop_step_parity(ext):
    torch.manual_seed(2)
    n = 100_000
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda") * 50  # force clipping
    ms = torch.rand(n, device="cuda")
    p_ref, ms_ref = _cpu_rmsprop_golden(p.cpu(), g.cpu(), ms.cpu(),
                                        40.0, 1e-3, 0.99, 0.1)
    ext.rmsprop_step(p, g, ms, 40.0, 1e-3, 0.99, 0.1)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-6)
    assert torch.allclose(ms.cpu(), ms_ref, atol=1e-6)


def test_adam_step_parity(ext):
    torch.manual_seed(3)
    n = 50_000
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    lr_t = 1e-3 * (1 - 0.999) ** 0.5 / (1 - 0.9)
    # no clip
    gc = g.cpu()
    m_ref = 0.1 * gc
    v_ref = 0.001 * gc * gc
    p_ref = p.cpu() - lr_t * m_ref / (v_ref.sqrt() + 1e-8)
    ext.adam_step(p, g, m, v, -1.0, lr_t, 0.9, 0.999, 1e-8)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-6)
    assert torch.allclose(m.cpu(), m_ref, atol=1e-7)
    assert torch.allclose(v.cpu(), v_ref, atol=1e-7)


def test_fused_optimizer_gpu_matches_cpu_fallback():
    """Whole FusedRMSProp step on GPU vs the CPU golden implementation."""
    from distributed_reinforcement_learning_amd.ops.optim import FusedRMSProp
    torch.manual_seed(4)
    lin_cpu = torch.nn.Linear(64, 64)
    lin_gpu = torch.nn.Linear(64, 64)
    lin_gpu.load_state_dict(lin_cpu.state_dict())
    lin_gpu = lin_gpu.cuda()
    opt_cpu = FusedRMSProp(lin_cpu.parameters(), lr=1e-3, clip_norm=1.0)
    opt_gpu = FusedRMSProp(lin_gpu.parameters(), lr=1e-3, clip_norm=1.0)
    x = torch.randn(32, 64)
    for _ in range(3):
        opt_cpu.zero_grad()
        lin_cpu(x).pow(2).sum().backward()
        opt_cpu.step()
        opt_gpu.zero_grad()
        lin_gpu(x.cuda()).pow(2).sum().backward()
        opt_gpu.step()
    assert torch.allclose(opt_gpu.flat_params.cpu(), opt_cpu.flat_params,
                          atol=1e-4, rtol=1e-4)


def test_impala_gpu_train_step_finite():
    from distributed_reinforcement_learning_amd.agents import impala
    B, T, A, H = 4, 8, 18, 64
    agent = impala.Agent(
        trajectory=T, input_shape=[84, 84, 4], num_action=A,
        lstm_hidden_size=H, discount_factor=0.99, start_learning_rate=6e-4,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.05, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cuda:0", seed=0)
    rng = np.random.default_rng(0)
    for _ in range(2):
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


def test_impala_gpu_matches_cpu_reference_forward():
    """GPU bf16 unroll vs CPU fp32 unroll of the same weights: policies close."""
    import torch
    from distributed_reinforcement_learning_amd.models import ImpalaActorCritic
    torch.manual_seed(5)
    m_cpu = ImpalaActorCritic([84, 84, 4], 6, 32).eval()
    m_gpu = ImpalaActorCritic([84, 84, 4], 6, 32).eval()
    m_gpu.load_state_dict(m_cpu.state_dict())
    m_gpu = m_gpu.cuda()
    B, T = 2, 4
    s = torch.rand(B, T, 84, 84, 4)
    pa = torch.randint(0, 6, (B, T))
    h = torch.randn(B, T, 32) * 0.1
    c = torch.randn(B, T, 32) * 0.1
    with torch.no_grad():
        P_cpu, V_cpu = m_cpu.unroll(s, pa, h, c)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            P_gpu, V_gpu = m_gpu.unroll(s.cuda(), pa.cuda(), h.cuda(),
                                        c.cuda())
    assert torch.allclose(P_cpu, P_gpu.float().cpu(), atol=0.03)
    assert torch.allclose(V_cpu, V_gpu.float().cpu(), atol=0.15, rtol=0.1)


def test_embed_backward_matches_torch(ext):
    torch.manual_seed(9)
    A, H, N = 18, 256, 640
    table1 = torch.randn(A, H, device="cuda", requires_grad=True)
    table2 = table1.detach().clone().requires_grad_(True)
    idx = torch.randint(0, A, (N,), device="cuda")
    go = torch.randn(N, H, device="cuda")
    from distributed_reinforcement_learning_amd.ops.embed_op import embed_lookup
    embed_lookup(table1, idx).backward(go)
    torch.nn.functional.embedding(idx, table2).backward(go)
    assert torch.allclose(table1.grad, table2.grad, atol=1e-4, rtol=1e-4)
    # bf16 table path
    t3 = torch.randn(A, H, device="cuda").bfloat16().requires_grad_(True)
    embed_lookup(t3, idx).float().sum().backward()
    assert t3.grad.dtype == torch.bfloat16


def test_mixed_precision_rmsprop_master(ext):
    """bf16-model / fp32-master fused RMSProp vs manual fp32 math."""
    torch.manual_seed(7)
    n = 10_000
    master_ref = torch.randn(n)
    p = master_ref.bfloat16().cuda()
    master = master_ref.clone().cuda()
    g = (torch.randn(n) * 10).bfloat16().cuda()
    ms = torch.rand(n).cuda()
    lr_buf = torch.full((1,), 1e-3, device="cuda")
    ms_ref = ms.cpu().clone()
    ext.rmsprop_step_bf16_t(p, g, master, ms, 40.0, lr_buf, 0.99, 0.1,
                            None)
    # golden: clip on bf16 grads, update fp32 master, round to bf16
    gf = g.float().cpu()
    norm = gf.norm()
    scale = float(40.0 / torch.clamp(norm, min=40.0))
    gc = gf * scale
    ms2 = 0.99 * ms_ref + 0.01 * gc * gc
    master2 = master_ref - 1e-3 * gc / (ms2 + 0.1).sqrt()
    assert torch.allclose(master.cpu(), master2, atol=1e-5, rtol=1e-5)
    assert torch.allclose(ms.cpu(), ms2, atol=1e-5, rtol=1e-5)
    assert torch.equal(p.cpu(), master2.bfloat16())


def test_bf16_model_agents_train_on_gpu():
    """bf16-native agents: one train step each for apex and r2d2 on GPU."""
    from distributed_reinforcement_learning_amd.agents import apex, r2d2
    rng = np.random.default_rng(3)
    ag = apex.Agent(input_shape=[84, 84, 4], num_action=4,
                    discount_factor=0.99, gradient_clip_norm=40.0,
                    reward_clipping="abs_one", start_learning_rate=1e-4,
                    end_learning_rate=0.0, learning_frame=10 ** 9,
                    device="cuda:0", seed=0)
    assert next(ag.model.parameters()).dtype == torch.bfloat16
    N = 8
    loss, td = ag.train(
        rng.integers(0, 255, (N, 84, 84, 4), dtype=np.uint8),
        rng.integers(0, 255, (N, 84, 84, 4), dtype=np.uint8),
        rng.integers(0, 4, N), rng.integers(0, 4, N),
        rng.normal(size=N).astype(np.float32), np.zeros(N, bool))
    assert np.isfinite(loss) and np.isfinite(td).all()

    r = r2d2.Agent(seq_len=6, burn_in=2, input_shape=[84, 84, 1],
                   num_action=4, lstm_size=8, discount_factor=0.997,
                   start_learning_rate=1e-4, end_learning_rate=0.0,
                   learning_frame=10 ** 9, gradient_clip_norm=40.0,
                   device="cuda:0", seed=0)
    B, L = 2, 6
    loss, td = r.train(
        state=rng.integers(0, 255, (B, L, 84, 84, 1), dtype=np.uint8),
        previous_action=rng.integers(0, 4, (B, L)).astype(np.int32),
        action=rng.integers(0, 4, (B, L)).astype(np.int32),
        h=np.zeros((B, L, 8), np.float32), c=np.zeros((B, L, 8), np.float32),
        reward=rng.normal(size=(B, L)).astype(np.float32),
        done=np.zeros((B, L), bool), weight=np.ones(B, np.float32))
    assert np.isfinite(loss) and np.isfinite(td).all()


def test_lstm_seq_kernel_matches_torch_loop():
    """drla_lstm_seq_fwd (whole no-grad unroll in one kernel) vs the
    per-step torch loop with the same bf16 weights."""
    from distributed_reinforcement_learning_amd.models import R2D2LstmQ
    torch.manual_seed(11)
    m = R2D2LstmQ([84, 84, 1], 4, 64).cuda().bfloat16()
    B, L = 3, 9
    s = torch.randint(0, 256, (B, L, 84, 84, 1), dtype=torch.uint8,
                      device="cuda")
    pa = torch.randint(0, 4, (B, L), device="cuda")
    h0 = torch.randn(B, 64, device="cuda") * 0.1
    c0 = torch.randn(B, 64, device="cuda") * 0.1
    done = torch.zeros(B, L, dtype=torch.bool, device="cuda")
    done[:, 3] = True
    with torch.no_grad():
        q_kernel = m.unroll_sequence(s, pa, h0, c0, done)  # seq kernel
    # torch loop: force the eager path by enabling grad
    h0g = h0.clone().requires_grad_(True)
    q_loop = m.unroll_sequence(s, pa, h0g, c0, done)
    assert torch.allclose(q_kernel.float(), q_loop.detach().float(),
                          atol=5e-2, rtol=5e-2), \
        f"max err {(q_kernel.float()-q_loop.detach().float()).abs().max()}"

    # burn-in state recompute agrees with the carry of the loop
    with torch.no_grad():
        hb, cb = m.burn_in_states(s, pa, h0, c0, done)
    # recompute carry via loop
    feat = m.features(s.reshape(B * L, 84, 84, 1).contiguous(),
                      pa.reshape(-1)).reshape(B, L, -1)
    h, c = h0, c0
    for i in range(L):
        h, c = m.lstm(feat[:, i], h, c)
        keep = (~done[:, i]).to(h.dtype).unsqueeze(1)
        h, c = h * keep, c * keep
    assert torch.allclose(hb, h.float(), atol=3e-2)
    assert torch.allclose(cb, c.float(), atol=3e-2)


def test_fused_action_embed_parity(ext):
    from distributed_reinforcement_learning_amd.ops.embed_op import (
        fused_action_embed,
    )
    torch.manual_seed(11)
    N, A = 163, 18
    idx = torch.randint(0, A, (N,), device="cuda")
    mk = lambda *s: (torch.randn(*s, device="cuda") * 0.5).to(
        torch.bfloat16).requires_grad_(True)
    t1, b1, w2, b2 = mk(A, 256), mk(256), mk(256, 256), mk(256)
    t2 = t1.detach().clone().requires_grad_(True)
    b1r = b1.detach().clone().requires_grad_(True)
    w2r = w2.detach().clone().requires_grad_(True)
    b2r = b2.detach().clone().requires_grad_(True)

    out_f = fused_action_embed(idx, t1, b1, w2, b2)
    ref = torch.relu(torch.nn.functional.linear(
        torch.relu(t2.index_select(0, idx) + b1r), w2r, b2r))
    assert torch.allclose(out_f.float(), ref.float(), atol=0.05, rtol=0.05)

    g = torch.randn(N, 256, device="cuda")
    out_f.backward(g.to(torch.bfloat16))
    ref.backward(g.to(torch.bfloat16))
    for a, b, tol in [(t1.grad, t2.grad, 0.6), (b1.grad, b1r.grad, 0.5),
                      (w2.grad, w2r.grad, 0.6), (b2.grad, b2r.grad, 0.5)]:
        assert torch.allclose(a.float(), b.float(), atol=tol, rtol=0.05), \
            (a.float() - b.float()).abs().max()


@pytest.mark.parametrize("L", [9, 40, 80])  # 80 = BASELINE seq_len (config #4)
def test_lstm_seq_train_parity(ext, L):
    from distributed_reinforcement_learning_amd.ops.lstm_op import (
        lstm_seq_train,
    )
    torch.manual_seed(21)
    B, H = 5, 64
    xg = (torch.randn(B, L, 4 * H, device="cuda") * 0.5).to(
        torch.bfloat16).requires_grad_(True)
    wh = (torch.randn(H, 4 * H, device="cuda") * 0.2).to(
        torch.bfloat16).requires_grad_(True)
    h0 = torch.randn(B, H, device="cuda").requires_grad_(True)
    c0 = torch.randn(B, H, device="cuda").requires_grad_(True)
    done = torch.rand(B, L, device="cuda") < 0.15

    h_out, h_fin, c_fin = lstm_seq_train(xg, wh, h0, c0, done, 1.0)
    g = torch.randn_like(h_out)
    (h_out * g).sum().backward()

    # python reference (fp32 math on the same bf16 inputs)
    xg2 = xg.detach().clone().requires_grad_(True)
    wh2 = wh.detach().clone().requires_grad_(True)
    h02 = h0.detach().clone().requires_grad_(True)
    c02 = c0.detach().clone().requires_grad_(True)
    h, c = h02, c02
    outs = []
    for t in range(L):
        gates = xg2[:, t].float() + h.float() @ wh2.float()
        i, gg, f, o = gates.chunk(4, dim=1)
        c_new = torch.sigmoid(f + 1.0) * c + torch.sigmoid(i) * torch.tanh(gg)
        h_new = torch.sigmoid(o) * torch.tanh(c_new)
        outs.append(h_new)
        keep = (~done[:, t]).float().unsqueeze(1)
        h, c = h_new * keep, c_new * keep
    ref = torch.stack(outs, dim=1)
    (ref * g).sum().backward()

    assert torch.allclose(h_out, ref, atol=2e-2, rtol=2e-2)
    assert torch.allclose(h_fin, h, atol=2e-2, rtol=2e-2)
    assert torch.allclose(c_fin, c, atol=3e-2, rtol=3e-2)
    assert torch.allclose(xg.grad.float(), xg2.grad.float(), atol=0.05,
                          rtol=0.05)
    assert torch.allclose(wh.grad.float(), wh2.grad.float(), atol=0.35,
                          rtol=0.1), (wh.grad.float() -
                                      wh2.grad.float()).abs().max()
    assert torch.allclose(h0.grad, h02.grad, atol=0.05, rtol=0.05)
    assert torch.allclose(c0.grad, c02.grad, atol=0.05, rtol=0.05)


def test_aug_gate_gemm_matches_addmm(ext):
    """blocks._AugGateWeight: the ones-column + adjacent-slot augmented
    weight view must reproduce addmm(bias, xh, W) exactly (fwd) and route
    dbias through the augmented dW row (bwd)."""
    from distributed_reinforcement_learning_amd.models.blocks import (
        LSTMCellTF, _AugGateWeight,
    )
    from distributed_reinforcement_learning_amd.ops.optim import (
        FusedRMSProp,
    )
    torch.manual_seed(3)
    cell = LSTMCellTF(24, 16).cuda().bfloat16()
    # heads follow the cell in the real models — the augmented view reads
    # PAD-1 alias rows past the bias, so the flat buffer must extend
    # (the storage guard rejects a cell-only buffer, by design)
    tail = torch.nn.Linear(64, 64).cuda().bfloat16()
    opt = FusedRMSProp(list(cell.parameters()) + list(tail.parameters()),
                       lr=1e-3)  # re-homes into one flat buffer
    assert cell._aug_weight_ok()
    N = 33
    xh = (torch.randn(N, 40, device="cuda") * 0.5).to(torch.bfloat16)
    pad = cell._ones_col(N, torch.bfloat16, xh.device)
    xh_aug = torch.cat([xh, pad], dim=1)
    g_aug = torch.mm(xh_aug, _AugGateWeight.apply(cell.weight, cell.bias))
    g_ref = torch.addmm(cell.bias, xh, cell.weight)
    assert torch.allclose(g_aug.float(), g_ref.float(), atol=2e-2,
                          rtol=2e-2)
    # backward: dbias must land in the bias param via the augmented row
    opt.enable_scatter_grads()
    g_aug.float().sum().backward()
    assert cell.bias.grad is not None
    assert cell.bias.grad.shape == cell.bias.shape
    assert torch.allclose(cell.bias.grad.float(),
                          torch.full_like(cell.bias.grad.float(), N),
                          rtol=0.05)
