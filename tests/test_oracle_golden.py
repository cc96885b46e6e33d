"""Pin the CPU oracle against golden vectors generated from the reference.

Goldens come from tests/golden/make_golden.py, which executes the
reference's own eager CPU code (build container only). Integer/index paths
must be bit-exact; fp paths are fp32-vs-fp32 and must match tightly.
"""

import torch

from oracle import losses, moe, norms


def test_scatter_index_bitexact(golden):
    for name in ("small", "tiny", "one_expert", "full"):
        idx = golden[f"scatter/{name}/expert_index"]
        E = int(golden[f"scatter/{name}/num_experts"])
        sorted_order, scatter_index = moe.compute_expert_scatter_index(idx)
        assert torch.equal(sorted_order, golden[f"scatter/{name}/sorted_order"]), name
        assert torch.equal(scatter_index, golden[f"scatter/{name}/scatter_index"]), name
        hist = moe.expert_histogram(idx, E)
        assert torch.equal(hist, golden[f"scatter/{name}/histogram"]), name


def test_eager_moe_matches_reference(golden):
    out = moe.eager_moe_forward(
        golden["moe/hidden"], golden["moe/top_i"], golden["moe/top_w"],
        golden["moe/gate_up_proj"], golden["moe/down_proj"],
    )
    torch.testing.assert_close(out, golden["moe/out"], rtol=1e-6, atol=1e-6)


def test_eager_moe_backward_matches_reference(golden):
    hidden = golden["moe/hidden"].clone().requires_grad_(True)
    gup = golden["moe/gate_up_proj"].clone().requires_grad_(True)
    down = golden["moe/down_proj"].clone().requires_grad_(True)
    out = moe.eager_moe_forward(hidden, golden["moe/top_i"], golden["moe/top_w"], gup, down)
    out.backward(golden["moe/dy"])
    torch.testing.assert_close(hidden.grad, golden["moe/dhidden"], rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(gup.grad, golden["moe/dgate_up"], rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(down.grad, golden["moe/ddown"], rtol=1e-5, atol=1e-6)


def test_router_matches_reference(golden):
    logits, top_w, top_i = moe.router(
        golden["moe/hidden"], golden["moe/router_weight"], top_k=2, norm_topk_prob=True
    )
    torch.testing.assert_close(logits, golden["moe/router_logits"])
    torch.testing.assert_close(top_w, golden["moe/top_w"])
    assert torch.equal(top_i, golden["moe/top_i"])


def test_fused_order_vs_eager_order(golden):
    """Fused math (weights before fc2) == eager math (weights after down_proj)
    in fp32 up to rounding — the documented operator-ordering equivalence
    (ref tests/ops/test_fused_moe_split_vs_merged.py:30-38)."""
    out = moe.fused_moe_forward(
        num_experts=golden["moe/gate_up_proj"].shape[0],
        routing_weights=golden["moe/top_w"],
        selected_experts=golden["moe/top_i"],
        hidden_states=golden["moe/hidden"],
        fc1_1_2_weight=golden["moe/gate_up_proj"],
        fc2_weight=golden["moe/down_proj"],
    )
    torch.testing.assert_close(out, golden["moe/out"], rtol=1e-4, atol=1e-5)


def test_rmsnorm_matches_reference(golden):
    y = norms.rms_norm(golden["rmsnorm/x"], golden["rmsnorm/w"], 1e-6)
    torch.testing.assert_close(y, golden["rmsnorm/y"])
    dx, dw = norms.rms_norm_bwd(golden["rmsnorm/dy"], golden["rmsnorm/x"], golden["rmsnorm/w"], 1e-6)
    torch.testing.assert_close(dx, golden["rmsnorm/dx"], rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(dw, golden["rmsnorm/dw"], rtol=1e-5, atol=1e-6)


def test_rope_matches_reference(golden):
    qe, ke = norms.apply_rotary_pos_emb(
        golden["rope/q"], golden["rope/k"], golden["rope/cos"], golden["rope/sin"]
    )
    torch.testing.assert_close(qe, golden["rope/qe"])
    torch.testing.assert_close(ke, golden["rope/ke"])


def test_cross_entropy_matches_reference(golden):
    # chunked fused-linear CE (reference default backend) vs unchunked oracle
    hs = golden["ce/hs"].clone().requires_grad_(True)
    w = golden["ce/w"].clone().requires_grad_(True)
    loss = losses.causal_lm_loss(hs, w, golden["ce/labels"], shift=True)
    torch.testing.assert_close(loss, golden["ce/loss"], rtol=1e-5, atol=1e-6)
    loss.backward()
    torch.testing.assert_close(hs.grad, golden["ce/dhs"], rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(w.grad, golden["ce/dw"], rtol=1e-4, atol=1e-6)

    # plain eager CE on logits
    lab2 = golden["ce/labels2"]
    l2 = losses.fixed_cross_entropy(
        golden["ce/logits"].float(), lab2, (lab2 != -100).sum()
    )
    torch.testing.assert_close(l2, golden["ce/loss2"])


def test_load_balancing_matches_reference(golden):
    gl = tuple(golden[f"lbl/gate_logits_{i}"] for i in range(3))
    torch.testing.assert_close(losses.load_balancing_loss(gl, 8, 2), golden["lbl/loss"])
    torch.testing.assert_close(
        losses.load_balancing_loss(gl, 8, 2, golden["lbl/mask"]), golden["lbl/loss_masked"]
    )


def test_sp_collator_matches_reference(golden):
    for sp in (2, 4):
        for rank in range(sp):
            ids = golden[f"spcol/{sp}/{rank}/ids"]
            lab_out = losses.sp_shift_pad_slice(ids.clone(), sp, rank, is_labels=True)
            ids_out = losses.sp_shift_pad_slice(ids.clone(), sp, rank, is_labels=False)
            assert torch.equal(lab_out, golden[f"spcol/{sp}/{rank}/labels_out"]), (sp, rank)
            assert torch.equal(ids_out, golden[f"spcol/{sp}/{rank}/ids_out"]), (sp, rank)


def test_grouped_gemm_oracle_selfconsistent(golden):
    """group_gemm_same_nk/mn oracle vs direct per-expert matmul and autograd."""
    torch.manual_seed(11)
    G, M, N, K = 4, 64, 24, 16
    counts = torch.tensor([10, 0, 34, 20])
    cumsum = counts.cumsum(0)
    a = torch.randn(M, K)
    b = torch.randn(G, N, K)
    c = moe.group_gemm_same_nk(a, b, cumsum, transpose_b=True)
    start = 0
    for g in range(G):
        end = int(cumsum[g])
        torch.testing.assert_close(c[start:end], a[start:end] @ b[g].t())
        start = end
    # same_mn wgrad vs autograd of the blockwise matmul
    gout = torch.randn(M, N)
    dW = moe.group_gemm_same_mn(gout, a, cumsum)  # dW[g] = gout_g^T a_g -> [G,N,K]
    start = 0
    for g in range(G):
        end = int(cumsum[g])
        torch.testing.assert_close(dW[g], gout[start:end].t() @ a[start:end])
        start = end
