"""nn synchronization layer (reference test/nn.lua semantics — SURVEY.md
section 4: same minibatch on all ranks => synchronized grads equal
W x single-rank grads; synchronizeParameters equalizes replicas)."""
import torch

from .util import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 8), torch.nn.ReLU(),
        torch.nn.Linear(8, 4),
    )


def _sync_params_body(tm):
    # Different seeds -> different replicas; after synchronize_parameters
    # all replicas must equal rank 0's.
    model = _make_model(seed=tm.rank())
    tm.nn.synchronize_parameters(model)
    ref = _make_model(seed=0)
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p.data, q.data)


def test_synchronize_parameters():
    run_distributed(_sync_params_body, world=4)


def _sync_grads_body(tm):
    W = tm.size()
    model = _make_model(seed=0)
    torch.manual_seed(42)  # same minibatch everywhere
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    single = [p.grad.clone() for p in model.parameters()]
    tm.nn.synchronize_gradients(model, average=False)
    for p, g in zip(model.parameters(), single):
        assert torch.allclose(p.grad, g * W, rtol=1e-5), \
            (p.grad.flatten()[:3], (g * W).flatten()[:3])


def test_synchronize_gradients():
    run_distributed(_sync_grads_body, world=4)


def _overlap_engine_body(tm):
    from torchmpi_amd.nn import GradientSynchronizer
    W, r = tm.size(), tm.rank()
    model = _make_model(seed=0)
    sync = GradientSynchronizer(model, bucket_mb=1, average=True)
    # Rank-dependent minibatch; compare against the manual average of grads.
    torch.manual_seed(100 + r)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)

    ref_model = _make_model(seed=0)
    grads_per_rank = []
    for k in range(W):
        torch.manual_seed(100 + k)
        xk = torch.randn(8, 16)
        yk = torch.randn(8, 4)
        ref_model.zero_grad()
        torch.nn.functional.mse_loss(ref_model(xk), yk).backward()
        grads_per_rank.append([p.grad.clone() for p in ref_model.parameters()])
    expect = [torch.stack([g[i] for g in grads_per_rank]).mean(0)
              for i in range(len(grads_per_rank[0]))]

    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    sync.finish()
    for p, e in zip(model.parameters(), expect):
        assert torch.allclose(p.grad, e, rtol=1e-4, atol=1e-6), \
            (p.grad.flatten()[:3], e.flatten()[:3])

    # Second step must work with the same bucket state (zero_grad keeps
    # tensors in place).
    for p in model.parameters():
        p.grad.zero_()
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    sync.finish()
    for p, e in zip(model.parameters(), expect):
        assert torch.allclose(p.grad, e, rtol=1e-4, atol=1e-6)


def test_gradient_synchronizer_overlap():
    run_distributed(_overlap_engine_body, world=2)


def _accumulation_body(tm):
    """no_sync() gradient accumulation: K-1 local backwards + 1 synced
    backward == allreduce of the K-batch summed grads."""
    from torchmpi_amd.nn import GradientSynchronizer
    W, r = tm.size(), tm.rank()
    K = 3
    model = _make_model(seed=0)
    sync = GradientSynchronizer(model, average=False)

    # reference: sum grads of all K*W microbatches on one replica
    ref_model = _make_model(seed=0)
    for k in range(W * K):
        torch.manual_seed(500 + k)
        xk, yk = torch.randn(8, 16), torch.randn(8, 4)
        torch.nn.functional.mse_loss(ref_model(xk), yk).backward()

    for k in range(K):
        torch.manual_seed(500 + r * K + k)
        x, y = torch.randn(8, 16), torch.randn(8, 4)
        loss = torch.nn.functional.mse_loss(model(x), y)
        if k < K - 1:
            with sync.no_sync():
                loss.backward()
        else:
            loss.backward()
    sync.finish()
    for p, q in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p.grad, q.grad, rtol=1e-4, atol=1e-6), \
            (p.grad.flatten()[:3], q.grad.flatten()[:3])


def test_gradient_accumulation_no_sync():
    run_distributed(_accumulation_body, world=2)


def _double_backward_error_body(tm):
    """Accumulating without no_sync() must raise loudly, not silently
    compute wrong gradients (advisor finding, round 1)."""
    from torchmpi_amd.nn import GradientSynchronizer
    model = _make_model(seed=0)
    GradientSynchronizer(model, average=False)
    x, y = torch.randn(8, 16), torch.randn(8, 4)
    torch.nn.functional.mse_loss(model(x), y).backward()
    try:
        torch.nn.functional.mse_loss(model(x), y).backward()
    except RuntimeError as e:
        assert "no_sync" in str(e), e
        return
    raise AssertionError("second backward without no_sync did not raise")


def test_double_backward_raises():
    run_distributed(_double_backward_error_body, world=2)


def _e2e_training_body(tm):
    """End-to-end DP-SGD equivalence (reference stack 3.5, examples/mnist):
    W-rank data-parallel training with averaged grads == single-process
    training on the concatenated batch."""
    from torchmpi_amd.nn import GradientSynchronizer, synchronize_parameters
    W, r = tm.size(), tm.rank()
    model = _make_model(seed=7)
    synchronize_parameters(model)
    sync = GradientSynchronizer(model, average=True)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)

    ref = _make_model(seed=7)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)

    for step in range(3):
        torch.manual_seed(1000 + step)
        xs = torch.randn(W, 8, 16)
        ys = torch.randn(W, 8, 4)
        # distributed step
        opt.zero_grad(set_to_none=False)
        loss = torch.nn.functional.mse_loss(model(xs[r]), ys[r])
        loss.backward()
        sync.finish()
        opt.step()
        # reference step on the full batch
        ref_opt.zero_grad()
        ref_loss = torch.nn.functional.mse_loss(
            ref(xs.view(-1, 16)), ys.view(-1, 4))
        ref_loss.backward()
        ref_opt.step()

    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, q.data, rtol=1e-4, atol=1e-6)


def test_e2e_dp_training_matches_single_process():
    run_distributed(_e2e_training_body, world=2)


def _bf16_comm_dtype_body(tm):
    """The bench's actual configuration: fp32 params, bf16 bucket compression
    (cast-fused pack/unpack), averaged — CPU-path equivalence with tolerance."""
    from torchmpi_amd.nn import GradientSynchronizer
    W, r = tm.size(), tm.rank()
    model = _make_model(seed=3)
    sync = GradientSynchronizer(model, average=True, comm_dtype=torch.bfloat16)
    torch.manual_seed(50 + r)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    fp32_grads = [p.grad.clone() for p in model.parameters()]
    # reference: average of each rank's fp32 grads (recompute locally)
    ref_model = _make_model(seed=3)
    acc = None
    for k in range(W):
        torch.manual_seed(50 + k)
        xk, yk = torch.randn(8, 16), torch.randn(8, 4)
        ref_model.zero_grad()
        torch.nn.functional.mse_loss(ref_model(xk), yk).backward()
        g = [p.grad.clone() for p in ref_model.parameters()]
        acc = g if acc is None else [a + b for a, b in zip(acc, g)]
    expect = [a / W for a in acc]
    sync.finish()
    for p, e in zip(model.parameters(), expect):
        # bf16 wire: ~3 rounding steps (pack, reduce, unpack)
        assert torch.allclose(p.grad, e, rtol=0.1, atol=0.05), \
            (p.grad.flatten()[:3], e.flatten()[:3])



def test_bf16_comm_dtype():
    run_distributed(_bf16_comm_dtype_body, world=2)


def _llama_dp_equivalence_body(tm):
    """Flagship-transformer DP step equivalence: W-rank training with
    averaged grads == single-process training on the concatenated batch
    (reference stack 3.5, applied to the Llama model family)."""
    from torchmpi_amd.models import LlamaModel, llama_debug
    from torchmpi_amd.nn import GradientSynchronizer, synchronize_parameters
    W, r = tm.size(), tm.rank()
    cfg = llama_debug()
    cfg.n_layers = 2
    torch.manual_seed(11)
    model = LlamaModel(cfg)
    synchronize_parameters(model)
    sync = GradientSynchronizer(model, average=True)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)

    torch.manual_seed(11)
    ref = LlamaModel(cfg)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)

    for step in range(2):
        torch.manual_seed(300 + step)
        toks = torch.randint(0, cfg.vocab_size, (W, 2, 32))
        tgts = torch.randint(0, cfg.vocab_size, (W, 2, 32))
        opt.zero_grad(set_to_none=False)
        loss = model(toks[r], tgts[r])
        loss.backward()
        sync.finish()
        opt.step()

        ref_opt.zero_grad()
        ref_loss = sum(ref(toks[k], tgts[k]) for k in range(W)) / W
        ref_loss.backward()
        ref_opt.step()

    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, q.data, atol=1e-5, rtol=1e-4), \
            (p.shape, (p.data - q.data).abs().max())


def test_llama_dp_equivalence():
    run_distributed(_llama_dp_equivalence_body, world=2)
