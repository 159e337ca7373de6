"""DAG engine unit test — the reference's own DAG test shape
(main.cpp:80-116: logistic-regression graph, 20 fwd/bwd iterations,
decreasing loss)."""

import pytest
import torch

from lightctr_amd.engine.dag import (ActivationOp, AddOp, AggregateNode,
                                     DAGPipeline, LossOp, MatmulOp,
                                     MultiplyOp, SourceNode, TerminusNode,
                                     TrainableNode)


def test_dag_lr_converges():
    g = torch.Generator().manual_seed(0)
    X = torch.randn(128, 4, generator=g)
    w_true = torch.tensor([[1.0], [-2.0], [0.5], [1.5]])
    y = (torch.sigmoid(X @ w_true) > 0.5).float()

    w = TrainableNode(torch.zeros(4, 1), lr=0.5)
    b = TrainableNode(torch.zeros(1, 1), lr=0.5)
    x = SourceNode(X)
    label = SourceNode(y)
    z = AddOp(MatmulOp(x, w), b)
    p = ActivationOp(z, "sigmoid")
    loss_node = LossOp(p, label, "logistic")
    term = TerminusNode(loss_node)
    pipe = DAGPipeline().add_flow(x, w, b, label, term)

    losses = []
    for _ in range(20):
        losses.append(float(pipe.step()))
    assert losses[-1] < losses[0] * 0.6, losses
    # learned weights correlate with truth
    corr = torch.cosine_similarity(w.value_init.flatten(),
                                   w_true.flatten(), dim=0)
    assert corr > 0.9


def test_dag_fanout_computes_once():
    calls = {"n": 0}

    class CountingOp(AddOp):
        def forward_compute(self, a, b):
            calls["n"] += 1
            return super().forward_compute(a, b)

    a = SourceNode(torch.ones(3))
    b = SourceNode(torch.ones(3) * 2)
    shared = CountingOp(a, b)  # fan-out: consumed twice
    m = MultiplyOp(shared, shared)
    term = TerminusNode(m)
    DAGPipeline().add_flow(a, b, term)
    out = term.run_flow()
    assert torch.allclose(out, torch.full((3,), 9.0))
    assert calls["n"] == 1  # deal_flag semantics: one execution per flow


def test_dag_hadamard_and_aggregate_grads():
    a = TrainableNode(torch.tensor([2.0, 3.0]), lr=0.0)
    bsrc = SourceNode(torch.tensor([4.0, 5.0]))
    m = MultiplyOp(a, bsrc)
    agg = AggregateNode(m, a)
    term = TerminusNode(agg)
    pipe = DAGPipeline().add_flow(a, bsrc, term)
    out = pipe.run_forward()
    assert torch.allclose(out, torch.tensor([10.0, 18.0]))
    pipe.run_backward()
    # d(out)/da = b + 1
    assert torch.allclose(a.grad, torch.tensor([5.0, 6.0]))


def test_dag_softmax_backward_matches_autograd():
    g = torch.Generator().manual_seed(1)
    X = torch.randn(5, 4, generator=g)
    xt = X.clone().requires_grad_(True)
    y_ref = torch.softmax(xt, dim=-1)
    dy = torch.randn(5, 4, generator=g)
    y_ref.backward(dy)

    src = TrainableNode(X, lr=0.0)
    sm = ActivationOp(src, "softmax")
    term = TerminusNode(sm)
    pipe = DAGPipeline().add_flow(src, term)
    pipe.run_forward()
    term.accumulate_grad(dy)
    term.backward_run()
    assert torch.allclose(src.grad, xt.grad, atol=1e-6)


def test_deep_and_wide_graph_no_deadlock():
    """Graphs deeper/wider than the executor pool must complete (the
    naive submit-all-and-wait scheduling deadlocked at depth > workers;
    forward_run now runs its first input inline and work-steals queued
    siblings)."""
    import torch

    from lightctr_amd.engine.dag import AddOp, AggregateNode, SourceNode

    # chain of 60 adds (depth >> 8 workers)
    node = SourceNode(torch.ones(2))
    one = SourceNode(torch.ones(2))
    for _ in range(60):
        node = AddOp(node, one)
    out = node.forward_run()
    assert torch.allclose(out, torch.full((2,), 61.0))

    # wide fan-in of 24 branches, each a depth-4 chain
    branches = []
    for b in range(24):
        n = SourceNode(torch.full((2,), float(b)))
        for _ in range(4):
            n = AddOp(n, one)
        branches.append(n)
    agg = AggregateNode(*branches)
    out = agg.forward_run()
    expect = sum(float(b) + 4.0 for b in range(24))
    assert torch.allclose(out, torch.full((2,), expect))

    # backward through the deep chain completes and reaches the source
    node.accumulate_grad(torch.ones(2))
    node.backward_run()
    assert hasattr(node, "grad")


def test_graph_executor_matches_threaded():
    """The sequential topo GraphExecutor (GPU scheduler, hipGraph-captured
    on device) computes the same losses/updates as the threaded
    future/promise engine on an identical LR graph."""
    import torch

    from lightctr_amd.engine.dag import (ActivationOp, AddOp, DAGPipeline,
                                         GraphExecutor, LossOp, MatmulOp,
                                         SourceNode, TerminusNode,
                                         TrainableNode)

    g = torch.Generator().manual_seed(7)
    X = torch.randn(64, 3, generator=g)
    y = (X @ torch.tensor([[1.0], [-1.0], [0.5]]) > 0).float()

    def build():
        w = TrainableNode(torch.zeros(3, 1), lr=0.5)
        b = TrainableNode(torch.zeros(1, 1), lr=0.5)
        t = TerminusNode(LossOp(ActivationOp(AddOp(MatmulOp(SourceNode(X),
                                                            w), b),
                                             "sigmoid"),
                                SourceNode(y), "logistic"))
        return w, b, t

    w1, b1, t1 = build()
    pipe = DAGPipeline().add_flow(t1)
    thr = [float(pipe.step(engine="threaded")) for _ in range(10)]

    w2, b2, t2 = build()
    ex = GraphExecutor(t2, capture=False)
    seq = [float(ex.step()) for _ in range(10)]
    for a, b in zip(thr, seq):
        assert abs(a - b) < 1e-6, (a, b)
    assert torch.allclose(w1.value_init, w2.value_init, atol=1e-6)
    assert torch.allclose(b1.value_init, b2.value_init, atol=1e-6)
