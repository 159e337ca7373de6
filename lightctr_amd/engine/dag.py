"""DAG autograd engine (gen-2) — future/promise dataflow over tensors.

Capability parity with the reference's dag/ engine
(/root/reference/LightCTR/dag/: node_abst.h forward_run/backward_run
CAS-guarded single execution with child futures and fan-out promises
:57-87,166-198; source_node.h TrainableNode applies its updater in backward
:63-77; operators add/multiply(Hadamard w/ div-based per-input delta,
multiply_op.h:36-69)/matmul/activations/loss; dag_pipeline.h
addAutogradFlow wiring :33-37).

MI355X redesign: node results are torch tensors (GPU-resident when the
graph lives on a GPU device); independent nodes execute concurrently on a
ThreadPoolExecutor — on ROCm each runner thread issues kernels onto the
same HIP stream, so the concurrency model (futures + once-latches) matches
the reference while the math stays on-device. Every node computes exactly
once per flow (the deal_flag / CAS semantics) no matter how many consumers
fan out.
"""

from __future__ import annotations

import threading
from concurrent.futures import ThreadPoolExecutor

import torch

_EXECUTOR = ThreadPoolExecutor(max_workers=8)


class Node:
    """Base dataflow node. Subclasses implement forward_compute(*inputs)
    and backward_compute(grad) -> tuple of per-input grads."""

    def __init__(self, *inputs: "Node"):
        self.inputs = list(inputs)
        self.consumers: list[Node] = []
        for i in self.inputs:
            i.consumers.append(self)
        self.value: torch.Tensor | None = None
        self._fwd_done = False
        self._lock = threading.Lock()
        self._grad_parts: list[torch.Tensor] = []
        self._bwd_done = False

    # ---- forward (future/promise style, once-latch = deal_flag) ----
    def forward_run(self):
        with self._lock:
            if self._fwd_done:
                return self.value
            vals = []
            if self.inputs:
                # Run the FIRST input on this thread (graph depth then
                # consumes the caller's stack, not pool workers) and pool
                # only the siblings; a pooled task that has not started is
                # cancelled and run inline. Every blocked result() is
                # therefore on a RUNNING task, so a saturated pool can
                # never deadlock the flow (the naive submit-all-and-wait
                # version deadlocks once depth or width exceeds the pool).
                futures = [_EXECUTOR.submit(i.forward_run)
                           for i in self.inputs[1:]]
                vals.append(self.inputs[0].forward_run())
                for f, node in zip(futures, self.inputs[1:]):
                    if f.cancel():
                        vals.append(node.forward_run())
                    else:
                        vals.append(f.result())
            self.value = self.forward_compute(*vals)
            self._fwd_done = True
            return self.value

    def forward_compute(self, *vals):  # pragma: no cover - abstract
        raise NotImplementedError

    # ---- backward (mirror recursion; grads fan in from consumers) ----
    def accumulate_grad(self, g: torch.Tensor):
        with self._lock:
            self._grad_parts.append(g)

    def grad_ready(self) -> bool:
        return len(self._grad_parts) == max(1, len(self.consumers))

    def backward_run(self):
        """Called once all consumers contributed their grad parts."""
        with self._lock:
            if self._bwd_done:
                return
            self._bwd_done = True
        grad = self._grad_parts[0]
        for g in self._grad_parts[1:]:
            grad = grad + g
        self.grad = grad
        in_grads = self.backward_compute(grad)
        for inp, g in zip(self.inputs, in_grads):
            inp.accumulate_grad(g)
            if inp.grad_ready():
                inp.backward_run()

    def backward_compute(self, grad):  # pragma: no cover - abstract
        raise NotImplementedError

    def reset_flow(self):
        """init_forward_Flow equivalent: clear deal flags over the graph."""
        self._fwd_done = False
        self._bwd_done = False
        self._grad_parts = []
        for i in self.inputs:
            if i._fwd_done or i._grad_parts or i._bwd_done:
                i.reset_flow()


class SourceNode(Node):
    def __init__(self, value: torch.Tensor):
        super().__init__()
        self.value_init = value

    def forward_compute(self):
        return self.value_init

    def backward_compute(self, grad):
        return ()


class TrainableNode(SourceNode):
    """Parameter node: applies its (Adagrad) updater when its gradient
    arrives (reference source_node.h:63-77)."""

    def __init__(self, value: torch.Tensor, lr: float = 0.1,
                 eps: float = 1e-8):
        super().__init__(value.clone())
        self.lr, self.eps = lr, eps
        self.accum = torch.zeros_like(value)

    def backward_compute(self, grad):
        self.accum += grad * grad
        self.value_init -= self.lr * grad / (self.accum + self.eps).sqrt()
        return ()


class AggregateNode(Node):
    """Sums its inputs (fan-in aggregation)."""

    def forward_compute(self, *vals):
        out = vals[0]
        for v in vals[1:]:
            out = out + v
        return out

    def backward_compute(self, grad):
        return tuple(grad for _ in self.inputs)


class TerminusNode(Node):
    def forward_compute(self, v):
        return v

    def backward_compute(self, grad):
        return (grad,)

    def run_flow(self):
        self.reset_flow()
        out = self.forward_run()
        return out


def _reduce_to_shape(grad: torch.Tensor, shape) -> torch.Tensor:
    """Sum grad over dims broadcast during forward so it matches `shape`."""
    while grad.dim() > len(shape):
        grad = grad.sum(dim=0)
    for d, s in enumerate(shape):
        if s == 1 and grad.shape[d] != 1:
            grad = grad.sum(dim=d, keepdim=True)
    return grad


class AddOp(Node):
    def forward_compute(self, a, b):
        self._sa, self._sb = a.shape, b.shape
        return a + b

    def backward_compute(self, grad):
        return (_reduce_to_shape(grad, self._sa),
                _reduce_to_shape(grad, self._sb))


class MultiplyOp(Node):
    """Hadamard product (reference multiply_op.h)."""

    def forward_compute(self, a, b):
        self._a, self._b = a, b
        return a * b

    def backward_compute(self, grad):
        return (_reduce_to_shape(grad * self._b, self._a.shape),
                _reduce_to_shape(grad * self._a, self._b.shape))


class MatmulOp(Node):
    def forward_compute(self, a, b):
        self._a, self._b = a, b
        return a @ b

    def backward_compute(self, grad):
        return (grad @ self._b.t(), self._a.t() @ grad)


class ActivationOp(Node):
    def __init__(self, inp: Node, kind: str = "sigmoid"):
        super().__init__(inp)
        self.kind = kind

    def forward_compute(self, x):
        if self.kind == "sigmoid":
            y = torch.sigmoid(torch.clamp(x, -16, 16))
        elif self.kind == "relu":
            y = torch.relu(x)
        elif self.kind == "tanh":
            y = torch.tanh(x)
        elif self.kind == "softmax":
            y = torch.softmax(x, dim=-1)
        else:
            raise ValueError(self.kind)
        self._y = y
        return y

    def backward_compute(self, grad):
        y = self._y
        if self.kind == "sigmoid":
            return (grad * y * (1 - y),)
        if self.kind == "relu":
            return (grad * (y > 0).float(),)
        if self.kind == "tanh":
            return (grad * (1 - y * y),)
        # softmax Jacobian-vector
        s = (grad * y).sum(dim=-1, keepdim=True)
        return (y * (grad - s),)


class LossOp(Node):
    """kind: logistic | square. Inputs: (pred, label-as-SourceNode)."""

    def __init__(self, pred: Node, label: Node, kind: str = "logistic"):
        super().__init__(pred, label)
        self.kind = kind

    def forward_compute(self, p, y):
        self._p, self._y = p, y
        if self.kind == "logistic":
            pc = p.clamp(1e-7, 1 - 1e-7)
            return -(y * pc.log() + (1 - y) * (1 - pc).log()).mean()
        return ((p - y) ** 2).mean()

    def backward_compute(self, grad):
        n = self._p.numel()
        if self.kind == "logistic":
            pc = self._p.clamp(1e-7, 1 - 1e-7)
            g = (pc - self._y) / (pc * (1 - pc)) / n
        else:
            g = 2 * (self._p - self._y) / n
        return (grad * g, torch.zeros_like(self._y))


class GraphExecutor:
    """GPU-credible scheduler for a STATIC dag (round-2):
    one topological order computed once, forward + backward issued
    sequentially with no thread pool and no per-node locks (kernels queue
    asynchronously on the HIP stream — Python never blocks on device
    work), and after a warmup step the whole train step is captured into
    a hipGraph (torch.cuda.CUDAGraph) and replayed, making the step
    launch-bound instead of Python-bound. The threaded future/promise
    path above stays as the dynamic-graph capability-parity engine
    (reference node_abst.h:57-87 semantics)."""

    def __init__(self, terminus: "TerminusNode", capture: bool = True):
        self.terminus = terminus
        self.order = self._topo(terminus)
        self.capture = capture
        self._graph = None
        self._calls = 0

    @staticmethod
    def _topo(root: Node) -> list[Node]:
        order, seen = [], set()

        def visit(n: Node):
            if id(n) in seen:
                return
            seen.add(id(n))
            for i in n.inputs:
                visit(i)
            order.append(n)

        visit(root)
        return order

    def _run_seq(self):
        for n in self.order:
            n.value = n.forward_compute(*[i.value for i in n.inputs])
        # backward in reverse topo order: when a node is reached, all of
        # its consumers (later in topo order) have contributed grads
        grads: dict[int, torch.Tensor] = {
            id(self.terminus): torch.ones_like(self.terminus.value)}
        for n in reversed(self.order):
            g = grads.pop(id(n), None)
            if g is None:
                continue  # node not on a grad path
            n.grad = g
            in_grads = n.backward_compute(g)
            for inp, ig in zip(n.inputs, in_grads):
                prev = grads.get(id(inp))
                grads[id(inp)] = ig if prev is None else prev + ig
        return self.terminus.value

    def step(self):
        dev = self.terminus_device()
        if self._graph is not None:
            self._graph.replay()
            return self.terminus.value
        loss = self._run_seq()
        self._calls += 1
        if (self.capture and dev is not None and dev.type == "cuda"
                and self._calls == 2):
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            try:
                with torch.cuda.graph(g):
                    self._run_seq()
                self._graph = g
                # capture records without executing: replay once so this
                # call still performs a real training step
                g.replay()
                loss = self.terminus.value
            except RuntimeError:
                self.capture = False  # capture-unsupported op: stay eager
        return loss

    def terminus_device(self):
        v = self.terminus.value
        if v is None:
            for n in self.order:
                if isinstance(n, SourceNode):
                    return n.value_init.device
            return None
        return v.device


class DAGPipeline:
    """Wiring helper (reference dag_pipeline.h:33-37)."""

    def __init__(self):
        self.sources: list[Node] = []
        self.terminus: TerminusNode | None = None
        self._executor: GraphExecutor | None = None

    def add_flow(self, *nodes):
        for n in nodes:
            if isinstance(n, SourceNode) and n not in self.sources:
                self.sources.append(n)
            if isinstance(n, TerminusNode):
                self.terminus = n
        return self

    def run_forward(self):
        return self.terminus.run_flow()

    def run_backward(self):
        """Seed grad at terminus and propagate to all sources (applies the
        TrainableNodes' updaters)."""
        self.terminus.accumulate_grad(
            torch.ones_like(self.terminus.value))
        self.terminus.backward_run()

    def step(self, engine: str = "auto"):
        """engine: "auto" (GraphExecutor with hipGraph capture on GPU,
        threaded elsewhere), "graph", or "threaded"."""
        if engine == "threaded":
            loss = self.run_forward()
            self.run_backward()
            return loss
        if self._executor is None:
            self._executor = GraphExecutor(self.terminus)
        use_graph = engine == "graph" or (
            self._executor.terminus_device() is not None
            and self._executor.terminus_device().type == "cuda")
        if engine == "auto" and not use_graph:
            loss = self.run_forward()
            self.run_backward()
            return loss
        return self._executor.step()
