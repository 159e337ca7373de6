"""GBM — second-order gradient-boosted trees (xgboost-style), MI355X-native.

Capability parity with the reference GBM
(/root/reference/LightCTR/gbm_algo_abst.h + train/train_gbm_algo.{h,cpp}):
second-order grad/hess boosting, gain-thresholded splits with L1/L2
regularization, NaN default-direction trick (train_gbm_algo.cpp:215-328),
multiclass = K trees per round via softmax (:66-94), row/feature
subsampling 0.7 (train_gbm_algo.h:72-86).

GPU-native redesign: the reference's per-feature parallel split search over
sorted columns becomes histogram-based split finding — features are
quantile-bucketized to uint8 once, each level builds per-(node, feature)
(grad, hess) histograms with the LDS-staged HIP kernel
(ops/csrc/gbm_kernels.hip), and the bin scan/gain argmax is tensor algebra.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops._extension import require_hip_ops

NAN_BIN = 255  # reserved bucket for missing values


@dataclass
class GBMHyper:
    n_rounds: int = 20
    max_depth: int = 5
    learning_rate: float = 0.3
    reg_lambda: float = 1.0  # L2 on leaf weights
    gamma: float = 0.0  # min split gain (the reference's L1 threshold role)
    min_child_weight: float = 1.0
    subsample: float = 0.7
    colsample: float = 0.7
    n_bins: int = 255  # data bins (255 kept for NaN)
    n_classes: int = 2  # 2 = binary (1 tree/round), >2 = softmax K trees
    seed: int = 1234


@dataclass
class Tree:
    feature: torch.Tensor  # [n_nodes] int32 (-1 = leaf)
    threshold: torch.Tensor  # [n_nodes] int32 (bin id; go left if bin<=thr)
    nan_left: torch.Tensor  # [n_nodes] bool (missing goes left?)
    left: torch.Tensor  # [n_nodes] int32
    right: torch.Tensor  # [n_nodes] int32
    value: torch.Tensor  # [n_nodes] float (leaf weight)


class GBMModel:
    def __init__(self, hyper: GBMHyper, device: str = "cpu"):
        self.h = hyper
        self.device = torch.device(device)
        self.trees: list[list[Tree]] = []  # [round][class]
        self.bin_edges = None  # [D, n_bins-1]
        self.base_score = 0.0
        self._gpu = self.device.type == "cuda"
        self._gen = torch.Generator().manual_seed(hyper.seed)

    # ---- binning ----
    def _fit_bins(self, X: torch.Tensor):
        D = X.shape[1]
        qs = torch.linspace(0, 1, self.h.n_bins + 1)[1:-1]
        edges = []
        for f in range(D):
            col = X[:, f]
            col = col[~torch.isnan(col)]
            if col.numel() == 0:
                edges.append(torch.zeros(self.h.n_bins - 1))
                continue
            e = torch.quantile(col.float().cpu(), qs)
            edges.append(e)
        self.bin_edges = torch.stack(edges).to(self.device)

    def _binize(self, X: torch.Tensor) -> torch.Tensor:
        D = X.shape[1]
        out = torch.empty(X.shape, dtype=torch.uint8, device=self.device)
        for f in range(D):
            col = X[:, f].to(self.device).contiguous()
            b = torch.bucketize(col, self.bin_edges[f]).clamp(
                0, self.h.n_bins - 1)
            b = torch.where(torch.isnan(col),
                            torch.full_like(b, NAN_BIN), b)
            out[:, f] = b.to(torch.uint8)
        return out

    # ---- histogram ----
    def _hist(self, bins, grad, hess, node_of_row, n_nodes):
        if self._gpu:
            return require_hip_ops().gbm_hist(bins, grad, hess, node_of_row,
                                              n_nodes)
        N, D = bins.shape
        hist = torch.zeros(n_nodes, D, 256, 2)
        act = node_of_row >= 0
        rows = act.nonzero(as_tuple=True)[0]
        nd = node_of_row[rows].long()
        b = bins[rows].long()  # [n_act, D]
        for f in range(D):
            idx = (nd * D + f) * 256 + b[:, f]
            hist.view(-1, 2)[:, 0].index_add_(0, idx, grad[rows])
            hist.view(-1, 2)[:, 1].index_add_(0, idx, hess[rows])
        return hist

    # ---- one tree ----
    def _grow_tree(self, bins, grad, hess, col_mask) -> Tree:
        h = self.h
        N, D = bins.shape
        dev = self.device
        feature, thr, nanl, left, right, value = [], [], [], [], [], []

        sub = torch.rand(N, generator=self._gen) < h.subsample
        node_of_row = torch.where(sub.to(dev),
                                  torch.zeros(N, dtype=torch.int32,
                                              device=dev),
                                  torch.full((N,), -1, dtype=torch.int32,
                                             device=dev))
        # level-local node ids; global arrays grow as nodes are emitted
        level_global = [self._emit(feature, thr, nanl, left, right, value)]
        for depth in range(h.max_depth):
            nl = len(level_global)
            hist = self._hist(bins, grad, hess, node_of_row, nl)
            # exclude masked-out columns by zeroing their gain later
            g_all = hist[..., 0]  # [nl, D, 256]
            h_all = hist[..., 1]
            G = g_all.sum(dim=2)  # [nl, D] (same for every f; use f0)
            H = h_all.sum(dim=2)
            Gtot = G[:, 0:1]
            Htot = H[:, 0:1]
            # missing-right: cumsum over data bins only
            GLr = torch.cumsum(g_all[:, :, :NAN_BIN], dim=2)
            HLr = torch.cumsum(h_all[:, :, :NAN_BIN], dim=2)
            # missing-left: NaN bucket counted into the left side
            GLl = GLr + g_all[:, :, NAN_BIN:NAN_BIN + 1]
            HLl = HLr + h_all[:, :, NAN_BIN:NAN_BIN + 1]

            def gain_of(GL, HL):
                GR = Gtot.unsqueeze(2) - GL
                HR = Htot.unsqueeze(2) - HL
                ok = (HL >= h.min_child_weight) & (HR >= h.min_child_weight)
                lam = h.reg_lambda
                parent = (Gtot * Gtot / (Htot + lam)).unsqueeze(2)
                g = GL * GL / (HL + lam) + GR * GR / (HR + lam) - parent
                return torch.where(ok, g, torch.full_like(g, -1e30))

            gain_r = gain_of(GLr, HLr)
            gain_l = gain_of(GLl, HLl)
            use_left = gain_l > gain_r
            gain = torch.maximum(gain_r, gain_l)  # [nl, D, n_bins]
            gain = torch.where(col_mask.view(1, D, 1), gain,
                               torch.full_like(gain, -1e30))
            flat = gain.view(nl, -1)
            best = flat.argmax(dim=1)
            best_gain = flat.gather(1, best.unsqueeze(1)).squeeze(1)
            bf = (best // gain.shape[2]).int()
            bt = (best % gain.shape[2]).int()
            bnl = use_left.view(nl, -1).gather(1, best.unsqueeze(1)) \
                .squeeze(1)

            # nodes at the deepest level must close as leaves (their values
            # come from this level's histograms)
            split = (best_gain > h.gamma) & (depth < h.max_depth - 1)
            # one device->host copy per level (not per node)
            split_h = split.cpu()
            bf_h, bt_h, bnl_h = bf.cpu(), bt.cpu(), bnl.cpu()
            G_h, H_h = G[:, 0].cpu(), H[:, 0].cpu()
            new_level = []
            child_of = torch.full((nl, 2), -1, dtype=torch.int32)
            for j in range(nl):
                gid = level_global[j]
                if bool(split_h[j]):
                    feature[gid] = int(bf_h[j])
                    thr[gid] = int(bt_h[j])
                    nanl[gid] = bool(bnl_h[j])
                    lgid = self._emit(feature, thr, nanl, left, right, value)
                    rgid = self._emit(feature, thr, nanl, left, right, value)
                    left[gid], right[gid] = lgid, rgid
                    child_of[j, 0] = len(new_level)
                    new_level.append(lgid)
                    child_of[j, 1] = len(new_level)
                    new_level.append(rgid)
                else:
                    lam = h.reg_lambda
                    value[gid] = float(-G_h[j] / (H_h[j] + lam)
                                       * h.learning_rate)
            if not new_level:
                break
            # route rows to level-local child ids
            act = node_of_row >= 0
            rows = act.nonzero(as_tuple=True)[0]
            nd = node_of_row[rows].long()
            fsel = bf.to(dev)[nd].long()
            tsel = bt.to(dev)[nd]
            nlsel = bnl.to(dev)[nd]
            rb = bins[rows, fsel].int()
            is_nan = rb == NAN_BIN
            go_left = torch.where(is_nan, nlsel, rb <= tsel)
            co = child_of.to(dev)
            new_ids = torch.where(go_left, co[nd, 0], co[nd, 1])
            was_split = split.to(dev)[nd]
            node_of_row[rows] = torch.where(was_split, new_ids,
                                            torch.full_like(new_ids, -1))
            level_global = new_level
        # any remaining open nodes at max depth -> leaves (computed above in
        # the loop via split=False at depth==max_depth-1... ensure closure)
        for gid in level_global:
            if feature[gid] == -1 and value[gid] is None:
                value[gid] = 0.0
        vals = [0.0 if v is None else v for v in value]
        return Tree(torch.tensor(feature, dtype=torch.int32),
                    torch.tensor(thr, dtype=torch.int32),
                    torch.tensor(nanl, dtype=torch.bool),
                    torch.tensor(left, dtype=torch.int32),
                    torch.tensor(right, dtype=torch.int32),
                    torch.tensor(vals, dtype=torch.float32))

    @staticmethod
    def _emit(feature, thr, nanl, left, right, value) -> int:
        feature.append(-1)
        thr.append(0)
        nanl.append(False)
        left.append(-1)
        right.append(-1)
        value.append(None)
        return len(feature) - 1

    def _route(self, bins: torch.Tensor, tree: Tree) -> torch.Tensor:
        """Vectorized root-to-leaf routing; returns leaf value per row."""
        N = bins.shape[0]
        dev = self.device
        node = torch.zeros(N, dtype=torch.long, device=dev)
        feat = tree.feature.long().to(dev)
        thr = tree.threshold.to(dev)
        nanl = tree.nan_left.to(dev)
        lft = tree.left.long().to(dev)
        rgt = tree.right.long().to(dev)
        val = tree.value.to(dev)
        for _ in range(self.h.max_depth + 1):
            is_leaf = feat[node] < 0
            if bool(is_leaf.all()):
                break
            f = feat[node].clamp(min=0)
            b = bins[torch.arange(N, device=dev), f].int()
            go_left = torch.where(b == NAN_BIN, nanl[node],
                                  b <= thr[node])
            nxt = torch.where(go_left, lft[node], rgt[node])
            node = torch.where(is_leaf, node, nxt)
        return val[node]

    # ---- boosting ----
    def fit(self, X: torch.Tensor, y: torch.Tensor, log=None):
        h = self.h
        N, D = X.shape
        self._fit_bins(X)
        bins = self._binize(X)
        y = y.to(self.device)
        K = 1 if h.n_classes <= 2 else h.n_classes
        margins = torch.zeros(N, K, device=self.device)
        for rnd in range(h.n_rounds):
            if K == 1:
                p = torch.sigmoid(margins[:, 0])
                grad = (p - y).contiguous()
                hess = (p * (1 - p)).clamp(min=1e-6).contiguous()
                grads = [grad]
                hesss = [hess]
            else:
                p = torch.softmax(margins, dim=1)
                yk = torch.nn.functional.one_hot(y.long(), K).float()
                grads = [(p[:, k] - yk[:, k]).contiguous() for k in range(K)]
                hesss = [(p[:, k] * (1 - p[:, k])).clamp(min=1e-6)
                         .contiguous() for k in range(K)]
            col_mask = (torch.rand(D, generator=self._gen)
                        < h.colsample).to(self.device)
            if not bool(col_mask.any()):
                col_mask[0] = True
            round_trees = []
            for k in range(K):
                tree = self._grow_tree(bins, grads[k], hesss[k], col_mask)
                round_trees.append(tree)
                margins[:, k] += self._route(bins, tree)
            self.trees.append(round_trees)
            if log:
                if K == 1:
                    p = torch.sigmoid(margins[:, 0]).clamp(1e-7, 1 - 1e-7)
                    loss = float(-(y * p.log()
                                   + (1 - y) * (1 - p).log()).mean())
                else:
                    loss = float(torch.nn.functional.cross_entropy(
                        margins, y.long()))
                log(f"round {rnd}: train loss={loss:.5f}")
        return self

    def predict_margin(self, X: torch.Tensor) -> torch.Tensor:
        bins = self._binize(X)
        K = 1 if self.h.n_classes <= 2 else self.h.n_classes
        out = torch.zeros(X.shape[0], K, device=self.device)
        for round_trees in self.trees:
            for k, tree in enumerate(round_trees):
                out[:, k] += self._route(bins, tree)
        return out

    def predict_proba(self, X: torch.Tensor) -> torch.Tensor:
        m = self.predict_margin(X)
        if m.shape[1] == 1:
            return torch.sigmoid(m[:, 0])
        return torch.softmax(m, dim=1)

    # ---- checkpoint (reference saveModel is a stub; ours is real) ----
    def save(self, path: str):
        torch.save({"hyper": self.h.__dict__,
                    "bin_edges": self.bin_edges,
                    "trees": [[t.__dict__ for t in rt]
                              for rt in self.trees]}, path)

    def load(self, path: str):
        d = torch.load(path, map_location=self.device, weights_only=False)
        self.bin_edges = d["bin_edges"].to(self.device)
        self.trees = [[Tree(**td) for td in rt] for rt in d["trees"]]
