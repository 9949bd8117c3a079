"""lpx_sklearn — RandomForest inference + pairwise distances on gfx950
(recipe: scikit-learn; BASELINE.json config 4 — the headline metric workload).

The forest is flattened to a SoA node table (one int4 per node) uploaded once;
inference is the hand-written traversal kernel. Matches
sklearn semantics: `X[:, f] <= threshold` goes left; predict_proba averages
per-tree leaf class distributions; predict argmaxes (ties -> lowest class).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

try:
    import lpx_ops as _ops
except ImportError:
    from lambdipy_amd import ops as _ops

_DEV = "cuda"


MAX_PACKED_FEAT = 2046   # feat field is 11 bits (0x7FF = leaf sentinel)
MAX_PACKED_DELTA = (1 << 21) - 1


class ForestSoA:
    """Device-resident SoA forest.

    Two node layouts (the HIP kernels accept both):
      - classic: int32 [n,4] {feat, thresh-bits, left, right}; leaf feat=-1,
        left = leaf_values row.
      - packed (default when it fits): int32 [n,2] {meta, val} in DEPTH-FIRST
        order — meta = feat<<21 | right_delta, left child implicit at nd+1,
        leaf meta feat==0x7FF and val = leaf row. Half the node bytes and the
        '<=' chain is cacheline-local (the traversal kernel is L2-BW-bound).
    """

    def __init__(self, nodes: torch.Tensor, leaf_values: torch.Tensor,
                 roots: torch.Tensor, n_trees: int, n_classes: int,
                 classes: Optional[np.ndarray] = None, packed: bool = False,
                 layout: Optional[str] = None, depth: Optional[int] = None):
        self.nodes = nodes
        self.leaf_values = leaf_values
        self.roots = roots
        self.n_trees = n_trees
        self.n_classes = n_classes
        self.layout = layout or ("packed" if packed else "classic")
        self.depth = depth
        self.classes = classes if classes is not None else np.arange(n_classes)

    @property
    def packed(self) -> bool:
        return self.layout == "packed"

    def to(self, device) -> "ForestSoA":
        return ForestSoA(self.nodes.to(device), self.leaf_values.to(device),
                         self.roots.to(device), self.n_trees, self.n_classes,
                         self.classes, layout=self.layout, depth=self.depth)

    @property
    def nbytes(self) -> int:
        return (self.nodes.numel() * 4 + self.leaf_values.numel() * 4 +
                self.roots.numel() * 4)

    # ---- construction ----------------------------------------------------

    @classmethod
    def from_sklearn(cls, clf, device=_DEV) -> "ForestSoA":
        """Flatten a fitted sklearn RandomForest{Classifier,Regressor}."""
        estimators = clf.estimators_
        is_clf = hasattr(clf, "classes_")
        n_classes = int(clf.n_classes_) if is_clf else 1
        all_nodes = []
        all_leaves = []
        roots = []
        offset = 0
        leaf_off = 0
        for est in estimators:
            t = est.tree_
            n = t.node_count
            feat = t.feature.astype(np.int32).copy()
            thr = t.threshold.astype(np.float32)
            left = t.children_left.astype(np.int32)
            right = t.children_right.astype(np.int32)
            is_leaf = left < 0
            nodes = np.empty((n, 4), dtype=np.int32)
            nodes[:, 0] = np.where(is_leaf, -1, feat)
            nodes[:, 1] = thr.view(np.int32)
            nodes[:, 2] = np.where(is_leaf, 0, left + offset)
            nodes[:, 3] = np.where(is_leaf, 0, right + offset)
            # leaf value rows
            leaf_idx = np.flatnonzero(is_leaf)
            vals = t.value[leaf_idx, 0, :].astype(np.float64)  # [L, C] or [L, 1]
            if is_clf:
                s = vals.sum(axis=1, keepdims=True)
                s[s == 0] = 1.0
                vals = vals / s
            nodes[leaf_idx, 2] = np.arange(len(leaf_idx), dtype=np.int32) + leaf_off
            all_nodes.append(nodes)
            all_leaves.append(vals.astype(np.float32))
            roots.append(offset)
            offset += n
            leaf_off += len(leaf_idx)
        nodes = torch.from_numpy(np.concatenate(all_nodes)).contiguous()
        leaves = torch.from_numpy(np.concatenate(all_leaves)).contiguous()
        rt = torch.tensor(roots, dtype=torch.int32)
        f = cls(nodes, leaves, rt, len(estimators), n_classes,
                clf.classes_ if is_clf else None)
        f = f.pack()  # depth-first 8-B layout when it fits
        return f.to(device) if device else f

    def pack(self) -> "ForestSoA":
        """Convert classic [n,4] nodes to packed depth-first [n,2] nodes.

        Returns self unchanged when the forest does not fit the packed
        limits (F > 2046, subtree > 2^21 nodes) or is already packed."""
        if self.packed:
            return self
        nodes = self.nodes.cpu().numpy()
        n = nodes.shape[0]
        feat = nodes[:, 0]
        if feat.max(initial=-1) > MAX_PACKED_FEAT:
            return self
        left, right = nodes[:, 2], nodes[:, 3]
        # subtree sizes: sklearn/our layouts put children after parents per
        # tree, so a reverse sweep sees children first
        size = np.ones(n, dtype=np.int64)
        internal = feat >= 0
        order = np.arange(n - 1, -1, -1)
        for i in order:
            if internal[i]:
                size[i] = 1 + size[left[i]] + size[right[i]]
        new_nodes = np.empty((n, 2), dtype=np.uint32)
        new_roots = np.empty(self.n_trees, dtype=np.int32)
        pos = 0
        roots = self.roots.cpu().numpy()
        thr_col = nodes[:, 1]
        thr_bits = (thr_col.view(np.uint32) if thr_col.flags['C_CONTIGUOUS']
                    else np.ascontiguousarray(thr_col).view(np.uint32))
        for t, root in enumerate(roots):
            new_roots[t] = pos
            stack = [int(root)]
            while stack:
                i = stack.pop()
                if internal[i]:
                    delta = int(size[left[i]]) + 1
                    if delta > MAX_PACKED_DELTA:
                        return self
                    new_nodes[pos, 0] = (int(feat[i]) << 21) | delta
                    new_nodes[pos, 1] = thr_bits[i]  # f32 threshold bits
                    stack.append(int(right[i]))
                    stack.append(int(left[i]))
                else:
                    new_nodes[pos, 0] = 0x7FF << 21
                    new_nodes[pos, 1] = nodes[i, 2]  # leaf row
                pos += 1
        assert pos == n
        new_nodes = new_nodes.view(np.int32)
        return ForestSoA(torch.from_numpy(new_nodes).contiguous(),
                         self.leaf_values, torch.from_numpy(new_roots),
                         self.n_trees, self.n_classes, self.classes,
                         layout="packed")

    @classmethod
    def alloc(cls, n_trees: int, depth: int, n_features: int, n_classes: int,
              device=_DEV) -> "ForestSoA":
        """Empty forest with the shapes random(...) produces (for non-root
        ranks receiving the model via broadcast)."""
        n_leaves = n_trees * (2 ** depth)
        if n_features <= MAX_PACKED_FEAT and depth <= 20:
            n_internal = n_trees * (2 ** depth - 1)
            return cls(torch.empty(n_internal, 2, dtype=torch.int32, device=device),
                       torch.empty(n_leaves, n_classes, dtype=torch.float32,
                                   device=device),
                       torch.empty(n_trees, dtype=torch.int32, device=device),
                       n_trees, n_classes, layout="heap", depth=depth)
        n_nodes = n_trees * (2 ** (depth + 1) - 1)
        return cls(torch.empty(n_nodes, 4, dtype=torch.int32, device=device),
                   torch.empty(n_leaves, n_classes, dtype=torch.float32,
                               device=device),
                   torch.empty(n_trees, dtype=torch.int32, device=device),
                   n_trees, n_classes, layout="classic")

    @classmethod
    def random(cls, n_trees: int, depth: int, n_features: int, n_classes: int,
               seed: int = 0, device=_DEV) -> "ForestSoA":
        """Synthetic full binary forest (bench/smoke: random-init weights).

        Generated DIRECTLY in the packed depth-first layout (vectorized —
        no per-node Python loop; the pack() conversion was ~250 ms of
        cold-start for the bench forest) when the packed limits allow."""
        if n_features <= MAX_PACKED_FEAT and depth <= 20:
            return cls._random_heap(n_trees, depth, n_features, n_classes,
                                    seed, device)
        return cls._random_classic(n_trees, depth, n_features, n_classes,
                                   seed, device)

    @classmethod
    def _random_heap(cls, n_trees, depth, n_features, n_classes, seed,
                     device):
        """Complete forest in level-order (heap) layout: internal nodes only
        ({u32 feat, f32 thresh-bits}); children implicit at 2i+1/2i+2, leaf
        row = heap index - (2^depth - 1). Generation shared with the
        torch-free runtime (lpx_rt.random_heap_forest_np)."""
        try:
            from lpx_rt import random_heap_forest_np
        except ImportError:
            from .lpx_rt import random_heap_forest_np
        nodes, lv = random_heap_forest_np(n_trees, depth, n_features,
                                          n_classes, seed)
        roots = np.arange(n_trees, dtype=np.int32) * ((1 << depth) - 1)
        f = cls(torch.from_numpy(nodes).contiguous(),
                torch.from_numpy(lv).contiguous(),
                torch.from_numpy(roots).contiguous(), n_trees, n_classes,
                layout="heap", depth=depth)
        return f.to(device) if device else f

    @classmethod
    def _random_classic(cls, n_trees: int, depth: int, n_features: int,
                        n_classes: int, seed: int = 0, device=_DEV) -> "ForestSoA":
        """Synthetic full binary forest in the classic [n,4] layout."""
        rng = np.random.default_rng(seed)
        n_internal = 2**depth - 1
        n_leaves = 2**depth
        n = n_internal + n_leaves
        all_nodes = np.empty((n_trees * n, 4), dtype=np.int32)
        all_leaves = np.empty((n_trees * n_leaves, n_classes), dtype=np.float32)
        roots = np.arange(n_trees, dtype=np.int32) * n
        for t in range(n_trees):
            o = t * n
            nodes = np.empty((n, 4), dtype=np.int32)
            # internal nodes 0..n_internal-1 in breadth-first order
            feat = rng.integers(0, n_features, n_internal).astype(np.int32)
            thr = rng.standard_normal(n_internal).astype(np.float32)
            nodes[:n_internal, 0] = feat
            nodes[:n_internal, 1] = thr.view(np.int32)
            idx = np.arange(n_internal, dtype=np.int32)
            nodes[:n_internal, 2] = 2 * idx + 1 + o
            nodes[:n_internal, 3] = 2 * idx + 2 + o
            nodes[n_internal:, 0] = -1
            nodes[n_internal:, 1] = 0
            nodes[n_internal:, 2] = np.arange(n_leaves, dtype=np.int32) + t * n_leaves
            nodes[n_internal:, 3] = 0
            all_nodes[o : o + n] = nodes
            lv = rng.random((n_leaves, n_classes)).astype(np.float32)
            lv /= lv.sum(axis=1, keepdims=True)
            all_leaves[t * n_leaves : (t + 1) * n_leaves] = lv
        f = cls(torch.from_numpy(all_nodes).contiguous(),
                torch.from_numpy(all_leaves).contiguous(),
                torch.from_numpy(roots).contiguous(), n_trees, n_classes)
        return f.to(device) if device else f

    # ---- inference -------------------------------------------------------

    _chunk_cache = None

    def _plan_chunks(self, F: int):
        """Greedy whole-tree chunking for the forest-in-LDS kernel.

        Returns (chunk_tree_begin int32 device tensor [n_chunks+1],
        max_chunk_nodes) or None when a single tree exceeds the LDS budget
        (fall back to the global-memory kernel)."""
        if not self.packed:
            return None
        if self._chunk_cache is not None and self._chunk_cache[0] == F:
            return self._chunk_cache[1]
        xbytes = 256 * (F + 1) * 4 if F <= 63 else 0
        cap = (160 * 1024 - xbytes - 2048) // 8  # nodes per chunk
        roots = self.roots.cpu().numpy()
        total = self.nodes.shape[0]
        counts = np.diff(np.append(roots, total))
        plan = None
        if counts.max(initial=0) <= cap and cap > 0:
            begins = [0]
            cur = 0
            for i, c in enumerate(counts):
                if cur + c > cap:
                    begins.append(i)
                    cur = 0
                cur += int(c)
            begins.append(len(counts))
            ctb = torch.tensor(begins, dtype=torch.int32,
                               device=self.nodes.device)
            starts = np.array(begins[:-1])
            ends = np.array(begins[1:])
            node_b = roots[starts]
            node_e = np.where(ends < len(roots), 
                              np.append(roots, total)[ends], total)
            max_nodes = int((node_e - node_b).max())
            plan = (ctb, max_nodes)
        self._chunk_cache = (F, plan)
        return plan

    # forest-in-LDS routing measured SLOWER on MI355X (latency-bound at one
    # 4-wave block/CU + per-chunk X restaging) — global packed kernel is the
    # default; set use_lds_chunks=True to A/B the LDS variant.
    use_lds_chunks = False

    def _predict_proba_t_cpu(self, X: torch.Tensor) -> torch.Tensor:
        """Plain-torch heap-forest reference (CPU tensors only): the same
        f32 traversal semantics as the gfx950 kernel (right iff x > thr).
        Lets the multi-process bench/fanout path run under gloo on CPU —
        it is NOT a GPU fallback (ops.api stays fail-loud on GPU boxes)."""
        assert self.layout == "heap", "CPU reference supports heap forests"
        d = self.depth
        stride = (1 << d) - 1
        T = self.nodes.shape[0] // stride
        nh = self.nodes.view(T, stride, 2)
        feat = nh[..., 0].long()
        thr = nh[..., 1].contiguous().view(torch.float32)
        idx = torch.zeros(T, X.shape[0], dtype=torch.long)
        for _ in range(d):
            f = feat.gather(1, idx)                      # [T,S]
            t = thr.gather(1, idx)
            xv = X.gather(1, f.t()).t()                  # X[s, f[tr,s]]
            idx = 2 * idx + 1 + (xv > t).long()
        leaf = idx - stride
        C = self.leaf_values.shape[1]
        lv = self.leaf_values.view(T, 1 << d, C)
        gathered = lv.gather(1, leaf.unsqueeze(-1).expand(T, -1, C))
        return gathered.sum(dim=0) / float(self.n_trees)

    def predict_proba_t(self, X: torch.Tensor) -> torch.Tensor:
        if self.nodes.device.type == "cpu" and self.layout == "heap":
            return self._predict_proba_t_cpu(X)
        if self.layout == "heap":
            sums = _ops.rf_predict_proba_sum(self.nodes, self.leaf_values,
                                             self.roots, X, layout="heap",
                                             depth=self.depth)
        else:
            chunks = (self._plan_chunks(X.shape[1])
                      if (self.packed and self.use_lds_chunks) else None)
            sums = _ops.rf_predict_proba_sum(self.nodes, self.leaf_values,
                                             self.roots, X, packed=self.packed,
                                             chunks=chunks)
        return sums / float(self.n_trees)

    def predict_t(self, X: torch.Tensor) -> torch.Tensor:
        return self.predict_proba_t(X).argmax(dim=1)

    def predict_proba(self, X: np.ndarray) -> np.ndarray:
        t = torch.from_numpy(np.ascontiguousarray(X, dtype=np.float32))
        t = t.to(self.nodes.device)
        return self.predict_proba_t(t).cpu().numpy()

    def predict(self, X: np.ndarray) -> np.ndarray:
        proba = self.predict_proba(X)
        return np.asarray(self.classes)[proba.argmax(axis=1)]


def pairwise_sqeuclidean(X: np.ndarray, Y: Optional[np.ndarray] = None) -> np.ndarray:
    """sklearn.metrics.pairwise.euclidean_distances(..., squared=True) analog."""
    tx = torch.from_numpy(np.ascontiguousarray(X, dtype=np.float32)).to(_DEV)
    ty = tx if Y is None else torch.from_numpy(
        np.ascontiguousarray(Y, dtype=np.float32)).to(_DEV)
    return _ops.pairwise_sqeuclidean(tx, ty).cpu().numpy()


def pairwise_sqeuclidean_t(X: torch.Tensor, Y: torch.Tensor) -> torch.Tensor:
    return _ops.pairwise_sqeuclidean(X, Y)
