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


class ForestSoA:
    """Device-resident SoA forest: nodes int32 [n,4] {feat, thresh-bits,
    left, right}, leaf feat=-1 and left = row into leaf_values [L, C]."""

    def __init__(self, nodes: torch.Tensor, leaf_values: torch.Tensor,
                 roots: torch.Tensor, n_trees: int, n_classes: int,
                 classes: Optional[np.ndarray] = None):
        self.nodes = nodes
        self.leaf_values = leaf_values
        self.roots = roots
        self.n_trees = n_trees
        self.n_classes = n_classes
        self.classes = classes if classes is not None else np.arange(n_classes)

    def to(self, device) -> "ForestSoA":
        return ForestSoA(self.nodes.to(device), self.leaf_values.to(device),
                         self.roots.to(device), self.n_trees, self.n_classes,
                         self.classes)

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
        return f.to(device) if device else f

    @classmethod
    def random(cls, n_trees: int, depth: int, n_features: int, n_classes: int,
               seed: int = 0, device=_DEV) -> "ForestSoA":
        """Synthetic full binary forest (bench/smoke: random-init weights)."""
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

    def predict_proba_t(self, X: torch.Tensor) -> torch.Tensor:
        sums = _ops.rf_predict_proba_sum(self.nodes, self.leaf_values,
                                         self.roots, X)
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
