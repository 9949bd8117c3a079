"""Config-4 handler: sklearn RandomForest inference with RCCL batch fan-out.

THE metric workload (BASELINE.json:2 'cold-start->first-inference ms, sklearn
RF 1/2/4/8 GPU'). Single-GPU: build forest, predict one batch. Multi-GPU
(WORLD_SIZE>1 env, one process per GPU): rank 0 broadcasts the forest over
RCCL/xGMI, the batch is fanned out (broadcast + local slice), each rank runs
the traversal kernel on its shard, predictions all_gather back (SURVEY.md
§3.4 / §2.2 N15).

Model: synthetic random forest (random-init — no network for datasets) with
sklearn-matching semantics; set LPX_RF_PICKLE to a joblib/pickle of a fitted
sklearn forest to use a real model instead.
"""

from __future__ import annotations

import json
import os
import time


def main() -> dict:
    t0 = time.perf_counter()
    phases = []

    def mark(name, start):
        phases.append({"phase": name, "dur_ms": (time.perf_counter() - start) * 1e3})

    # Wait for the layer BEFORE importing torch: torch itself imports numpy,
    # and a half-extracted layer numpy on PYTHONPATH would poison the import
    # (the torch-free rf_lite handler keeps the full unzip/import overlap).
    ready = os.environ.get("LPX_READY_FILE")
    if ready:
        t = time.perf_counter()
        while not os.path.exists(ready):
            time.sleep(0.002)
        mark("wait_layer_extract", t)

    t = time.perf_counter()
    import torch
    mark("import_torch", t)

    t = time.perf_counter()
    try:
        import lpx_sklearn
    except ImportError:
        from lambdipy_amd.shims import lpx_sklearn
    try:
        import lpx_fanout
    except ImportError:
        from lambdipy_amd.parallel import fanout as lpx_fanout
    mark("import_shim", t)

    n_trees = int(os.environ.get("LPX_RF_TREES", "256"))
    depth = int(os.environ.get("LPX_RF_DEPTH", "10"))
    n_features = int(os.environ.get("LPX_RF_FEATURES", "32"))
    n_classes = int(os.environ.get("LPX_RF_CLASSES", "8"))
    batch = int(os.environ.get("LPX_RF_BATCH", "1048576"))

    t = time.perf_counter()
    ctx = lpx_fanout.init_fanout()
    mark("comm_init", t)

    t = time.perf_counter()
    use_gpu = torch.cuda.is_available()
    device = ctx.device
    pickle_path = os.environ.get("LPX_RF_PICKLE")
    if pickle_path:
        import pickle
        with open(pickle_path, "rb") as fh:
            clf = pickle.load(fh)
        forest = lpx_sklearn.ForestSoA.from_sklearn(clf, device=device)
    else:
        # rank 0 builds; others allocate empty and receive via broadcast
        if ctx.is_root:
            forest = lpx_sklearn.ForestSoA.random(
                n_trees, depth, n_features, n_classes, seed=17, device=device)
        else:
            forest = lpx_sklearn.ForestSoA.alloc(
                n_trees, depth, n_features, n_classes, device=device)
    if use_gpu:
        torch.cuda.synchronize()
    mark("model_build", t)

    t = time.perf_counter()
    lpx_fanout.broadcast_tensors(ctx, [forest.nodes, forest.leaf_values, forest.roots])
    if use_gpu:
        torch.cuda.synchronize()
    mark("model_broadcast", t)

    t = time.perf_counter()
    if ctx.is_root:
        gen = torch.Generator(device="cpu").manual_seed(23)
        X = torch.randn(batch, n_features, generator=gen, dtype=torch.float32)
        X = X.to(device)
    else:
        X = torch.empty(batch, n_features, dtype=torch.float32, device=device)
    if use_gpu:
        torch.cuda.synchronize()
    mark("batch_data", t)

    t = time.perf_counter()
    shard, counts = lpx_fanout.shard_rows(ctx, X)
    proba_local = forest.predict_proba_t(shard)
    proba = lpx_fanout.all_gather_rows(ctx, proba_local, counts)
    pred = proba.argmax(dim=1)
    if use_gpu:
        torch.cuda.synchronize()
    mark("first_inference", t)

    out = {
        "handler": "rf",
        "rank": ctx.rank,
        "world_size": ctx.world_size,
        "model": {"trees": n_trees, "depth": depth, "features": n_features,
                  "classes": n_classes, "model_mb": round(forest.nbytes / 2**20, 2)},
        "batch": batch,
        "checksum": float(pred.float().sum()),
        "total_ms": (time.perf_counter() - t0) * 1e3,
        "phases": phases,
    }
    if ctx.is_root:
        print("LPX_RESULT " + json.dumps(out), flush=True)
    lpx_fanout.destroy()
    return out


if __name__ == "__main__":
    main()
