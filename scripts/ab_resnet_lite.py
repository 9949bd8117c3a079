"""ResNet50Lite serving throughput: eager ctypes launches vs hipGraph
replay, vs the torch-extension path (lpx_torch) at the same batch sizes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lambdipy_amd.shims import lpx_nn, lpx_rt  # noqa: E402


def main():
    lpx_rt.set_device(0)
    L = lpx_nn.lib()
    for batch in (32, 64, 128):
        m = lpx_nn.ResNet50Lite(n_classes=1000, seed=3)
        x = m.pool.alloc(batch * 224 * 224 * 3 * 4)
        lpx_nn._ck(L.lpxn_randn(x, batch * 224 * 224 * 3, 5, 1.0,
                                m.stream.h), "randn")
        m.forward(x, batch)          # calibration
        m.stream.sync()

        def run_eager(iters=10):
            t0 = time.perf_counter()
            for _ in range(iters):
                m.forward(x, batch)
            m.stream.sync()
            return (time.perf_counter() - t0) / iters

        run_eager(3)
        dt_e = run_eager()
        ok = m.capture(x, batch)
        m.stream.sync()

        def run_graph(iters=20):
            t0 = time.perf_counter()
            for _ in range(iters):
                m.forward(x, batch)
            m.stream.sync()
            return (time.perf_counter() - t0) / iters

        run_graph(5)
        dt_g = run_graph()
        print(f"bs={batch}: eager {batch/dt_e:8.0f} img/s ({dt_e*1e3:.2f} ms)"
              f"  graph {batch/dt_g:8.0f} img/s ({dt_g*1e3:.2f} ms)"
              f"  capture_ok={ok}")

    # torch-extension reference (same kernels through lpx_torch)
    try:
        import torch

        from lambdipy_amd.shims import lpx_torch

        for batch in (64, 128):
            model = lpx_torch.ResNet50FP8(n_classes=1000, device="cuda",
                                          seed=3)
            X = torch.randn(batch, 3, 224, 224, device="cuda")
            model(X)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                model(X)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 10
            print(f"torch-path bs={batch}: {batch/dt:8.0f} img/s "
                  f"({dt*1e3:.2f} ms)")
    except Exception as e:  # noqa: BLE001
        print("torch-path skipped:", e)


if __name__ == "__main__":
    main()
