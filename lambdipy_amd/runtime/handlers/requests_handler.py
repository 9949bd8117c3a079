"""Config-1 handler: pure-Python layer sanity (no GPU).

Imports the vendored `requests` from the layer and builds a Request object
(no network — BASELINE.json config 1 exercises resolver + zip plumbing)."""

from __future__ import annotations

import json
import os
import time


def main() -> dict:
    t0 = time.perf_counter()
    ready = os.environ.get("LPX_READY_FILE")
    if ready:  # overlapped cold start: layer may still be extracting
        while not os.path.exists(ready):
            time.sleep(0.002)
    import requests

    req = requests.Request("GET", "https://example.invalid/lpx").prepare()
    out = {
        "handler": "requests",
        "requests_version": requests.__version__,
        "prepared_url": req.url,
        "total_ms": (time.perf_counter() - t0) * 1e3,
        "phases": [],
    }
    print("LPX_RESULT " + json.dumps(out), flush=True)
    return out


if __name__ == "__main__":
    main()
