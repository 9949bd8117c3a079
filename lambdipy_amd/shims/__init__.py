# Shim modules bundled into recipe layers as TOP-LEVEL modules (lpx_numpy,
# lpx_scipy, lpx_sklearn, lpx_torch). In-repo they are importable as
# lambdipy_amd.shims.* for dev/tests; each is self-contained + lpx_ops only.
