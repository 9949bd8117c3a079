# Inference handlers shipped into layers under lpx_handlers/.
# Each module is standalone (imports only torch/numpy + the layer's shims)
# and exposes main() which runs ONE cold inference and prints a single
# "LPX_RESULT {json}" line (the cold-start harness times to that line).
