#!/bin/bash
# End-to-end serving validation: build every config's layer through the CLI
# on the box, then cold-start each handler from its zip (the full
# build->package->deploy->serve path, configs 1-5 of BASELINE.json).
set -e
cd /root/repo
W=/tmp/e2e; rm -rf $W; mkdir -p $W

run_cfg () {  # name, requirements-content, handler
  echo "== config $1 (handler $3) =="
  echo -e "$2" > $W/req_$1.txt
  timeout 300 python -m lambdipy_amd.cli build -r $W/req_$1.txt -z $W/$1.zip --handler $3 -o $W/build_$1 > $W/build_$1.log 2>&1
  grep -E "layer:" $W/build_$1.log | tail -1
  timeout 300 python -m lambdipy_amd.cli run $W/$1.zip --handler $3 --json 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('  cold-start->first-inference:', d.get('coldstart_ms'), 'ms; ok =', d.get('ok', d))"
}

run_lite () {  # name, lite-handler — serve the SAME zip torch-free
  echo "-- $1 via $2 (torch-free) --"
  timeout 300 python -m lambdipy_amd.cli run $W/$1.zip --handler $2 --json 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('  cold-start->first-inference:', d.get('coldstart_ms'), 'ms; ok =', d.get('ok', d))"
}

run_cfg requests "requests" requests
run_cfg numpy "numpy" matmul
run_lite numpy matmul_lite
run_cfg scipy "numpy\nscipy" fftspmv
run_lite scipy fftspmv_lite
run_cfg sklearn "numpy\nscipy\nscikit-learn" rf
run_lite sklearn rf_lite
run_cfg torch "torch" resnet50
run_lite torch resnet50_lite
echo "ALL CONFIGS SERVED"
