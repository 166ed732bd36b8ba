#!/bin/bash
# Round-2 GPU call 3: NHWC BN validation + channels_last fused-BN bench.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

# clean steady-state profile FIRST (before any SEARCH pollutes the find-db):
# NCHW fused-BN eager-dispatch flagship
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --output-format rocpd \
    -d gpurun_out/prof_r02 -o r02 \
    -- python bench.py --steps 12 --warmup 6 --no-hip-graph \
    > gpurun_out/prof_bench.log 2>&1
echo "prof rc=$?"
DB=$(find gpurun_out/prof_r02 -name '*.db' | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/prof_steady_r02.txt 2>&1 || true
rm -rf gpurun_out/prof_r02

timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu3.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/pytest_gpu3.log

# the headline candidate: channels_last + NHWC fused BN + SEARCH
timeout 600 python bench.py --steps 30 --warmup 10 --channels-last \
    > gpurun_out/b_cl_fused.log 2>&1
echo "cl_fused rc=$?"
# graph on top of channels_last
timeout 420 python bench.py --steps 30 --warmup 10 --channels-last --hip-graph \
    > gpurun_out/b_cl_fused_graph.log 2>&1
echo "cl_fused_graph rc=$?"
grep -h '"metric"' gpurun_out/b_cl_fused.log gpurun_out/b_cl_fused_graph.log
