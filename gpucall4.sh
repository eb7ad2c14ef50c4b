cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_ops_gpu.py -q -k "packed or avgpool or class_rank or build_wT" > gpurun_out/pytest_new.log 2>&1; echo "new-op tests rc=$?"; tail -4 gpurun_out/pytest_new.log
timeout 400 python -m pytest tests/test_train_gpu.py tests/test_conv_gpu.py::test_resnet_native_conv_step -q > gpurun_out/pytest_train.log 2>&1; echo "train tests rc=$?"; tail -4 gpurun_out/pytest_train.log
timeout 240 python bench.py --steps 50 --warmup 15 --amp bf16 --channels-last > gpurun_out/bench_cl2.json 2>/dev/null; echo "bcl rc=$?"; cat gpurun_out/bench_cl2.json
timeout 240 python bench.py --steps 50 --warmup 15 --amp bf16 > gpurun_out/bench_nchw2.json 2>/dev/null; echo "bnchw rc=$?"; cat gpurun_out/bench_nchw2.json
export TMPDIR=/tmp; cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_cl2" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 20 --warmup 5 --amp bf16 --channels-last > "$GRAFT_REPO_ROOT/gpurun_out/rocprof_cl2.log" 2>&1; echo "prof rc=$?"
