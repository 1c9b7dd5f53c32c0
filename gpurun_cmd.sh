cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
SB=sboxgates_amd/sboxes/rijndael.txt
cat > /tmp/svc_micro.py <<'PY'
import sys, time, os
sys.path.insert(0, '.')
from sboxgates_amd import models
from sboxgates_amd.ops import make_engine, mask_for_inputs
eng = make_engine(gpu="force", seed=1)
sbox, n = models.load("rijndael")
eng.set_sbox(sbox, n)
st = eng.initial_state()
st.grow_pool_random(80, 7)
t = eng.target(0); mask = mask_for_inputs(8)
mode = "nosvc" if os.environ.get("SBOXGATES_NO_SVC") else "svc"
for rng in (512, 2048, 16384, 82160):
    for _ in range(30): eng.scan_pool(4, st, t, mask, 0, rng, 1, True)
    N = 2000
    t0 = time.perf_counter()
    for i in range(N):
        f, r, ev = eng.scan_pool(4, st, t, mask, 0, rng, i, True)
        assert ev == rng
    dt = (time.perf_counter() - t0) / N
    print(f"[{mode}] k4 range={rng}: {dt*1e6:.1f} us/call (python-inclusive)", flush=True)
# pool-delta path: alternate pools to exercise prefix sync
st2 = eng.initial_state(); st2.grow_pool_random(80, 8)
t0 = time.perf_counter(); N = 1000
for i in range(N):
    f, r, ev = eng.scan_pool(4, st if i % 2 else st2, t, mask, 0, 2048, i, True)
    assert ev == 2048
print(f"[{mode}] k4 alternating pools range=2048: {(time.perf_counter()-t0)/N*1e6:.1f} us/call", flush=True)
PY
echo "=== micro svc ==="
timeout 240 python -u /tmp/svc_micro.py 2>&1 | tail -6
echo "=== micro nosvc ==="
SBOXGATES_NO_SVC=1 timeout 240 python -u /tmp/svc_micro.py 2>&1 | tail -6
echo "=== k4 parity tests ==="
timeout 600 python -m pytest tests/test_gpu.py::test_scan4_gpu_matches_cpu tests/test_gpu.py::test_gpu_window_fuzz_soak tests/test_gpu.py::test_random_window_count_parity -x -q 2>&1 | tail -3
echo "=== gate-mode AES bit 0 (auto hybrid) ==="
timeout 300 bash -c "time ./bin/sboxgates -o 0 --seed 11 -v $SB" > gpurun_out/gate_auto.log 2>&1
echo "rc=$?"; tail -5 gpurun_out/gate_auto.log
echo "=== gate-mode AES bit 0 (auto, no service) ==="
SBOXGATES_NO_SVC=1 timeout 300 bash -c "time ./bin/sboxgates -o 0 --seed 11 -v $SB" > gpurun_out/gate_auto_nosvc.log 2>&1
echo "rc=$?"; tail -5 gpurun_out/gate_auto_nosvc.log
echo ALL_DONE
