cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
SB=sboxgates_amd/sboxes/rijndael.txt
echo "=== gate-mode AES bit 0 (hybrid, after host-loop optimization) ==="
timeout 150 bash -c "time ./bin/sboxgates -o 0 --seed 11 -v $SB" 2>&1 | tail -4
echo "=== config 2 continuation ==="
mkdir -p gpurun_out/aes_gate8
cp results/scratch_aes_gate8/*.xml gpurun_out/aes_gate8/ 2>/dev/null
timeout 120 bash -c "./bin/sboxgates --resume-dir gpurun_out/aes_gate8 --beam 1 --jobs 8 --seed 11 -v $SB" > gpurun_out/aes_gate8_d.log 2>&1
echo "rc=$?"; grep -E "Resuming|Found|outputs" gpurun_out/aes_gate8_d.log | tail -3
ls gpurun_out/aes_gate8/ | sort | tail -3
echo ALL_DONE
