cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
SB=sboxgates_amd/sboxes/rijndael.txt
mkdir -p gpurun_out/aes_gate8
cp results/scratch_aes_gate8/*.xml gpurun_out/aes_gate8/ 2>/dev/null
echo "=== config 2 continuation (resume from 2 outputs, jobs 8) ==="
timeout 620 bash -c "time ./bin/sboxgates --resume-dir gpurun_out/aes_gate8 --beam 1 --jobs 8 --seed 11 -v $SB" > gpurun_out/aes_gate8_c.log 2>&1
echo "rc=$?"
grep -E "Resuming|Found|outputs\.|No solution|fault" gpurun_out/aes_gate8_c.log | tail -10
ls gpurun_out/aes_gate8/ | sort | tail -6
echo ALL_DONE
