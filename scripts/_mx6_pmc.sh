cd /tmp && export TMPDIR=/tmp
set -x
timeout 420 rocprofv3 --pmc SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_INSTS_LDS SQ_INSTS_MFMA -d /root/repo/gpurun_out/pmc_mx6 -o runc --output-format csv -- bash -c 'cd /root/repo && GEMM_N=16384 python scripts/mx6_pmc.py' 2>&1 | tail -3
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/mx6_trace -o tr --output-format csv -- bash -c 'cd /root/repo && GEMM_N=16384 python scripts/mx6_pmc.py' 2>&1 | tail -4
cd /root/repo
MX6_ST_SWEEP=1 timeout 600 python scripts/mx6_ab.py 2>&1 | tail -20
