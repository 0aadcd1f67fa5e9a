cd /tmp && export TMPDIR=/tmp
for P in mx8 mx6 mx4; do
  timeout 420 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/tcc_$P -o f --output-format csv -- bash -c "cd /root/repo && MX_PREC=$P python scripts/mx6_pmc.py" 2>&1 | tail -1
done
echo TCC-DONE
