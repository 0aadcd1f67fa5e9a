cd /root/repo
set -e
for N in 256 512 4096; do
  MATREL_MX4_V=1 RING_N=$N timeout 300 python scripts/mx4_ring_ab.py
  MATREL_MX4_V=3 RING_N=$N timeout 300 python scripts/mx4_ring_ab.py
  python - <<PYEOF
import torch
a = torch.load("/tmp/ring_v1_$N.pt"); b = torch.load("/tmp/ring_v3_$N.pt")
assert torch.equal(a, b), "MISMATCH v1 vs v3 at $N: %g" % (a-b).abs().max().item()
print("bit-exact v1 == v3 at $N")
PYEOF
done
for V in 1 3 1 3; do
  MATREL_MX4_V=$V RING_N=16384 RING_REPS=2 timeout 300 python scripts/mx4_ring_ab.py
done
MATREL_MX4_V=3 RING_N=32768 RING_REPS=2 timeout 300 python scripts/mx4_ring_ab.py
