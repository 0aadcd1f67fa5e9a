cd /root/repo
set -e
for N in 512 4096; do
  for V in 1 2 3; do
    MATREL_MX4_V=$V RING_N=$N timeout 300 python scripts/mx4_ring_ab.py
  done
  python - <<PYEOF
import torch
a = torch.load("/tmp/ring_v1_$N.pt")
for v in (2, 3):
    b = torch.load(f"/tmp/ring_v{v}_$N.pt")
    assert torch.equal(a, b), f"MISMATCH v1 vs v{v} at $N"
print("bit-exact v1 == v2 == v3 at $N")
PYEOF
done
for V in 1 2 1 2; do
  MATREL_MX4_V=$V RING_N=16384 RING_REPS=2 timeout 300 python scripts/mx4_ring_ab.py
done
MATREL_MX4_V=2 RING_N=32768 RING_REPS=2 timeout 300 python scripts/mx4_ring_ab.py
MATREL_MX4_V=1 RING_N=32768 RING_REPS=2 timeout 300 python scripts/mx4_ring_ab.py
