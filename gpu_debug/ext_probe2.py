"""Stage/frag/bhi-blo introspection for the KT=2 EXT bug."""

import torch

import flink_ms_amd._hip_ops as hip
from flink_ms_amd.data.blocked import CSR

dev = torch.device("cuda:0")
st = lambda: torch.cuda.current_stream().cuda_stream

k = 32
SP = 64
n = 32
G = torch.eye(n)                       # [n][k]
r = torch.arange(1, n + 1, dtype=torch.float32)
csr = CSR(torch.tensor([0, n], dtype=torch.int64),
          torch.arange(n, dtype=torch.int32), r, 1, n).to(dev)
fac = G.to(torch.bfloat16).to(dev)

# --- staged LDS image ---
out = torch.zeros(32 * SP, dtype=torch.bfloat16, device=dev)
hip.dbg_stage_dump(csr.indptr, csr.indices, csr.values, fac, out, st())
torch.cuda.synchronize()
img = out.cpu().float().view(32, SP)
print("stage check: factor region identity?",
      bool(torch.equal(img[:, :32], torch.eye(32))))
print("ext col hi (should be 1..32):", img[:, 32].tolist())
print("ext col lo (should be 0):", img[:, 33].abs().max().item())
print("ext cols 34..47 max:", img[:, 34:48].abs().max().item())
print("pad cols 48..63 max:", img[:, 48:].abs().max().item())

# --- fragment dump ---
fr = torch.zeros(4 * 3 * 64 * 8, dtype=torch.bfloat16, device=dev)
hip.dbg_frag_dump(csr.indptr, csr.indices, csr.values, fac, fr, st())
torch.cuda.synchronize()
fr = fr.cpu().float().view(4, 3, 64, 8)
# expected frag[t][lane][j] = stage_elem[(lane>>4)*8+j][t*16 + (lane&15)]
lanes = torch.arange(64)
exp = torch.zeros(3, 64, 8)
for t in range(3):
    for l in range(64):
        for j in range(8):
            exp[t, l, j] = img[(l // 16) * 8 + j, t * 16 + (l % 16)]
for w in range(4):
    for t in range(3):
        diff = (fr[w, t] - exp[t]).abs()
        if diff.max() > 0:
            bad = torch.nonzero(diff)
            print(f"FRAG MISMATCH wave {w} tile {t}: nbad={len(bad)}; "
                  f"first: {[ (int(a), int(b), float(fr[w,t,a,b]), float(exp[t,a,b])) for a,b in bad[:8] ]}")
        else:
            print(f"frag wave {w} tile {t}: OK")

# --- bhi / blo ---
A = torch.zeros(1, k, k, device=dev)
bhi = torch.zeros(1, k, device=dev)
blo = torch.zeros(1, k, device=dev)
hip.dbg_gramian(csr.indptr, csr.indices, csr.values, fac, A, bhi, blo,
                0.5, st())
torch.cuda.synchronize()
print("bhi (expect 1..32):", bhi[0].cpu().tolist())
print("blo (expect 0):", blo[0].cpu().tolist())
A0 = A[0].cpu()
eA = torch.eye(k)
print("A check (no reg in dbg):", (A0 - eA).abs().max().item())
