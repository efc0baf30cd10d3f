"""Diagnose the q8 pack bitwise mismatch between the CPU torch reference and
the GPU kernel at large n.

Ground truth: a strict-IEEE f32 numpy emulation of the documented scheme
(per-256-block m = max|x|; inv = f32div(127, m); q = clamp(rint(x*inv));
scale = f32div(m, 127)).  Prints, for every mismatching byte between CPU and
GPU payloads, which of the two differs from the strict emulation and the
local values, so the faulty side is identified in one run.
"""
import numpy as np
import torch

from ps_pytorch_amd.ops import functional as F

BLK = 256


def strict_pack(x32: np.ndarray):
    n = x32.size
    nblk = (n + BLK - 1) // BLK
    xp = np.zeros(nblk * BLK, dtype=np.float32)
    xp[:n] = x32
    xb = xp.reshape(nblk, BLK)
    m = np.abs(xb).max(axis=1)  # exact
    inv = np.where(m > 0, np.divide(np.float32(127.0), m, dtype=np.float32),
                   np.float32(0.0)).astype(np.float32)
    prod = (xb * inv[:, None]).astype(np.float32)
    q = np.clip(np.rint(prod), -127, 127).astype(np.int8)
    scales = np.divide(m, np.float32(127.0), dtype=np.float32)
    return q.reshape(-1)[:n], scales, m, inv, prod


def run(n, dtype):
    g = torch.Generator().manual_seed(n)
    x = (torch.randn(n, generator=g) * 0.01).to(dtype)
    qb, tot = F.q8_layout(n)
    nblk = (n + BLK - 1) // BLK
    p_cpu = torch.zeros(tot, dtype=torch.uint8)
    F.pack_q8(p_cpu, x)
    p_gpu = torch.zeros(tot, dtype=torch.uint8, device='cuda')
    F.pack_q8(p_gpu, x.cuda())
    torch.cuda.synchronize()
    p_gpu = p_gpu.cpu()
    x32 = x.to(torch.float32).numpy()
    sq, ss, m, inv, prod = strict_pack(x32)

    cq = p_cpu[:n].numpy().view(np.int8)
    gq = p_gpu[:n].numpy().view(np.int8)
    cs = p_cpu[qb:qb + 4 * nblk].numpy().view(np.float32)
    gs = p_gpu[qb:qb + 4 * nblk].numpy().view(np.float32)

    dq = np.nonzero(cq != gq)[0]
    ds = np.nonzero(cs.view(np.uint32) != gs.view(np.uint32))[0]
    print(f"=== n={n} dtype={dtype}: quant mismatches {dq.size}, "
          f"scale mismatches {ds.size}")
    print(f"    cpu-vs-strict: quant {np.count_nonzero(cq != sq)}, "
          f"scale {np.count_nonzero(cs.view(np.uint32) != ss.view(np.uint32))}")
    print(f"    gpu-vs-strict: quant {np.count_nonzero(gq != sq)}, "
          f"scale {np.count_nonzero(gs.view(np.uint32) != ss.view(np.uint32))}")
    for i in dq[:8]:
        b = i // BLK
        v = x32[i]
        p64 = np.float64(v) * np.float64(inv[b])
        print(f"  q[{i}] blk={b} cpu={cq[i]} gpu={gq[i]} strict={sq[i]} "
              f"v={v!r}({np.float32(v).view(np.uint32):08x}) "
              f"m={m[b]!r}({m[b].view(np.uint32):08x}) "
              f"inv={inv[b]!r} prod32={prod[b, i % BLK]!r} prod64={p64!r}")
    for b in ds[:8]:
        print(f"  s[{b}] cpu={cs[b]!r}({cs[b].view(np.uint32):08x}) "
              f"gpu={gs[b]!r}({gs[b].view(np.uint32):08x}) "
              f"strict={ss[b]!r}({ss[b].view(np.uint32):08x}) m={m[b]!r}")


if __name__ == "__main__":
    for n in (1 << 20, (1 << 20) + 13):
        for dt in (torch.float32, torch.bfloat16):
            run(n, dt)
