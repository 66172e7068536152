import os, sys, torch
sys.path.insert(0, "/root/repo")
from torchrec_amd import ops
ops.hip_ops()
torch.manual_seed(0)
for (B, F) in [(4096, 2), (8192, 26)]:
    lengths = torch.full((F * B,), 1, dtype=torch.int64)
    offsets = torch.zeros(F * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    vals = torch.cat([torch.randint(0, 1 << 20, (B,)) + f * (1 << 20) for f in range(F)]).cuda()
    print("start", B, F, flush=True)
    s, p, _ = torch.ops.trec_amd.seg_sort_pairs_2level(vals, offsets.cuda(), B, F, 26, B)
    torch.cuda.synchronize()
    ok = True
    for f in range(F):
        seg = vals[f*B:(f+1)*B].cpu()
        rv, ri = torch.sort(seg, stable=True)
        if not torch.equal(s[f*B:(f+1)*B].cpu(), rv):
            ok = False; print("KEYS MISMATCH seg", f, flush=True); break
        if not torch.equal(p[f*B:(f+1)*B].cpu().long() - f*B, ri):
            ok = False; print("PERM MISMATCH seg", f, flush=True); break
    print("done", B, F, "ok" if ok else "BAD", flush=True)
