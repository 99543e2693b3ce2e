import torch, time
torch.manual_seed(0)
dev = "cuda"
B, T, H = 8, 4096, 4096
M = B * T
x = torch.randn(M, H, dtype=torch.bfloat16, device=dev)
# qkv: 4096->4096,1024,1024  fused 6144
wq = torch.randn(4096, H, dtype=torch.bfloat16, device=dev)
wk = torch.randn(1024, H, dtype=torch.bfloat16, device=dev)
wv = torch.randn(1024, H, dtype=torch.bfloat16, device=dev)
wqkv = torch.cat([wq, wk, wv], 0)
# mlp: 4096->14336 x2 fused 28672
wg = torch.randn(14336, H, dtype=torch.bfloat16, device=dev)
wu = torch.randn(14336, H, dtype=torch.bfloat16, device=dev)
wgu = torch.cat([wg, wu], 0)

def timeit(fn, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e3

t_sep_qkv = timeit(lambda: (x@wq.t(), x@wk.t(), x@wv.t()))
t_fus_qkv = timeit(lambda: x@wqkv.t())
t_sep_mlp = timeit(lambda: (x@wg.t(), x@wu.t()))
t_fus_mlp = timeit(lambda: x@wgu.t())
print(f"qkv separate {t_sep_qkv:.3f} ms  fused {t_fus_qkv:.3f} ms")
print(f"mlp separate {t_sep_mlp:.3f} ms  fused {t_fus_mlp:.3f} ms")
# backward-ish: dgrad (grad@W) and wgrad (grad^T@x)
gq = torch.randn(M, 6144, dtype=torch.bfloat16, device=dev)
gsep = gq.split([4096,1024,1024], 1)
gsepc = [g.contiguous() for g in gsep]
t_dg_sep = timeit(lambda: (gsepc[0]@wq, gsepc[1]@wk, gsepc[2]@wv))
t_dg_fus = timeit(lambda: gq@wqkv)
t_wg_sep = timeit(lambda: (gsepc[0].t()@x, gsepc[1].t()@x, gsepc[2].t()@x))
t_wg_fus = timeit(lambda: gq.t()@x)
print(f"qkv dgrad separate {t_dg_sep:.3f} fused {t_dg_fus:.3f}")
print(f"qkv wgrad separate {t_wg_sep:.3f} fused {t_wg_fus:.3f}")
