import torch, sys, time
sys.path.insert(0, "/root/repo")
from oktopk_amd import _hip_ops as H

torch.manual_seed(0)
n = 10_000_000
for with_g in (False, True):
    t = torch.randn(n, device="cuda")
    r = torch.randn(n, device="cuda") * 0.05
    g = torch.randn(n, device="cuda").bfloat16() if with_g else None
    restored = (g.float() + r) if with_g else (t + r)
    tau0 = torch.kthvalue(restored.abs().cpu(), n - n//1000).values.item()
    taus = [tau0 * 0.97 * 1.03 ** i for i in range(4)]
    r0 = r.clone(); t0 = t.clone()
    idx, val, chosen, cnt = H.compact_adaptive_ef(t, r, g, taus, 4*(n//1000)//3)
    torch.cuda.synchronize()
    print(f"with_g={with_g} chosen={int(chosen)} cnt={int(cnt)} idx={idx.numel()}")
    # invariants
    ok_r = torch.allclose(r, restored, atol=1e-6)
    ok_t = torch.equal(t, t0)
    ok_val = torch.allclose(val, restored[idx.long()], atol=1e-6)
    sel = (restored.abs() > taus[int(chosen)]).sum().item()
    print("  r==restored:", ok_r, " t untouched:", ok_t, " val==restored[idx]:", ok_val,
          " count match:", sel == idx.numel(), sel)
    # timing
    def timeit(f, it=10):
        for _ in range(3): f()
        torch.cuda.synchronize(); s=time.perf_counter()
        for _ in range(it): f()
        torch.cuda.synchronize(); return (time.perf_counter()-s)/it*1000
    r.copy_(r0)
    ms = timeit(lambda: (r.copy_(r0), H.compact_adaptive_ef(t, r, g, taus, 4*(n//1000)//3)))
    ms_reset = timeit(lambda: r.copy_(r0))
    print(f"  op ms (n=10M, minus reset): {ms - ms_reset:.3f}")
