import torch, time
print("fp8 dtypes:", hasattr(torch, "float8_e4m3fn"), hasattr(torch, "float8_e4m3fnuz"))
dev = "cuda"
M, K, N = 8192, 4096, 4096
a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
# bf16 baseline
for _ in range(3): c = a @ b.t()
torch.cuda.synchronize(); t0 = time.time()
for _ in range(20): c = a @ b.t()
torch.cuda.synchronize()
tb = (time.time() - t0) / 20
print(f"bf16: {2*M*K*N/tb/1e12:.0f} TF")
for dt_name in ("float8_e4m3fn", "float8_e4m3fnuz"):
    if not hasattr(torch, dt_name):
        continue
    dt = getattr(torch, dt_name)
    try:
        a8 = a.to(dt); b8 = b.to(dt)
        sa = torch.tensor(1.0, device=dev); sb = torch.tensor(1.0, device=dev)
        c8 = torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16)
        torch.cuda.synchronize(); t0 = time.time()
        for _ in range(20):
            c8 = torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16)
        torch.cuda.synchronize()
        t8 = (time.time() - t0) / 20
        err = (c8.float() - c.float()).abs().max() / c.float().abs().max()
        print(f"{dt_name}: {2*M*K*N/t8/1e12:.0f} TF rel_err {err:.3f}")
    except Exception as e:
        print(f"{dt_name}: FAIL {type(e).__name__}: {str(e)[:140]}")
