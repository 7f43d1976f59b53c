import torch
print("fp8 dtypes:", hasattr(torch, "float8_e4m3fn"), hasattr(torch, "float8_e4m3fnuz"))
for dt in ("float8_e4m3fn", "float8_e4m3fnuz"):
    try:
        d = getattr(torch, dt)
        a = (torch.randn(256, 512, device="cuda") * 0.1).to(d)
        b = (torch.randn(1024, 512, device="cuda") * 0.1).to(d)
        sa = torch.tensor(1.0, device="cuda")
        out = torch._scaled_mm(a, b.t(), scale_a=sa, scale_b=sa,
                               out_dtype=torch.bfloat16)
        ref = a.to(torch.float32) @ b.to(torch.float32).t()
        err = (out.float() - ref).abs().max().item()
        print(dt, "OK shape", tuple(out.shape), "maxerr", round(err, 4))
    except Exception as e:
        print(dt, "FAIL:", str(e)[:160])
