import sys, time, torch
sys.path.insert(0, ".")
from llm_instance_gateway_amd import ops
st = torch.cuda.current_stream().cuda_stream
for (N, ctx, ns) in ((4, 300, 2), (16, 1024, 4), (128, 2048, 4), (3, 7, 3), (128, 2048, 1)):
    torch.manual_seed(0)
    HKV, HQ, D, BS = 8, 32, 128, 16
    bps = (ctx + BS - 1)//BS
    nb = N*bps
    k = torch.randn(nb+1, HKV, BS, D, dtype=torch.bfloat16, device="cuda")*0.2
    v = torch.randn(nb+1, HKV, BS, D, dtype=torch.bfloat16, device="cuda")*0.2
    q = torch.randn(N, HQ, D, dtype=torch.bfloat16, device="cuda")*0.3
    bt = torch.arange(nb, dtype=torch.int32, device="cuda").reshape(N, bps)
    cl = torch.full((N,), ctx, dtype=torch.int32, device="cuda"); cl[0] = max(1, ctx//2)
    ref = ops.paged_attention_decode(q, k, v, bt, cl, D**-0.5)
    out = torch.empty_like(ref)
    scr = torch.empty(N, HQ, ns, D+2, dtype=torch.float32, device="cuda")
    ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
    torch.cuda.synchronize()
    err = (out.float()-ref.float()).abs().max().item()
    assert err < 0.02, (N, ctx, ns, err)
    # quick timing at the long-ctx shape
    if ctx == 2048 and N == 128 and ns == 4:
        for _ in range(10): ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(50): ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
        torch.cuda.synchronize(); t_split=(time.perf_counter()-t0)/50*1e6
        for _ in range(10): ops.paged_attention_decode(q, k, v, bt, cl, D**-0.5)
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(50): ops.paged_attention_decode(q, k, v, bt, cl, D**-0.5)
        torch.cuda.synchronize(); t_base=(time.perf_counter()-t0)/50*1e6
        print(f"ctx=2048 N=128: split(ns=4) {t_split:.1f}us vs base {t_base:.1f}us")
    print("ok", N, ctx, ns, round(err,4))
print("SPLIT OK")
