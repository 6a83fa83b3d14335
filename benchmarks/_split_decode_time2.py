import sys, time, torch
sys.path.insert(0, ".")
from llm_instance_gateway_amd import ops
st = torch.cuda.current_stream().cuda_stream
torch.manual_seed(0)
HKV, HQ, D, BS = 8, 32, 128, 16
N, ctx = 4, 16384
bps = ctx//BS; nb = N*bps
k = torch.randn(nb+1, HKV, BS, D, dtype=torch.bfloat16, device="cuda")*0.2
v = torch.randn(nb+1, HKV, BS, D, dtype=torch.bfloat16, device="cuda")*0.2
q = torch.randn(N, HQ, D, dtype=torch.bfloat16, device="cuda")*0.3
bt = torch.arange(nb, dtype=torch.int32, device="cuda").reshape(N, bps)
cl = torch.full((N,), ctx, dtype=torch.int32, device="cuda")
ref = ops.paged_attention_decode(q, k, v, bt, cl, D**-0.5)
for ns in (1, 4, 8, 16, 32):
    out = torch.empty_like(ref)
    scr = torch.empty(N, HQ, ns, D+2, dtype=torch.float32, device="cuda")
    ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
    torch.cuda.synchronize()
    err = (out.float()-ref.float()).abs().max().item(); assert err < 0.02, (ns, err)
    for _ in range(5): ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(30): ops._EXT.paged_attention_decode_split(out, q, k, v, bt, cl, scr, D**-0.5, ns, st)
    torch.cuda.synchronize(); print(f"ns={ns:2d}: {(time.perf_counter()-t0)/30*1e6:7.1f} us")
torch.cuda.synchronize(); t0=time.perf_counter()
for _ in range(30): ops.paged_attention_decode(q, k, v, bt, cl, D**-0.5)
torch.cuda.synchronize(); print(f"base : {(time.perf_counter()-t0)/30*1e6:7.1f} us  (N=4 ctx=16384)")
