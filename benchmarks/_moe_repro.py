import sys, torch
sys.path.insert(0, ".")
from llm_instance_gateway_amd.models.config import get_model_config
from llm_instance_gateway_amd.models.mixtral import MoEBlock
from llm_instance_gateway_amd import ops

def sync(tag):
    torch.cuda.synchronize()
    print("ok:", tag, flush=True)

mc = get_model_config("mixtral-8x7b")
torch.manual_seed(0)
blk = MoEBlock(mc, device="cuda:0", dtype=torch.bfloat16)
sync("init")
for T in (32, 256, 4096):
    x = torch.randn(T, mc.hidden_size, device="cuda:0", dtype=torch.bfloat16) * 0.1
    # manual step-through of forward
    logits = blk.gate(x); sync(f"gate T={T}")
    w, e = torch.topk(logits, blk.top_k, dim=-1)
    w = torch.softmax(w.float(), dim=-1).to(x.dtype)
    fe = e.reshape(-1); ft = torch.arange(T, device=x.device).unsqueeze(1).expand(T, blk.top_k).reshape(-1)
    fw = w.reshape(-1)
    order = torch.argsort(fe, stable=True)
    se, st_, sw = fe[order], ft[order], fw[order]
    counts = torch.bincount(se, minlength=blk.num_experts)
    cap = int(counts.max()); print("cap", cap, counts.tolist(), flush=True)
    starts = torch.cumsum(counts, 0) - counts
    E = blk.num_experts
    pos = torch.arange(se.numel(), device=x.device) - starts.index_select(0, se)
    gi = torch.full((E*cap,), T, dtype=torch.long, device=x.device)
    dest = se * cap + pos
    gi.index_copy_(0, dest, st_); sync("index_copy")
    x_pad = torch.cat([x, x.new_zeros(1, mc.hidden_size)], dim=0)
    xe = x_pad.index_select(0, gi).view(E, cap, mc.hidden_size); sync("gather")
    b1 = torch.bmm(xe, blk.w13); sync("bmm1")
    act = ops.silu_and_mul(b1.view(E*cap, -1)); sync("silu")
    ye = torch.bmm(act.view(E, cap, -1), blk.w2).view(E*cap, mc.hidden_size); sync("bmm2")
    wp = torch.zeros(E*cap, 1, dtype=x.dtype, device=x.device)
    wp.index_copy_(0, dest, sw.unsqueeze(1))
    out = torch.zeros_like(x)
    out.index_add_(0, gi.clamp(max=T-1), ye * wp); sync("scatter")
    full = blk(x); sync("full-forward")
    print("diff", (full - out).abs().max().item(), flush=True)
print("ALL OK", flush=True)
