import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from genrec_amd.ops import eager
from genrec_amd.ops.attention import fused_attention
DEV="cuda:0"
torch.manual_seed(7)
B,H,L,D = 3,2,61,64
q = torch.randn(B,H,L,D, device=DEV, dtype=torch.bfloat16, requires_grad=True)
k = torch.randn_like(q, requires_grad=True)
v = torch.randn_like(q, requires_grad=True)
dout = torch.randn(B,H,L,D, device=DEV, dtype=torch.bfloat16)
bias = torch.randn(B,H,L,L, device=DEV, requires_grad=True)
kw = dict(bias=bias, causal=True, score_act="silu", scale=1.0)
out = fused_attention(q,k,v,**kw); out.backward(dout)
q2,k2,v2 = [t.detach().float().requires_grad_(True) for t in (q,k,v)]
b2 = bias.detach().float().requires_grad_(True)
ref = eager.fused_attention(q2,k2,v2,bias=b2,causal=True,score_act="silu",scale=1.0)
ref.backward(dout.float())
for name, a, b in [("out", out.float(), ref.detach()), ("dq", q.grad.float(), q2.grad),
                   ("dk", k.grad.float(), k2.grad), ("dv", v.grad.float(), v2.grad),
                   ("dbias", bias.grad.float(), b2.grad)]:
    d = (a-b).abs()
    rel = (d/(b.abs()+1)).max()
    print(f"{name}: maxabs={d.max().item():.4f} rel={rel.item():.4f} refmax={b.abs().max().item():.1f}")
