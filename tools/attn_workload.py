"""Fixed attention fwd+bwd workload at TIGER encoder shapes (for PMC runs)."""
import torch

from genrec_amd.ops.attention import fused_attention

def main():
    torch.manual_seed(0)
    B, H, L, D = 256, 6, 61, 64
    q = torch.randn(B, H, L, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    bias = torch.randn(H, L, L, device="cuda", dtype=torch.float32,
                       requires_grad=True)
    for _ in range(30):
        out = fused_attention(q, k, v, bias=bias, causal=False)
        out.sum().backward()
        q.grad = k.grad = v.grad = bias.grad = None
    torch.cuda.synchronize()
    print("done")

if __name__ == "__main__":
    main()
