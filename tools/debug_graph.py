"""Bisect hipGraph capture of the TIGER train step: capture progressively
larger regions and report which stage fails."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from genrec_amd.models.tiger import Tiger  # noqa: E402


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    dropout = float(os.environ.get("DBG_DROPOUT", "0.1"))
    model = Tiger(embedding_dim=128, attn_dim=384, dropout=dropout,
                  num_heads=6, n_layers=8, num_item_embeddings=256,
                  num_user_embeddings=10000, sem_id_dim=3).to(dev)
    model.train()
    B = int(os.environ.get("DBG_B", "64"))
    L = 60
    batch = dict(
        user_input_ids=torch.randint(0, 10000, (B, 1), device=dev),
        item_input_ids=torch.randint(0, 256, (B, L), device=dev),
        token_type_ids=(torch.arange(L, device=dev) % 3).unsqueeze(0)
        .expand(B, -1).contiguous(),
        target_input_ids=torch.randint(0, 256, (B, 3), device=dev),
        target_token_type_ids=torch.arange(3, device=dev).unsqueeze(0)
        .expand(B, -1).contiguous(),
        seq_mask=torch.ones(B, L, dtype=torch.long, device=dev),
    )
    if os.environ.get("DBG_RAGGED", "0") == "1":
        lens = torch.randint(5, 21, (B,))
        for i in range(0, B, 3):
            batch["seq_mask"][i, lens[i] * 3:] = 0
    amp = torch.autocast(device_type="cuda", dtype=torch.bfloat16)
    params = [p for p in model.parameters() if p.requires_grad]
    opt = torch.optim.AdamW(params, lr=1e-4, capturable=True, foreach=True)

    with amp:
        model(**batch).loss.backward()
    flat = torch.zeros(sum(p.numel() for p in params), device=dev)
    off = 0
    for p in params:
        p.grad = flat[off:off + p.numel()].view_as(p)
        off += p.numel()

    stages = {
        "fwd": lambda: model(**batch).loss,
        "fwd_bwd": None,
        "fwd_bwd_clip": None,
        "full": None,
    }

    def fwd():
        with amp:
            return model(**batch).loss

    def fwd_bwd():
        flat.zero_()
        loss = fwd()
        loss.backward()
        return loss

    def fwd_bwd_clip():
        loss = fwd_bwd()
        n = flat.norm()
        flat.mul_(torch.clamp(1.0 / (n + 1e-6), max=1.0))
        return loss

    def full():
        loss = fwd_bwd_clip()
        opt.step()
        return loss

    for name, fn in [("fwd", fwd), ("fwd_bwd", fwd_bwd),
                     ("fwd_bwd_clip", fwd_bwd_clip), ("full", full)]:
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    fn()
            torch.cuda.current_stream().wait_stream(s)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                fn()
            for r in range(4):
                g.replay()
                torch.cuda.synchronize()  # bench-style sync between replays
                print(f"  {name}: replay {r} ok", flush=True)
            print(f"stage {name}: OK", flush=True)
        except Exception as e:
            torch.cuda.synchronize()
            print(f"stage {name}: FAIL {type(e).__name__}: {e}", flush=True)
            break


if __name__ == "__main__":
    main()
