"""Bisect which part of the COBRA step breaks hipGraph capture.

Run on GPU: python tools/repro_cobra_capture.py
Captures progressively larger pieces of the forward and prints the first
failure with a full traceback.
"""

import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def try_capture(name, fn):
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                fn()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"OK   {name}")
        return True
    except Exception:
        print(f"FAIL {name}")
        traceback.print_exc()
        torch.cuda.synchronize()
        return False


def main():
    from genrec_amd.models.cobra import Cobra

    torch.manual_seed(0)
    dev = "cuda:0"
    m = Cobra(encoder_n_layers=2, encoder_hidden_dim=384,
              encoder_num_heads=6, encoder_vocab_size=32128,
              id_vocab_size=256, n_codebooks=3, d_model=384,
              decoder_n_layers=4, decoder_num_heads=6,
              decoder_dropout=0.1).to(dev).to(torch.bfloat16)
    m.train()
    m.static_infonce = True
    B, T, Ltxt = 32, 21, 16
    ids = torch.randint(0, 256, (B, T * 3), device=dev)
    enc = torch.randint(1, 32128, (B, T, Ltxt), device=dev)

    with torch.no_grad():
        vecs = m.encoder(enc)
        seq_mask = m.interleave_seq_mask(ids != m.pad_id, m.C)
        emb = m.cobra_emb(ids, vecs, seq_mask)
    try_capture("encoder", lambda: m.encoder(enc))
    try_capture("interleave+embed",
                lambda: m.cobra_emb(ids, vecs, seq_mask))
    try_capture("decoder",
                lambda: m.decoder(emb, tgt_key_padding_mask=~seq_mask))
    try_capture("full fwd", lambda: m(ids, enc).loss)
    # fwd+bwd must not hold stale autograd state from the phases above
    # (AccumulateGrad nodes pinned to the default stream break capture)
    del vecs, emb
    for p in m.parameters():
        p.grad = None

    def fwd_bwd():
        out = m(ids, enc)
        (out.loss_sparse + out.loss_dense).backward()

    try_capture("fwd+bwd", fwd_bwd)


if __name__ == "__main__":
    main()
