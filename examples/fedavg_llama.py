"""Cross-party FedAvg on Llama-3-8B-sized gradients — the flagship GPU flow.

Each party holds a 16 GB bf16 flat gradient buffer on its GPU; every round
both parties push their gradients to each other over the device-IPC lane
(zero-copy receive: the peer's slabs are combined directly by the fused
HIP combine+verify kernel) and apply the weighted average.  Run one
process per party on the same node:

    python examples/fedavg_llama.py alice
    python examples/fedavg_llama.py bob

Shrink with --layers/--vocab for smoke runs (works on CPU too).
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import rayfed_amd as fed
from rayfed_amd.parallel.fedavg import weighted_combine_


def grad_shapes(layers, vocab):
    shapes = [(vocab, 4096)]
    for _ in range(layers):
        shapes += [
            (4096, 4096), (1024, 4096), (1024, 4096), (4096, 4096),
            (14336, 4096), (14336, 4096), (4096, 14336), (4096,), (4096,),
        ]
    shapes += [(4096,), (vocab, 4096)]
    return shapes


def main():
    p = argparse.ArgumentParser()
    p.add_argument("party", choices=["alice", "bob"])
    p.add_argument("--rounds", type=int, default=5)
    p.add_argument("--layers", type=int, default=32)
    p.add_argument("--vocab", type=int, default=128256)
    args = p.parse_args()

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    addresses = {"alice": "127.0.0.1:11022", "bob": "127.0.0.1:11021"}
    fed.init(addresses=addresses, party=args.party,
             config={"gpu_data_plane": {"lazy_ipc": True}})

    torch.manual_seed(0 if args.party == "alice" else 1)
    total = sum(
        int(torch.prod(torch.tensor(s))) for s in grad_shapes(args.layers,
                                                              args.vocab)
    )
    flat = torch.empty(total, dtype=torch.bfloat16, device=dev).uniform_(-1, 1)

    @fed.remote
    class Combiner:
        def combine(self, peer, local):
            out = torch.empty_like(local)
            # peer may be a zero-copy LazyIpcTensor — the fused HIP kernel
            # reads the sender's slabs directly and verifies in-pass.
            weighted_combine_(out, [local, peer], [0.5, 0.5])
            return float(out[:2].float().sum())

    combiners = {p_: Combiner.party(p_).remote() for p_ in addresses}

    @fed.remote
    def produce(_round):
        return flat

    import time

    for r in range(args.rounds):
        t0 = time.perf_counter()
        fa = produce.party("alice").remote(r)
        fb = produce.party("bob").remote(r)
        ca = combiners["alice"].combine.remote(fb, fa)
        cb = combiners["bob"].combine.remote(fa, fb)
        fed.get([ca, cb])
        dt = time.perf_counter() - t0
        gb = 2 * total * 2 / 1e9
        print(f"round {r}: {gb:.1f} GB exchanged+combined in "
              f"{dt*1e3:.1f} ms ({gb/dt:.0f} GB/s)")

    fed.shutdown()


if __name__ == "__main__":
    main()
