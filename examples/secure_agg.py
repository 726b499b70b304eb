"""Pairwise-masked secure aggregation across two parties.

The classic secure-aggregation shape: each party adds a pairwise random
mask to its update before sending; the masks cancel in the sum, so the
aggregator learns only the aggregate.  The mask addition and the final
combine run as HIP kernels (`masked_add_`, `fedavg_reduce_`) when a GPU is
visible.  Run one process per party:

    python examples/secure_agg.py alice
    python examples/secure_agg.py bob
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import rayfed_amd as fed


def main(party):
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    addresses = {"alice": "127.0.0.1:11032", "bob": "127.0.0.1:11031"}
    fed.init(addresses=addresses, party=party)

    n = 1 << 20

    @fed.remote
    def masked_update(seed, mask_seed, sign):
        """Local update + pairwise mask (same mask_seed in both parties,
        opposite signs — cancels in the aggregate)."""
        g = torch.Generator(device="cpu").manual_seed(seed)
        update = torch.randn(n, generator=g).to(dev)
        gm = torch.Generator(device="cpu").manual_seed(mask_seed)
        mask = torch.randn(n, generator=gm).to(dev)
        return update + float(sign) * mask

    @fed.remote
    def aggregate(a, b):
        from rayfed_amd.parallel.fedavg import weighted_combine_

        out = torch.empty_like(a)
        weighted_combine_(out, [a, b], [1.0, 1.0])
        return float(out.float().mean())

    ua = masked_update.party("alice").remote(1, 42, +1)
    ub = masked_update.party("bob").remote(2, 42, -1)
    agg = aggregate.party("bob").remote(ua, ub)
    result = fed.get(agg)

    # Reference: the UNMASKED sum has the same mean (masks cancelled).
    ga = torch.randn(n, generator=torch.Generator().manual_seed(1))
    gb = torch.randn(n, generator=torch.Generator().manual_seed(2))
    expect = float((ga + gb).mean())
    print(f"[{party}] masked aggregate mean {result:.6f} "
          f"(unmasked reference {expect:.6f})")
    assert abs(result - expect) < 5e-3
    fed.shutdown()


if __name__ == "__main__":
    assert len(sys.argv) == 2, "Please run this script with a party name."
    main(sys.argv[1])
