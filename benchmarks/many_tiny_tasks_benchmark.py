"""many_tiny_tasks benchmark — CLI parity with the reference harness.

Reference: /root/reference/benchmarks/many_tiny_tasks_benchmark.py:35-67
(same loop shape: per iteration one actor call per party, one cross-party
2-arg aggregate in alice, one fed.get; prints total ms and per-task
overhead).  Run one process per party:

    python benchmarks/many_tiny_tasks_benchmark.py alice &
    python benchmarks/many_tiny_tasks_benchmark.py bob

``bench.py`` at the repo root wraps the same loop under the driver's
measurement contract; this script is the interactive/parity harness.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import rayfed_amd as fed  # noqa: E402


@fed.remote
class MyActor:
    def run(self):
        return None


@fed.remote
class Aggregator:
    def aggr(self, val1, val2):
        return None


def main(party: str, num_calls: int = 10000):
    addresses = {
        "alice": "127.0.0.1:11010",
        "bob": "127.0.0.1:11011",
    }
    fed.init(addresses=addresses, party=party, logging_level="warning")

    actor_alice = MyActor.party("alice").remote()
    actor_bob = MyActor.party("bob").remote()
    aggregator = Aggregator.party("alice").remote()

    start = time.time()
    for i in range(num_calls):
        val_alice = actor_alice.run.remote()
        val_bob = actor_bob.run.remote()
        sum_val_obj = aggregator.aggr.remote(val_alice, val_bob)
        fed.get(sum_val_obj)
        if i % 100 == 0:
            print(f"Running {i}th call")
    print(f"num calls: {num_calls}")
    print("total time (ms) = ", (time.time() - start) * 1000)
    print("per task overhead (ms) =", (time.time() - start) * 1000 / num_calls)

    fed.shutdown()


if __name__ == "__main__":
    assert len(sys.argv) >= 2, "Please run this script with a party name."
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 10000
    main(sys.argv[1], n)
