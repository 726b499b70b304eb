"""The reference README's two-party demo, on this engine.

Parity: /root/reference/README.md:124-168 (`demo.py`) — same actor/aggregate
shape and the same run protocol (the identical script runs once per party):

    python examples/demo.py alice   # terminal 1
    python examples/demo.py bob     # terminal 2

No Ray is needed: `fed.init` brings up this party's in-process runtime.
Add TLS with ``--tls`` (self-signed certs are generated on the fly).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import rayfed_amd as fed


@fed.remote
class MyActor:
    def __init__(self, value):
        self.value = value

    def inc(self, num):
        self.value = self.value + num
        return self.value


@fed.remote
def aggregate(val1, val2):
    return val1 + val2


def main(party, tls=False):
    addresses = {
        "alice": "127.0.0.1:11012",
        "bob": "127.0.0.1:11011",
    }
    tls_config = None
    if tls:
        import os

        sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
        from tool.generate_tls_certs import generate

        tls_config = generate("/tmp/rayfed_amd/demo-certs")
        tls_config["target_name_override"] = "localhost"
    fed.init(addresses=addresses, party=party, tls_config=tls_config)

    actor_alice = MyActor.party("alice").remote(1)
    actor_bob = MyActor.party("bob").remote(1)

    val_alice = actor_alice.inc.remote(1)
    val_bob = actor_bob.inc.remote(2)

    sum_val_obj = aggregate.party("bob").remote(val_alice, val_bob)
    result = fed.get(sum_val_obj)
    print(f"The result in party {party} is {result}")

    fed.shutdown()


if __name__ == "__main__":
    assert len(sys.argv) >= 2, "Please run this script with a party name."
    main(sys.argv[1], tls="--tls" in sys.argv)
