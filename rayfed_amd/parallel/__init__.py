"""Intra-party parallelism: RCCL-over-xGMI process groups and FedAvg
aggregation."""
