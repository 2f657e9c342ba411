"""Distributed-execution helpers: gang utilities, bucketed DDP over RCCL,
and the sharded checkpoint path."""
