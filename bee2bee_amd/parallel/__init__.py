"""Multi-GPU parallelism over torch.distributed (RCCL on ROCm).

The WS mesh is control-plane only; activations move over RCCL/xGMI:
  * rendezvous.py — mesh peer-ids -> torch.distributed ranks (DHT records)
  * pp.py         — pipeline parallelism: contiguous layer shards, bf16
                    hidden-state hops via send/recv (Llama-3-70B across 8
                    peers, BASELINE config 4)
  * planner.py    — layer-shard planning sized to HBM (pieces.py semantics
                    generalized, reference hf.py:180-205 layer ranges)
"""
