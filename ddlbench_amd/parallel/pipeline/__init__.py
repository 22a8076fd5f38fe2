"""Pipeline-parallel engines.

gpipe    — synchronous micro-batch pipeline, single process × N devices
           (the reference's torchgpipe path)
graph    — layer-graph IR + partitioning algorithms (ref §2.7)
profiler — hook-based per-layer profiler (ref §2.8, no autograd patch)
partition— pipeline-stage partitioner (ref §2.9)
runtime  — asynchronous 1F1B runtime with weight stashing over RCCL p2p
           (ref §2.10)
"""
