"""Multi-GPU data-parallel serving over RCCL/xGMI.

The reference's only parallelism is CPU sentence fan-out on a rayon pool
(crates/sonata/synth/src/lib.rs:316-320).  The MI355X-native equivalent
 is one process per GPU with torch.distributed
(backend "nccl" IS RCCL on ROCm): voice weights broadcast once at load,
utterances sharded across ranks, waveforms gathered to the serving rank
with a single length-aware all-gather per round.
"""

from .dp import (  # noqa: F401
    DistributedSynthesizer,
    broadcast_module,
    gather_audio_to_rank0,
    init_distributed,
    shard_round_robin,
)
