#!/usr/bin/env python3
"""Generate synthetic profile JSONs for CPU tests (no GPU, no reference).

Produces a 10-layer GPT-ish profile set for two device types (MI355X and a
clock-capped MI355X_LC at 0.5x throughput), tp in {1,2,4} x bs in {1,2,4},
from a simple analytic timing model. Values are synthetic but shaped like
real profiles (embedding layer cheap, transformer layers equal, head layer
vocab-heavy; times scale ~linearly in bs and ~1/tp with a fixed overhead).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from metis_amd.profiles import ProfileStore

NUM_LAYERS = 10   # embed + 8 transformer + head
HIDDEN = 4096
SEQ = 1024
VOCAB = 51200

# parameters per layer (bytes, fp16-ish): embed = vocab*hidden*2, transformer
# ~= 12*hidden^2*2, head tied to embed
P_EMBED = VOCAB * HIDDEN * 2
P_BLOCK = 12 * HIDDEN * HIDDEN * 2
PARAMS = [P_EMBED] + [P_BLOCK] * (NUM_LAYERS - 2) + [P_EMBED]


def layer_times(speed: float, tp: int, bs: int):
    """ms per layer: t = work/(speed*tp) + overhead."""
    t_block = (20.0 * bs) / (speed * tp) + 1.5
    t_embed = (3.0 * bs) / (speed * tp) + 0.5
    t_head = (8.0 * bs) / (speed * tp) + 0.8
    return [t_embed] + [t_block] * (NUM_LAYERS - 2) + [t_head]


def layer_memory(tp: int, bs: int):
    m_block = 900.0 * bs / tp + 300.0
    m_embed = 420.0 / tp + 80.0 * bs
    m_head = 900.0 / tp + 160.0 * bs
    return [m_embed] + [m_block] * (NUM_LAYERS - 2) + [m_head]


def main(out_dir: str, device_speeds=None) -> None:
    os.makedirs(out_dir, exist_ok=True)
    for dtype, speed in device_speeds or (("MI355X", 1.0), ("MI355X_LC", 0.5)):
        for tp in (1, 2, 4):
            for bs in (1, 2, 4):
                times = layer_times(speed, tp, bs)
                mems = layer_memory(tp, bs)
                fwd_bwd = sum(times) * 1.08  # fb_sync residual = 8%
                ProfileStore.write_profile_json(
                    os.path.join(out_dir, f"DeviceType.{dtype}_tp{tp}_bs{bs}.json"),
                    model_name="GPT-synth",
                    parameters_per_layer_bytes=[float(p) for p in PARAMS],
                    total_time_ms=fwd_bwd + 12.0,
                    forward_backward_time_ms=fwd_bwd,
                    batch_generator_time_ms=1.0,
                    layernorm_grads_all_reduce_time_ms=0.4,
                    embedding_grads_all_reduce_time_ms=0.9,
                    optimizer_time_ms=10.0 / tp / speed,
                    layer_compute_total_ms=times,
                    total_memory_mb=sum(mems),
                    layer_memory_total_mb=mems,
                )
    print(f"wrote synthetic profiles to {out_dir}")


if __name__ == "__main__":
    out = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "tests", "data", "profiles_synth",
    )
    main(out)
