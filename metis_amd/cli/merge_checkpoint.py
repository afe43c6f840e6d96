"""Merge a distributed (dp/tp/pp) checkpoint into one tp=pp=1 state dict.

Closes the train -> serve loop: cli/train.py writes one shard per rank
(rank{r}.pt with the plan embedded); this tool reassembles the full
model so cli/serve.py (or a tp=1 fine-tune) can load it.

  python -m metis_amd.cli.merge_checkpoint \
      --model gpt2-small --dir ckpts/step_1000 --out merged.pt

Merge rules (GPT family; inverse of the Megatron-style sharding in
models/gpt.py):
- column-parallel (fc1, head): concatenate output rows across tp ranks;
  qkv keeps its [q|k|v] block layout, so each block's head-slices are
  concatenated per tp rank within each of q/k/v;
- row-parallel (proj, fc2): concatenate input columns; bias replicated;
- norms/embeddings/everything else: replicated, take tp rank 0;
- pp (and interleaved vpp) stages: each virtual stage contributes its
  blocks at its layer_partition offset; embeddings from stage 0, head
  from the last stage.
"""

from __future__ import annotations

import argparse
import os
from typing import Dict

import torch

from metis_amd.models.gpt import GPTModel, MODEL_SPECS


def _merge_tp(key: str, parts, num_heads: int) -> torch.Tensor:
    """Merge one parameter's tp shards (parts in tp-rank order)."""
    if len(parts) == 1:
        return parts[0]
    name = key.rsplit(".", 2)[-2:]  # (module, param)
    mod, param = name[0], name[1]
    if mod == "qkv":
        # per-rank layout [q|k|v] blocks of hp rows each: regroup so the
        # merged tensor is [all q | all k | all v]
        merged = []
        for blk in range(3):
            for p in parts:
                hp = p.size(0) // 3
                merged.append(p[blk * hp:(blk + 1) * hp])
        return torch.cat(merged, dim=0)
    if mod in ("fc1", "head"):
        return torch.cat(parts, dim=0)           # column-parallel rows
    if mod in ("proj", "fc2"):
        if param == "weight":
            return torch.cat(parts, dim=1)       # row-parallel columns
        return parts[0]                          # replicated bias
    return parts[0]                              # replicated


def merge_checkpoint(model_name: str, ckpt_dir: str) -> Dict[str, torch.Tensor]:
    spec = MODEL_SPECS[model_name]
    ranks = {}
    for f in sorted(os.listdir(ckpt_dir)):
        if f.startswith("rank") and f.endswith(".pt"):
            state = torch.load(os.path.join(ckpt_dir, f),
                               map_location="cpu", weights_only=True)
            ranks[state["rank"]] = state
    assert ranks, f"no rank*.pt files in {ckpt_dir}"
    plan = ranks[0]["plan"]
    dp, tp, pp = plan["dp"], plan["tp"], plan["pp"]
    vpp = plan.get("vpp", 1)
    partition = plan["layer_partition"]
    total = spec.profile_num_layers

    out: Dict[str, torch.Tensor] = {}
    for p in range(pp):
        # dp replicas are identical; take dp_rank 0, all tp ranks of stage p
        stage_tp = [ranks[(p * dp + 0) * tp + t]["model"] for t in range(tp)]
        for c in range(vpp):
            vs = c * pp + p
            start, end = partition[vs], partition[vs + 1]
            prefix = f"{c}." if vpp > 1 else ""
            first_block = max(start - 1, 0)
            keys = {k for k in stage_tp[0] if k.startswith(prefix)}
            for k in sorted(keys):
                bare = k[len(prefix):]
                parts = [sd[k] for sd in stage_tp]
                merged = _merge_tp(bare, parts, spec.num_heads)
                if bare.startswith("blocks."):
                    _, idx, rest = bare.split(".", 2)
                    gidx = first_block + int(idx)
                    out[f"blocks.{gidx}.{rest}"] = merged
                else:
                    out[bare] = merged
    # sanity: the merged dict must load into a tp=1 full model
    full = GPTModel(spec, tp=1, dtype=next(iter(out.values())).dtype)
    full.load_state_dict(out)
    return out


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", required=True, choices=sorted(MODEL_SPECS))
    p.add_argument("--dir", required=True, help="step_N checkpoint directory")
    p.add_argument("--out", required=True)
    args = p.parse_args()
    state = merge_checkpoint(args.model, args.dir)
    torch.save({"model": state,
                "plan": {"dp": 1, "tp": 1, "pp": 1, "mbs": 1, "gbs": 1}},
               args.out)
    n = sum(v.numel() for v in state.values())
    print(f"wrote {args.out} ({len(state)} tensors, {n:,} params)")


if __name__ == "__main__":
    main()
