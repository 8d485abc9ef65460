"""Reconstruct a full fp32 state_dict from ZeRO checkpoint shards.

Parity: reference `deepspeed/utils/zero_to_fp32.py`
(`get_fp32_state_dict_from_zero_checkpoint:563`,
`convert_zero_checkpoint_to_fp32_state_dict:628`). Works offline on a
checkpoint directory written by deepspeed_amd (stages 1/2/3).
"""
import argparse
import glob
import os
from collections import OrderedDict

import torch


def _read_tag(ckpt_dir, tag):
    if tag is None:
        latest = os.path.join(ckpt_dir, "latest")
        if os.path.exists(latest):
            with open(latest) as f:
                tag = f.read().strip()
        else:
            raise FileNotFoundError(f"no 'latest' file in {ckpt_dir}")
    return os.path.join(ckpt_dir, str(tag))


def _load_zero_shards(dirpath, mp_rank=None):
    all_files = glob.glob(os.path.join(
        dirpath, "zero_pp_rank_*_mp_rank_*_optim_states.pt"))
    mp_ranks = sorted({os.path.basename(f).split("_")[6]
                       for f in all_files})
    if mp_rank is None:
        if len(mp_ranks) > 1:
            # TP shards hold DIFFERENT parameters: dp-concat across mp
            # ranks would interleave garbage. Offline TP-slice merging
            # is not implemented — reconstruct per mp rank.
            raise NotImplementedError(
                f"checkpoint has {len(mp_ranks)} tensor-parallel shards;"
                " pass mp_rank= to reassemble one TP shard at a time")
        mp_rank = int(mp_ranks[0]) if mp_ranks else 0
    files = sorted((f for f in all_files
                    if os.path.basename(f).split("_")[6]
                    == f"{mp_rank:02d}"),
                   key=lambda f: int(os.path.basename(f).split("_")[3]))
    if not files:
        if glob.glob(os.path.join(dirpath, "*_pp_rank_*.pt")):
            raise NotImplementedError(
                "pipeline-parallel checkpoint (stage-qualified files): "
                "offline cross-stage reassembly is not implemented — "
                "resume with the same pipeline topology instead")
        raise FileNotFoundError(f"no zero shard files in {dirpath}")
    return [torch.load(f, map_location="cpu", weights_only=False)
            ["optimizer_state_dict"] for f in files]


def _reassemble(shards, key="flat"):
    """Yield (name, fp32_tensor) from per-rank optimizer state dicts."""
    layout = shards[0]["shard_layout"]
    if layout["kind"] == "bucket":  # stages 1/2
        flats_per_rank = [sd["single_partition_of_fp32_groups"]
                          for sd in shards]
        for bi, binfo in enumerate(layout["buckets"]):
            if binfo.get("expert"):
                # Expert buckets are partitioned over the expert-DP group
                # (each EP rank holds DIFFERENT experts): concatenating
                # across all dp-rank files would interleave unrelated
                # experts. Those params are sourced from the
                # expert_ep_rank model-state files instead.
                continue
            full = torch.cat([flats[bi].detach().float()
                              for flats in flats_per_rank])
            for name, off, numel, shape in binfo["params"]:
                yield name, full[off:off + numel].reshape(shape).clone()
    elif layout["kind"] == "subgroup":  # stage 3
        import re
        W = len(shards)
        expat = re.compile(r"(.*deepspeed_experts\.)(\d+)(\..*)")
        emitted = set()
        for r, sd in enumerate(shards):
            lay = sd["shard_layout"]
            # local-expert count per Experts path (for global expert ids)
            counts = {}
            for g in lay["subgroups"]:
                for e in g["params"]:
                    m = expat.match(e[0])
                    if m:
                        counts.setdefault(m.group(1),
                                          set()).add(int(m.group(2)))
            counts = {k: len(v) for k, v in counts.items()}
            for gi, ginfo in enumerate(lay["subgroups"]):
                for e in ginfo["params"]:
                    name, off, shard_numel, full_numel, shape = e[:5]
                    gw = e[5] if len(e) > 5 else W
                    if gw >= W:  # dense: same on every rank, emit once
                        if r != 0:
                            continue
                        ranks = range(W)
                        out = name
                    else:
                        # expert param: sharded over the expert-DP group
                        # [epoff, epoff+ep, ...]; module-local expert ids
                        # map to GLOBAL ids (epoff*num_local + local)
                        ep = W // gw
                        epoff = r % ep
                        if r // ep != 0:
                            continue  # first member of the group emits
                        ranks = [epoff + k * ep for k in range(gw)]
                        m = expat.match(name)
                        out = (f"{m.group(1)}"
                               f"{epoff * counts[m.group(1)] + int(m.group(2))}"
                               f"{m.group(3)}") if m else f"{name}@ep{epoff}"
                    if out in emitted:
                        continue
                    emitted.add(out)
                    pieces = [shards[rr]["fp32_flat_groups"][gi].detach()
                              .float()[off:off + shard_numel]
                              for rr in ranks]
                    yield out, torch.cat(pieces)[:full_numel] \
                        .reshape(shape).clone()
    else:
        raise ValueError(f"unknown layout kind {layout['kind']}")


def get_fp32_state_dict_from_zero_checkpoint(checkpoint_dir, tag=None):
    dirpath = _read_tag(checkpoint_dir, tag)
    shards = _load_zero_shards(dirpath)
    state_dict = OrderedDict()
    for name, tensor in _reassemble(shards):
        state_dict[name] = tensor
    # merge non-sharded buffers from the model states file
    model_files = glob.glob(os.path.join(dirpath,
                                         "mp_rank_*_model_states.pt"))
    if model_files:
        ms = torch.load(model_files[0], map_location="cpu",
                        weights_only=False)
        for bname in ms.get("buffer_names", []):
            if bname in ms["module"]:
                state_dict[bname] = ms["module"][bname]
    # MoE (stages 1/2): expert params come from the per-EP-rank expert
    # files — each carries its own experts' 16-bit weights (ref
    # _save_moe_checkpoint, engine.py:4921)
    for efile in sorted(glob.glob(os.path.join(
            dirpath, "expert_ep_rank_*_model_states.pt"))):
        esd = torch.load(efile, map_location="cpu", weights_only=False)
        for name, t in esd["module"].items():
            state_dict[name] = t.detach().float()
    return state_dict


def convert_zero_checkpoint_to_fp32_state_dict(checkpoint_dir,
                                               output_file, tag=None):
    sd = get_fp32_state_dict_from_zero_checkpoint(checkpoint_dir, tag)
    torch.save(sd, output_file)
    print(f"saved fp32 state_dict ({len(sd)} entries) to {output_file}")
    return sd


def main():
    p = argparse.ArgumentParser()
    p.add_argument("checkpoint_dir")
    p.add_argument("output_file")
    p.add_argument("-t", "--tag", default=None)
    args = p.parse_args()
    convert_zero_checkpoint_to_fp32_state_dict(args.checkpoint_dir,
                                               args.output_file, args.tag)


if __name__ == "__main__":
    main()
