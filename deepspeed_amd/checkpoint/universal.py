"""Universal checkpoint: topology-independent per-parameter format.

Parity: reference `deepspeed/checkpoint/ds_to_universal.py`
(`extract_zero_shards:124`, stage-3 variant @164) and
`universal_checkpoint.py:149` (load_hp_checkpoint_state).

Format: <out>/zero/<param_name>/{fp32,exp_avg,exp_avg_sq}.pt + meta.pt —
convert once offline, then resume into ANY dp world size (elastic resume).
"""
import os
from collections import OrderedDict

import torch

from ..utils.zero_to_fp32 import _load_zero_shards, _read_tag

# Adam's moments by default; other fused optimizers (Lion: exp_avg only,
# Adagrad: sq_accum, Muon: momentum) are discovered dynamically from the
# base optimizer state at convert/load time.
_DEFAULT_STATE_KEYS = ("exp_avg", "exp_avg_sq")


def _discover_state_keys(base_states):
    keys = set()
    for bs in base_states:
        for st in bs.get("state", {}).values():
            keys.update(k for k, v in st.items()
                        if torch.is_tensor(v) and v.dim() >= 1)
    return tuple(sorted(keys)) or _DEFAULT_STATE_KEYS


def _iter_params_with_states(shards):
    """Yield (name, {'fp32': t, 'exp_avg': t, 'exp_avg_sq': t})."""
    layout = shards[0]["shard_layout"]
    base_states = [sd["base_optimizer_state"] for sd in shards]
    state_keys = _discover_state_keys(base_states)

    def master_state(rank, master_idx, key):
        st = base_states[rank]["state"].get(master_idx, {})
        return st.get(key)

    if layout["kind"] == "bucket":
        W = len(shards)
        flats_per_rank = [sd["single_partition_of_fp32_groups"]
                          for sd in shards]

        def assemble(bi, ranks, rbinfo, qual=""):
            full = {"fp32": torch.cat([flats_per_rank[r][bi].detach()
                                       .float() for r in ranks])}
            for key in state_keys:
                parts = [master_state(r, bi, key) for r in ranks]
                if all(p is not None for p in parts):
                    full[key] = torch.cat([p.float() for p in parts])
            for name, off, numel, shape in rbinfo["params"]:
                yield name + qual, {k: v[off:off + numel].reshape(shape)
                                    .clone() for k, v in full.items()}

        for bi, binfo in enumerate(layout["buckets"]):
            if binfo.get("expert"):
                # expert buckets partition over the expert-DP group
                # (stride ep); each EP offset holds DIFFERENT experts
                # under the SAME module-local names, so entries are
                # qualified "@ep<off>" (load resolves by rank % ep)
                edp = binfo.get("world", W)
                ep = max(W // edp, 1)
                for off in range(ep):
                    ranks = [off + k * ep for k in range(edp)]
                    rbinfo = shards[ranks[0]]["shard_layout"]["buckets"][bi]
                    yield from assemble(bi, ranks, rbinfo, f"@ep{off}")
            else:
                yield from assemble(bi, list(range(W)), binfo)
    else:  # subgroup / stage 3
        W = len(shards)
        flats_per_rank = [sd["fp32_flat_groups"] for sd in shards]

        def emit(gi, ranks, entry, qual=""):
            name, off, shard_numel, full_numel, shape = entry[:5]
            out = {"fp32": torch.cat(
                [flats_per_rank[r][gi].detach()
                 .float()[off:off + shard_numel]
                 for r in ranks])[:full_numel].reshape(shape).clone()}
            for key in state_keys:
                parts = [master_state(r, gi, key) for r in ranks]
                if all(p is not None for p in parts):
                    out[key] = torch.cat(
                        [p.float()[off:off + shard_numel]
                         for p in parts])[:full_numel] \
                        .reshape(shape).clone()
            return name + qual, out

        for r, sd in enumerate(shards):
            lay = sd["shard_layout"]
            for gi, ginfo in enumerate(lay["subgroups"]):
                for entry in ginfo["params"]:
                    gw = entry[5] if len(entry) > 5 else W
                    if gw >= W:  # dense param: emit once, from rank 0
                        if r == 0:
                            yield emit(gi, range(W), entry)
                    else:
                        # expert param: sharded over the expert-DP group
                        # (stride ep); qualified "@ep<off>" like the
                        # stage-1/2 expert buckets
                        ep = W // gw
                        epoff = r % ep
                        if r // ep != 0:
                            continue
                        ranks = [epoff + k * ep for k in range(gw)]
                        yield emit(gi, ranks, entry, f"@ep{epoff}")


def ds_to_universal(checkpoint_dir, output_dir, tag=None):
    """Convert a (dp-sharded) checkpoint into universal format."""
    dirpath = _read_tag(checkpoint_dir, tag)
    shards = _load_zero_shards(dirpath)
    zero_dir = os.path.join(output_dir, "zero")
    os.makedirs(zero_dir, exist_ok=True)
    names = []
    for name, tensors in _iter_params_with_states(shards):
        pdir = os.path.join(zero_dir, name)
        os.makedirs(pdir, exist_ok=True)
        for key, t in tensors.items():
            torch.save(t, os.path.join(pdir, f"{key}.pt"))
        names.append(name)
    # meta: step counts per group
    steps = [g.get("step", 0)
             for g in shards[0]["base_optimizer_state"]["param_groups"]]
    keys = _discover_state_keys(
        [sd["base_optimizer_state"] for sd in shards])
    torch.save({"param_names": names, "group_steps": steps,
                "state_keys": list(keys),
                "source_world": shards[0].get("partition_count", 1)},
               os.path.join(output_dir, "meta.pt"))
    return names


def _load_param(universal_dir, name, state_keys=_DEFAULT_STATE_KEYS):
    pdir = os.path.join(universal_dir, "zero", name)
    out = {}
    for key in ("fp32",) + tuple(state_keys):
        f = os.path.join(pdir, f"{key}.pt")
        if os.path.exists(f):
            out[key] = torch.load(f, map_location="cpu", weights_only=False)
    return out


def load_universal_into_optimizer(optimizer, universal_dir):
    """Scatter a universal checkpoint into the CURRENT topology.

    Works for ZeroStage12Optimizer (buckets) and ZeroStage3Optimizer
    (subgroups) at any dp world size.
    """
    meta = torch.load(os.path.join(universal_dir, "meta.pt"),
                      map_location="cpu", weights_only=False)
    state_keys = tuple(meta.get("state_keys", _DEFAULT_STATE_KEYS))
    layout = optimizer.shard_layout()
    world = layout["world"]
    base = optimizer.optimizer

    def set_state(master, key, flat_shard):
        st = base.state.setdefault(master, {})
        st[key] = flat_shard.to(master.device)

    if layout["kind"] == "bucket":
        rank = optimizer.rank
        for bi, (b, binfo) in enumerate(zip(optimizer.buckets,
                                            layout["buckets"])):
            shard_rank, qual = rank, ""
            if binfo.get("expert"):
                # this bucket shards over the expert-DP group: pick the
                # "@ep<off>" entries for this rank's EP offset and slice
                # by its position WITHIN the expert-DP group
                edp = binfo.get("world", world)
                ep = max(world // edp, 1)
                qual = f"@ep{rank % ep}"
                shard_rank = rank // ep
            full = {"fp32": torch.zeros(binfo["numel_padded"])}
            for key in state_keys:
                full[key] = torch.zeros(binfo["numel_padded"])
            for name, off, numel, shape in binfo["params"]:
                t = _load_param(universal_dir, name + qual, state_keys)
                if not t and qual:  # pre-qualifier universal dirs
                    t = _load_param(universal_dir, name, state_keys)
                for key in full:
                    if key in t:
                        full[key][off:off + numel] = t[key].reshape(-1)
            lo = shard_rank * binfo["shard_numel"]
            hi = lo + binfo["shard_numel"]
            b.master32.data.copy_(full["fp32"][lo:hi])
            master = b.master32
            for key in state_keys:
                set_state(master, key, full[key][lo:hi])
            b.shard16.copy_(b.master32.detach().to(b.shard16.dtype))
        from .. import comm as dist
        for b in optimizer.buckets:
            # expert buckets gather over their expert-DP group
            dist.all_gather_into_tensor(b.flat16, b.shard16,
                                        group=b.pg or optimizer.dp_group)
    else:  # stage 3 subgroups
        rank = optimizer.rank
        for gi, (sg, ginfo) in enumerate(zip(optimizer.sub_groups,
                                             layout["subgroups"])):
            fp32 = torch.zeros(sg.numel)
            states = {key: torch.zeros(sg.numel) for key in state_keys}
            for entry in ginfo["params"]:
                name, off, shard_numel, full_numel, shape = entry[:5]
                gw = entry[5] if len(entry) > 5 else world
                shard_rank, qual = rank, ""
                if gw < world:  # expert param: slice by expert-DP pos
                    ep = world // gw
                    qual = f"@ep{rank % ep}"
                    shard_rank = rank // ep
                t = _load_param(universal_dir, name + qual, state_keys)
                if not t and qual:
                    t = _load_param(universal_dir, name, state_keys)
                if "fp32" not in t:
                    continue
                flat = t["fp32"].reshape(-1)
                lo = shard_rank * shard_numel
                hi = min(lo + shard_numel, full_numel)
                if hi > lo:
                    fp32[off:off + hi - lo] = flat[lo:hi]
                for key in state_keys:
                    if key in t:
                        states[key][off:off + hi - lo] = \
                            t[key].reshape(-1)[lo:hi]
            sg.master32.data.copy_(fp32.to(sg.master32.device))
            for key in state_keys:
                set_state(sg.master32, key, states[key])
            sg.copy_master_to_shards()
        optimizer._refresh_persistent_params()

    # restore per-group step counters
    for g, step in zip(base.param_groups, meta.get("group_steps", [])):
        if step:
            g["step"] = step
