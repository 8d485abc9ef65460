"""Memory-needs estimators for ZeRO configs on MI355X (288 GB HBM3E).

Parity: reference `deepspeed/runtime/zero/stage3.py`
(estimate_zero3_model_states_mem_needs_all_live) and stage_1_and_2
estimator. Prints per-config device/host bytes for model states only
(params+grads+optimizer); activations are workload-dependent.
"""

GB = 1 << 30


def _fmt(nbytes):
    return f"{nbytes / GB:7.2f}GB"


def estimate_zero2_model_states_mem_needs(total_params, num_gpus_per_node=8,
                                          num_nodes=1, cpu_offload=False,
                                          additional_buffer_factor=1.5):
    world = num_gpus_per_node * num_nodes
    if cpu_offload:
        gpu = 2 * total_params  # bf16 params (+transient grads)
        cpu = total_params * 12 / world * world  # fp32 master+m+v on host
    else:
        gpu = 2 * total_params + 2 * total_params + \
            total_params * 12 / world
        cpu = total_params * additional_buffer_factor  # pinned staging
    return int(gpu), int(cpu)


def estimate_zero3_model_states_mem_needs(total_params, largest_layer_params,
                                          num_gpus_per_node=8, num_nodes=1,
                                          cpu_offload=False,
                                          cpu_offload_params=False,
                                          additional_buffer_factor=1.5):
    world = num_gpus_per_node * num_nodes
    gathered = 2 * largest_layer_params * 2  # live + prefetch
    if cpu_offload:
        gpu = 2 * total_params / world + gathered
        cpu = total_params * (12 if not cpu_offload_params else 14)
    else:
        gpu = (2 + 2 + 12) * total_params / world + gathered
        cpu = total_params * additional_buffer_factor / world
    return int(gpu), int(cpu)


def estimate_zero3_model_states_mem_needs_all_live(
        model, num_gpus_per_node=8, num_nodes=1,
        additional_buffer_factor=1.5):
    total = sum(p.numel() for p in model.parameters())
    largest = max((sum(p.numel() for p in m.parameters(recurse=False))
                   for m in model.modules()), default=0)
    print(f"Estimates for {total/1e9:.2f}B params "
          f"({largest/1e6:.0f}M largest layer) on "
          f"{num_nodes}x{num_gpus_per_node} MI355X (288 GB HBM3E each):")
    header = f"{'config':<34}{'per-GPU HBM':>14}{'host DRAM':>14}"
    print(header)
    for offload, offp, name in ((False, False, "zero3"),
                                (True, False, "zero3 + offload_optimizer"),
                                (True, True, "zero3 + offload opt+params")):
        gpu, cpu = estimate_zero3_model_states_mem_needs(
            total, largest, num_gpus_per_node, num_nodes,
            cpu_offload=offload, cpu_offload_params=offp,
            additional_buffer_factor=additional_buffer_factor)
        flag = "" if gpu < 288 * GB else "  [exceeds 288 GB]"
        print(f"{name:<34}{_fmt(gpu):>14}{_fmt(cpu):>14}{flag}")
    return total, largest
