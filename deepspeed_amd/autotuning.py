"""Autotuner: searches ZeRO stage / micro-batch space with short profiled
runs.

Parity: reference `autotuning/autotuner.py:42` (Autotuner.tune,
run_tuning_micro_batch_sizes). In-process variant: each candidate config
trains `steps` steps on synthetic data and is ranked by throughput;
returns (best_config, results).
"""
import copy
import gc
import time

import torch

from .utils.logging import log_dist


class Autotuner:
    def __init__(self, model_fn, data_fn, base_config, steps=5, warmup=2):
        """model_fn() -> fresh nn.Module; data_fn() -> (inputs, labels) or
        a dict of kwargs for engine(**kwargs)."""
        self.model_fn = model_fn
        self.data_fn = data_fn
        self.base_config = base_config
        self.steps = steps
        self.warmup = warmup
        self.results = []

    def _run_one(self, config):
        import deepspeed_amd
        model = self.model_fn()
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        try:
            def step():
                batch = self.data_fn(config)
                if isinstance(batch, dict):
                    loss = engine(**batch)
                else:
                    loss = engine(*batch)
                engine.backward(loss)
                engine.step()

            for _ in range(self.warmup):
                step()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(self.steps):
                step()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            elapsed = (time.time() - t0) / self.steps
            return elapsed
        finally:
            engine.destroy()
            del engine, model
            gc.collect()
            if torch.cuda.is_available():
                torch.cuda.empty_cache()

    def tune(self, micro_batches=(1, 2, 4, 8), stages=(1, 2, 3)):
        best = None
        for stage in stages:
            for mb in micro_batches:
                cfg = copy.deepcopy(self.base_config)
                cfg["train_micro_batch_size_per_gpu"] = mb
                cfg.setdefault("zero_optimization", {})["stage"] = stage
                try:
                    sec = self._run_one(cfg)
                    tput = mb / sec
                    self.results.append(
                        {"stage": stage, "micro_batch": mb,
                         "sec_per_step": sec, "samples_per_sec": tput})
                    log_dist(f"autotune: stage={stage} mb={mb} "
                             f"{sec*1000:.1f} ms/step", ranks=[0])
                    if best is None or tput > best[0]:
                        best = (tput, cfg)
                except (RuntimeError, torch.cuda.OutOfMemoryError) as e:
                    self.results.append({"stage": stage, "micro_batch": mb,
                                         "error": str(e)[:200]})
                    if torch.cuda.is_available():
                        torch.cuda.empty_cache()
        if best is None:
            raise RuntimeError("autotuning: no candidate succeeded")
        return best[1], self.results
