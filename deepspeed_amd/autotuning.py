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


def estimate_memory_per_gpu(n_params, stage, world=1, micro_batch=1,
                            seq_len=2048, hidden=4096, n_layers=32,
                            offload_optimizer=False, act_ckpt=True):
    """ZeRO memory model (ref autotuner model_info + memory estimators):
    bytes of HBM for params+grads+optimizer+activations per GPU."""
    P = n_params
    opt = 12 * P  # fp32 master + 2 Adam moments
    if stage >= 3:
        weights = 2 * P / world
        grads = 4 * P / world   # fp32 shard accumulators
        opt = opt / world
    elif stage == 2:
        weights = 2 * P
        grads = 2 * P / world + 2 * P
        opt = opt / world
    elif stage == 1:
        weights = 2 * P
        grads = 2 * P
        opt = opt / world
    else:
        weights = 2 * P
        grads = 2 * P
        opt = opt
    if offload_optimizer:
        opt = 0
    act = micro_batch * seq_len * hidden * 2 * \
        (2 if act_ckpt else n_layers)
    return weights + grads + opt + act


class ModelBasedTuner:
    """Saturating-throughput cost model over micro-batch (ref
    tuner/model_based_tuner.py role, without xgboost): fit
    tput(mb) = mb / (a + b*mb) from observed runs — a = fixed per-step
    overhead, b = per-sample cost — and propose the best untried mb."""

    def __init__(self):
        self.obs = []  # (mb, sec_per_step)

    def record(self, mb, sec):
        self.obs.append((float(mb), float(sec)))

    def fit(self):
        # sec(mb) = a + b*mb — least squares on the observations
        import numpy as np
        if len(self.obs) < 2:
            return None
        x = np.array([o[0] for o in self.obs])
        y = np.array([o[1] for o in self.obs])
        A = np.stack([np.ones_like(x), x], axis=1)
        (a, b), *_ = np.linalg.lstsq(A, y, rcond=None)
        return float(a), float(b)

    def predict_tput(self, mb):
        f = self.fit()
        if f is None:
            return None
        a, b = f
        return mb / max(a + b * mb, 1e-9)

    def propose(self, candidates):
        """Untried candidate with the best predicted throughput."""
        tried = {o[0] for o in self.obs}
        untried = [c for c in candidates if float(c) not in tried]
        if not untried:
            return None
        f = self.fit()
        if f is None:
            return untried[0]
        return max(untried, key=self.predict_tput)


class AutotunerFull(Autotuner):
    """Memory-pruned, cost-model-guided search with on-disk experiment
    records (ref autotuner.py:42 + scheduler.py roles)."""

    def __init__(self, model_fn, data_fn, base_config, steps=5, warmup=2,
                 results_dir=None, model_info=None):
        super().__init__(model_fn, data_fn, base_config, steps, warmup)
        self.results_dir = results_dir
        self.model_info = model_info or {}

    def _feasible(self, stage, mb):
        info = self.model_info
        if not info.get("num_params"):
            return True
        if torch.cuda.is_available():
            cap = torch.cuda.get_device_properties(0).total_memory * 0.92
        else:
            cap = info.get("memory_per_gpu", 64e9)
        need = estimate_memory_per_gpu(
            info["num_params"], stage,
            world=info.get("world", 1), micro_batch=mb,
            seq_len=info.get("seq_len", 2048),
            hidden=info.get("hidden_size", 4096),
            n_layers=info.get("num_layers", 32),
            offload_optimizer=bool(
                self.base_config.get("zero_optimization", {})
                .get("offload_optimizer")))
        return need <= cap

    def _record_exp(self, rec):
        self.results.append(rec)
        if self.results_dir:
            import json
            import os
            os.makedirs(self.results_dir, exist_ok=True)
            i = len(self.results)
            with open(f"{self.results_dir}/exp_{i:03d}.json", "w") as f:
                json.dump(rec, f)

    def tune(self, micro_batches=(1, 2, 4, 8), stages=(1, 2, 3),
             max_experiments=None):
        best = None
        n_run = 0
        for stage in stages:
            tuner = ModelBasedTuner()
            # seed with the smallest and largest feasible micro-batch,
            # then follow the cost model
            feas = [mb for mb in micro_batches if self._feasible(stage, mb)]
            for mb in feas:
                if mb not in (feas[0], feas[-1]) and len(feas) > 2:
                    continue
                if max_experiments and n_run >= max_experiments:
                    break
                self._try(stage, mb, tuner)
                n_run += 1
            while True:
                if max_experiments and n_run >= max_experiments:
                    break
                nxt = tuner.propose(feas)
                if nxt is None:
                    break
                self._try(stage, nxt, tuner)
                n_run += 1
            for r in self.results:
                if r.get("stage") == stage and "samples_per_sec" in r:
                    if best is None or r["samples_per_sec"] > best[0]:
                        cfg = copy.deepcopy(self.base_config)
                        cfg["train_micro_batch_size_per_gpu"] = \
                            r["micro_batch"]
                        cfg.setdefault("zero_optimization",
                                       {})["stage"] = stage
                        best = (r["samples_per_sec"], cfg)
        if best is None:
            raise RuntimeError("autotuning: no candidate succeeded")
        return best[1], self.results

    def _try(self, stage, mb, tuner):
        cfg = copy.deepcopy(self.base_config)
        cfg["train_micro_batch_size_per_gpu"] = mb
        cfg.setdefault("zero_optimization", {})["stage"] = stage
        try:
            sec = self._run_one(cfg)
            tuner.record(mb, sec)
            self._record_exp({"stage": stage, "micro_batch": mb,
                              "sec_per_step": sec,
                              "samples_per_sec": mb / sec})
        except (RuntimeError, torch.cuda.OutOfMemoryError) as e:
            self._record_exp({"stage": stage, "micro_batch": mb,
                              "error": str(e)[:200]})
